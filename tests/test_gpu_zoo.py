"""GPU model-zoo smoke: every model trains 2 steps on MI355X in bf16."""
import pytest
import torch

pytestmark = pytest.mark.gpu

from deeprec_amd.data.synthetic import CriteoSyntheticDataset  # noqa: E402
from deeprec_amd.models import MODEL_REGISTRY, SEQUENCE_MODELS  # noqa: E402
from deeprec_amd.optimizers import AdamAsyncOptimizer  # noqa: E402

DEV = "cuda:0"
NON_SEQ = sorted(set(MODEL_REGISTRY) - SEQUENCE_MODELS - {"wide_and_deep"})


@pytest.mark.parametrize("name", NON_SEQ)
def test_model_gpu(name):
    torch.manual_seed(0)
    m = MODEL_REGISTRY[name](device=DEV, bf16=True)
    ds = CriteoSyntheticDataset(batch_size=512, seed=3, device=DEV,
                                matrix_format=True)
    opt = AdamAsyncOptimizer(params=m.parameters(),
                             embedding_variables=m.embedding_variables())
    for i in range(2):
        dense, ids, labels = ds.next_batch()
        loss = m.loss_fn(m(dense, ids), labels)
        opt.zero_grad()
        loss.backward()
        opt.step()
        assert torch.isfinite(loss), name


@pytest.mark.parametrize("name", sorted(SEQUENCE_MODELS))
def test_sequence_model_gpu(name):
    torch.manual_seed(0)
    m = MODEL_REGISTRY[name](device=DEV, bf16=True)
    ds = CriteoSyntheticDataset(batch_size=256, seed=4, device=DEV)
    opt = AdamAsyncOptimizer(params=m.parameters(),
                             embedding_variables=m.embedding_variables())
    for i in range(2):
        dense, ids, seq, target, labels = ds.next_seq_batch(seq_len=50)
        loss = m.loss_fn(m(dense, ids[:, :m.num_sparse], seq, target),
                         labels)
        opt.zero_grad()
        loss.backward()
        opt.step()
        assert torch.isfinite(loss), name


def test_checkpoint_roundtrip_gpu(tmp_path):
    from deeprec_amd.checkpoint.saver import Saver, latest_checkpoint
    from deeprec_amd.models.dlrm import DLRM
    torch.manual_seed(0)
    m = DLRM(device=DEV, bf16=True)
    ds = CriteoSyntheticDataset(batch_size=256, seed=5, device=DEV,
                                matrix_format=True)
    opt = AdamAsyncOptimizer(params=m.parameters(),
                             embedding_variables=m.embedding_variables())
    for _ in range(3):
        dense, ids, labels = ds.next_batch()
        loss = m.loss_fn(m(dense, ids), labels)
        opt.zero_grad()
        loss.backward()
        opt.step()
    saver = Saver(module=m, embedding_variables=m.embedding_variables(),
                  optimizer=opt)
    from deeprec_amd.embedding.variable import GLOBAL_STEP
    path = saver.save(str(tmp_path), GLOBAL_STEP.value)

    m2 = DLRM(device=DEV, bf16=True, name_prefix="dlrm_restored")
    m2.collection.name = m.collection.name
    # the default-value matrix comes from the initializer, not the ckpt
    # (reference semantics); align it so unseen keys compare equal too
    m2.collection.storage.default_values.copy_(
        m.collection.storage.default_values)
    saver2 = Saver(module=m2,
                   embedding_variables=m2.embedding_variables())
    saver2.restore(path)
    dense, ids, labels = ds.next_batch()
    out1 = m(dense, ids, train=False)
    out2 = m2(dense, ids, train=False)
    torch.testing.assert_close(out1, out2)
