"""Model-zoo smoke tests: 2 training steps per model on CPU."""
import pytest
import torch

from deeprec_amd.data.synthetic import CriteoSyntheticDataset
from deeprec_amd.models import MODEL_REGISTRY, SEQUENCE_MODELS
from deeprec_amd.optimizers import AdagradOptimizer

NON_SEQ = sorted(set(MODEL_REGISTRY) - SEQUENCE_MODELS - {"wide_and_deep"})


@pytest.mark.parametrize("name", NON_SEQ)
def test_model_trains(name):
    torch.manual_seed(0)
    m = MODEL_REGISTRY[name](device="cpu", bf16=False)
    ds = CriteoSyntheticDataset(batch_size=32, seed=3, matrix_format=True)
    opt = AdagradOptimizer(params=m.parameters(),
                           embedding_variables=m.embedding_variables(),
                           learning_rate=0.01)
    losses = []
    for i in range(2):
        dense, ids, labels = ds.next_batch()
        out = m(dense, ids)
        loss = m.loss_fn(out, labels)
        opt.zero_grad()
        loss.backward()
        opt.step()
        assert torch.isfinite(loss), name
        losses.append(float(loss))
    assert m.collection.size() > 0


@pytest.mark.parametrize("name", sorted(SEQUENCE_MODELS))
def test_sequence_model_trains(name):
    torch.manual_seed(0)
    m = MODEL_REGISTRY[name](device="cpu", bf16=False)
    ds = CriteoSyntheticDataset(batch_size=16, seed=4)
    opt = AdagradOptimizer(params=m.parameters(),
                           embedding_variables=m.embedding_variables(),
                           learning_rate=0.01)
    for i in range(2):
        dense, ids, seq, target, labels = ds.next_seq_batch(seq_len=20)
        out = m(dense, ids[:, :m.num_sparse], seq, target)
        loss = m.loss_fn(out, labels)
        opt.zero_grad()
        loss.backward()
        opt.step()
        assert torch.isfinite(loss), name
    assert m.item_ev.size() > 0


def test_dien_auxiliary_loss():
    """DIEN trains with the reference's auxiliary next-item loss
    (modelzoo/dien/train.py:231-251): enabled by default, masked to
    valid step pairs, added to the main objective, gradients flow
    through the GRU states; disabled => plain BCE."""
    import torch
    from deeprec_amd.data.synthetic import CriteoSyntheticDataset
    from deeprec_amd.models.sequence import DIEN

    torch.manual_seed(0)
    ds = CriteoSyntheticDataset(batch_size=16, seed=2)
    dense, ids, seq, target, labels = ds.next_seq_batch(seq_len=12)
    ids = ids[:, :10]

    m = DIEN(device="cpu", bf16=False)
    logits = m(dense, ids, seq, target)
    assert m._aux_loss is not None
    assert float(m._aux_loss.detach()) > 0
    loss_aux = m.loss_fn(logits, labels)
    assert m._aux_loss is None  # consumed by loss_fn

    torch.manual_seed(0)
    m2 = DIEN(device="cpu", bf16=False, use_aux_loss=False)
    logits2 = m2(dense, ids, seq, target)
    loss_plain = m2.loss_fn(logits2, labels)
    assert float(loss_aux) > float(loss_plain)  # aux term really added

    loss_aux.backward()
    assert any(p.grad is not None and p.grad.abs().sum() > 0
               for p in m.aux_net.parameters())
    assert any(p.grad is not None for p in m.gru.parameters())

    # eval path: no aux computation, no stale carryover
    m.zero_grad(set_to_none=True) if hasattr(m, "zero_grad") else None
    _ = m(dense, ids, seq, target, train=False)
    assert m._aux_loss is None
