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
