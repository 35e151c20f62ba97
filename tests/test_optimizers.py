"""Sparse-optimizer semantics tests (every optimizer × EV), modeled on the
reference's embedding_variable_ops_test.py optimizer sweep."""
import math

import pytest
import torch

from deeprec_amd import EmbeddingVariable, RaggedIds, embedding_lookup_sparse
from deeprec_amd.optimizers import (
    AdagradDecayOptimizer, AdagradOptimizer, AdamAsyncOptimizer,
    AdamOptimizer, AdamWOptimizer, FtrlOptimizer, GradientDescentOptimizer,
    make_optimizer,
)

ALL_OPTS = [
    ("sgd", GradientDescentOptimizer, {}),
    ("adagrad", AdagradOptimizer, {}),
    ("adagraddecay", AdagradDecayOptimizer, {"accumulator_decay_step": 2}),
    ("adam", AdamOptimizer, {}),
    ("adamasync", AdamAsyncOptimizer, {}),
    ("adamw", AdamWOptimizer, {}),
    ("ftrl", FtrlOptimizer, {}),
]


def _one_step(opt_cls, kw, steps=3, dim=4):
    ev = EmbeddingVariable(f"ev_{opt_cls.__name__}_{steps}", dim)
    opt = opt_cls(embedding_variables=[ev], learning_rate=0.1, **kw)
    sp = RaggedIds.from_lists([[1, 2], [2, 3]])
    w0 = ev.gather(torch.tensor([1, 2, 3])).clone()
    for _ in range(steps):
        out = embedding_lookup_sparse(ev, sp, combiner="sum")
        loss = (out ** 2).sum()
        loss.backward()
        opt.step()
    w1 = ev.gather(torch.tensor([1, 2, 3]))
    return w0, w1


@pytest.mark.parametrize("name,cls,kw", ALL_OPTS)
def test_optimizer_updates_weights(name, cls, kw):
    w0, w1 = _one_step(cls, kw)
    assert not torch.allclose(w0, w1), f"{name} made no update"
    assert torch.isfinite(w1).all()


def test_adagrad_formula():
    ev = EmbeddingVariable("ev_ag_formula", 2)
    opt = AdagradOptimizer(embedding_variables=[ev], learning_rate=0.1,
                           initial_accumulator_value=0.1)
    sp = RaggedIds.from_lists([[5]])
    out = embedding_lookup_sparse(ev, sp, combiner="sum")
    w0 = ev.gather(torch.tensor([5])).clone()
    out.sum().backward()  # grad = 1
    opt.step()
    w1 = ev.gather(torch.tensor([5]))
    expected = w0 - 0.1 * 1.0 / math.sqrt(0.1 + 1.0)
    torch.testing.assert_close(w1, expected)


def test_adam_formula():
    ev = EmbeddingVariable("ev_adam_formula", 2)
    opt = AdamOptimizer(embedding_variables=[ev], learning_rate=0.01)
    sp = RaggedIds.from_lists([[5]])
    out = embedding_lookup_sparse(ev, sp, combiner="sum")
    w0 = ev.gather(torch.tensor([5])).clone()
    out.sum().backward()
    opt.step()
    w1 = ev.gather(torch.tensor([5]))
    # t=1: m=(1-b1), v=(1-b2); lr_t = lr*sqrt(1-b2)/(1-b1); update = lr_t*m/(sqrt(v)+eps)
    m, v = 0.1, 0.001
    lr_t = 0.01 * math.sqrt(1 - 0.999) / (1 - 0.9)
    expected = w0 - lr_t * m / (math.sqrt(v) + 1e-8)
    torch.testing.assert_close(w1, expected)


def test_dense_and_sparse_together():
    ev = EmbeddingVariable("ev_mix", 4)
    lin = torch.nn.Linear(4, 1)
    opt = AdamOptimizer(params=lin.parameters(), embedding_variables=[ev],
                        learning_rate=0.01)
    sp = RaggedIds.from_lists([[1], [2]])
    w_dense0 = lin.weight.detach().clone()
    out = lin(embedding_lookup_sparse(ev, sp, combiner="sum"))
    out.sum().backward()
    opt.step()
    assert not torch.allclose(lin.weight.detach(), w_dense0)


def test_unadmitted_keys_skipped():
    from deeprec_amd import CounterFilter, EmbeddingVariableOption
    opt_ev = EmbeddingVariableOption(filter_option=CounterFilter(filter_freq=10))
    ev = EmbeddingVariable("ev_filter_skip", 4, ev_option=opt_ev)
    opt = AdagradOptimizer(embedding_variables=[ev], learning_rate=0.1)
    sp = RaggedIds.from_lists([[1]])
    out = embedding_lookup_sparse(ev, sp, combiner="sum")
    out.sum().backward()
    opt.step()  # must not crash; key 1 not admitted
    assert ev.size() == 0


def test_make_optimizer_factory():
    for name, cls, _ in ALL_OPTS:
        o = make_optimizer(name, learning_rate=0.1)
        assert isinstance(o, cls)


def test_multiple_lookups_same_ev_per_step():
    ev = EmbeddingVariable("ev_multi", 4)
    opt = GradientDescentOptimizer(embedding_variables=[ev], learning_rate=1.0)
    sp = RaggedIds.from_lists([[7]])
    w0 = ev.gather(torch.tensor([7])).clone()
    out1 = embedding_lookup_sparse(ev, sp, combiner="sum")
    out2 = embedding_lookup_sparse(ev, sp, combiner="sum")
    (out1.sum() + out2.sum()).backward()
    opt.step()
    w1 = ev.gather(torch.tensor([7]))
    torch.testing.assert_close(w1, w0 - 2.0)  # two applies of grad 1


def test_ftrl_v2_l2_shrinkage():
    """FtrlV2: l2_shrinkage feeds the linear term but not the accumulator
    (reference: KvResourceSparseApplyFtrlV2)."""
    import torch
    from deeprec_amd.embedding import EmbeddingVariable
    from deeprec_amd.optimizers import FtrlOptimizer

    torch.manual_seed(0)

    def run(shrink):
        from deeprec_amd.embedding.variable import reset_registry
        reset_registry()
        ev = EmbeddingVariable(f"ftrlv2_{shrink}", 4, device="cpu")
        opt = FtrlOptimizer(embedding_variables=[ev], learning_rate=0.1,
                            l2_shrinkage_regularization_strength=shrink)
        ids = torch.tensor([1, 2, 3])
        for _ in range(3):
            from deeprec_amd.embedding import embedding_lookup
            out = embedding_lookup(ev, ids, train=True)
            loss = (out ** 2).sum()
            loss.backward()
            opt.step()
        k, v, _, _ = ev.export()
        return v[torch.argsort(k)]

    v0 = run(0.0)
    v1 = run(0.5)
    assert not torch.allclose(v0, v1), \
        "l2_shrinkage must change the trajectory"


def test_gradient_clipping_by_norm():
    """clip_norm: per-gradient tf.clip_by_norm semantics on both the
    dense params and the sparse EV grad rows (reference: DIN/DIEN
    train.py wraps compute_gradients with clip_by_norm(grad, 5))."""
    import torch
    from deeprec_amd.embedding import (EmbeddingVariable, RaggedIds,
                                       embedding_lookup_sparse)
    from deeprec_amd.optimizers import GradientDescentOptimizer

    ev = EmbeddingVariable("clip/ev", 4)
    dense = torch.nn.Linear(4, 1, bias=False)
    opt = GradientDescentOptimizer(params=dense.parameters(),
                                   embedding_variables=[ev],
                                   learning_rate=1.0, clip_norm=0.5)
    ids = RaggedIds(torch.tensor([1, 2, 3, 4]), torch.arange(0, 5))
    before = ev.gather(torch.tensor([1, 2, 3, 4])).clone()
    w_before = dense.weight.detach().clone()
    out = dense(embedding_lookup_sparse(ev, ids, combiner="sum"))
    (out.sum() * 100).backward()  # huge grads
    opt.step()
    # dense update bounded by lr * clip_norm
    assert float((dense.weight.detach() - w_before).norm()) <= 0.5 + 1e-5
    # each sparse row moved, but the whole grad tensor norm was clipped
    after = ev.gather(torch.tensor([1, 2, 3, 4]))
    moved = (after - before).norm()
    assert 0 < float(moved) <= 0.5 + 1e-5
    # attribute form works for subclasses with custom __init__
    from deeprec_amd.optimizers import AdagradOptimizer
    o2 = AdagradOptimizer(embedding_variables=[ev], learning_rate=0.1)
    o2.clip_norm = 5.0
    assert o2.clip_norm == 5.0
