"""EmbeddingCollection (multi-table composite-key) tests vs per-EV path."""
import torch

from deeprec_amd import (
    EmbeddingVariable, EmbeddingVariableOption, RaggedIds,
    embedding_lookup_sparse,
)
from deeprec_amd.embedding.collection import EmbeddingCollection
from deeprec_amd.embedding.options import InitializerOption


def _const_opt(v=1.0, dvd=4):
    return EmbeddingVariableOption(
        init_option=InitializerOption(initializer=v, default_value_dim=dvd))


def test_collection_matches_per_ev_lookup():
    torch.manual_seed(0)
    n_tables, dim, batch = 3, 8, 16

    def init(t):
        gen = torch.Generator().manual_seed(42)
        t.normal_(0, 1, generator=gen)

    # collection with one big default matrix; per-EV references use the
    # matching block of the same matrix
    opt_c = EmbeddingVariableOption(
        init_option=InitializerOption(initializer=init, default_value_dim=4))
    coll = EmbeddingCollection("coll", [f"t{i}" for i in range(n_tables)],
                               dim, ev_option=opt_c,
                               combiners=["mean", "sum", "sqrtn"])
    evs = []
    for t in range(n_tables):
        block = coll.storage.default_values[t * 4:(t + 1) * 4].clone()

        def init_t(x, block=block):
            x.copy_(block)

        opt_e = EmbeddingVariableOption(
            init_option=InitializerOption(initializer=init_t,
                                          default_value_dim=4))
        evs.append(EmbeddingVariable(f"ref_t{t}", dim, ev_option=opt_e))

    sp_list = [RaggedIds.from_lists(
        [torch.randint(0, 30, (int(torch.randint(0, 5, ())),)).tolist()
         for _ in range(batch)]) for _ in range(n_tables)]
    out = coll.lookup(sp_list)
    assert out.shape == (batch, n_tables * dim)
    for t, combiner in enumerate(["mean", "sum", "sqrtn"]):
        ref = embedding_lookup_sparse(evs[t], sp_list[t], combiner=combiner)
        torch.testing.assert_close(out[:, t * dim:(t + 1) * dim], ref,
                                   rtol=1e-5, atol=1e-6)


def test_collection_training_matches_per_ev():
    from deeprec_amd.optimizers import AdagradOptimizer
    torch.manual_seed(1)
    dim, batch = 4, 8
    coll = EmbeddingCollection("coll_t", ["a", "b"], dim,
                               ev_option=_const_opt(),
                               combiners=["sum", "sum"])
    ev_a = EmbeddingVariable("ref_a", dim, ev_option=_const_opt())
    ev_b = EmbeddingVariable("ref_b", dim, ev_option=_const_opt())
    opt_c = AdagradOptimizer(embedding_variables=[coll], learning_rate=0.1)
    opt_e = AdagradOptimizer(embedding_variables=[ev_a, ev_b],
                             learning_rate=0.1)
    for step in range(3):
        lists_a = [torch.randint(0, 10, (2,)).tolist() for _ in range(batch)]
        lists_b = [torch.randint(0, 10, (3,)).tolist() for _ in range(batch)]
        sp_a, sp_b = RaggedIds.from_lists(lists_a), RaggedIds.from_lists(lists_b)
        out = coll.lookup([sp_a, sp_b])
        loss_c = (out ** 2).sum()
        loss_c.backward()
        opt_c.step()
        oa = embedding_lookup_sparse(ev_a, sp_a, combiner="sum")
        ob = embedding_lookup_sparse(ev_b, sp_b, combiner="sum")
        loss_e = (torch.cat([oa, ob], dim=1) ** 2).sum()
        loss_e.backward()
        opt_e.step()
        torch.testing.assert_close(loss_c, loss_e)
    tabs = coll.export_tables()
    ka, va, fa, _ = tabs["a"]
    ref_keys, ref_vals, ref_freqs, _ = ev_a.export()
    order_c = torch.argsort(ka)
    order_r = torch.argsort(ref_keys)
    torch.testing.assert_close(ka[order_c], ref_keys[order_r])
    torch.testing.assert_close(va[order_c], ref_vals[order_r],
                               rtol=1e-5, atol=1e-6)
    assert torch.equal(fa[order_c], ref_freqs[order_r])


def test_collection_weighted():
    coll = EmbeddingCollection("coll_w", ["a"], 4, ev_option=_const_opt(),
                               combiners=["mean"])
    ev = EmbeddingVariable("ref_w", 4, ev_option=_const_opt())
    w = torch.tensor([0.5, 2.0, 3.0])
    sp = RaggedIds(torch.tensor([1, 2, 3]),
                   torch.tensor([0, 2, 3], dtype=torch.int32), weights=w)
    out = coll.lookup([sp])
    ref = embedding_lookup_sparse(ev, sp, combiner="mean")
    torch.testing.assert_close(out, ref, rtol=1e-5, atol=1e-6)


def test_collection_export_restore_roundtrip():
    coll = EmbeddingCollection("coll_rt", ["a", "b"], 4,
                               ev_option=_const_opt())
    sp = RaggedIds.from_lists([[1, 2, 3]])
    coll.lookup([sp, sp])
    tabs = coll.export_tables()
    coll2 = EmbeddingCollection("coll_rt2", ["a", "b"], 4,
                                ev_option=_const_opt())
    for t, name in enumerate(["a", "b"]):
        coll2.restore_table(t, *tabs[name])
    out1 = coll.lookup([sp, sp], train=False)
    out2 = coll2.lookup([sp, sp], train=False)
    torch.testing.assert_close(out1, out2)


def test_dlrm_with_collection_cpu():
    from deeprec_amd.data.synthetic import CriteoSyntheticDataset
    from deeprec_amd.models.dlrm import DLRM
    from deeprec_amd.optimizers import AdamOptimizer
    torch.manual_seed(0)
    m = DLRM(device="cpu", bf16=False, use_collection=True)
    assert m.collection is not None
    ds = CriteoSyntheticDataset(batch_size=64, seed=7)
    opt = AdamOptimizer(params=m.parameters(),
                        embedding_variables=m.embedding_variables())
    for i, (dense, sparse, labels) in enumerate(ds):
        if i >= 3:
            break
        loss = m.loss_fn(m(dense, sparse), labels)
        opt.zero_grad()
        loss.backward()
        opt.step()
        assert torch.isfinite(loss)
    assert m.collection.size() > 0


def test_lookup_matrix_matches_ragged():
    torch.manual_seed(5)
    coll = EmbeddingCollection("coll_m", ["a", "b", "c"], 8,
                               ev_option=_const_opt(dvd=8))
    ids = torch.randint(0, 50, (16, 3))
    out_m = coll.lookup_matrix(ids)
    sp_list = [RaggedIds.from_dense(ids[:, t:t + 1]) for t in range(3)]
    out_r = coll.lookup(sp_list, train=False)
    torch.testing.assert_close(out_m, out_r)


def test_lookup_matrix_training():
    from deeprec_amd.optimizers import AdagradOptimizer
    coll = EmbeddingCollection("coll_mt", ["a", "b"], 4,
                               ev_option=_const_opt())
    opt = AdagradOptimizer(embedding_variables=[coll], learning_rate=0.1)
    ids = torch.tensor([[1, 2], [1, 3]])
    out = coll.lookup_matrix(ids)
    out.sum().backward()
    opt.step()
    out2 = coll.lookup_matrix(ids, train=False)
    assert not torch.allclose(out, out2)
