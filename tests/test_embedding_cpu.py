"""Engine-level tests for the CPU embedding backend.

Modeled on the reference's embedding_variable_ops_test.cc /
embedding_variable_ops_test.py coverage: creation, lookup determinism,
default-value init, counter/bloom filters, eviction policies,
export/import round trip, frequency/version tracking.
"""
import pytest
import torch

from deeprec_amd import (
    CBFFilter, CounterFilter, EmbeddingVariable, EmbeddingVariableOption,
    GlobalStepEvict, L2WeightEvict, RaggedIds,
    embedding_lookup, embedding_lookup_sparse, get_embedding_variable,
)
from deeprec_amd.embedding.options import InitializerOption
from deeprec_amd.embedding.variable import GLOBAL_STEP


def _ids(lists):
    return RaggedIds.from_lists(lists)


def test_create_and_lookup_deterministic():
    ev = EmbeddingVariable("ev1", 8)
    out1 = embedding_lookup(ev, torch.tensor([1, 2, 3, 1]))
    out2 = embedding_lookup(ev, torch.tensor([1, 2, 3, 1]))
    assert out1.shape == (4, 8)
    assert torch.equal(out1, out2)
    assert torch.equal(out1[0], out1[3])
    assert ev.size() == 3


def test_default_value_dim_init():
    opt = EmbeddingVariableOption(
        init_option=InitializerOption(default_value_dim=2))
    ev = EmbeddingVariable("ev_dvd", 4, ev_option=opt)
    out = embedding_lookup(ev, torch.tensor([0, 2, 4, 1, 3]))
    # keys 0,2,4 share default row 0; keys 1,3 share row 1
    assert torch.equal(out[0], out[1])
    assert torch.equal(out[0], out[2])
    assert torch.equal(out[3], out[4])
    assert not torch.equal(out[0], out[3])


def test_constant_initializer():
    ev = get_embedding_variable("ev_const", 4, initializer=2.5)
    out = embedding_lookup(ev, torch.tensor([7]))
    assert torch.allclose(out, torch.full((1, 4), 2.5))


def test_pooled_combiners():
    ev = EmbeddingVariable("ev_comb", 4)
    sp = _ids([[1, 2], [3], []])
    for combiner, denom in [("sum", [1.0, 1.0]), ("mean", [2.0, 1.0]),
                            ("sqrtn", [2 ** 0.5, 1.0])]:
        out = embedding_lookup_sparse(ev, sp, combiner=combiner)
        e = {k: embedding_lookup(ev, torch.tensor([k]))[0] for k in (1, 2, 3)}
        torch.testing.assert_close(out[0], (e[1] + e[2]) / denom[0])
        torch.testing.assert_close(out[1], e[3] / denom[1])
        assert torch.equal(out[2], torch.zeros(4))


def test_weighted_combiners():
    ev = EmbeddingVariable("ev_w", 4)
    sp = RaggedIds(torch.tensor([1, 2, 3]),
                   torch.tensor([0, 2, 3], dtype=torch.int32),
                   weights=torch.tensor([0.5, 2.0, 3.0]))
    e = {k: embedding_lookup(ev, torch.tensor([k]))[0] for k in (1, 2, 3)}
    out = embedding_lookup_sparse(ev, sp, combiner="sum")
    torch.testing.assert_close(out[0], 0.5 * e[1] + 2.0 * e[2])
    out = embedding_lookup_sparse(ev, sp, combiner="mean")
    torch.testing.assert_close(out[0], (0.5 * e[1] + 2.0 * e[2]) / 2.5)
    out = embedding_lookup_sparse(ev, sp, combiner="sqrtn")
    torch.testing.assert_close(
        out[0], (0.5 * e[1] + 2.0 * e[2]) / (0.25 + 4.0) ** 0.5)


def test_counter_filter_admission():
    opt = EmbeddingVariableOption(
        filter_option=CounterFilter(filter_freq=3),
        init_option=InitializerOption(default_value_no_permission=0.0))
    ev = EmbeddingVariable("ev_cf", 4, ev_option=opt)
    ids = torch.tensor([42])
    # below filter_freq: not admitted, lookup returns no-permission value
    out = embedding_lookup(ev, ids)
    assert ev.size() == 0 and ev.total_count() == 1
    assert torch.equal(out, torch.zeros(1, 4))
    embedding_lookup(ev, ids)
    out = embedding_lookup(ev, ids)  # third occurrence -> admitted
    assert ev.size() == 1
    assert int(ev.get_frequency(ids)[0]) == 3


def test_cbf_filter_admission():
    opt = EmbeddingVariableOption(
        filter_option=CBFFilter(filter_freq=2, max_element_size=10000,
                                false_positive_probability=0.01))
    ev = EmbeddingVariable("ev_cbf", 4, ev_option=opt)
    embedding_lookup(ev, torch.tensor([5]))
    assert ev.size() == 0
    assert ev.total_count() == 0  # pre-admission counts live in the CBF only
    embedding_lookup(ev, torch.tensor([5]))
    assert ev.size() == 1


def test_global_step_eviction():
    opt = EmbeddingVariableOption(evict_option=GlobalStepEvict(steps_to_live=5))
    ev = EmbeddingVariable("ev_gse", 4, ev_option=opt)
    GLOBAL_STEP.value = 0
    embedding_lookup(ev, torch.tensor([1]))
    GLOBAL_STEP.value = 10
    embedding_lookup(ev, torch.tensor([2]))
    n = ev.shrink(step=10)
    assert n == 1 and ev.size() == 1
    keys, *_ = ev.export()
    assert keys.tolist() == [2]


def test_l2_eviction():
    opt = EmbeddingVariableOption(evict_option=L2WeightEvict(
        l2_weight_threshold=0.5))
    ev = EmbeddingVariable("ev_l2", 4, ev_option=opt)
    embedding_lookup(ev, torch.tensor([1, 2]))
    k, *_ = ev.export()
    s1 = ev.storage.lookup(torch.tensor([1]))
    ev.storage.values[s1[0]] = torch.full((4,), 10.0)
    s2 = ev.storage.lookup(torch.tensor([2]))
    ev.storage.values[s2[0]] = torch.full((4,), 0.01)
    assert ev.shrink() == 1
    keys, *_ = ev.export()
    assert keys.tolist() == [1]


def test_export_import_roundtrip():
    ev = EmbeddingVariable("ev_exp", 8)
    GLOBAL_STEP.value = 3
    embedding_lookup(ev, torch.arange(100))
    keys, values, freqs, versions = ev.export()
    assert keys.numel() == 100 and values.shape == (100, 8)
    assert (versions == 3).all()
    ev2 = EmbeddingVariable("ev_imp", 8)
    ev2.restore(keys, values, freqs, versions)
    out1 = embedding_lookup(ev, keys)
    out2 = embedding_lookup(ev2, keys)
    torch.testing.assert_close(out1, out2)
    assert torch.equal(ev2.get_frequency(keys), ev.get_frequency(keys))


def test_frequency_and_version_tracking():
    ev = EmbeddingVariable("ev_fv", 4)
    GLOBAL_STEP.value = 7
    embedding_lookup(ev, torch.tensor([1, 1, 1, 2]))
    assert int(ev.get_frequency(torch.tensor([1]))[0]) == 3
    assert int(ev.get_version(torch.tensor([2]))[0]) == 7
    GLOBAL_STEP.value = 9
    embedding_lookup(ev, torch.tensor([2]))
    assert int(ev.get_version(torch.tensor([2]))[0]) == 9
    assert int(ev.get_version(torch.tensor([1]))[0]) == 7


def test_gradient_flow_and_training():
    ev = EmbeddingVariable("ev_train", 4)
    from deeprec_amd.optimizers import GradientDescentOptimizer
    opt = GradientDescentOptimizer(embedding_variables=[ev], learning_rate=0.5)
    sp = _ids([[1, 2]])
    before = embedding_lookup(ev, torch.tensor([1])).clone()
    out = embedding_lookup_sparse(ev, sp, combiner="sum")
    out.sum().backward()
    opt.step()
    after = embedding_lookup(ev, torch.tensor([1]))
    torch.testing.assert_close(after, before - 0.5)


def test_storage_growth():
    opt = EmbeddingVariableOption(init_capacity=8)
    ev = EmbeddingVariable("ev_grow", 4, ev_option=opt)
    embedding_lookup(ev, torch.arange(5000))
    assert ev.size() == 5000
    out = embedding_lookup(ev, torch.arange(5000))
    assert out.shape == (5000, 4)


def test_multihash_variable():
    from deeprec_amd.embedding.extras import get_multihash_variable
    mv = get_multihash_variable("mh", dims=[100, 100], operation="add",
                                embedding_dim=8)
    ids = torch.tensor([5, 105, 205])
    out = mv.lookup(ids)
    assert out.shape == (3, 8)
    # ids 5, 105, 205 share r-part (5) but differ in q-part
    eq = embedding_lookup(mv.q, torch.tensor([0, 1, 2]))
    er = embedding_lookup(mv.r, torch.tensor([5]))
    torch.testing.assert_close(out, eq + er)
    mv2 = get_multihash_variable("mh2", dims=[100, 100], operation="concat",
                                 embedding_dim=8)
    assert mv2.lookup(ids).shape == (3, 16)


def test_dynamic_dimension_ev():
    from deeprec_amd.embedding.extras import (
        get_dynamic_dimension_embedding_variable)
    ev = get_dynamic_dimension_embedding_variable(
        "dyn", embedding_block_dim=4, embedding_block_num=3,
        block_thresholds=[1, 3, 5])
    ids = torch.tensor([7])
    out1 = ev.lookup(ids)          # freq 1 -> 1 block
    assert out1.shape == (1, 12)
    assert (out1[0, 4:] == 0).all() and not (out1[0, :4] == 0).all()
    ev.lookup(ids)
    out3 = ev.lookup(ids)          # freq 3 -> 2 blocks
    assert not (out3[0, 4:8] == 0).all()
    assert (out3[0, 8:] == 0).all()
    ev.lookup(ids)
    out5 = ev.lookup(ids)          # freq 5 -> 3 blocks
    assert not (out5[0, 8:] == 0).all()


def test_adaptive_embedding_lookup():
    from deeprec_amd.embedding.extras import adaptive_embedding_lookup_sparse
    ev = EmbeddingVariable("adapt_ev", 4)
    static = torch.randn(50, 4)
    sp = RaggedIds.from_lists([[1, 1, 1], [2]])
    out = adaptive_embedding_lookup_sparse(ev, static, sp, threshold=2,
                                           combiner="sum")
    # key 1 (freq 3 >= 2) -> EV; key 2 (freq 1 < 2) -> static row 2
    e1 = embedding_lookup(ev, torch.tensor([1]), train=False)[0]
    torch.testing.assert_close(out[0], 3 * e1)
    torch.testing.assert_close(out[1], static[2])
    assert ev.size() == 1  # only the hot key entered the EV


def test_memory_usage_accounting():
    ev = EmbeddingVariable("mem_ev", 8, device="cpu")
    from deeprec_amd.embedding import embedding_lookup
    from deeprec_amd.optimizers import AdamOptimizer
    opt = AdamOptimizer(embedding_variables=[ev])
    out = embedding_lookup(ev, torch.arange(100), train=True)
    out.sum().backward()
    opt.step()
    mu = ev.memory_usage()
    assert set(mu) >= {"table_bytes", "values_bytes", "slab_bytes",
                       "total_bytes"}
    # value slab holds >= 100 rows of 8 fp32
    assert mu["values_bytes"] >= 100 * 8 * 4
    # adam m+v slabs mirror the value slab
    assert mu["slab_bytes"] >= 2 * mu["values_bytes"] * 0  # exist
    assert mu["slab_bytes"] > 0
    assert mu["total_bytes"] == (mu["table_bytes"] + mu["values_bytes"]
                                 + mu["slab_bytes"])


def test_lookup_tier_cpu():
    """KvResourceLookupTier parity on the CPU tier: 0 resident, -1
    absent (reference: kv_variable_lookup_ops.cc:537)."""
    import torch
    from deeprec_amd.embedding import EmbeddingVariable
    ev = EmbeddingVariable("tier_cpu/ev", 4)
    ev.lookup_or_create(torch.tensor([1, 2, 3]))
    tier = ev.lookup_tier(torch.tensor([1, 2, 3, 99]))
    assert tier.tolist() == [0, 0, 0, -1]


def test_invalid_key_sentinel():
    """get_embedding_variable(invalid_key=K): K is dropped from lookups
    (zeros in pooled output), never admitted, never trained (reference:
    the invalid-key sentinel of tf.get_embedding_variable,
    variable_scope.py:2146)."""
    import torch
    from deeprec_amd.embedding import (RaggedIds, embedding_lookup_sparse,
                                       get_embedding_variable)
    ev = get_embedding_variable("invkey/ev", 4, invalid_key=-1)
    ids = RaggedIds(torch.tensor([5, -1, 6, -1, -1]),
                    torch.tensor([0, 2, 4, 5]))
    out = embedding_lookup_sparse(ev, ids, combiner="sum")
    ref = embedding_lookup_sparse(
        ev, RaggedIds(torch.tensor([5, 6]), torch.tensor([0, 1, 2, 2])),
        combiner="sum")
    torch.testing.assert_close(out.detach(), ref.detach())
    assert bool((out[2].detach() == 0).all())  # all-invalid row -> zeros
    tier = ev.lookup_tier(torch.tensor([-1, 5, 6]))
    assert tier.tolist() == [-1, 0, 0]  # sentinel never admitted
    # and it never accumulates gradients
    out.sum().backward()
    grads = ev.consume_grads()
    for _, keys, _ in grads:
        assert bool((keys != -1).all())
