"""MI355X engine tests: HIP hash table + fused kernels vs the fp32 CPU
reference backend (ops/cpu_backend.py). All marked gpu."""
import math

import pytest
import torch

pytestmark = pytest.mark.gpu

from deeprec_amd import (  # noqa: E402
    CounterFilter, EmbeddingVariable, EmbeddingVariableOption,
    GlobalStepEvict, L2WeightEvict, RaggedIds,
    embedding_lookup, embedding_lookup_sparse,
)
from deeprec_amd.embedding.options import InitializerOption  # noqa: E402
from deeprec_amd.embedding.variable import GLOBAL_STEP  # noqa: E402

DEV = "cuda:0"


def _pair(name, dim=16, **opt_kw):
    """Matching GPU and CPU EVs with identical default values."""
    opt_g = EmbeddingVariableOption(**opt_kw)
    opt_c = EmbeddingVariableOption(**opt_kw)
    gen = torch.Generator().manual_seed(99)
    ev_g = EmbeddingVariable(f"{name}_g", dim, ev_option=opt_g, device=DEV,
                             generator=gen)
    gen2 = torch.Generator().manual_seed(99)
    ev_c = EmbeddingVariable(f"{name}_c", dim, ev_option=opt_c, device="cpu",
                             generator=gen2)
    torch.testing.assert_close(ev_g.storage.default_values.cpu(),
                               ev_c.storage.default_values)
    return ev_g, ev_c


def test_hash_insert_lookup_roundtrip():
    ev, _ = _pair("ht_rt")
    keys = torch.randperm(100000, device=DEV)[:50000].to(torch.int64)
    slots = ev.storage.lookup_or_create(
        keys, torch.ones_like(keys, dtype=torch.int32), step=1)
    assert int((slots >= 0).sum()) == keys.numel()
    assert slots.unique().numel() == keys.numel()  # distinct slots
    slots2 = ev.storage.lookup(keys)
    assert torch.equal(slots.cpu(), slots2.cpu())
    missing = ev.storage.lookup(torch.tensor([10**12], device=DEV))
    assert int(missing[0]) == -1
    assert ev.size() == keys.numel()


def test_gather_matches_cpu():
    ev_g, ev_c = _pair("gather")
    ids = torch.randint(0, 1000, (4096,), dtype=torch.int64)
    out_g = embedding_lookup(ev_g, ids.to(DEV))
    out_c = embedding_lookup(ev_c, ids)
    torch.testing.assert_close(out_g.cpu(), out_c, rtol=1e-6, atol=1e-6)


@pytest.mark.parametrize("combiner", ["sum", "mean", "sqrtn"])
def test_pooled_forward_matches_cpu(combiner):
    ev_g, ev_c = _pair(f"pool_{combiner}")
    torch.manual_seed(0)
    lists = [torch.randint(0, 500, (int(torch.randint(0, 8, ())),)).tolist()
             for _ in range(512)]
    sp_c = RaggedIds.from_lists(lists)
    sp_g = sp_c.to(DEV)
    out_g = embedding_lookup_sparse(ev_g, sp_g, combiner=combiner)
    out_c = embedding_lookup_sparse(ev_c, sp_c, combiner=combiner)
    torch.testing.assert_close(out_g.cpu(), out_c, rtol=1e-5, atol=1e-5)


def test_pooled_weighted_matches_cpu():
    ev_g, ev_c = _pair("pool_w")
    vals = torch.randint(0, 100, (300,))
    offs = torch.arange(0, 301, 3, dtype=torch.int32)
    w = torch.rand(300) + 0.1
    for combiner in ("sum", "mean", "sqrtn"):
        out_c = embedding_lookup_sparse(
            ev_c, RaggedIds(vals, offs, w), combiner=combiner)
        out_g = embedding_lookup_sparse(
            ev_g, RaggedIds(vals, offs, w).to(DEV), combiner=combiner)
        torch.testing.assert_close(out_g.cpu(), out_c, rtol=1e-5, atol=1e-5)


def test_bf16_gather_output():
    ev, _ = _pair("bf16")
    sp = RaggedIds.from_lists([[1, 2], [3]]).to(DEV)
    out = embedding_lookup_sparse(ev, sp, combiner="mean",
                                  out_dtype=torch.bfloat16)
    assert out.dtype == torch.bfloat16
    ref = embedding_lookup_sparse(ev, sp, combiner="mean")
    torch.testing.assert_close(out.float().cpu(), ref.cpu(), rtol=1e-2,
                               atol=1e-2)


OPTS = ["sgd", "adagrad", "adagrad_decay", "adam", "adam_async", "adamw",
        "ftrl"]


@pytest.mark.parametrize("name", OPTS)
def test_sparse_apply_matches_cpu(name):
    from deeprec_amd.ops import sparse_optim_cpu as cpu_apply
    from deeprec_amd.ops.hip_backend import sparse_apply

    ev_g, ev_c = _pair(f"apply_{name}", dim=8)
    keys = torch.arange(64, dtype=torch.int64)
    counts = torch.ones(64, dtype=torch.int32)
    slots_g = ev_g.storage.lookup_or_create(keys.to(DEV), counts.to(DEV), 1)
    slots_c = ev_c.storage.lookup_or_create(keys, counts, 1)
    hyper_all = {
        "sgd": dict(lr=0.1),
        "adagrad": dict(lr=0.1, initial_accumulator=0.1, epsilon=0.0),
        "adagrad_decay": dict(lr=0.1, global_step=7, initial_accumulator=0.1,
                              accumulator_decay_step=2,
                              accumulator_decay_rate=0.9,
                              accumulator_baseline=0.01, epsilon=0.0),
        "adam": dict(lr=0.01, step_t=1, beta1=0.9, beta2=0.999,
                     epsilon=1e-8),
        "adam_async": dict(lr=0.01, beta1_power=0.9, beta2_power=0.999,
                           beta1=0.9, beta2=0.999, epsilon=1e-8,
                           sparse_rmsprop=False),
        "adamw": dict(lr=0.01, step_t=1, beta1=0.9, beta2=0.999,
                      epsilon=1e-8, weight_decay=0.02),
        "ftrl": dict(lr=0.1, l1=0.1, l2=0.01, lr_power=-0.5),
    }
    hyper = hyper_all[name]
    torch.manual_seed(1)
    for step in range(3):
        grad = torch.randn(64, 8)
        if name in ("adam", "adamw"):
            hyper["step_t"] = step + 1
        if name == "adagrad_decay":
            hyper["global_step"] = step * 3
        sparse_apply(name, ev_g.storage, slots_g, grad.to(DEV), dict(hyper))
        getattr(cpu_apply, f"apply_{name}")(ev_c.storage, slots_c, grad,
                                            **hyper)
    w_g = ev_g.storage.gather(keys.to(DEV), slots_g).cpu()
    w_c = ev_c.storage.gather(keys, slots_c)
    torch.testing.assert_close(w_g, w_c, rtol=1e-5, atol=1e-6)


def test_counter_filter_on_gpu():
    opt = EmbeddingVariableOption(filter_option=CounterFilter(filter_freq=3))
    ev = EmbeddingVariable("gpu_cf", 4, ev_option=opt, device=DEV)
    ids = torch.tensor([42], device=DEV)
    embedding_lookup(ev, ids)
    embedding_lookup(ev, ids)
    assert ev.size() == 0 and ev.total_count() == 1
    embedding_lookup(ev, ids)
    assert ev.size() == 1
    assert int(ev.get_frequency(ids)[0]) == 3


def test_growth_under_load():
    opt = EmbeddingVariableOption(init_capacity=1024)
    ev = EmbeddingVariable("gpu_grow", 8, ev_option=opt, device=DEV)
    for chunk in torch.arange(200000, dtype=torch.int64).split(37777):
        embedding_lookup(ev, chunk.to(DEV))
    assert ev.size() == 200000
    # lookups after growth still find the same values
    out = embedding_lookup(ev, torch.tensor([5, 123456], device=DEV))
    assert torch.isfinite(out).all()


def test_eviction_gpu():
    opt = EmbeddingVariableOption(evict_option=GlobalStepEvict(steps_to_live=5))
    ev = EmbeddingVariable("gpu_evict", 4, ev_option=opt, device=DEV)
    GLOBAL_STEP.value = 0
    embedding_lookup(ev, torch.arange(100, device=DEV))
    GLOBAL_STEP.value = 20
    embedding_lookup(ev, torch.arange(100, 150, device=DEV))
    n = ev.shrink(step=20)
    assert n == 100
    assert ev.size() == 50
    keys, values, freqs, versions = ev.export()
    assert sorted(keys.cpu().tolist()) == list(range(100, 150))


def test_export_import_roundtrip_gpu():
    ev_g, _ = _pair("gpu_exp")
    GLOBAL_STEP.value = 5
    embedding_lookup(ev_g, torch.arange(1000, device=DEV))
    keys, values, freqs, versions = ev_g.export()
    assert keys.numel() == 1000
    ev2 = EmbeddingVariable("gpu_imp", 16, device=DEV)
    ev2.restore(keys, values, freqs, versions)
    out1 = embedding_lookup(ev_g, keys)
    out2 = embedding_lookup(ev2, keys)
    torch.testing.assert_close(out1.cpu(), out2.cpu())
    torch.testing.assert_close(ev2.get_frequency(keys).cpu(),
                               ev_g.get_frequency(keys).cpu())


def test_end_to_end_training_matches_cpu():
    """3 DLRM-ish steps on GPU vs CPU with identical data and init."""
    from deeprec_amd.optimizers import AdagradOptimizer

    ev_g, ev_c = _pair("e2e", dim=8)
    opt_g = AdagradOptimizer(embedding_variables=[ev_g], learning_rate=0.1)
    opt_c = AdagradOptimizer(embedding_variables=[ev_c], learning_rate=0.1)
    torch.manual_seed(3)
    for step in range(3):
        lists = [torch.randint(0, 50, (3,)).tolist() for _ in range(16)]
        sp = RaggedIds.from_lists(lists)
        out_g = embedding_lookup_sparse(ev_g, sp.to(DEV), combiner="mean")
        out_c = embedding_lookup_sparse(ev_c, sp, combiner="mean")
        (out_g ** 2).sum().backward()
        (out_c ** 2).sum().backward()
        opt_g.step()
        opt_c.step()
    keys = torch.arange(50, dtype=torch.int64)
    w_g = ev_g.gather(keys.to(DEV)).cpu()
    w_c = ev_c.gather(keys)
    torch.testing.assert_close(w_g, w_c, rtol=1e-4, atol=1e-5)


@pytest.mark.gpu
def test_memory_usage_gpu():
    from deeprec_amd.embedding import EmbeddingVariable, embedding_lookup
    from deeprec_amd.optimizers import AdamAsyncOptimizer

    ev = EmbeddingVariable("mem_gpu", 16, device="cuda")
    opt = AdamAsyncOptimizer(embedding_variables=[ev])
    out = embedding_lookup(ev, torch.arange(500, device="cuda"),
                           train=True)
    out.sum().backward()
    opt.step()
    mu = ev.memory_usage()
    assert mu["values_bytes"] >= 500 * 16 * 4
    assert mu["slab_bytes"] >= 2 * mu["values_bytes"] // 2
    assert mu["total_bytes"] == sum(
        v for k, v in mu.items() if k != "total_bytes")


@pytest.mark.gpu
def test_bulk_insert_lookup_beyond_grid_cap():
    """Regression: kernels whose grid caps at 65535 blocks must
    grid-stride — a 120M-key rebalance rebuild silently lost every key
    past 16.7M (insert_bulk/lookup had no stride loop)."""
    from deeprec_amd.embedding.options import EmbeddingVariableOption
    from deeprec_amd.ops.hip_backend import HbmStorage

    n = 20_000_000  # > 65535 * 256
    st = HbmStorage(8, EmbeddingVariableOption(init_capacity=1 << 26),
                    device="cuda")
    st._grow_slots(n)
    keys = torch.arange(n, dtype=torch.int64, device="cuda")
    slots = torch.arange(n, dtype=torch.int32, device="cuda")
    st.ext.ht_insert_bulk(keys, slots, torch.Tensor(), torch.Tensor(),
                          st.ht_keys, st.ht_slot, st.ht_freq,
                          st.ht_version, st.entry_counter, st.error_flag)
    st._check_error()
    assert st.total_count() == n
    # probe the TAIL (the region the capped grid used to drop)
    probe = torch.arange(n - 1_000_000, n, dtype=torch.int64,
                         device="cuda")
    got = st.lookup(probe)
    assert bool((got >= 0).all()), "tail keys lost by bulk insert"
    torch.testing.assert_close(got.long(), probe - (n - 1_000_000)
                               + (n - 1_000_000))
