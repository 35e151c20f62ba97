"""HBM_DRAM multi-tier storage tests: tiny hot tier forces cold-tier use;
results must match the single-tier CPU reference."""
import pytest
import torch

pytestmark = pytest.mark.gpu

from deeprec_amd import (  # noqa: E402
    EmbeddingVariable, EmbeddingVariableOption, RaggedIds, StorageOption,
    StorageType, embedding_lookup, embedding_lookup_sparse,
)
from deeprec_amd.embedding.options import InitializerOption  # noqa: E402

DEV = "cuda:0"


def _mt_option(hot_rows=64, dim=8, init=None, dvd=4):
    return EmbeddingVariableOption(
        storage_option=StorageOption(
            storage_type=StorageType.HBM_DRAM,
            storage_size=[hot_rows * dim * 4]),
        init_option=InitializerOption(initializer=init,
                                      default_value_dim=dvd))


def _pair(name, dim=8, hot_rows=64):
    def init(t):
        g = torch.Generator().manual_seed(13)
        t.normal_(0, 1, generator=g)

    ev_g = EmbeddingVariable(f"{name}_g", dim,
                             ev_option=_mt_option(hot_rows, dim, init),
                             device=DEV)
    opt_c = EmbeddingVariableOption(
        init_option=InitializerOption(initializer=init, default_value_dim=4))
    ev_c = EmbeddingVariable(f"{name}_c", dim, ev_option=opt_c, device="cpu")
    return ev_g, ev_c


def test_cold_tier_engaged_and_correct():
    ev_g, ev_c = _pair("mt1", hot_rows=64)
    ids = torch.arange(1000, dtype=torch.int64)  # 1000 keys >> 64 hot rows
    out_g = embedding_lookup(ev_g, ids.to(DEV))
    out_c = embedding_lookup(ev_c, ids)
    torch.testing.assert_close(out_g.cpu(), out_c, rtol=1e-6, atol=1e-6)
    st = ev_g.storage
    assert st.hot_rows == 64
    assert ev_g.size() == 1000  # 64 hot + 936 cold


def test_multitier_training_matches_cpu():
    from deeprec_amd.optimizers import AdagradOptimizer
    ev_g, ev_c = _pair("mt2", hot_rows=32)
    og = AdagradOptimizer(embedding_variables=[ev_g], learning_rate=0.1)
    oc = AdagradOptimizer(embedding_variables=[ev_c], learning_rate=0.1)
    torch.manual_seed(0)
    for step in range(3):
        lists = [torch.randint(0, 100, (3,)).tolist() for _ in range(32)]
        sp = RaggedIds.from_lists(lists)
        out_g = embedding_lookup_sparse(ev_g, sp.to(DEV), combiner="mean")
        out_c = embedding_lookup_sparse(ev_c, sp, combiner="mean")
        (out_g ** 2).sum().backward()
        (out_c ** 2).sum().backward()
        og.step()
        oc.step()
    keys = torch.arange(100, dtype=torch.int64)
    w_g = ev_g.gather(keys.to(DEV)).cpu()
    w_c = ev_c.gather(keys)
    torch.testing.assert_close(w_g, w_c, rtol=1e-4, atol=1e-5)


def test_multitier_export_import_shrink():
    from deeprec_amd.embedding.options import GlobalStepEvict
    from deeprec_amd.embedding.variable import GLOBAL_STEP
    opt = _mt_option(hot_rows=16, dim=4)
    opt.evict_option = GlobalStepEvict(steps_to_live=5)
    ev = EmbeddingVariable("mt3", 4, ev_option=opt, device=DEV)
    GLOBAL_STEP.value = 0
    embedding_lookup(ev, torch.arange(100, device=DEV))
    GLOBAL_STEP.value = 20
    embedding_lookup(ev, torch.arange(100, 140, device=DEV))
    keys, values, freqs, versions = ev.export()
    assert keys.numel() == 140
    n = ev.shrink(step=20)
    assert n == 100
    keys2, values2, *_ = ev.export()
    assert sorted(keys2.cpu().tolist()) == list(range(100, 140))
    # values preserved across the tier rebuild
    order1 = torch.argsort(keys.cpu())
    kept = keys.cpu()[order1] >= 100
    order2 = torch.argsort(keys2.cpu())
    torch.testing.assert_close(values2.cpu()[order2],
                               values.cpu()[order1][kept])


def test_multitier_ssd_path(tmp_path):
    """storage_path -> cold slabs are mmap files (SSD tier)."""
    import os
    opt = _mt_option(hot_rows=16, dim=4)
    opt.storage_option.storage_path = str(tmp_path / "emb")
    ev = EmbeddingVariable("mt_ssd", 4, ev_option=opt, device=DEV)
    embedding_lookup(ev, torch.arange(500, device=DEV))
    assert ev.size() == 500
    files = os.listdir(tmp_path / "emb")
    assert any(f.startswith("values-") for f in files)
    # values persisted through the mmap round-trip
    out1 = embedding_lookup(ev, torch.arange(500, device=DEV))
    assert torch.isfinite(out1).all()


@pytest.mark.gpu
def test_lfu_rebalance_promotes_hot_keys():
    """After rebalance(), the highest-frequency keys must occupy the HBM
    tier (slot < hot_rows) and all values must survive the repack."""
    from deeprec_amd.embedding.options import (EmbeddingVariableOption,
                                               StorageOption, StorageType)
    from deeprec_amd.ops.hbm_dram_backend import HbmDramStorage

    torch.manual_seed(0)
    dim = 8
    hot_rows = 32
    opt = EmbeddingVariableOption(storage_option=StorageOption(
        storage_type=StorageType.HBM_DRAM,
        storage_size=[hot_rows * dim * 4]))
    st = HbmDramStorage(dim, opt, device="cuda")
    n = 96
    keys = torch.arange(n, dtype=torch.int64, device="cuda")
    # slot order WITHIN one batched insert is race-ordered, so force the
    # misplacement deterministically: insert 64 keys first (they own the
    # whole 32-row hot tier + 32 cold), then the 32 soon-to-be-hot keys
    # (guaranteed cold slots 64..95)
    st.lookup_or_create(keys[:64],
                        torch.ones(64, dtype=torch.int32, device="cuda"),
                        step=0)
    st.lookup_or_create(keys[64:],
                        torch.ones(32, dtype=torch.int32, device="cuda"),
                        step=0)
    slots0 = st.lookup(keys)
    assert bool((slots0[64:] >= hot_rows).all())
    before = st.materialize(keys, slots0).cpu()
    # make the LAST 32 keys the hottest (high counts -> high freq)
    hotkeys = keys[64:]
    for _ in range(5):
        st.lookup_or_create(hotkeys,
                            torch.full((32,), 50, dtype=torch.int32,
                                       device="cuda"), step=1)
    moved = st.rebalance()
    assert moved == 64  # 32 promoted + 32 demoted
    slots1 = st.lookup(keys)
    # hottest keys now in the HBM tier
    assert bool((slots1[64:] < hot_rows).all())
    # values survive the repack exactly
    after = st.materialize(keys, slots1).cpu()
    torch.testing.assert_close(before, after)
    # frequencies preserved too
    f = st.frequencies(keys)
    assert int(f[64:].min()) > int(f[:64].max())


def test_multitier_filter_restore_roundtrip():
    """Checkpoint round trip with a frequency admission filter: restored
    rows must come back trained, not re-initialized (regression: ADVICE
    r1 high — import_ routed keys through admission with counts=1)."""
    from deeprec_amd.embedding.options import CounterFilter
    from deeprec_amd.ops.hbm_dram_backend import HbmDramStorage

    dim, hot_rows = 8, 16
    opt = EmbeddingVariableOption(
        storage_option=StorageOption(storage_type=StorageType.HBM_DRAM,
                                     storage_size=[hot_rows * dim * 4]),
        filter_option=CounterFilter(filter_freq=3),
        init_option=InitializerOption(initializer=1.0))
    st = HbmDramStorage(dim, opt, device=DEV)
    keys = torch.arange(64, dtype=torch.int64, device=DEV)
    # admit everything (3 sightings), spilling past the 16-row hot tier
    for _ in range(3):
        st.lookup_or_create(keys, torch.ones(64, dtype=torch.int32,
                                             device=DEV), step=0)
    slots = st.lookup(keys)
    assert bool((slots >= 0).all()) and bool((slots >= hot_rows).any())
    # "train": write recognizable values into every row
    trained = torch.arange(64, dtype=torch.float32,
                           device=DEV)[:, None].repeat(1, dim) + 100.0
    hot = slots < hot_rows
    st.values[slots[hot].long()] = trained[hot]
    cold_idx = (slots[~hot].cpu().long() - hot_rows)
    st.values_cold[cold_idx] = trained[~hot].cpu()
    k, v, f, ver = st.export()
    order = torch.argsort(k)
    # restore into a FRESH storage with the same filter
    st2 = HbmDramStorage(dim, opt, device=DEV)
    st2.import_(k, v, f.to(torch.int32), ver)
    slots2 = st2.lookup(keys)
    assert bool((slots2 >= 0).all()), \
        "restored keys must be admitted regardless of the filter"
    got = st2.materialize(keys, slots2)
    torch.testing.assert_close(got, trained)
    # frequencies survive the round trip
    torch.testing.assert_close(st2.frequencies(keys).cpu(),
                               st.frequencies(keys).cpu())


def test_background_maintenance_no_step_spike():
    """EvictionManager: scoring on a worker thread + bounded chunk
    application must promote hot keys WITHOUT any multi-second step
    (the blocking rebalance() on the same shape costs seconds)."""
    import time

    from deeprec_amd.embedding.maintenance import EvictionManager
    from deeprec_amd.ops import hip_backend

    dim, hot_rows = 16, 4096
    opt = EmbeddingVariableOption(
        storage_option=StorageOption(storage_type=StorageType.HBM_DRAM,
                                     storage_size=[hot_rows * dim * 4]),
        init_option=InitializerOption(initializer=1.0))
    ev = EmbeddingVariable("bg_maint", dim, ev_option=opt, device=DEV)
    st = ev.storage
    st._grow_slots(200_000)
    st.get_slab("adagrad_accum", dim, 0.1)
    # populate 100k ids; the first 4096 slots (hot tier) land on
    # insert order, NOT on the hot id set
    all_ids = torch.arange(100_000, dtype=torch.int64, device=DEV)
    st.lookup_or_create(all_ids, torch.ones(100_000, dtype=torch.int32,
                                            device=DEV), step=0)
    mgr = EvictionManager(interval_steps=5, chunk_rows=8192)
    mgr.register(ev)
    hot_ids = torch.arange(90_000, 94_096, dtype=torch.int64,
                           device=DEV)  # 4096 ids, inserted cold
    hyper = {"lr": 0.01, "initial_accumulator": 0.1, "epsilon": 1e-8}
    step_times = []
    for step in range(60):
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        uniq, inverse, counts, slots = st.dedup_lookup(hot_ids, step)
        emb = st.gather(uniq, slots)
        st.apply_split("adagrad", slots, emb * 1e-3, dict(hyper))
        mgr.step(step)
        torch.cuda.synchronize()
        step_times.append(time.perf_counter() - t0)
    mgr.wait_idle()
    for step in range(60, 70):  # drain remaining chunks
        mgr.step(step)
    # the hot ids must end up in the HBM tier
    final_slots = st.lookup(hot_ids)
    frac_hot = float((final_slots < st.hot_rows).float().mean())
    assert frac_hot > 0.9, f"only {frac_hot:.2f} promoted"
    assert mgr.stats["rows_promoted"] > 3000
    # no step may stall on maintenance (blocking rebalance here ~seconds)
    assert max(step_times) < 0.25, f"step spike {max(step_times):.3f}s"
    # values survive the tier swaps exactly (all-ones init + decay grads)
    vals = st.materialize(hot_ids, final_slots)
    assert torch.isfinite(vals).all()
    st._check_error()


def test_three_tier_hbm_dram_ssd(tmp_path):
    """HBM_DRAM_SSD: rows spill HBM -> pinned DRAM -> append-only SSD
    files; lookups, training applies and checkpoint round-trips span all
    three tiers (reference capability: hbm_dram_ssd_storage.h)."""
    from deeprec_amd.optimizers import AdagradOptimizer

    dim = 8
    opt = EmbeddingVariableOption(
        storage_option=StorageOption(
            storage_type=StorageType.HBM_DRAM_SSD,
            storage_size=[32 * dim * 4, 64 * dim * 4],  # 32 hot, 64 dram
            storage_path=str(tmp_path / "ssd")),
        init_option=InitializerOption(initializer=1.0))
    ev = EmbeddingVariable("3tier", dim, ev_option=opt, device=DEV)
    st = ev.storage
    ids = torch.arange(200, dtype=torch.int64, device=DEV)  # 200 >> 96
    out = embedding_lookup(ev, ids)
    torch.testing.assert_close(out, torch.ones(200, dim, device=DEV))
    assert st.memory_usage()["ssd_rows"] >= 104  # 200 - 32 - 64
    # train: gradients hit all three tiers
    og = AdagradOptimizer(embedding_variables=[ev], learning_rate=0.1)
    out = embedding_lookup(ev, ids, train=True)
    (out ** 2).sum().backward()
    og.step()
    after = embedding_lookup(ev, ids)
    assert bool((after < out.detach()).all())  # every row trained
    # the same update math on every tier
    torch.testing.assert_close(after.min(), after.max())
    # checkpoint round trip through a fresh 3-tier storage
    k, v, f, ver = st.export()
    ev2 = EmbeddingVariable("3tier_b", dim, ev_option=EmbeddingVariableOption(
        storage_option=StorageOption(
            storage_type=StorageType.HBM_DRAM_SSD,
            storage_size=[32 * dim * 4, 64 * dim * 4],
            storage_path=str(tmp_path / "ssd2")),
        init_option=InitializerOption(initializer=1.0)), device=DEV)
    ev2.storage.import_(k, v, f.to(torch.int32), ver)
    got = embedding_lookup(ev2, ids)
    order = torch.argsort(ids)
    torch.testing.assert_close(got[order], after[order], rtol=1e-5,
                               atol=1e-6)
    # SSD compaction round-trips under the storage API
    st.compact_ssd(sync=True)
    torch.testing.assert_close(embedding_lookup(ev, ids), after)


def test_lookup_tier_reports_placement():
    """KvResourceLookupTier parity (reference:
    kv_variable_lookup_ops.cc:537): -1 absent, 0 HBM, 1 DRAM cold."""
    from deeprec_amd.embedding.options import (EmbeddingVariableOption,
                                               StorageOption, StorageType)
    from deeprec_amd.ops.hbm_dram_backend import HbmDramStorage

    dim, hot_rows = 8, 32
    opt = EmbeddingVariableOption(storage_option=StorageOption(
        storage_type=StorageType.HBM_DRAM,
        storage_size=[hot_rows * dim * 4]))
    st = HbmDramStorage(dim, opt, device="cuda")
    keys = torch.arange(64, dtype=torch.int64, device="cuda")
    st.lookup_or_create(keys, torch.ones(64, dtype=torch.int32,
                                         device="cuda"), step=0)
    probe = torch.cat([keys, torch.tensor([1 << 40], device="cuda")])
    tier = st.lookup_tier(probe)
    assert int(tier[-1]) == -1                      # never inserted
    counts = torch.bincount(tier[:-1].clamp(min=0), minlength=2)
    assert int(counts[0]) == hot_rows               # hot tier full
    assert int(counts[1]) == 64 - hot_rows          # rest in DRAM
    assert bool((tier[:-1] >= 0).all())
