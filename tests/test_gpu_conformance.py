"""GPU conformance matrix: every HIP engine cell (optimizer x variant x
filter x storage) against the CPU oracle trained on the same data, plus
the hash-table concurrency stress (reference:
embedding_variable_ops_gpu_test.py + TestFeatureFilterParallel,
embedding_variable_ops_test.cc:717)."""
import itertools

import pytest
import torch

pytestmark = pytest.mark.gpu

from tests.test_conformance_matrix import (  # noqa: E402
    FILTERS, OPTIMIZERS, _make, _snapshot, _step, run_cell)

DEV = "cuda:0"


@pytest.mark.parametrize(
    "opt_name,variant,filt",
    list(itertools.product(OPTIMIZERS, ["ev", "collection"],
                           ["none", "counter"])))
def test_matrix_gpu_vs_cpu_oracle(opt_name, variant, filt):
    """Same data through the HIP engine and the CPU oracle: identical
    admitted keys/freqs; fp32-close values."""
    from deeprec_amd.embedding.variable import GLOBAL_STEP

    evs = {}
    for device in ("cpu", DEV):
        ev = _make(variant, filt, device, f"g_{opt_name}_{variant}_"
                   f"{filt}_{device.replace(':', '')}")
        opt = OPTIMIZERS[opt_name]([ev])
        base = GLOBAL_STEP.value
        for i in range(4):
            GLOBAL_STEP.value = base + i
            _step(ev, variant, i, device)
            opt.step(increment_global_step=False)
        GLOBAL_STEP.value = base
        evs[device] = ev
    kc, vc, fc, _ = _snapshot(evs["cpu"])
    kg, vg, fg, _ = _snapshot(evs[DEV])
    torch.testing.assert_close(kc, kg)
    torch.testing.assert_close(fc, fg)
    torch.testing.assert_close(vc, vg, rtol=1e-4, atol=1e-5)


@pytest.mark.parametrize("opt_name", ["adagrad", "adam_async", "ftrl"])
def test_matrix_gpu_save_restore(opt_name, tmp_path):
    """Full save/restore/continue cells on the HIP engine."""
    run_cell(opt_name, "collection", "counter", DEV, tmp_path=tmp_path)
    run_cell(opt_name, "ev", "none", DEV, tmp_path=tmp_path)


def test_hash_table_concurrency_stress():
    """Concurrent read probes (own HIP streams) against a table being
    mutated by interleaved dedup/insert/shrink storms on the main
    stream; the dict oracle checks every settled state (the GPU analog
    of TestFeatureFilterParallel)."""
    import threading

    from deeprec_amd.embedding import (EmbeddingVariable,
                                       EmbeddingVariableOption)
    from deeprec_amd.embedding.options import (GlobalStepEvict,
                                               InitializerOption)

    torch.manual_seed(0)
    opt = EmbeddingVariableOption(
        init_option=InitializerOption(initializer=1.0),
        evict_option=GlobalStepEvict(steps_to_live=50))
    ev = EmbeddingVariable("stress", 8, ev_option=opt, device=DEV)
    st = ev.storage
    stop = threading.Event()
    probe_errs = []

    def prober():
        # read-only probes on a separate stream while the main thread
        # mutates: misses are legal mid-insert; found slots must be
        # in-bounds and non-negative
        stream = torch.cuda.Stream()
        try:
            with torch.cuda.stream(stream):
                g = torch.Generator("cuda").manual_seed(99)
                while not stop.is_set():
                    keys = torch.randint(0, 5000, (2048,), device=DEV,
                                         generator=g)
                    slots = st.lookup(keys)
                    assert int(slots.max()) < st.max_slots
        except Exception as e:  # noqa: BLE001
            probe_errs.append(e)

    threads = [threading.Thread(target=prober) for _ in range(2)]
    for t in threads:
        t.start()
    oracle = {}
    try:
        for step in range(60):
            g = torch.Generator("cuda").manual_seed(step)
            keys = torch.randint(0, 5000, (4096,), device=DEV,
                                 generator=g)
            uniq, inverse, counts, slots = st.dedup_lookup(keys, step)
            for k in uniq.cpu().tolist():
                oracle[k] = step
            if step % 20 == 19:
                evicted = ev.shrink(step)
                dead = [k for k, v in oracle.items()
                        if v < step - 50]
                assert evicted == len(dead)
                for k in dead:
                    del oracle[k]
                # oracle state must match the table exactly after shrink
                kk, ss, ff, vv = st._export_entries()
                adm = (ss >= 0)
                assert sorted(kk[adm].cpu().tolist()) == sorted(oracle)
    finally:
        stop.set()
        for t in threads:
            t.join()
    st._check_error()
    assert not probe_errs, probe_errs


def test_dedup_insert_interleave_storm():
    """Alternating fused-dedup and bulk-insert batches with overlapping
    key ranges; the table must stay exact vs a python-dict oracle."""
    from deeprec_amd.embedding import (EmbeddingVariable,
                                       EmbeddingVariableOption)
    from deeprec_amd.embedding.options import InitializerOption

    ev = EmbeddingVariable(
        "storm", 4,
        ev_option=EmbeddingVariableOption(
            init_option=InitializerOption(initializer=2.0)),
        device=DEV)
    st = ev.storage
    freq_oracle = {}
    for step in range(40):
        g = torch.Generator("cuda").manual_seed(1000 + step)
        keys = torch.randint(0, 800, (1024,), device=DEV, generator=g)
        if step % 2 == 0:
            uniq, inverse, counts, slots = st.dedup_lookup(keys, step)
            for k, c in zip(uniq.cpu().tolist(), counts.cpu().tolist()):
                freq_oracle[k] = freq_oracle.get(k, 0) + c
        else:
            uniq = torch.unique(keys)
            counts = torch.zeros_like(uniq, dtype=torch.int32)
            cnt = torch.bincount((keys.cpu()).long(), minlength=800)
            counts = cnt[uniq.cpu().long()].to(torch.int32).to(DEV)
            st.lookup_or_create(uniq, counts, step)
            for k, c in zip(uniq.cpu().tolist(), counts.cpu().tolist()):
                freq_oracle[k] = freq_oracle.get(k, 0) + c
    kk, ss, ff, vv = st._export_entries()
    got = dict(zip(kk.cpu().tolist(), ff.cpu().tolist()))
    assert got == freq_oracle
    st._check_error()
