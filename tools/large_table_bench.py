"""100-GB-class EmbeddingVariable benchmark — the scale check for the
north-star config #5 (100B-row-class tables across 288 GB HBM + DRAM cold
tier; reference capability: hbm_dram_storage.h:412-435 batched staging,
multi_tier_storage.cu.cc:42-109).

Phases:
  hbm  — single-tier HBM EV pre-sized so the engine-resident bytes
         (value slab + optimizer slab + hash table) exceed 100 GB;
         populates the full id space, then measures training-lookup
         steps/s on a hot/uniform id mix.
  tier — HBM_DRAM EV with the hot tier capped far below the live set;
         measures training steps/s with cold-tier traffic, the H2D
         staging bandwidth of a pure cold materialize, and the cost of
         an LFU rebalance.

Every allocation is pre-sized exactly (pow2) BEFORE population so no
growth-doubling transient can spike device memory. Prints one JSON line
per phase.

Usage:
  python tools/large_table_bench.py --phase hbm  --ids 220000000 --dim 64
  python tools/large_table_bench.py --phase tier --ids 60000000 --dim 64 \
      --hot-gb 4
"""
import argparse
import json
import sys
import time

import os
import sys

import torch

# runnable from anywhere: the repo root is the import root
_REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
if _REPO_ROOT not in sys.path:
    sys.path.insert(0, _REPO_ROOT)

sys.path.insert(0, ".")


def _pow2(n):
    return 1 << (n - 1).bit_length()


def log(msg):
    print(f"[large_table] {msg}", flush=True)


def make_ev(name, dim, ids, hot_bytes=None, optimizer="adagrad"):
    from deeprec_amd.embedding import EmbeddingVariable
    from deeprec_amd.embedding.options import (EmbeddingVariableOption,
                                               InitializerOption,
                                               StorageOption, StorageType)
    cap = _pow2(int(ids / 0.55) + 1)
    so = None
    if hot_bytes is not None:
        so = StorageOption(storage_type=StorageType.HBM_DRAM,
                           storage_size=[hot_bytes])
    opt = EmbeddingVariableOption(
        init_capacity=cap,
        storage_option=so or StorageOption(),
        init_option=InitializerOption(default_value_dim=4096))
    ev = EmbeddingVariable(name, dim, ev_option=opt, device="cuda")
    return ev


def plan_check(bytes_needed):
    free, total = torch.cuda.mem_get_info()
    log(f"planned resident bytes: {bytes_needed/1e9:.1f} GB; "
        f"device free {free/1e9:.1f} / total {total/1e9:.1f} GB")
    if bytes_needed > free * 0.92:
        raise SystemExit("refusing: plan exceeds 92% of free HBM")


def populate(ev, ids, chunk=1 << 23, optimizer="adagrad"):
    """Insert the full id space (spread via a multiplicative hash so the
    insert order exercises the whole table)."""
    t0 = time.perf_counter()
    n = 0
    step = 0
    while n < ids:
        m = min(chunk, ids - n)
        keys = torch.arange(n, n + m, dtype=torch.int64, device="cuda")
        keys = (keys * 2654435761) % ids  # permutation-ish spread
        counts = torch.ones(m, dtype=torch.int32, device="cuda")
        ev.storage.lookup_or_create(keys, counts, step=0, train=True)
        n += m
        step += 1
        if step % 8 == 0:
            torch.cuda.synchronize()
            log(f"populated {n/1e6:.0f}M / {ids/1e6:.0f}M ids "
                f"({n/(time.perf_counter()-t0)/1e6:.1f}M ids/s)")
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    log(f"population done: {ids/1e6:.0f}M ids in {dt:.1f}s "
        f"({ids/dt/1e6:.1f}M ids/s)")
    return dt


def batch_ids(batch, per_sample, ids, hot_frac=0.8, hot_space=1 << 20,
              gen=None):
    n = batch * per_sample
    hot = torch.randint(0, hot_space, (n,), device="cuda", generator=gen)
    cold = torch.randint(0, ids, (n,), device="cuda", generator=gen)
    pick = torch.rand(n, device="cuda", generator=gen) < hot_frac
    return torch.where(pick, hot, cold)


def debug_lookup(ev, args, label=""):
    """One finely-timed dedup_lookup to localize host/device cost."""
    st = ev.storage
    gen = torch.Generator("cuda").manual_seed(123)
    keys = batch_ids(args.batch, args.per_sample, args.ids, gen=gen)
    torch.cuda.synchronize()
    marks = [("start", time.perf_counter())]

    def mark(name):
        torch.cuda.synchronize()
        marks.append((name, time.perf_counter()))

    nnz = keys.numel()
    prev = st._watermark()
    mark("watermark")
    st._ensure_capacity(nnz)
    mark("ensure_cap")
    st._epoch += 1
    uniq_buf = torch.empty(nnz, dtype=torch.int64, device=st.device)
    centry_buf = torch.empty(nnz, dtype=torch.int64, device=st.device)
    m_counter = torch.zeros(1, dtype=torch.int32, device=st.device)
    mark("alloc")
    st.ext.ht_dedup_a(keys, st.ht_keys, st.ht_freq, st.ht_version,
                      st.ht_epoch, st.ht_compact, st._epoch, 0,
                      st.entry_counter, m_counter, uniq_buf, centry_buf,
                      st.error_flag)
    mark("pass_a")
    c = torch.cat([m_counter, st.entry_counter, st.slot_counter]).cpu()
    m = int(c[0])
    mark("sync_m")
    inverse, counts, rank = st.ext.ht_dedup_c(keys, st.ht_keys,
                                              st.ht_compact, m)
    mark("pass_c")
    slots = st.ext.ht_dedup_b(
        centry_buf[:m], uniq_buf[:m], st.ht_slot, st.ht_freq,
        st.ht_version, counts, 0, st.slot_counter, st.max_slots,
        st.values, st.default_values, st.dvd_per_table, st.key_bits,
        st._init_limit(), st.filter_freq, st.error_flag)
    mark("pass_b")
    cold_new = (slots >= st.hot_rows) & (slots >= max(st.hot_rows, prev))
    n_cold_new = int(cold_new.sum())
    mark("cold_mask")
    st._init_cold_rows(uniq_buf[:m], slots, prev)
    mark("cold_init")
    parts = "  ".join(f"{n}={1000*(t1-t0):.2f}ms" for (_, t0), (n, t1)
                      in zip(marks, marks[1:]))
    log(f"{label} lookup breakdown (m={m}, prev={prev}, "
        f"slot_counter={int(st.slot_counter.cpu())}, "
        f"max_slot={int(slots.max())}, n_cold_new={n_cold_new}): {parts}")


def train_steps(ev, args, optimizer="adagrad", label=""):
    """Training-shaped engine steps: dedup+probe+admit, gather, grad
    scatter-equivalent (rows += g), fused sparse apply."""
    from deeprec_amd.ops import hip_backend
    gen = torch.Generator("cuda").manual_seed(7)
    hyper = {"lr": 0.01, "initial_accumulator": 0.1, "epsilon": 1e-8}
    times = []
    parts = [0.0, 0.0, 0.0]  # lookup, gather, apply (timed steps only)

    def tick():
        torch.cuda.synchronize()
        return time.perf_counter()

    for step in range(args.warmup + args.steps):
        timed_step = step >= args.warmup
        t0 = tick()
        keys = batch_ids(args.batch, args.per_sample, args.ids, gen=gen)
        uniq, inverse, counts, slots = ev.storage.dedup_lookup(keys, step)
        t1 = tick()
        emb = ev.storage.gather(uniq, slots)
        t2 = tick()
        grad = emb * 1e-4  # decay-shaped grads keep values bounded
        if hasattr(ev.storage, "apply_split"):
            ev.storage.apply_split(optimizer, slots, grad, dict(hyper))
        else:
            hip_backend.sparse_apply(optimizer, ev.storage,
                                     slots.to(torch.int32), grad,
                                     dict(hyper))
        t3 = tick()
        times.append(t3 - t0)
        if timed_step:
            parts[0] += t1 - t0
            parts[1] += t2 - t1
            parts[2] += t3 - t2
    timed = times[args.warmup:]
    ms = sum(timed) / len(timed) * 1000
    n = len(timed)
    sps = args.batch / (ms / 1000)
    log(f"{label} step: {ms:.2f} ms  ({sps/1e6:.2f}M samples/s, "
        f"nnz/step={args.batch * args.per_sample}; "
        f"lookup {parts[0]/n*1000:.2f} / gather {parts[1]/n*1000:.2f} / "
        f"apply {parts[2]/n*1000:.2f} ms)")
    return ms, sps


def phase_hbm(args):
    dim = args.dim
    cap = _pow2(int(args.ids / 0.55) + 1)
    slab_rows = _pow2(args.ids)
    # values + adagrad accum + table arrays
    plan = slab_rows * dim * 4 * 2 + cap * 26
    plan_check(plan)
    ev = make_ev("big_hbm", dim, args.ids)
    st = ev.storage
    st._grow_slots(slab_rows)
    st.get_slab("adagrad_accum", dim, 0.1)  # pre-size: no growth later
    populate(ev, args.ids)
    ms, sps = train_steps(ev, args, label="hbm")
    mu = st.memory_usage()
    out = {"phase": "hbm", "ids": args.ids, "dim": dim,
           "ms_per_step": round(ms, 3), "samples_per_sec": int(sps),
           "nnz_per_step": args.batch * args.per_sample,
           "engine_bytes": mu["total_bytes"],
           "engine_gb": round(mu["total_bytes"] / 1e9, 1),
           "values_gb": round(mu["values_bytes"] / 1e9, 1),
           "slab_gb": round(mu["slab_bytes"] / 1e9, 1),
           "table_gb": round(mu["table_bytes"] / 1e9, 1),
           "live_rows": args.ids}
    print(json.dumps(out), flush=True)


def phase_tier_manager(args):
    """Background-maintenance variant: instead of the stop-the-world
    rebalance, the EvictionManager promotes hot rows in bounded chunks
    WHILE training; reports the step-time trajectory and the final hot
    hit fraction."""
    from deeprec_amd.embedding.maintenance import EvictionManager
    from deeprec_amd.ops import hip_backend
    dim = args.dim
    hot_bytes = int(args.hot_gb * (1 << 30))
    plan = hot_bytes * 2 + _pow2(int(args.ids / 0.55) + 1) * 26 + (1 << 30)
    plan_check(plan)
    ev = make_ev("big_tier_mgr", dim, args.ids, hot_bytes=hot_bytes)
    st = ev.storage
    st._grow_slots(args.ids + 1024)
    st.get_slab("adagrad_accum", dim, 0.1)
    populate(ev, args.ids)
    mgr = EvictionManager(interval_steps=10, chunk_rows=1 << 20)
    mgr.register(ev)
    gen = torch.Generator("cuda").manual_seed(7)
    hyper = {"lr": 0.01, "initial_accumulator": 0.1, "epsilon": 1e-8}
    times = []
    for step in range(args.steps):
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        keys = batch_ids(args.batch, args.per_sample, args.ids, gen=gen)
        uniq, inverse, counts, slots = ev.storage.dedup_lookup(keys, step)
        emb = ev.storage.gather(uniq, slots)
        hip_backend.sparse_apply("adagrad", ev.storage,
                                 slots.to(torch.int32), emb * 1e-4,
                                 dict(hyper)) if not hasattr(
            ev.storage, "apply_split") else ev.storage.apply_split(
            "adagrad", slots, emb * 1e-4, dict(hyper))
        mgr.step(step)
        torch.cuda.synchronize()
        times.append(time.perf_counter() - t0)
        if step % 20 == 19:
            hot_frac = float((slots < st.hot_rows).float().mean())
            log(f"step {step}: {times[-1]*1000:.2f} ms, "
                f"hot-slot fraction of batch uniques {hot_frac:.3f}, "
                f"promoted {mgr.stats['rows_promoted']/1e6:.2f}M")
    mgr.wait_idle()
    first = sum(times[2:12]) / 10 * 1000
    last = sum(times[-10:]) / 10 * 1000
    out = {"phase": "tier_manager", "ids": args.ids, "dim": dim,
           "hot_gb": args.hot_gb, "steps": args.steps,
           "ms_first10": round(first, 2), "ms_last10": round(last, 2),
           "ms_max": round(max(times[2:]) * 1000, 2),
           "rows_promoted": int(mgr.stats["rows_promoted"]),
           "chunks": int(mgr.stats["chunks_applied"])}
    print(json.dumps(out), flush=True)


def phase_tier(args):
    dim = args.dim
    hot_bytes = int(args.hot_gb * (1 << 30))
    hot_rows = hot_bytes // (dim * 4)
    cap = _pow2(int(args.ids / 0.55) + 1)
    plan = hot_bytes * 2 + cap * 26 + (1 << 30)
    plan_check(plan)
    ev = make_ev("big_tier", dim, args.ids, hot_bytes=hot_bytes)
    st = ev.storage
    # pre-size the cold pinned slabs (growth would re-allocate + copy
    # tens of GB of pinned memory repeatedly)
    st._grow_slots(args.ids + 1024)
    st.get_slab("adagrad_accum", dim, 0.1)
    populate(ev, args.ids)
    cold_rows = args.ids - st.hot_rows
    log(f"hot rows {st.hot_rows/1e6:.1f}M, cold rows ~{cold_rows/1e6:.1f}M")
    ms, sps = train_steps(ev, args, label="tier")
    debug_lookup(ev, args, label="tier")
    # staging bandwidth: materialize a pure-cold batch
    gen = torch.Generator("cuda").manual_seed(9)
    m = 1 << 20
    cold_keys = torch.randint(args.ids // 2, args.ids, (m,),
                              device="cuda", generator=gen)
    cold_keys = (cold_keys * 2654435761) % args.ids
    slots = st.lookup(cold_keys)
    sel = slots >= st.hot_rows
    ck, cs = cold_keys[sel], slots[sel]
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    st.materialize(ck, cs)
    torch.cuda.synchronize()
    stage_dt = time.perf_counter() - t0
    stage_bytes = int(ck.numel()) * dim * 4
    stage_bw = stage_bytes / stage_dt / 1e9
    log(f"staging: {ck.numel()/1e6:.2f}M cold rows "
        f"({stage_bytes/1e6:.0f} MB) in {stage_dt*1000:.1f} ms "
        f"-> {stage_bw:.2f} GB/s")
    # rebalance cost
    t0 = time.perf_counter()
    moved = ev.rebalance()
    torch.cuda.synchronize()
    reb_dt = time.perf_counter() - t0
    log(f"rebalance: {moved/1e6:.2f}M rows changed tier in {reb_dt:.2f}s")
    ms2, sps2 = train_steps(ev, args, label="tier(after-rebalance)")
    debug_lookup(ev, args, label="tier(after-rebalance)")
    mu = st.memory_usage()
    out = {"phase": "tier", "ids": args.ids, "dim": dim,
           "hot_gb": args.hot_gb, "hot_rows": int(st.hot_rows),
           "cold_rows": int(cold_rows),
           "ms_per_step": round(ms, 3), "samples_per_sec": int(sps),
           "ms_per_step_after_rebalance": round(ms2, 3),
           "staging_gb_per_sec": round(stage_bw, 2),
           "staged_mb": round(stage_bytes / 1e6, 1),
           "rebalance_rows_moved": int(moved),
           "rebalance_sec": round(reb_dt, 2),
           "cold_bytes_gb": round(mu.get("cold_bytes", 0) / 1e9, 1),
           "engine_gb": round(mu["total_bytes"] / 1e9, 1)}
    print(json.dumps(out), flush=True)


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--phase", choices=["hbm", "tier", "tier_manager"],
                   required=True)
    p.add_argument("--ids", type=int, default=220_000_000)
    p.add_argument("--dim", type=int, default=64)
    p.add_argument("--batch", type=int, default=8192)
    p.add_argument("--per-sample", type=int, default=26)
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--hot-gb", type=float, default=4.0)
    args = p.parse_args()
    torch.cuda.init()
    if args.phase == "hbm":
        phase_hbm(args)
    elif args.phase == "tier_manager":
        phase_tier_manager(args)
    else:
        phase_tier(args)


if __name__ == "__main__":
    main()
