"""Background multi-tier maintenance.

Capability parity with the reference EvictionManager
(eviction_manager.h:39-82 + cache_thread_pool_creator.h): a worker
thread drives hot/cold placement for multi-tier EVs so training steps
never pay the maintenance latency. Re-designed for this engine's
slot-tier scheme:

- SCORING runs fully on the worker thread (snapshot export + frequency
  ranking on a side HIP stream; approximate against concurrent training
  by design — the reference's cache bookkeeping is equally racy and
  equally benign);
- MUTATION is split into bounded chunks applied at step boundaries by
  the training thread (promote/demote chunk_rows rows per step): the
  hash table is only ever touched between steps, so no probe races, and
  a step's maintenance overhead is a few hundred microseconds instead
  of the multi-second stop-the-world rebalance;
- chunk application re-validates each key's slot right before the swap,
  so admissions that happened after scoring are skipped, not corrupted.

The blocking `storage.rebalance()` remains available for offline
repacks; shrink (feature eviction) stays hook-driven at checkpoint save,
matching the reference's shrink-on-save policy.
"""
from __future__ import annotations

import threading
from typing import List, Optional

import torch


class EvictionManager:
    def __init__(self, interval_steps: int = 200,
                 chunk_rows: int = 1 << 17):
        self.interval_steps = interval_steps
        self.chunk_rows = chunk_rows
        self._evs: List = []
        self._plans = {}          # id(storage) -> (promote_k, demote_k)
        self._lock = threading.Lock()
        self._worker: Optional[threading.Thread] = None
        self._last_kick = -1
        self._interval = interval_steps
        self.stats = {"chunks_applied": 0, "rows_promoted": 0,
                      "scores_computed": 0}

    def register(self, ev):
        st = getattr(ev, "storage", ev)
        if hasattr(st, "values_cold"):
            self._evs.append(ev)

    # ---------------- training-thread API ----------------
    def step(self, global_step: int):
        """Call once per training step (cheap). Kicks background scoring
        every interval; applies at most one bounded chunk."""
        if not self._evs:
            return
        if (global_step - self._last_kick >= self._interval
                and (self._worker is None or not self._worker.is_alive())):
            self._last_kick = global_step
            self._worker = threading.Thread(target=self._score_all,
                                            daemon=True)
            self._worker.start()
        for ev in self._evs:
            st = getattr(ev, "storage", ev)
            with self._lock:
                plan = self._plans.get(id(st))
            if plan is not None:
                self._apply_chunk(st, plan)

    def wait_idle(self):
        t = self._worker
        if t is not None:
            t.join()

    def pending_rows(self) -> int:
        with self._lock:
            return sum(int(p[0].numel()) for p in self._plans.values())

    # ---------------- worker thread ----------------
    def _score_all(self):
        stream = torch.cuda.Stream() if torch.cuda.is_available() else None
        for ev in self._evs:
            st = getattr(ev, "storage", ev)
            try:
                if stream is not None:
                    with torch.cuda.stream(stream):
                        self._score(st)
                    stream.synchronize()
                else:
                    self._score(st)
            except Exception:  # noqa: BLE001 — scoring must never kill training
                pass

    def _score(self, st):
        """Rank keys by frequency (LFU; version/LRU when configured) and
        emit promote/demote key lists (CPU tensors)."""
        keys, slots, freqs, versions = st._export_entries()
        adm = slots >= 0
        if int(adm.sum()) == 0:
            return
        keys, slots = keys[adm], slots[adm]
        score = freqs[adm].long()
        from deeprec_amd.embedding.options import CacheStrategy
        so = st.ev_option.storage_option
        if so is not None and so.cache_strategy == CacheStrategy.LRU:
            score = versions[adm]
        hot_rows = st.hot_rows
        is_hot = slots < hot_rows
        # the coldest hot rows and the hottest cold rows
        hot_scores = score[is_hot]
        cold_scores = score[~is_hot]
        if cold_scores.numel() == 0 or hot_scores.numel() == 0:
            return
        n = min(hot_scores.numel(), cold_scores.numel())
        cold_order = torch.argsort(cold_scores, descending=True)
        hot_order = torch.argsort(hot_scores)
        k = min(n, cold_order.numel(), hot_order.numel())
        promote_k = keys[~is_hot][cold_order[:k]]
        demote_k = keys[is_hot][hot_order[:k]]
        # keep only genuinely-misplaced pairs (cold score > hot score)
        ps = cold_scores[cold_order[:k]]
        ds = hot_scores[hot_order[:k]]
        keep = ps > ds
        promote_k, demote_k = promote_k[keep], demote_k[keep]
        self.stats["scores_computed"] += 1
        if promote_k.numel() == 0:
            # placement converged: back the scoring cadence off
            # exponentially (a 120M-row score costs ~tens of ms of GPU
            # contention; re-running it every interval after convergence
            # taxed steady-state steps ~5x)
            self._interval = min(self._interval * 2,
                                 self.interval_steps * 64)
            return
        self._interval = self.interval_steps
        with self._lock:
            self._plans[id(st)] = (promote_k.cpu(), demote_k.cpu())

    # ---------------- bounded chunk application ----------------
    def _apply_chunk(self, st, plan):
        promote_k, demote_k = plan
        take = min(self.chunk_rows, promote_k.numel())
        pk = promote_k[:take].to(st.device)
        dk = demote_k[:take].to(st.device)
        rest = (promote_k[take:], demote_k[take:])
        with self._lock:
            if rest[0].numel():
                self._plans[id(st)] = rest
            else:
                self._plans.pop(id(st), None)
        # re-validate: the plan is a snapshot; only swap keys whose tier
        # still matches it
        ps = st.lookup(pk).long()
        ds = st.lookup(dk).long()
        ok = (ps >= st.hot_rows) & (ds >= 0) & (ds < st.hot_rows)
        ssd_base = getattr(st, "ssd_base", None)
        if ssd_base is not None:
            # SSD-resident rows are key-addressed, not slab rows: the
            # HBM<->DRAM swap only applies to the in-memory tiers
            ok &= ps < ssd_base
        n = int(ok.sum())
        if n == 0:
            return
        pk, ps = pk[ok], ps[ok]
        dk, ds = dk[ok], ds[ok]
        cold_idx = ps - st.hot_rows
        # swap value rows (and every optimizer slab) between tiers
        pairs = [(st.values, st.values_cold)] + [
            (st.slabs[nm], st.cold_slabs[nm]) for nm in st.slabs]
        for hot_t, cold_t in pairs:
            hot_rows_data = hot_t[ds].clone()
            cold_rows_data = st.ext.gather_host_rows(cold_t, cold_idx)
            hot_t[ds] = cold_rows_data
            st.ext.scatter_host_rows(hot_rows_data, cold_idx, cold_t)
        # swap slot assignments (existing keys: insert_bulk updates
        # ht_slot in place; freq/version untouched)
        st.ext.ht_insert_bulk(
            torch.cat([pk, dk]),
            torch.cat([ds, ps]).to(torch.int32),
            torch.Tensor(), torch.Tensor(),
            st.ht_keys, st.ht_slot, st.ht_freq, st.ht_version,
            st.entry_counter, st.error_flag)
        self.stats["chunks_applied"] += 1
        self.stats["rows_promoted"] += n


class MaintenanceHook:
    """Session hook driving an EvictionManager once per step
    (SessionRunHook-shaped: begin/before_run/after_run/end)."""

    def __init__(self, manager: EvictionManager, variables=None):
        self.manager = manager
        self._variables = variables
        self._n = 0

    def begin(self, session=None):
        if self._variables is not None:
            for ev in self._variables:
                self.manager.register(ev)
        else:
            from deeprec_amd.embedding.variable import (
                all_embedding_variables)
            for ev in all_embedding_variables():
                self.manager.register(ev)

    def before_run(self, session=None):
        pass

    def after_run(self, *a, **kw):
        self._n += 1
        self.manager.step(self._n)

    def end(self, session=None):
        self.manager.wait_idle()
