"""HBM (GPU) storage backend: Python wrapper over the HIP embedding engine.

Same interface/semantics as ops/cpu_backend.CpuStorage; all hot-path work
runs in the hand-written gfx950 kernels (ops/hip/ev_kernels.hip). Growth
and shrink/compaction are host-coordinated (rare, off the hot path) —
mirroring the reference's coarse multi-tier design choice (SURVEY.md §7
hard parts: host-coordinated rehash).
"""
from __future__ import annotations

import math
import torch

from deeprec_amd.embedding.options import (
    CBFFilter,
    CounterFilter,
    EmbeddingVariableOption,
    GlobalStepEvict,
    L2WeightEvict,
)
from deeprec_amd.ops.build_ext import require_extension

_COMBINER_ID = {"sum": 0, "mean": 1, "sqrtn": 2}
_LOAD_FACTOR = 0.6


def _pow2(n: int) -> int:
    return 1 << max(10, (n - 1).bit_length())


class HbmStorage:
    """Single-tier HBM KV storage (reference capability: HbmStorage,
    single_tier_storage.h:387 + GPUHashMapKV, gpu_hash_map_kv.h:29)."""

    def __init__(self, dim: int, ev_option: EmbeddingVariableOption,
                 value_dtype=torch.float32, device=None, generator=None):
        assert value_dtype == torch.float32, \
            "master values are fp32; bf16 is produced by the fused gather"
        self.ext = require_extension()
        self.dim = dim
        self.device = torch.device(device or "cuda")
        self.value_dtype = value_dtype
        self.ev_option = ev_option

        io = ev_option.init_option
        self.default_value_dim = max(1, io.default_value_dim)
        dv = torch.empty(self.default_value_dim, dim, dtype=torch.float32)
        if io.initializer is None:
            dv.normal_(0.0, 1.0 / math.sqrt(dim), generator=generator)
        elif callable(io.initializer):
            io.initializer(dv)
        else:
            dv.fill_(float(io.initializer))
        self.default_values = dv.to(self.device)

        fo = ev_option.filter_option
        self.filter_freq = 0
        if isinstance(fo, (CounterFilter, CBFFilter)):
            # CBF semantics on GPU: per-entry counters are cheap in 288 GB
            # HBM, so the CBF option maps to exact counter admission. The
            # memory-bounded probabilistic pre-admission store is a CPU-tier
            # concern (cpu_backend implements a true counting bloom).
            self.filter_freq = fo.filter_freq

        # composite-key support (EmbeddingCollection): 0 = plain keys
        self.key_bits = 0
        self.dvd_per_table = self.default_value_dim

        cap = _pow2(ev_option.init_capacity)
        self._alloc_table(cap)
        self._alloc_slabs(max(1024, cap // 2))
        self.slabs = {}
        self._slab_init = {}
        self.slot_counter = torch.zeros(1, dtype=torch.int32,
                                        device=self.device)
        self.entry_counter = torch.zeros(1, dtype=torch.int32,
                                         device=self.device)
        self.error_flag = torch.zeros(1, dtype=torch.int32,
                                      device=self.device)
        # host-side mirrors, refreshed lazily (avoid per-step D2H sync)
        self._entries_hint = 0
        self._slots_hint = 0
        # adaptive dedup-vs-sort choice: hash dedup wins on duplication-
        # heavy batches (zipf id streams); sort-based unique wins when keys
        # are mostly distinct (long uniform sequences). Track the observed
        # unique ratio and pick per step.
        self._uniq_ratio = None

    # ---------------- allocation ----------------
    def _alloc_table(self, capacity: int):
        self.capacity = capacity
        self.ht_keys = torch.full((capacity,), -(2 ** 63), dtype=torch.int64,
                                  device=self.device)
        self.ht_slot = torch.full((capacity,), -1, dtype=torch.int32,
                                  device=self.device)
        self.ht_freq = torch.zeros(capacity, dtype=torch.int32,
                                   device=self.device)
        self.ht_version = torch.full((capacity,), -1, dtype=torch.int64,
                                     device=self.device)
        # per-entry epoch stamp + compact index for the fused dedup path
        self.ht_epoch = torch.zeros(capacity, dtype=torch.int32,
                                    device=self.device)
        self.ht_compact = torch.zeros(capacity, dtype=torch.int32,
                                      device=self.device)
        self._epoch = getattr(self, "_epoch", 0)

    def _alloc_slabs(self, max_slots: int):
        self.values = torch.empty(max_slots, self.dim, dtype=torch.float32,
                                  device=self.device)

    @property
    def max_slots(self) -> int:
        return self.values.shape[0]

    def _init_limit(self) -> int:
        """Rows the insert kernel may default-initialize; the HBM_DRAM
        subclass caps this at the hot-tier size."""
        return self.max_slots

    def _sync_counters(self):
        c = torch.stack([self.slot_counter, self.entry_counter]).cpu()
        self._slots_hint = int(c[0])
        self._entries_hint = int(c[1])

    def _ensure_capacity(self, incoming: int):
        """Grow hash table / value slab before a batch that could overflow.
        Uses host-side hints (upper bounds) so the hot path stays sync-free;
        hints are refreshed only when a growth decision is near."""
        if (self._entries_hint + incoming > self.capacity * _LOAD_FACTOR or
                self._slots_hint + incoming > self.max_slots):
            self._sync_counters()
            if self._entries_hint + incoming > self.capacity * _LOAD_FACTOR:
                self._rehash(_pow2(
                    int((self._entries_hint + incoming) / _LOAD_FACTOR) + 1))
            if self._slots_hint + incoming > self.max_slots:
                self._grow_slots(self._slots_hint + incoming)
        # track the upper bound without a device sync
        self._entries_hint += incoming
        self._slots_hint += incoming

    def _grow_slots(self, need: int):
        new_cap = self.max_slots
        while new_cap < need:
            new_cap *= 2
        nv = torch.empty(new_cap, self.dim, dtype=torch.float32,
                         device=self.device)
        nv[: self.max_slots] = self.values
        for name, t in list(self.slabs.items()):
            nt = torch.full((new_cap, t.shape[1]), self._slab_init[name],
                            dtype=t.dtype, device=self.device)
            nt[: t.shape[0]] = t
            self.slabs[name] = nt
        self.values = nv

    def _rehash(self, new_capacity: int):
        keys, slots, freqs, versions = self._export_entries()
        self._alloc_table(new_capacity)
        self.entry_counter.zero_()
        self.ext.ht_insert_bulk(keys, slots, freqs, versions, self.ht_keys,
                                self.ht_slot, self.ht_freq, self.ht_version,
                                self.entry_counter, self.error_flag)
        self._check_error()

    def _export_entries(self):
        n = int(self.entry_counter.cpu())
        keys, slots, freqs, versions = self.ext.ht_export(
            self.ht_keys, self.ht_slot, self.ht_freq, self.ht_version, n)
        # the padded sharded exchange admits one PAD_KEY sentinel entry
        # (wire padding); it is engine-internal and never exported
        if n and bool((keys == self.ext.PAD_KEY).any()):
            keep = keys != self.ext.PAD_KEY
            keys, slots = keys[keep], slots[keep]
            freqs, versions = freqs[keep], versions[keep]
        # the scan compacts via an atomic cursor, so raw order differs
        # between CALLS; key-sort it so export()/export_slabs() (separate
        # calls) stay row-aligned in checkpoints
        if n:
            order = torch.argsort(keys)
            keys, slots = keys[order], slots[order]
            freqs, versions = freqs[order], versions[order]
        return keys, slots, freqs, versions

    def _check_error(self):
        err = int(self.error_flag.cpu())
        if err != 0:
            raise RuntimeError(
                f"HBM hash table error {err} (1=slab overflow, "
                "2=table full, 3=CSR scatter out of bounds, "
                "4=padded all-to-all peer cap overflow — re-capture with "
                "a larger pad_cap) — engine invariant violated")

    # ---------------- public interface ----------------
    def get_slab(self, name, width, init_value, dtype=torch.float32):
        if name not in self.slabs:
            self.slabs[name] = torch.full(
                (self.max_slots, width), init_value, dtype=dtype,
                device=self.device)
            self._slab_init[name] = init_value
        return self.slabs[name]

    def lookup_or_create(self, keys, counts, step, train=True):
        if train:
            self._ensure_capacity(keys.numel())
        counts_i32 = (counts.to(torch.int32)
                      if counts is not None else torch.Tensor())
        return self.ext.ht_lookup_insert(
            keys, counts_i32, self.ht_keys, self.ht_slot, self.ht_freq,
            self.ht_version, self.slot_counter, self.entry_counter,
            self.values, self.default_values, self.max_slots,
            self.dvd_per_table, self.key_bits, self._init_limit(),
            self.filter_freq, step, train, self.error_flag)

    def lookup(self, keys):
        slots, _ = self.ext.ht_lookup(keys, self.ht_keys, self.ht_slot, False)
        return slots

    def lookup_tier(self, keys) -> torch.Tensor:
        """Per-key storage tier (reference: KvResourceLookupTier,
        kernels/kv_variable_lookup_ops.cc:537): −1 = not present,
        0 = HBM. Multi-tier storages override with 1 = DRAM, 2 = SSD."""
        slots = self.lookup(keys).long()
        hot = getattr(self, "hot_rows", None)
        tier = torch.where(slots >= 0,
                           torch.zeros_like(slots),
                           torch.full_like(slots, -1))
        if hot is not None:
            tier = torch.where(slots >= hot, torch.ones_like(slots), tier)
            ssd_base = getattr(self, "ssd_base", None)
            if ssd_base is not None:
                tier = torch.where(slots >= ssd_base,
                                   torch.full_like(slots, 2), tier)
        return tier

    def enable_graph_mode(self, expected_entries: int, expected_slots: int):
        """hipGraph capture prep: pre-size the table/slabs (no growth can
        happen inside a captured step) and move the dedup epoch/step
        counters to device scalars bumped inside the capture."""
        if self.capacity < _pow2(int(expected_entries / _LOAD_FACTOR) + 1):
            self._rehash(_pow2(int(expected_entries / _LOAD_FACTOR) + 1))
        if self.max_slots < expected_slots:
            self._grow_slots(expected_slots)
        self.graph_mode = True
        self._epoch_dev = torch.tensor([self._epoch], dtype=torch.int32,
                                       device=self.device)
        from deeprec_amd.embedding.variable import get_global_step
        self._step_dev = torch.tensor([get_global_step()],
                                      dtype=torch.int64, device=self.device)

    def dedup_lookup_capture(self, values_cat: torch.Tensor):
        """Sync-free, fixed-shape dedup+probe for hipGraph capture: all
        outputs are nnz-padded; the true unique count lives in a device
        counter consumed by the padded pass B (tail slots = -1)."""
        nnz = values_cat.numel()
        self.ext.bump_epoch(self._epoch_dev, self._step_dev)
        uniq_buf = torch.empty(nnz, dtype=torch.int64, device=self.device)
        centry_buf = torch.empty(nnz, dtype=torch.int64, device=self.device)
        occ_entry = torch.empty(nnz, dtype=torch.int64, device=self.device)
        m_counter = torch.zeros(1, dtype=torch.int32, device=self.device)
        self._last_m_dev = m_counter
        self.ext.ht_dedup_a_dev(values_cat, self.ht_keys, self.ht_freq,
                                self.ht_version, self.ht_epoch,
                                self.ht_compact, self._epoch_dev,
                                self._step_dev, self.entry_counter,
                                m_counter, uniq_buf, centry_buf,
                                occ_entry, self.error_flag)
        # pass C first: its exact per-batch counts feed pass B's single
        # per-unique freq update (pass A is claim-only — hot keys no
        # longer serialize per-occurrence atomics; index-direct: no
        # table re-probe)
        inverse, counts, rank = self.ext.ht_dedup_c_idx(
            occ_entry, self.ht_compact, nnz)
        self._last_rank = rank  # consumed by collection._prep_backward
        slots = self.ext.ht_dedup_b_padded(
            centry_buf, uniq_buf, m_counter, self.ht_slot, self.ht_freq,
            self.ht_version, counts, self._step_dev,
            self.slot_counter, self.max_slots, self.values,
            self.default_values, self.dvd_per_table, self.key_bits,
            self._init_limit(), self.filter_freq, self.error_flag)
        return uniq_buf, inverse, counts, slots

    def dedup_only_capture(self, values_cat: torch.Tensor):
        """Requester-side dedup for the padded sharded exchange: claim-only
        pass A + pass C against the local table (NO admission, NO
        freq/version writes — those happen once, on the owner). Non-owned
        keys become slot-less entries here (the table doubles as the dedup
        structure); they are cheap (~32 B) and never exported.

        Returns (uniq_buf [nnz] padded, inverse i32 [nnz], counts i32
        [nnz] padded, m_counter device scalar). Capture-safe: no host
        syncs, all shapes static."""
        nnz = values_cat.numel()
        self.ext.bump_epoch(self._epoch_dev, self._step_dev)
        uniq_buf = torch.empty(nnz, dtype=torch.int64, device=self.device)
        centry_buf = torch.empty(nnz, dtype=torch.int64, device=self.device)
        occ_entry = torch.empty(nnz, dtype=torch.int64, device=self.device)
        m_counter = torch.zeros(1, dtype=torch.int32, device=self.device)
        self._last_m_dev = m_counter
        self.ext.ht_dedup_a_dev(values_cat, self.ht_keys, self.ht_freq,
                                self.ht_version, self.ht_epoch,
                                self.ht_compact, self._epoch_dev,
                                self._step_dev, self.entry_counter,
                                m_counter, uniq_buf, centry_buf,
                                occ_entry, self.error_flag)
        inverse, counts, rank = self.ext.ht_dedup_c_idx(
            occ_entry, self.ht_compact, nnz)
        self._last_rank = rank
        return uniq_buf, inverse, counts, m_counter

    def dedup_lookup_capture_owner(self, recv_keys: torch.Tensor,
                                   recv_cnt: torch.Tensor):
        """Owner-side dedup+admission for the padded sharded exchange:
        full pipeline (A, C, B) over the w*cap received keys, except the
        freq bump uses the SUMMED true per-batch counts carried on the
        wire (a key arriving from several peers must count all of its
        occurrences, not the number of peers). PAD_KEY pads dedup into
        one engine-internal entry. Uses its own epoch bump (epoch-only:
        the step already advanced in the requester pass)."""
        nnz = recv_keys.numel()
        self.ext.bump_epoch_only(self._epoch_dev)
        uniq_buf = torch.empty(nnz, dtype=torch.int64, device=self.device)
        centry_buf = torch.empty(nnz, dtype=torch.int64, device=self.device)
        occ_entry = torch.empty(nnz, dtype=torch.int64, device=self.device)
        m_counter = torch.zeros(1, dtype=torch.int32, device=self.device)
        self.ext.ht_dedup_a_dev(recv_keys, self.ht_keys, self.ht_freq,
                                self.ht_version, self.ht_epoch,
                                self.ht_compact, self._epoch_dev,
                                self._step_dev, self.entry_counter,
                                m_counter, uniq_buf, centry_buf,
                                occ_entry, self.error_flag)
        # PAD-aware, index-direct pass C: pad elements get inverse -1
        # and never touch the shared counters (a naive path serialized
        # ~10^5 atomics on PAD_KEY's single cache line)
        inverse, _counts_c, _rank = self.ext.ht_dedup_c_idx_pad(
            recv_keys, occ_entry, self.ht_compact, nnz)
        cnt_sum = self.ext.cnt_sum_pad(inverse, recv_cnt, nnz)
        slots = self.ext.ht_dedup_b_padded(
            centry_buf, uniq_buf, m_counter, self.ht_slot, self.ht_freq,
            self.ht_version, cnt_sum, self._step_dev,
            self.slot_counter, self.max_slots, self.values,
            self.default_values, self.dvd_per_table, self.key_bits,
            self._init_limit(), self.filter_freq, self.error_flag)
        return uniq_buf, inverse, slots

    def prefers_dedup(self) -> bool:
        # hash dedup is O(nnz) probes; torch.unique is a full radix/merge
        # sort pipeline (~10 launches). Measured on the DIN seq lookup
        # (~200k ids, ~50% unique): sort path 204 us + 93 us lookup_insert
        # vs ~80 us dedup passes — dedup wins even at high unique ratios,
        # so only near-fully-unique streams (where pass B re-inserts
        # nearly every key) fall back to sort.
        return self._uniq_ratio is None or self._uniq_ratio < 0.95

    def observe_uniq_ratio(self, m: int, nnz: int):
        r = m / max(nnz, 1)
        self._uniq_ratio = (r if self._uniq_ratio is None
                            else 0.7 * self._uniq_ratio + 0.3 * r)

    def dedup_lookup(self, values_cat: torch.Tensor, step: int):
        """Fused unique+probe for a training step: raw (duplicated) keys in,
        (uniq, inverse i32, counts i32, slots i32) out. One host sync (the
        unique count) — the same sync torch.unique pays — and no sorts."""
        nnz = values_cat.numel()
        self._ensure_capacity(nnz)
        self._epoch += 1
        uniq_buf = torch.empty(nnz, dtype=torch.int64, device=self.device)
        centry_buf = torch.empty(nnz, dtype=torch.int64, device=self.device)
        occ_entry = torch.empty(nnz, dtype=torch.int64, device=self.device)
        m_counter = torch.zeros(1, dtype=torch.int32, device=self.device)
        self.ext.ht_dedup_a(values_cat, self.ht_keys, self.ht_freq,
                            self.ht_version, self.ht_epoch, self.ht_compact,
                            self._epoch, step, self.entry_counter, m_counter,
                            uniq_buf, centry_buf, occ_entry,
                            self.error_flag)
        # one D2H sync (as torch.unique pays); piggyback the real counters
        # so capacity hints don't inflate by nnz per step (which caused
        # needless rehashes on high-uniqueness workloads)
        c = torch.cat([m_counter, self.entry_counter,
                       self.slot_counter]).cpu()
        m = int(c[0])
        self._entries_hint = int(c[1])
        self._slots_hint = int(c[2]) + m  # pass B may admit up to m slots
        self.observe_uniq_ratio(m, nnz)
        uniq = uniq_buf[:m]
        # pass C first: exact counts feed pass B's per-unique freq update
        # (index-direct: pass A recorded each occurrence's entry)
        inverse, counts, rank = self.ext.ht_dedup_c_idx(
            occ_entry, self.ht_compact, m)
        self._last_rank = rank  # consumed by collection._prep_backward
        slots = self.ext.ht_dedup_b(
            centry_buf[:m], uniq, self.ht_slot, self.ht_freq,
            self.ht_version, counts, step,
            self.slot_counter, self.max_slots, self.values,
            self.default_values, self.dvd_per_table, self.key_bits,
            self._init_limit(), self.filter_freq, self.error_flag)
        return uniq, inverse, counts, slots

    def _use_no_permission(self) -> bool:
        return (self.filter_freq > 0 and
                self.ev_option.init_option.default_value_no_permission
                is not None)

    def _no_permission_value(self) -> float:
        v = self.ev_option.init_option.default_value_no_permission
        return float(v) if v is not None else 0.0

    def gather(self, keys, slots, out_dtype=None):
        return self.ext.ev_gather(
            self.values, self.default_values, keys, slots,
            self._no_permission_value(), self._use_no_permission(),
            out_dtype or torch.float32)

    def pooled_lookup(self, keys, slots, inverse, offsets, row_ids, combiner,
                      weights, out_dtype):
        return self.ext.pooled_fwd(
            self.values, self.default_values, keys, slots,
            inverse.to(torch.int32), offsets.to(torch.int32),
            weights.float() if weights is not None else torch.Tensor(),
            _COMBINER_ID[combiner], self._no_permission_value(),
            self._use_no_permission(), out_dtype or torch.float32)

    def pooled_grad(self, grad_out, inverse, offsets, row_ids, m, combiner,
                    weights):
        return self.ext.pooled_bwd(
            grad_out.contiguous(), inverse.to(torch.int32),
            offsets.to(torch.int32),
            weights.float() if weights is not None else torch.Tensor(),
            m, _COMBINER_ID[combiner])

    # ---------------- shrink / export / import ----------------
    def shrink(self, step: int) -> int:
        eo = self.ev_option.evict_option
        if eo is None:
            return 0
        keys, slots, freqs, versions = self._export_entries()
        if keys.numel() == 0:
            return 0
        if isinstance(eo, GlobalStepEvict) and eo.steps_to_live > 0:
            keep = versions >= (step - eo.steps_to_live)
        elif isinstance(eo, L2WeightEvict) and eo.l2_weight_threshold > 0:
            keep = torch.ones(keys.numel(), dtype=torch.bool,
                              device=self.device)
            has = slots >= 0
            norms = self.values[slots[has].long()].norm(dim=1)
            kk = keep[has].clone()
            kk &= norms >= eo.l2_weight_threshold
            keep[has] = kk
        else:
            return 0
        n_evicted = int((~keep).sum())
        if n_evicted == 0:
            return 0
        self._rebuild(keys[keep], slots[keep], freqs[keep], versions[keep])
        return n_evicted

    def _rebuild(self, keys, slots, freqs, versions):
        """Compact: renumber admitted slots densely, rewrite slabs/table."""
        adm = slots >= 0
        old_slots = slots[adm].long()
        n_adm = int(adm.sum())
        new_slots = torch.full_like(slots, -1)
        new_slots[adm] = torch.arange(n_adm, dtype=torch.int32,
                                      device=self.device)
        self.values[:n_adm] = self.values[old_slots]
        for name, t in self.slabs.items():
            t[:n_adm] = t[old_slots]
        self._alloc_table(self.capacity)
        self.entry_counter.zero_()
        self.slot_counter.fill_(n_adm)
        self.ext.ht_insert_bulk(keys, new_slots, freqs, versions,
                                self.ht_keys, self.ht_slot, self.ht_freq,
                                self.ht_version, self.entry_counter,
                                self.error_flag)
        self._check_error()
        self._sync_counters()

    def export(self, include_filtered: bool = False):
        keys, slots, freqs, versions = self._export_entries()
        adm = slots >= 0
        out = (keys[adm], self.values[slots[adm].long()].clone(),
               freqs[adm].to(torch.int64), versions[adm])
        if include_filtered:
            out = out + (keys[~adm], freqs[~adm].to(torch.int64))
        return out

    def export_slabs(self, names):
        keys, slots, freqs, versions = self._export_entries()
        adm = slots >= 0
        s = slots[adm].long()
        return [self.slabs[nm][s].clone() for nm in names]

    def import_(self, keys, values, freqs=None, versions=None,
                slab_rows=None):
        m = keys.numel()
        if m == 0:
            return
        keys = keys.to(self.device)
        values = values.to(self.device, torch.float32)
        # existing keys keep their slots; new keys get fresh ones
        existing = self.lookup(keys)
        is_new = existing < 0
        n_new = int(is_new.sum())
        self._ensure_capacity(m)
        self._sync_counters()
        base = self._slots_hint
        slots = existing.clone()
        slots[is_new] = base + torch.arange(n_new, dtype=torch.int32,
                                            device=self.device)
        self.slot_counter.fill_(base + n_new)
        self._slots_hint = base + n_new
        self.ext.ht_insert_bulk(
            keys, slots,
            freqs.to(self.device, torch.int32) if freqs is not None
            else torch.Tensor(),
            versions.to(self.device, torch.int64) if versions is not None
            else torch.Tensor(),
            self.ht_keys, self.ht_slot, self.ht_freq, self.ht_version,
            self.entry_counter, self.error_flag)
        self.values[slots.long()] = values
        if slab_rows is not None:
            for name, rows in slab_rows.items():
                self.get_slab(name, rows.shape[1], 0.0)[slots.long()] = \
                    rows.to(self.device)
        self._check_error()
        self._sync_counters()

    def import_filtered(self, keys, freqs):
        """Restore sub-threshold admission counters (reference capability:
        TF_EV_SAVE_FILTERED_FEATURES): hash entries with slot -1 carry
        the frequency until admission."""
        m = keys.numel()
        if m == 0:
            return
        keys = keys.to(self.device)
        self._ensure_capacity(m)
        slots = torch.full((m,), -1, dtype=torch.int32, device=self.device)
        self.ext.ht_insert_bulk(
            keys, slots, freqs.to(self.device, torch.int32),
            torch.Tensor(), self.ht_keys, self.ht_slot, self.ht_freq,
            self.ht_version, self.entry_counter, self.error_flag)
        self._check_error()
        self._sync_counters()

    def memory_usage(self) -> dict:
        """Byte accounting (reference capability:
        embedding_variable_memory_test.cc): hash-table arrays, value slab,
        optimizer slabs."""
        table = sum(t.numel() * t.element_size()
                    for t in (self.ht_keys, self.ht_slot, self.ht_freq,
                              self.ht_version, self.ht_epoch,
                              self.ht_compact))
        values = self.values.numel() * self.values.element_size()
        slabs = sum(t.numel() * t.element_size()
                    for t in self.slabs.values())
        return {"table_bytes": table, "values_bytes": values,
                "slab_bytes": slabs,
                "total_bytes": table + values + slabs}

    def size(self) -> int:
        _, slots, _, _ = self._export_entries()
        return int((slots >= 0).sum())

    def total_count(self) -> int:
        return int(self.entry_counter.cpu())

    def frequencies(self, keys):
        keys = keys.to(self.device)
        slots, entry = self.ext.ht_lookup(keys, self.ht_keys, self.ht_slot,
                                          True)
        out = torch.zeros(keys.numel(), dtype=torch.int64, device=self.device)
        found = entry >= 0
        out[found] = self.ht_freq[entry[found]].to(torch.int64)
        return out

    def versions(self, keys):
        keys = keys.to(self.device)
        slots, entry = self.ext.ht_lookup(keys, self.ht_keys, self.ht_slot,
                                          True)
        out = torch.full((keys.numel(),), -1, dtype=torch.int64,
                         device=self.device)
        found = entry >= 0
        out[found] = self.ht_version[entry[found]]
        return out


# ---------------- optimizer dispatch ----------------

def sparse_apply(name: str, storage: HbmStorage, slots, grad, hyper: dict):
    ext = storage.ext
    grad = grad.float().contiguous()
    slots = slots.to(torch.int32)
    w = storage.values
    d = storage.dim
    if name == "sgd":
        ext.apply_sgd(w, slots, grad, hyper["lr"])
    elif name == "adagrad":
        accum = storage.get_slab("adagrad_accum", d,
                                 hyper.get("initial_accumulator", 0.1))
        ext.apply_adagrad(w, accum, slots, grad, hyper["lr"],
                          hyper.get("epsilon", 0.0))
    elif name == "adagrad_decay":
        accum = storage.get_slab("adagrad_accum", d,
                                 hyper.get("initial_accumulator", 0.1))
        period = storage.get_slab("adagrad_decay_period", 1, 0.0)
        cur_period = float(hyper["global_step"]
                           // max(1, hyper["accumulator_decay_step"]))
        ext.apply_adagrad_decay(
            w, accum, period, slots, grad, hyper["lr"],
            hyper.get("epsilon", 0.0), cur_period,
            hyper["accumulator_decay_rate"], hyper["accumulator_baseline"])
    elif name in ("adam", "adamw"):
        import math as _m
        mom = storage.get_slab("adam_m", d, 0.0)
        vel = storage.get_slab("adam_v", d, 0.0)
        b1, b2 = hyper["beta1"], hyper["beta2"]
        t = hyper["step_t"]
        lr_t = hyper["lr"] * _m.sqrt(1 - b2 ** t) / (1 - b1 ** t)
        if name == "adam":
            ext.apply_adam(w, mom, vel, slots, grad, lr_t, b1, b2,
                           hyper["epsilon"])
        else:
            ext.apply_adamw(w, mom, vel, slots, grad, lr_t, hyper["lr"],
                            b1, b2, hyper["epsilon"], hyper["weight_decay"])
    elif name == "adam_async":
        import math as _m
        b1, b2 = hyper["beta1"], hyper["beta2"]
        if hyper.get("powers_dev") is not None:
            mom = storage.get_slab("adam_m", d, 0.0)
            vel = storage.get_slab("adam_v", d, 0.0)
            ext.apply_adam_dev(w, mom, vel, slots, grad, hyper["lr"],
                               b1, b2, hyper["epsilon"],
                               hyper["powers_dev"])
        elif hyper.get("sparse_rmsprop"):
            vel = storage.get_slab("adam_v", d, 0.0)
            ext.apply_rmsprop(w, vel, slots, grad, hyper["lr"], b2,
                              hyper["epsilon"])
        else:
            mom = storage.get_slab("adam_m", d, 0.0)
            vel = storage.get_slab("adam_v", d, 0.0)
            lr_t = hyper["lr"] * _m.sqrt(1 - hyper["beta2_power"]) / \
                (1 - hyper["beta1_power"])
            ext.apply_adam(w, mom, vel, slots, grad, lr_t, b1, b2,
                           hyper["epsilon"])
    elif name == "ftrl":
        n = storage.get_slab("ftrl_accum", d, 0.1)
        z = storage.get_slab("ftrl_linear", d, 0.0)
        ext.apply_ftrl(w, n, z, slots, grad, hyper["lr"], hyper["l1"],
                       hyper["l2"], hyper["lr_power"],
                       hyper.get("l2_shrinkage", 0.0))
    else:
        raise ValueError(f"unknown sparse optimizer {name!r}")


def group_pooled_lookup(evs, sp_ids_list, combiners, out_dtype):
    """Grouped lookup fast path — per-table loop for now; the fused
    multi-table kernel lands with the EmbeddingCollection."""
    from deeprec_amd.embedding.lookup import embedding_lookup_sparse
    return [embedding_lookup_sparse(ev, sp, c, out_dtype)
            for ev, sp, c in zip(evs, sp_ids_list, combiners)]
