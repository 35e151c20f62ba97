"""HBM_DRAM multi-tier storage: hot tier in 288 GB HBM, cold tier in host
pinned DRAM.

Capability parity with the reference's HbmDramStorage
(reference: hbm_dram_storage.h:37,83-105 — GPU hot values + CPU cold tier,
batched staging CopyEmbeddingsFromDramToHbm, multi_tier_storage.cu.cc:42),
re-designed for this engine's slot scheme:

- the hash table + per-key metadata live entirely in HBM (24 B/entry —
  cheap even for 10^10 keys);
- value slots are monotonic: slot < hot_rows -> HBM slab row, otherwise
  row (slot - hot_rows) of a pinned host slab. Optimizer slabs mirror the
  same split;
- a lookup materializes the step's unique embeddings [m, dim]: hot rows
  gathered on-GPU, cold rows gathered on host and staged in one batched
  pinned hipMemcpyAsync (the reference's staging design), then the fused
  group pooling runs on the materialized rows;
- sparse updates: hot rows via the fused HIP applies; cold rows via the
  fp32 host applies on the pinned slabs (the cold tier is the slow path by
  construction);
- promotion: at shrink/compaction the hottest keys by frequency are
  re-packed into the HBM tier (LFU placement, reference CacheStrategy);
- SSD tier (reference: ssd_hash_kv.h mmap emb files): when
  StorageOption.storage_path is set, the cold slabs are memory-mapped
  files instead of pinned DRAM — the HBM_DRAM_SSD composition with the
  OS page cache as the DRAM middle tier.
"""
from __future__ import annotations

import os

import torch

from deeprec_amd.embedding.options import EmbeddingVariableOption
from deeprec_amd.ops.hip_backend import _COMBINER_ID, HbmStorage

_GROW = 2


class _ColdView:
    """Adapter exposing the cold (host) slabs with the CpuStorage layout the
    fp32 reference applies expect (values / dim / get_slab)."""

    def __init__(self, parent: "HbmDramStorage"):
        self._p = parent
        self.dim = parent.dim

    @property
    def values(self):
        return self._p.values_cold

    def get_slab(self, name, width, init_value, dtype=torch.float32):
        return self._p._cold_slab(name, width, init_value, dtype)


class _ColdGpuView:
    """GPU-resident working copy of selected cold rows: the fused sparse
    applies run unmodified against it (values + slot-aligned slabs), then
    flush() scatters everything back into the pinned cold slabs."""

    def __init__(self, parent: "HbmDramStorage", cs: torch.Tensor):
        self._p = parent
        self._cs = cs  # int64 cuda, cold-slab row indices
        self.dim = parent.dim
        self.ext = parent.ext
        self.values = parent.ext.gather_host_rows(parent.values_cold, cs)
        self._gathered = {}

    def get_slab(self, name, width, init_value, dtype=torch.float32):
        if name not in self._gathered:
            host = self._p._cold_slab(name, width, init_value, dtype)
            self._gathered[name] = self._p.ext.gather_host_rows(host,
                                                                self._cs)
        return self._gathered[name]

    def flush(self):
        self._p.ext.scatter_host_rows(self.values, self._cs,
                                      self._p.values_cold)
        for name, rows in self._gathered.items():
            self._p.ext.scatter_host_rows(rows, self._cs,
                                          self._p.cold_slabs[name])


class HbmDramStorage(HbmStorage):
    def __init__(self, dim: int, ev_option: EmbeddingVariableOption,
                 value_dtype=torch.float32, device=None, generator=None):
        so = ev_option.storage_option
        hot_bytes = None
        if so.storage_size:
            hot_bytes = so.storage_size[0]
        super().__init__(dim, ev_option, value_dtype, device, generator)
        row_bytes = dim * 4
        self.hot_rows = max(16, (hot_bytes // row_bytes)
                            if hot_bytes else self.max_slots)
        # keep the HBM slab at exactly the hot-tier budget
        if self.max_slots > self.hot_rows:
            self.values = self.values[: self.hot_rows].clone()
        self.storage_path = so.storage_path
        if self.storage_path:
            os.makedirs(self.storage_path, exist_ok=True)
        self.values_cold = self._alloc_cold("values", 1024, dim)
        self.cold_slabs = {}
        self._cold_slab_init = {}
        self.default_values_cpu = self.default_values.cpu()

    def _alloc_cold(self, name: str, rows: int, width: int,
                    dtype=torch.float32) -> torch.Tensor:
        """Cold-slab allocation: pinned DRAM, or a memory-mapped file under
        storage_path (the SSD tier)."""
        if not self.storage_path:
            return torch.empty(rows, width, dtype=dtype, pin_memory=True)
        fn = os.path.join(self.storage_path, f"{name}-{rows}.emb")
        t = torch.from_file(fn, shared=True, size=rows * width,
                            dtype=dtype)
        return t.view(rows, width)

    # ---------------- tier plumbing ----------------
    def enable_graph_mode(self, *a, **kw):
        raise NotImplementedError(
            "hipGraph capture is not supported with the HBM_DRAM tier "
            "(cold-row initialization requires host work per step)")

    def _init_limit(self) -> int:
        return self.hot_rows

    def _grow_slots(self, need: int):
        """Hot slab is fixed at hot_rows; growth goes to the cold tier."""
        if need <= self.hot_rows:
            return
        cold_need = need - self.hot_rows
        cap = self.values_cold.shape[0]
        if cold_need <= cap:
            return
        new_cap = cap
        while new_cap < cold_need:
            new_cap *= _GROW
        nv = self._alloc_cold("values", new_cap, self.dim)
        nv[:cap] = self.values_cold
        self.values_cold = nv
        for name, t in list(self.cold_slabs.items()):
            nt = self._alloc_cold(name, new_cap, t.shape[1], t.dtype)
            nt.fill_(self._cold_slab_init[name])
            nt[: t.shape[0]] = t
            self.cold_slabs[name] = nt

    @property
    def max_slots(self) -> int:
        # logical slot space = hot + cold capacity
        base = self.values.shape[0]
        cold = self.values_cold.shape[0] if hasattr(self, "values_cold") \
            else 0
        return base + cold

    def _cold_slab(self, name, width, init_value, dtype=torch.float32):
        if name not in self.cold_slabs:
            t = self._alloc_cold(name, self.values_cold.shape[0], width,
                                 dtype)
            t.fill_(init_value)
            self.cold_slabs[name] = t
            self._cold_slab_init[name] = init_value
        return self.cold_slabs[name]

    def get_slab(self, name, width, init_value, dtype=torch.float32):
        if name not in self.slabs:
            self.slabs[name] = torch.full(
                (self.hot_rows, width), init_value, dtype=dtype,
                device=self.device)
            self._slab_init[name] = init_value
        self._cold_slab(name, width, init_value, dtype)
        return self.slabs[name]

    # ---------------- lookup/create ----------------
    def _watermark(self) -> int:
        # exact pre-call slot watermark: slots >= prev are new this call
        # (cold tier tolerates the D2H sync; the hot-only path never pays it
        # because new slots only exceed hot_rows once the hot tier is full)
        return (int(self.slot_counter.cpu())
                if self._slots_hint >= self.hot_rows else 0)

    def _init_cold_rows(self, keys, slots, prev):
        """Host-initialize freshly-created cold rows (kernels skip them)."""
        cold_new = (slots >= self.hot_rows) & (slots >= max(
            self.hot_rows, prev))
        if bool(cold_new.any()):
            ks = keys[cold_new].cpu()
            ss = (slots[cold_new].cpu().long() - self.hot_rows)
            rows = self._default_rows_cpu(ks)
            self.values_cold[ss] = self.default_values_cpu[rows]
            for name, t in self.cold_slabs.items():
                t[ss] = self._cold_slab_init[name]

    def lookup_or_create(self, keys, counts, step, train=True):
        prev = self._watermark()
        slots = super().lookup_or_create(keys, counts, step, train)
        if train:
            self._init_cold_rows(keys, slots, prev)
        return slots

    def dedup_lookup(self, values_cat, step):
        prev = self._watermark()
        uniq, inverse, counts, slots = super().dedup_lookup(values_cat, step)
        self._init_cold_rows(uniq, slots, prev)
        return uniq, inverse, counts, slots

    def _default_rows_cpu(self, keys):
        if self.key_bits > 0:
            mask = (1 << self.key_bits) - 1
            return ((keys >> self.key_bits) * self.dvd_per_table
                    + (keys & mask) % self.dvd_per_table)
        return keys % self.default_value_dim

    # ---------------- materialized gather (staging) ----------------
    def materialize(self, keys, slots) -> torch.Tensor:
        """[m, dim] fp32 rows on GPU: hot from HBM, cold staged H2D in one
        batched copy (≙ CopyEmbeddingsFromDramToHbm)."""
        m = keys.numel()
        out = torch.empty(m, self.dim, dtype=torch.float32,
                          device=self.device)
        hot = (slots >= 0) & (slots < self.hot_rows)
        cold = slots >= self.hot_rows
        none = slots < 0
        if bool(hot.any()):
            out[hot] = self.values[slots[hot].long()]
        if bool(cold.any()):
            cs = slots[cold].long() - self.hot_rows
            if self.values_cold.is_pinned():
                # zero-copy: the GPU gathers cold rows straight out of
                # pinned DRAM (one kernel at interconnect bandwidth;
                # the CPU-gather path measured 0.4 GB/s)
                out[cold] = self.ext.gather_host_rows(self.values_cold, cs)
            else:  # mmap SSD tier: page-cache-bound host gather
                staged = self.values_cold[cs.cpu()]
                out[cold] = staged.to(self.device, non_blocking=True)
        if bool(none.any()):
            ku = keys[none]
            out[none] = self.ext.ev_gather(
                self.values, self.default_values, ku,
                torch.full((ku.numel(),), -1, dtype=torch.int32,
                           device=self.device),
                self._no_permission_value(), self._use_no_permission(),
                torch.float32)
        return out

    def gather(self, keys, slots, out_dtype=None):
        out = self.materialize(keys, slots)
        return out.to(out_dtype) if out_dtype else out

    def pooled_lookup(self, keys, slots, inverse, offsets, row_ids, combiner,
                      weights, out_dtype):
        emb = self.materialize(keys, slots)
        # reuse the direct-rows fused pooling (single-table = 1 "table")
        comb = torch.tensor([_COMBINER_ID[combiner]], dtype=torch.int32,
                            device=self.device)
        return self.ext.group_pooled_fwd_direct(
            emb, keys, inverse.to(torch.int32), offsets.to(torch.int32),
            weights.float() if weights is not None else torch.Tensor(),
            comb, offsets.numel() - 1, 1, out_dtype or torch.float32)

    # ---------------- sparse apply split ----------------
    def apply_split(self, name, slots, grad, hyper):
        from deeprec_amd.ops import hip_backend, sparse_optim_cpu
        hot = (slots >= 0) & (slots < self.hot_rows)
        cold = slots >= self.hot_rows
        if bool(hot.any()):
            hip_backend.sparse_apply(name, self, slots[hot].to(torch.int32),
                                     grad[hot], dict(hyper))
        if bool(cold.any()):
            if self.values_cold.is_pinned():
                # zero-copy round trip: gather cold rows + optimizer
                # slabs to GPU, run the SAME fused apply kernels as the
                # hot tier, scatter back — the cold tier stays
                # interconnect-bound instead of CPU-bound
                cs = slots[cold].long() - self.hot_rows
                view = _ColdGpuView(self, cs)
                m = cs.numel()
                hip_backend.sparse_apply(
                    name, view,
                    torch.arange(m, dtype=torch.int32, device=self.device),
                    grad[cold], dict(hyper))
                view.flush()
            else:  # mmap SSD tier
                cold_slots = (slots[cold].cpu().long() - self.hot_rows)
                cold_grad = grad[cold].cpu().float()
                getattr(sparse_optim_cpu, f"apply_{name}")(
                    _ColdView(self), cold_slots, cold_grad, **hyper)

    # ---------------- export / import / shrink ----------------
    def shrink(self, step: int) -> int:
        from deeprec_amd.embedding.options import (GlobalStepEvict,
                                                   L2WeightEvict)
        eo = self.ev_option.evict_option
        if eo is None:
            return 0
        names = list(self.slabs.keys())
        keys, values, freqs, versions = self.export()
        if keys.numel() == 0:
            return 0
        slab_rows = dict(zip(names, self.export_slabs(names))) \
            if names else None
        if isinstance(eo, GlobalStepEvict) and eo.steps_to_live > 0:
            keep = versions >= (step - eo.steps_to_live)
        elif isinstance(eo, L2WeightEvict) and eo.l2_weight_threshold > 0:
            keep = values.norm(dim=1) >= eo.l2_weight_threshold
        else:
            return 0
        n_evicted = int((~keep).sum())
        if n_evicted == 0:
            return 0
        # rebuild both tiers from the surviving rows
        self._alloc_table(self.capacity)
        self.entry_counter.zero_()
        self.slot_counter.zero_()
        self._entries_hint = 0
        self._slots_hint = 0
        kc = keep.cpu()
        self.import_(keys[keep], values[keep], freqs[keep].to(torch.int32),
                     versions[keep],
                     {n: r[kc] for n, r in slab_rows.items()}
                     if slab_rows else None)
        return n_evicted

    def rebalance(self) -> int:
        """LFU promotion: repack so the hottest keys (by the engine's
        per-entry frequency counters) occupy the HBM tier and the long
        tail lives in DRAM/SSD (reference capability: CacheStrategy /
        LFUCache promotion, multi_tier_storage.h). Host-coordinated and
        off the hot path — run it periodically (RebalanceHook), like
        shrink/compaction. Returns the number of rows that changed tier."""
        names = list(self.slabs.keys())
        keys, slots, freqs, versions = self._export_entries()
        adm = slots >= 0
        n = int(adm.sum())
        if n == 0:
            return 0
        from deeprec_amd.embedding.options import CacheStrategy
        lru = (self.ev_option.storage_option is not None and
               self.ev_option.storage_option.cache_strategy
               == CacheStrategy.LRU)
        # LFU ranks by frequency counters, LRU by last-touched step
        score = versions[adm] if lru else freqs[adm].long()
        s_adm = slots[adm]
        order = torch.argsort(score, descending=True, stable=True)
        ranks = torch.empty(n, dtype=torch.int32, device=self.device)
        ranks[order] = torch.arange(n, dtype=torch.int32,
                                    device=self.device)
        moved = int(((s_adm < self.hot_rows)
                     != (ranks < self.hot_rows)).sum())
        if moved == 0:
            return 0
        # materialize current rows (old placement), then rewrite in rank
        # order: ranks [0, hot_rows) -> HBM, the rest -> cold
        values = self.materialize(keys[adm], s_adm)
        slab_rows = dict(zip(names, self.export_slabs(names))) \
            if names else {}
        vals_ranked = values[order]
        hot_n = min(n, self.hot_rows)
        self.values[:hot_n] = vals_ranked[:hot_n]
        if n > hot_n:
            self._grow_slots(n)
            self.values_cold[: n - hot_n] = vals_ranked[hot_n:].cpu()
        for name, rows in slab_rows.items():
            rows_ranked = rows[order.cpu()]
            self.slabs[name][:hot_n] = rows_ranked[:hot_n].to(self.device)
            if n > hot_n:
                self.cold_slabs[name][: n - hot_n] = rows_ranked[hot_n:]
        new_slots = torch.full_like(slots, -1)
        new_slots[adm] = ranks
        self._alloc_table(self.capacity)
        self.entry_counter.zero_()
        self.slot_counter.fill_(n)
        self.ext.ht_insert_bulk(keys, new_slots, freqs, versions,
                                self.ht_keys, self.ht_slot, self.ht_freq,
                                self.ht_version, self.entry_counter,
                                self.error_flag)
        self._check_error()
        self._sync_counters()
        return moved

    def memory_usage(self) -> dict:
        out = super().memory_usage()
        cold = (self.values_cold.numel() * self.values_cold.element_size()
                + sum(t.numel() * t.element_size()
                      for t in self.cold_slabs.values()))
        out["cold_bytes"] = cold
        out["total_bytes"] += cold
        return out

    def export(self, include_filtered: bool = False):
        keys, slots, freqs, versions = self._export_entries()
        adm = slots >= 0
        k, s = keys[adm], slots[adm]
        values = self.materialize(k, s)
        out = (k, values, freqs[adm].to(torch.int64), versions[adm])
        if include_filtered:
            out = out + (keys[~adm], freqs[~adm].to(torch.int64))
        return out

    def export_slabs(self, names):
        keys, slots, freqs, versions = self._export_entries()
        adm = slots >= 0
        s = slots[adm]
        hot = s < self.hot_rows
        outs = []
        for nm in names:
            t = torch.empty(s.numel(), self.slabs[nm].shape[1])
            t[hot.cpu()] = self.slabs[nm][s[hot].long()].cpu()
            cold_idx = (s[~hot].cpu().long() - self.hot_rows)
            t[(~hot).cpu()] = self.cold_slabs[nm][cold_idx]
            outs.append(t)
        return outs

    def import_(self, keys, values, freqs=None, versions=None,
                slab_rows=None):
        # route through lookup_or_create so tier placement stays consistent
        keys = keys.to(self.device)
        m = keys.numel()
        if m == 0:
            return
        # Admission must be unconditional for checkpointed rows: every
        # (key, value) pair in a checkpoint was admitted at save time, so
        # feed counts >= filter_freq (the true freqs when recorded) or the
        # frequency filter would assign slot=-1 and the restored values
        # would silently never be written (then re-initialized on next
        # lookup — a full round trip losing trained rows).
        if freqs is not None:
            counts = freqs.to(self.device, torch.int32).clamp(
                min=max(1, self.filter_freq))
        else:
            counts = torch.full((m,), max(1, self.filter_freq),
                                dtype=torch.int32, device=self.device)
        slots = self.lookup_or_create(keys, counts, step=0, train=True)
        values = values.to(self.device, torch.float32)
        hot = (slots >= 0) & (slots < self.hot_rows)
        cold = slots >= self.hot_rows
        if bool(hot.any()):
            self.values[slots[hot].long()] = values[hot]
        if bool(cold.any()):
            ci = (slots[cold].cpu().long() - self.hot_rows)
            self.values_cold[ci] = values[cold].cpu()
        if freqs is not None:
            self.ext.ht_insert_bulk(
                keys, slots, freqs.to(self.device, torch.int32),
                versions.to(self.device, torch.int64)
                if versions is not None else torch.Tensor(),
                self.ht_keys, self.ht_slot, self.ht_freq, self.ht_version,
                self.entry_counter, self.error_flag)
        if slab_rows:
            for name, rows in slab_rows.items():
                self.get_slab(name, rows.shape[1], 0.0)
                if bool(hot.any()):
                    self.slabs[name][slots[hot].long()] = \
                        rows[hot.cpu()].to(self.device)
                if bool(cold.any()):
                    ci = (slots[cold].cpu().long() - self.hot_rows)
                    self.cold_slabs[name][ci] = rows[cold.cpu()]
        self._sync_counters()


class HbmDramSsdStorage(HbmDramStorage):
    """Three-tier composition: HBM hot rows + pinned-DRAM middle tier +
    append-only SSD files with compaction (reference capability:
    HbmDramSsdStorage, hbm_dram_ssd_storage.h + ssd_hash_kv.h).

    Slot space: [0, hot_rows) = HBM slab, [hot_rows, hot_rows+dram_rows)
    = pinned DRAM slab, >= hot_rows+dram_rows = SSD-resident (sparse,
    keyed by the EV key in one SsdKv per column family: values + each
    optimizer slab). storage_size = [hbm_bytes, dram_bytes];
    storage_path = the SSD directory.
    """

    def __init__(self, dim, ev_option, value_dtype=torch.float32,
                 device=None, generator=None):
        import copy

        from deeprec_amd.embedding.ssd_kv import SsdKv
        so = ev_option.storage_option
        assert so.storage_path, "HBM_DRAM_SSD requires storage_path"
        assert so.storage_size and len(so.storage_size) >= 2, \
            "HBM_DRAM_SSD requires storage_size=[hbm_bytes, dram_bytes]"
        self._ssd_path = so.storage_path
        opt = copy.deepcopy(ev_option)
        # the base class builds the pinned-DRAM middle tier; the SSD dir
        # must not trigger its mmap-slab mode
        opt.storage_option.storage_path = None
        opt.storage_option.storage_size = [so.storage_size[0]]
        super().__init__(dim, opt, value_dtype, device, generator)
        self.dram_rows = max(16, so.storage_size[1] // (dim * 4))
        self.ssd_kvs = {"values": SsdKv(
            os.path.join(self._ssd_path, "values"), dim)}
        self._ssd_slab_init = {}

    # ---------------- tier plumbing ----------------
    @property
    def ssd_base(self) -> int:
        return self.hot_rows + self.dram_rows

    @property
    def max_slots(self) -> int:
        return (1 << 31) - 2  # SSD tier is unbounded (sparse, key-addressed)

    def _grow_slots(self, need: int):
        # DRAM middle tier caps at dram_rows; beyond that rows live on SSD
        super()._grow_slots(min(need, self.ssd_base))

    def _ssd_kv(self, name, width, init_value):
        from deeprec_amd.embedding.ssd_kv import SsdKv
        if name not in self.ssd_kvs:
            self.ssd_kvs[name] = SsdKv(
                os.path.join(self._ssd_path, name), width)
            self._ssd_slab_init[name] = init_value
        return self.ssd_kvs[name]

    def get_slab(self, name, width, init_value, dtype=torch.float32):
        out = super().get_slab(name, width, init_value, dtype)
        self._ssd_kv(name, width, init_value)
        self._ssd_slab_init[name] = init_value
        return out

    def _init_cold_rows(self, keys, slots, prev):
        dram_mask = (slots >= self.hot_rows) & (slots < self.ssd_base) & \
            (slots >= max(self.hot_rows, prev))
        if bool(dram_mask.any()):
            ks = keys[dram_mask].cpu()
            ss = (slots[dram_mask].cpu().long() - self.hot_rows)
            rows = self._default_rows_cpu(ks)
            self.values_cold[ss] = self.default_values_cpu[rows]
            for name, t in self.cold_slabs.items():
                t[ss] = self._cold_slab_init[name]
        ssd_new = (slots >= self.ssd_base) & (slots >= max(self.ssd_base,
                                                           prev))
        if bool(ssd_new.any()):
            ks = keys[ssd_new].cpu()
            rows = self._default_rows_cpu(ks)
            self.ssd_kvs["values"].write(
                ks, self.default_values_cpu[rows])
            # optimizer slabs default-init lazily on first read

    # ---------------- materialized gather ----------------
    def materialize(self, keys, slots) -> torch.Tensor:
        m = keys.numel()
        out = torch.empty(m, self.dim, dtype=torch.float32,
                          device=self.device)
        hot = (slots >= 0) & (slots < self.hot_rows)
        dram = (slots >= self.hot_rows) & (slots < self.ssd_base)
        ssd = slots >= self.ssd_base
        none = slots < 0
        if bool(hot.any()):
            out[hot] = self.values[slots[hot].long()]
        if bool(dram.any()):
            cs = slots[dram].long() - self.hot_rows
            out[dram] = self.ext.gather_host_rows(self.values_cold, cs)
        if bool(ssd.any()):
            rows = self.ssd_kvs["values"].read(keys[ssd])
            out[ssd] = rows.to(self.device)
        if bool(none.any()):
            ku = keys[none]
            out[none] = self.ext.ev_gather(
                self.values, self.default_values, ku,
                torch.full((ku.numel(),), -1, dtype=torch.int32,
                           device=self.device),
                self._no_permission_value(), self._use_no_permission(),
                torch.float32)
        return out

    # ---------------- sparse apply ----------------
    def apply_split(self, name, slots, grad, hyper):
        from deeprec_amd.ops import hip_backend
        in_mem = slots < self.ssd_base
        ssd = slots >= self.ssd_base
        if bool(in_mem.any()):
            super().apply_split(name, slots[in_mem], grad[in_mem], hyper)
        if bool(ssd.any()):
            # read-modify-write through the KV: the coldest tier is the
            # slow path by construction (reference semantics)
            keys = getattr(self, "_apply_keys", None)
            assert keys is not None, \
                "apply on SSD rows requires set_apply_keys(keys)"
            ks = keys[ssd]
            view = _SsdApplyView(self, ks)
            m = ks.numel()
            hip_backend.sparse_apply(
                name, view,
                torch.arange(m, dtype=torch.int32, device=self.device),
                grad[ssd], dict(hyper))
            view.flush(ks)

    def set_apply_keys(self, keys):
        """The SSD tier is key-addressed; optimizers call this before
        apply_split so SSD-resident rows can be located."""
        self._apply_keys = keys

    # ---------------- export / import ----------------
    def export(self, include_filtered: bool = False):
        keys, slots, freqs, versions = self._export_entries()
        adm = slots >= 0
        k, s = keys[adm], slots[adm]
        values = self.materialize(k, s)
        out = (k, values, freqs[adm].to(torch.int64), versions[adm])
        if include_filtered:
            out = out + (keys[~adm], freqs[~adm].to(torch.int64))
        return out

    def export_slabs(self, names):
        keys, slots, freqs, versions = self._export_entries()
        adm = slots >= 0
        k, s = keys[adm], slots[adm]
        hot = s < self.hot_rows
        dram = (s >= self.hot_rows) & (s < self.ssd_base)
        ssd = s >= self.ssd_base
        outs = []
        for nm in names:
            w = self.slabs[nm].shape[1]
            t = torch.empty(s.numel(), w)
            t[hot.cpu()] = self.slabs[nm][s[hot].long()].cpu()
            di = (s[dram].cpu().long() - self.hot_rows)
            t[dram.cpu()] = self.cold_slabs[nm][di]
            if bool(ssd.any()):
                t[ssd.cpu()] = self.ssd_kvs[nm].read(
                    k[ssd], default=self._ssd_slab_init.get(nm, 0.0))
            outs.append(t)
        return outs

    def shrink(self, step: int) -> int:
        raise NotImplementedError(
            "HBM_DRAM_SSD shrink: evict via EvictionManager / checkpoint "
            "repartition (full-rebuild shrink would need an SSD rewrite)")

    def rebalance(self) -> int:
        raise NotImplementedError(
            "HBM_DRAM_SSD rebalance: use the background EvictionManager "
            "(incremental promote/demote); a stop-the-world repack of an "
            "SSD-backed table is deliberately unsupported")

    def import_(self, keys, values, freqs=None, versions=None,
                slab_rows=None):
        keys = keys.to(self.device)
        m = keys.numel()
        if m == 0:
            return
        if freqs is not None:
            counts = freqs.to(self.device, torch.int32).clamp(
                min=max(1, self.filter_freq))
        else:
            counts = torch.full((m,), max(1, self.filter_freq),
                                dtype=torch.int32, device=self.device)
        slots = self.lookup_or_create(keys, counts, step=0, train=True)
        values = values.to(self.device, torch.float32)
        hot = (slots >= 0) & (slots < self.hot_rows)
        dram = (slots >= self.hot_rows) & (slots < self.ssd_base)
        ssd = slots >= self.ssd_base
        if bool(hot.any()):
            self.values[slots[hot].long()] = values[hot]
        if bool(dram.any()):
            ci = (slots[dram].cpu().long() - self.hot_rows)
            self.values_cold[ci] = values[dram].cpu()
        if bool(ssd.any()):
            self.ssd_kvs["values"].write(keys[ssd].cpu(),
                                         values[ssd].cpu())
        if freqs is not None:
            self.ext.ht_insert_bulk(
                keys, slots, freqs.to(self.device, torch.int32),
                versions.to(self.device, torch.int64)
                if versions is not None else torch.Tensor(),
                self.ht_keys, self.ht_slot, self.ht_freq, self.ht_version,
                self.entry_counter, self.error_flag)
        if slab_rows:
            for name, rows in slab_rows.items():
                self.get_slab(name, rows.shape[1], 0.0)
                if bool(hot.any()):
                    self.slabs[name][slots[hot].long()] = \
                        rows[hot.cpu()].to(self.device)
                if bool(dram.any()):
                    ci = (slots[dram].cpu().long() - self.hot_rows)
                    self.cold_slabs[name][ci] = rows[dram.cpu()]
                if bool(ssd.any()):
                    self.ssd_kvs[name].write(keys[ssd].cpu(),
                                             rows[ssd.cpu()])
        self._sync_counters()

    def memory_usage(self) -> dict:
        out = super().memory_usage()
        out["ssd_rows"] = sum(kv.size() for kv in self.ssd_kvs.values())
        out["ssd_files"] = sum(kv.file_count()
                               for kv in self.ssd_kvs.values())
        return out

    def compact_ssd(self, sync=True) -> int:
        return sum(kv.compact(sync=sync) for kv in self.ssd_kvs.values())


class _SsdApplyView:
    """GPU working copy of SSD-resident rows for the fused applies."""

    def __init__(self, parent: HbmDramSsdStorage, keys: torch.Tensor):
        self._p = parent
        self._keys = keys
        self.dim = parent.dim
        self.ext = parent.ext
        self.values = parent.ssd_kvs["values"].read(keys).to(parent.device)
        self._gathered = {}

    def get_slab(self, name, width, init_value, dtype=torch.float32):
        if name not in self._gathered:
            kv = self._p._ssd_kv(name, width, init_value)
            self._gathered[name] = kv.read(
                self._keys, default=init_value).to(self._p.device)
        return self._gathered[name]

    def flush(self, keys):
        self._p.ssd_kvs["values"].write(keys, self.values.cpu())
        for name, rows in self._gathered.items():
            self._p.ssd_kvs[name].write(keys, rows.cpu())
