"""CPU (DRAM) storage backend for EmbeddingVariable.

Functional reference implementation of the embedding engine: identical
semantics to the HIP/HBM backend (deeprec_amd/ops/hip/ev_kernels.hip) so
that (a) CPU-only CI validates the whole training stack and (b) GPU
numerics tests can compare the HIP kernels against this fp32 path.

Reference capability: DramStorage / LocklessHashMap
(reference: single_tier_storage.h:332, cpu_hash_map_kv.h:27) — here a
Python dict index over torch slab tensors; throughput on CPU is a
non-goal beyond the Wide&Deep CPU plumbing config.
"""
from __future__ import annotations

import math
import numpy as np
import torch

from deeprec_amd.embedding.options import (
    CBFFilter,
    CounterFilter,
    EmbeddingVariableOption,
    GlobalStepEvict,
    L2WeightEvict,
)

_GROW = 2


class _CountingBloom:
    """Counting bloom filter for CBF admission (reference: bloom_filter_policy.h)."""

    def __init__(self, max_elements: int, fpp: float, counter_dtype=np.uint16):
        max_elements = max(max_elements, 1024)
        nbits = int(-max_elements * math.log(max(fpp, 1e-9)) / (math.log(2) ** 2))
        self.nbits = max(1024, nbits)
        self.k = max(1, int(round(self.nbits / max_elements * math.log(2))))
        self.counters = np.zeros(self.nbits, dtype=counter_dtype)

    def _positions(self, keys: np.ndarray) -> np.ndarray:
        # double hashing: h_i = h1 + i*h2 (mod nbits)
        k64 = keys.astype(np.uint64)
        h1 = (k64 * np.uint64(0x9E3779B97F4A7C15)) >> np.uint64(17)
        h2 = (k64 ^ (k64 >> np.uint64(33))) * np.uint64(0xC2B2AE3D27D4EB4F) | np.uint64(1)
        i = np.arange(self.k, dtype=np.uint64)[:, None]
        return ((h1[None, :] + i * h2[None, :]) % np.uint64(self.nbits)).astype(np.int64)

    def add_and_estimate(self, keys: np.ndarray, counts: np.ndarray) -> np.ndarray:
        """Increment each key's k counters by its count; return min-counter estimate."""
        pos = self._positions(keys)  # [k, n]
        maxv = np.iinfo(self.counters.dtype).max
        for r in range(self.k):
            np.add.at(self.counters, pos[r],
                      np.minimum(counts, maxv).astype(self.counters.dtype))
        est = self.counters[pos].min(axis=0)
        return est.astype(np.int64)


class CpuStorage:
    """Single-tier DRAM KV storage: key -> (slot, freq, version) + value slabs."""

    def __init__(self, dim: int, ev_option: EmbeddingVariableOption,
                 value_dtype=torch.float32, device="cpu", generator=None):
        self.dim = dim
        self.device = torch.device(device)
        self.value_dtype = value_dtype
        self.ev_option = ev_option

        io = ev_option.init_option
        self.default_value_dim = max(1, io.default_value_dim)
        # composite-key support (EmbeddingCollection): 0 = plain keys
        self.key_bits = 0
        self.dvd_per_table = self.default_value_dim
        self.default_values = torch.empty(
            self.default_value_dim, dim, dtype=value_dtype)
        if io.initializer is None:
            self.default_values.normal_(0.0, 1.0 / math.sqrt(dim),
                                        generator=generator)
        elif callable(io.initializer):
            io.initializer(self.default_values)
        else:  # constant
            self.default_values.fill_(float(io.initializer))

        fo = ev_option.filter_option
        self.filter_freq = 0
        self._cbf = None
        if isinstance(fo, CounterFilter):
            self.filter_freq = fo.filter_freq
        elif isinstance(fo, CBFFilter):
            self.filter_freq = fo.filter_freq
            self._cbf = _CountingBloom(fo.max_element_size,
                                       fo.false_positive_probability)

        # entry-indexed metadata (an entry exists for every seen key unless CBF)
        self._key2entry: dict = {}
        cap = 1024
        self.entry_key = torch.full((cap,), 0, dtype=torch.int64)
        self.entry_slot = torch.full((cap,), -1, dtype=torch.int64)
        self.entry_freq = torch.zeros(cap, dtype=torch.int64)
        self.entry_version = torch.full((cap,), -1, dtype=torch.int64)
        self.entry_alive = torch.zeros(cap, dtype=torch.bool)
        self.n_entries = 0

        # slot-indexed value slabs
        self.values = torch.empty(cap, dim, dtype=value_dtype)
        self.slabs: dict = {}   # name -> tensor [max_slots, k]
        self._slab_init: dict = {}
        self.slot_count = 0
        self._free_slots: list = []

    # ---------------- internal growth ----------------
    def _grow_entries(self, need: int):
        cap = self.entry_key.numel()
        if self.n_entries + need <= cap:
            return
        new_cap = cap
        while new_cap < self.n_entries + need:
            new_cap *= _GROW

        def grow(t, fill):
            out = torch.full((new_cap, *t.shape[1:]), fill, dtype=t.dtype)
            out[:cap] = t
            return out

        self.entry_key = grow(self.entry_key, 0)
        self.entry_slot = grow(self.entry_slot, -1)
        self.entry_freq = grow(self.entry_freq, 0)
        self.entry_version = grow(self.entry_version, -1)
        alive = torch.zeros(new_cap, dtype=torch.bool)
        alive[:cap] = self.entry_alive
        self.entry_alive = alive

    def _grow_slots(self, need: int):
        cap = self.values.shape[0]
        if self.slot_count + need <= cap:
            return
        new_cap = cap
        while new_cap < self.slot_count + need:
            new_cap *= _GROW
        v = torch.empty(new_cap, self.dim, dtype=self.value_dtype)
        v[:cap] = self.values
        self.values = v
        for name, t in list(self.slabs.items()):
            nt = torch.full((new_cap, t.shape[1]), self._slab_init[name],
                            dtype=t.dtype)
            nt[:cap] = t
            self.slabs[name] = nt

    def _alloc_slot(self) -> int:
        if self._free_slots:
            s = self._free_slots.pop()
        else:
            self._grow_slots(1)
            s = self.slot_count
            self.slot_count += 1
        return s

    def _default_row(self, k: int) -> int:
        if self.key_bits > 0:
            mask = (1 << self.key_bits) - 1
            return ((k >> self.key_bits) * self.dvd_per_table
                    + (k & mask) % self.dvd_per_table)
        return k % self.default_value_dim

    def _default_rows(self, keys):
        if self.key_bits > 0:
            mask = (1 << self.key_bits) - 1
            return ((keys >> self.key_bits) * self.dvd_per_table
                    + (keys & mask) % self.dvd_per_table)
        return (keys % self.default_value_dim).clamp(min=0)

    # ---------------- public interface (mirrors HbmStorage) ----------------
    def get_slab(self, name: str, width: int, init_value: float,
                 dtype=torch.float32) -> torch.Tensor:
        """Optimizer state slab aligned with value slots (reference: slot_num
        blocks in embedding_config.h). Rows init to init_value."""
        if name not in self.slabs:
            t = torch.full((self.values.shape[0], width), init_value, dtype=dtype)
            self.slabs[name] = t
            self._slab_init[name] = init_value
        return self.slabs[name]

    def lookup_or_create(self, keys: torch.Tensor, counts: torch.Tensor,
                         step: int, train: bool = True) -> torch.Tensor:
        """keys: unique int64[m]; counts: occurrence count per key in batch.

        Returns slots int64[m]; -1 = not admitted (use default value).
        Single hash probe per step — the reference's `_OPT_` pointer-passing
        fusion (ops/kv_variable_ops.cc:636) is the model: downstream gather
        and sparse-apply reuse these slots with no second probe.
        """
        m = keys.numel()
        slots = torch.full((m,), -1, dtype=torch.int64)
        if m == 0:
            return slots
        keys_np = keys.numpy()
        counts_np = counts.numpy() if counts is not None else np.ones(m, np.int64)

        cbf_est = None
        if self._cbf is not None and train:
            cbf_est = self._cbf.add_and_estimate(keys_np, counts_np)

        self._grow_entries(m)
        k2e = self._key2entry
        for i in range(m):
            k = int(keys_np[i])
            e = k2e.get(k)
            if e is None:
                if not train:
                    continue
                if self._cbf is not None and cbf_est[i] < self.filter_freq:
                    continue  # pre-admission counts live only in the CBF
                e = self.n_entries
                self.n_entries += 1
                k2e[k] = e
                self.entry_key[e] = k
                self.entry_alive[e] = True
                # under CBF admission, seed freq from the bloom estimate so
                # the slot-admission check below sees the true count
                self.entry_freq[e] = (
                    int(cbf_est[i]) - int(counts_np[i])
                    if cbf_est is not None else 0)
            if train:
                self.entry_freq[e] += int(counts_np[i])
                self.entry_version[e] = step
            s = int(self.entry_slot[e])
            if s < 0 and train and int(self.entry_freq[e]) >= self.filter_freq:
                s = self._alloc_slot()
                self.entry_slot[e] = s
                self.values[s] = self.default_values[self._default_row(k)]
                for name, t in self.slabs.items():
                    t[s].fill_(self._slab_init[name])
            slots[i] = s
        return slots

    def lookup(self, keys: torch.Tensor) -> torch.Tensor:
        """Probe without insert (serving path)."""
        return self.lookup_or_create(keys, None, 0, train=False)

    def lookup_tier(self, keys: torch.Tensor) -> torch.Tensor:
        """Reference: KvResourceLookupTier
        (kernels/kv_variable_lookup_ops.cc:537): −1 = not present,
        0 = the (single) resident tier."""
        slots = self.lookup(keys).long()
        return torch.where(slots >= 0, torch.zeros_like(slots),
                           torch.full_like(slots, -1))

    def gather(self, keys: torch.Tensor, slots: torch.Tensor,
               out_dtype=None) -> torch.Tensor:
        """values[slots] with default-value fill for slot<0."""
        out_dtype = out_dtype or self.value_dtype
        admitted = slots >= 0
        out = self.default_values[self._default_rows(keys)].clone()
        if admitted.any():
            out[admitted] = self.values[slots[admitted]]
        if self.ev_option.init_option.default_value_no_permission is not None \
                and self.filter_freq > 0:
            out[~admitted] = self.ev_option.init_option.default_value_no_permission
        return out.to(out_dtype)

    # ---------------- shrink / export / import ----------------
    def shrink(self, step: int):
        """Eviction at checkpoint save (reference: shrink_policy.h)."""
        eo = self.ev_option.evict_option
        if eo is None or self.n_entries == 0:
            return 0
        alive_idx = self.entry_alive[: self.n_entries].nonzero().squeeze(1)
        if isinstance(eo, GlobalStepEvict) and eo.steps_to_live > 0:
            dead = alive_idx[
                self.entry_version[alive_idx] < (step - eo.steps_to_live)]
        elif isinstance(eo, L2WeightEvict) and eo.l2_weight_threshold > 0:
            slots = self.entry_slot[alive_idx]
            has = slots >= 0
            norms = torch.zeros(alive_idx.numel())
            norms[has] = self.values[slots[has]].float().norm(dim=1)
            dead = alive_idx[has & (norms < eo.l2_weight_threshold)]
        else:
            return 0
        for e in dead.tolist():
            k = int(self.entry_key[e])
            self._key2entry.pop(k, None)
            self.entry_alive[e] = False
            s = int(self.entry_slot[e])
            if s >= 0:
                self._free_slots.append(s)
            self.entry_slot[e] = -1
        return int(dead.numel())

    def export(self, include_filtered: bool = False):
        """-> (keys, values, freqs, versions) of admitted entries
        (+ optionally (filtered_keys, filtered_freqs))."""
        n = self.n_entries
        alive = self.entry_alive[:n]
        adm = alive & (self.entry_slot[:n] >= 0)
        idx = adm.nonzero().squeeze(1)
        keys = self.entry_key[idx].clone()
        values = self.values[self.entry_slot[idx]].clone()
        freqs = self.entry_freq[idx].clone()
        versions = self.entry_version[idx].clone()
        out = (keys, values, freqs, versions)
        if include_filtered:
            fidx = (alive & (self.entry_slot[:n] < 0)).nonzero().squeeze(1)
            out = out + (self.entry_key[fidx].clone(),
                         self.entry_freq[fidx].clone())
        return out

    def export_slabs(self, names):
        """Optimizer-state rows aligned with export() key order."""
        n = self.n_entries
        adm = self.entry_alive[:n] & (self.entry_slot[:n] >= 0)
        slots = self.entry_slot[:n][adm]
        return [self.slabs[nm][slots].clone() for nm in names]

    def import_(self, keys, values, freqs=None, versions=None, slab_rows=None):
        m = keys.numel()
        if m == 0:
            return
        self._grow_entries(m)
        self._grow_slots(m)
        for i in range(m):
            k = int(keys[i])
            e = self._key2entry.get(k)
            if e is None:
                e = self.n_entries
                self.n_entries += 1
                self._key2entry[k] = e
                self.entry_key[e] = k
                self.entry_alive[e] = True
                self.entry_slot[e] = -1
            if freqs is not None:
                self.entry_freq[e] = int(freqs[i])
            if versions is not None:
                self.entry_version[e] = int(versions[i])
            s = int(self.entry_slot[e])
            if s < 0:
                s = self._alloc_slot()
                self.entry_slot[e] = s
            self.values[s] = values[i]
            if slab_rows is not None:
                for name, rows in slab_rows.items():
                    self.get_slab(name, rows.shape[1], 0.0)[s] = rows[i]

    def import_filtered(self, keys, freqs):
        """Restore sub-threshold admission counters (reference capability:
        TF_EV_SAVE_FILTERED_FEATURES, embedding_var.h:533): entries exist
        with their frequency but no value slot until they cross the
        filter threshold."""
        m = keys.numel()
        if m == 0:
            return
        self._grow_entries(m)
        for i in range(m):
            k = int(keys[i])
            e = self._key2entry.get(k)
            if e is None:
                e = self.n_entries
                self.n_entries += 1
                self._key2entry[k] = e
                self.entry_key[e] = k
                self.entry_alive[e] = True
                self.entry_slot[e] = -1
            self.entry_freq[e] = int(freqs[i])

    def memory_usage(self) -> dict:
        values = self.values.numel() * self.values.element_size()
        slabs = sum(t.numel() * t.element_size()
                    for t in self.slabs.values())
        n = self.n_entries
        table = n * (8 + 4 + 4 + 8 + 1)  # key/slot/freq/version/alive
        return {"table_bytes": table, "values_bytes": values,
                "slab_bytes": slabs,
                "total_bytes": table + values + slabs}

    def size(self) -> int:
        n = self.n_entries
        return int((self.entry_alive[:n] & (self.entry_slot[:n] >= 0)).sum())

    def total_count(self) -> int:
        return int(self.entry_alive[: self.n_entries].sum())

    def frequencies(self, keys: torch.Tensor) -> torch.Tensor:
        out = torch.zeros(keys.numel(), dtype=torch.int64)
        for i, k in enumerate(keys.tolist()):
            e = self._key2entry.get(int(k))
            if e is not None:
                out[i] = self.entry_freq[e]
        return out

    def versions(self, keys: torch.Tensor) -> torch.Tensor:
        out = torch.full((keys.numel(),), -1, dtype=torch.int64)
        for i, k in enumerate(keys.tolist()):
            e = self._key2entry.get(int(k))
            if e is not None:
                out[i] = self.entry_version[e]
        return out
