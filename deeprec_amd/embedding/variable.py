"""EmbeddingVariable — dynamic-shape hash embedding.

The heart of the framework (reference capability: EmbeddingVar<K,V>,
embedding_var.h:53-800): a per-feature KV table keyed by int64 feature id,
value = trainable [dim] vector, with per-key frequency/version metadata,
feature admission filters, eviction policies, default-value initialization
(row k % default_value_dim of a default matrix) and optimizer-state slabs
aligned to value slots.

MI355X design: values live in a slot-indexed HBM slab behind an
open-addressing GPU hash table (ops/hip/ev_kernels.hip); one hash probe per
training step returns slot indices that the fused gather/pool, grad-scatter
and sparse-optimizer kernels all reuse (the reference's `_OPT_`
indices-as-pointers fusion, ops/kv_variable_ops.cc:636). The CPU backend
(ops/cpu_backend.py) implements identical semantics for CPU training and as
the GPU numerics reference.
"""
from __future__ import annotations

import threading
from typing import List, Optional

import torch

from deeprec_amd.embedding.options import EmbeddingVariableOption

_REGISTRY = {}
_REGISTRY_LOCK = threading.Lock()


class _GlobalStep:
    """Process-wide training step counter (reference: tf global_step)."""

    def __init__(self):
        self.value = 0

    def increment(self):
        self.value += 1


GLOBAL_STEP = _GlobalStep()


def get_global_step() -> int:
    return GLOBAL_STEP.value


class EmbeddingVariable:
    def __init__(self, name: str, embedding_dim: int,
                 value_dtype=torch.float32,
                 ev_option: Optional[EmbeddingVariableOption] = None,
                 device=None, generator=None, trainable: bool = True,
                 invalid_key: Optional[int] = None):
        self.name = name
        self.dim = embedding_dim
        # sentinel id for "no feature" (reference:
        # tf.get_embedding_variable invalid-key per dtype,
        # variable_scope.py:2146): lookups drop it (zeros in pooled
        # output), it is never admitted and never trains
        self.invalid_key = invalid_key
        self.value_dtype = value_dtype
        self.ev_option = ev_option or EmbeddingVariableOption()
        self.device = torch.device(device or "cpu")
        self.trainable = trainable

        if self.device.type == "cuda":
            from deeprec_amd.ops.hip_backend import HbmStorage
            from deeprec_amd.embedding.options import StorageType
            st = self.ev_option.storage_option.storage_type
            if st == StorageType.HBM_DRAM:
                from deeprec_amd.ops.hbm_dram_backend import HbmDramStorage
                self.storage = HbmDramStorage(embedding_dim, self.ev_option,
                                              value_dtype, self.device,
                                              generator)
            elif st == StorageType.HBM_DRAM_SSD:
                from deeprec_amd.ops.hbm_dram_backend import (
                    HbmDramSsdStorage)
                self.storage = HbmDramSsdStorage(
                    embedding_dim, self.ev_option, value_dtype,
                    self.device, generator)
            else:
                self.storage = HbmStorage(embedding_dim, self.ev_option,
                                          value_dtype, self.device, generator)
        else:
            from deeprec_amd.ops.cpu_backend import CpuStorage
            self.storage = CpuStorage(embedding_dim, self.ev_option,
                                      value_dtype, self.device, generator)

        # autograd anchor: lookups depend on it so backward fires and routes
        # sparse grads into _pending_grads (consumed by the optimizer step)
        self._anchor = torch.zeros(
            (), device=self.device, requires_grad=trainable)
        self._pending_grads: List = []
        # recorded ids since the last full/incremental save
        # (reference: RecordSparseIndices, incr_save_restore_ops.cc:22)
        self._recorded_ids: List[torch.Tensor] = []
        self._record_sparse_ids = False

    # ------------- training-step interface -------------
    def lookup_or_create(self, uniq_keys: torch.Tensor,
                         counts: torch.Tensor = None,
                         train: bool = True) -> torch.Tensor:
        slots = self.storage.lookup_or_create(
            uniq_keys, counts, get_global_step(), train=train)
        if train and self._record_sparse_ids:
            self._recorded_ids.append(uniq_keys.detach())
        return slots

    def lookup_tier(self, keys: torch.Tensor) -> torch.Tensor:
        """Per-key storage tier: −1 absent, 0 HBM/resident, 1 DRAM,
        2 SSD (reference: KvResourceLookupTier op,
        kernels/kv_variable_lookup_ops.cc:537)."""
        return self.storage.lookup_tier(keys)

    def gather(self, keys: torch.Tensor, out_dtype=None) -> torch.Tensor:
        """Inference read: no insert, default value for missing keys."""
        uniq, inverse = torch.unique(keys, return_inverse=True)
        slots = self.storage.lookup(uniq)
        vals = self.storage.gather(uniq, slots, out_dtype)
        return vals[inverse]

    def accumulate_grad(self, slots, keys, grad_unique):
        self._pending_grads.append((slots, keys, grad_unique))

    def consume_grads(self):
        out = self._pending_grads
        self._pending_grads = []
        return out

    def get_slab(self, name, width=None, init_value=0.0, dtype=torch.float32):
        return self.storage.get_slab(name, width or self.dim, init_value, dtype)

    # ------------- bookkeeping / checkpoint -------------
    def export(self, include_filtered: bool = False):
        return self.storage.export(include_filtered)

    def restore(self, keys, values, freqs=None, versions=None):
        self.storage.import_(keys, values, freqs, versions)

    def memory_usage(self) -> dict:
        return self.storage.memory_usage()

    def shrink(self, step: Optional[int] = None) -> int:
        return self.storage.shrink(
            step if step is not None else get_global_step())

    def rebalance(self) -> int:
        """LFU hot/cold repack for multi-tier storages; 0 elsewhere."""
        if hasattr(self.storage, "rebalance"):
            return self.storage.rebalance()
        return 0

    def start_sparse_recording(self):
        self._record_sparse_ids = True

    def consume_recorded_ids(self) -> torch.Tensor:
        if not self._recorded_ids:
            return torch.empty(0, dtype=torch.int64, device=self.device)
        ids = torch.unique(torch.cat(self._recorded_ids))
        self._recorded_ids = []
        return ids

    def total_count(self) -> int:
        return self.storage.total_count()

    def size(self) -> int:
        return self.storage.size()

    def get_frequency(self, keys) -> torch.Tensor:
        return self.storage.frequencies(keys)

    def get_version(self, keys) -> torch.Tensor:
        return self.storage.versions(keys)

    def __repr__(self):
        return (f"EmbeddingVariable(name={self.name!r}, dim={self.dim}, "
                f"device={self.device}, size={self.size()})")


def get_embedding_variable(name: str, embedding_dim: int,
                           value_dtype=torch.float32,
                           initializer=None,
                           ev_option: Optional[EmbeddingVariableOption] = None,
                           device=None, trainable: bool = True,
                           reuse: bool = True,
                           invalid_key: Optional[int] = None
                           ) -> EmbeddingVariable:
    """Create-or-reuse an EV by name (reference: tf.get_embedding_variable,
    python/ops/variable_scope.py:2147)."""
    with _REGISTRY_LOCK:
        if name in _REGISTRY:
            if not reuse:
                raise ValueError(f"EmbeddingVariable {name!r} already exists")
            ev = _REGISTRY[name]
            if ev.dim != embedding_dim:
                raise ValueError(
                    f"EmbeddingVariable {name!r} dim mismatch: "
                    f"{ev.dim} vs {embedding_dim}")
            return ev
        ev_option = ev_option or EmbeddingVariableOption()
        if initializer is not None:
            ev_option.init_option.initializer = initializer
        ev = EmbeddingVariable(name, embedding_dim, value_dtype, ev_option,
                               device, trainable=trainable,
                               invalid_key=invalid_key)
        _REGISTRY[name] = ev
        return ev


def all_embedding_variables():
    with _REGISTRY_LOCK:
        return list(_REGISTRY.values())


def reset_registry():
    """Test helper — clears the global EV registry."""
    with _REGISTRY_LOCK:
        _REGISTRY.clear()
