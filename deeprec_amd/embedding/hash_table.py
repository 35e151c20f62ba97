"""The reference's second (PAI) KV API: HashTable / DistributedHashTable.

Capability parity with python/ops/hash_table/hash_table.py:45 (
SimpleHashTable), :141 (HashTable), :388 (DistributedHashTable) and the
admit-strategy hooks (hash_table_lookup_with_admit_op,
hash_filter.py BloomFilterAdmitStrategy). This predates EmbeddingVariable
in the reference and coexists with it; here both APIs are facades over
the same MI355X engine (ops/hip_backend), so the whole op layer
(hash_training_ops.cc etc.) collapses into thin delegation.
"""
from __future__ import annotations

from typing import Optional, Sequence

import torch

from deeprec_amd.embedding.options import (
    CBFFilter,
    CounterFilter,
    EmbeddingVariableOption,
    InitializerOption,
)
from deeprec_amd.embedding.variable import EmbeddingVariable


class AdmitStrategy:
    """Base admission strategy (reference: hash_filter admit ops)."""

    def to_filter(self):
        return None


class BloomFilterAdmitStrategy(AdmitStrategy):
    """Admit a key after min_frequency sightings via a counting-bloom
    pre-table (reference: BloomFilterAdmitStrategy)."""

    def __init__(self, min_frequency: int, max_element_size: int = 1 << 20,
                 false_positive_probability: float = 0.01):
        self.min_frequency = min_frequency
        self.max_element_size = max_element_size
        self.fpp = false_positive_probability

    def to_filter(self):
        return CBFFilter(filter_freq=self.min_frequency,
                         max_element_size=self.max_element_size,
                         false_positive_probability=self.fpp)


class CounterAdmitStrategy(AdmitStrategy):
    def __init__(self, min_frequency: int):
        self.min_frequency = min_frequency

    def to_filter(self):
        return CounterFilter(filter_freq=self.min_frequency)


class HashTable:
    """Dynamic-shape hashed variable: keys int64 -> value rows of `shape`.

    lookup(keys, admit=True) returns trainable rows (creating/admitting
    as the strategy allows); admit=False is a read-only probe.
    """

    def __init__(self, shape: Sequence[int], dtype=torch.float32,
                 name: str = "hash_table", initializer=None,
                 admit_strategy: Optional[AdmitStrategy] = None,
                 device=None, trainable: bool = True):
        assert len(shape) == 1, "value rows are 1-D (dim,) like the EV"
        self.shape = tuple(shape)
        self.name = name
        opt = EmbeddingVariableOption(
            init_option=InitializerOption(initializer=initializer),
            filter_option=(admit_strategy.to_filter()
                           if admit_strategy else None))
        self._ev = EmbeddingVariable(name, int(shape[0]), ev_option=opt,
                                     device=device, trainable=trainable)

    @property
    def handle(self):
        return self._ev

    def lookup(self, keys: torch.Tensor, admit: bool = True):
        from deeprec_amd.embedding.lookup import embedding_lookup
        return embedding_lookup(self._ev, keys, train=admit)

    def lookup_with_admit(self, keys, frequencies=None):
        # frequency hints ride the engine's exact per-entry counters
        return self.lookup(keys, admit=True)

    def size(self) -> int:
        return self._ev.size()

    def frequencies(self, keys):
        return self._ev.storage.frequencies(keys.to(self._ev.device))

    def export(self):
        return self._ev.export()

    def embedding_variable(self) -> EmbeddingVariable:
        """Escape hatch to the primary API (optimizers/savers take EVs)."""
        return self._ev


class DistributedHashTable:
    """N-way partitioned HashTable; keys routed by key % N (reference:
    DistributedHashTable + fixed_size_partitioner mod routing,
    python/ops/embedding_ops.py:96-365)."""

    def __init__(self, shape, num_partitions: int = 2, dtype=torch.float32,
                 name: str = "dist_hash_table", initializer=None,
                 admit_strategy: Optional[AdmitStrategy] = None,
                 device=None):
        self.num_partitions = num_partitions
        self.parts = [
            HashTable(shape, dtype, f"{name}/part_{i}", initializer,
                      admit_strategy, device)
            for i in range(num_partitions)
        ]

    def lookup(self, keys: torch.Tensor, admit: bool = True):
        flat = keys.reshape(-1)
        dev = self.parts[0].handle.device
        out = torch.empty(flat.numel(), self.parts[0].shape[0], device=dev)
        owner = flat % self.num_partitions
        for i, part in enumerate(self.parts):
            mask = owner == i
            if bool(mask.any()):
                rows = part.lookup(flat[mask], admit=admit)
                out[mask] = rows.to(device=dev, dtype=out.dtype)
        return out.reshape(*keys.shape, -1)

    def size(self) -> int:
        return sum(p.size() for p in self.parts)

    def embedding_variables(self):
        return [p.embedding_variable() for p in self.parts]
