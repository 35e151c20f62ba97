"""Batch caches for hot-key tracking.

Capability parity with the reference's BatchCache hierarchy (cache.h:19
LRUCache:133 LFUCache:272 + CacheStrategy, config.proto:40-43): a cache
observes each training batch's ids and answers "which keys are hot"
(multi-tier placement) and "which k keys should leave" (eviction).

The engine's per-entry frequency/version counters already back the
DEFAULT placement policy (EvictionManager scores from table snapshots);
these classes are the standalone, bounded-memory variants for host-side
pipelines (admission caches, prefetch lists) and mirror the reference
API shape: add_to_cache / get_cached_ids / get_evict_ids / size.
"""
from __future__ import annotations

from collections import OrderedDict
from typing import Iterable

import torch


class BatchCache:
    def add_to_cache(self, ids: torch.Tensor):
        raise NotImplementedError

    def get_evict_ids(self, k: int) -> torch.Tensor:
        """Remove and return up to k coldest ids."""
        raise NotImplementedError

    def get_cached_ids(self, k: int) -> torch.Tensor:
        """Up to k hottest ids (prefetch list), most-hot first."""
        raise NotImplementedError

    def size(self) -> int:
        raise NotImplementedError

    def __contains__(self, key: int) -> bool:
        raise NotImplementedError


class LRUCache(BatchCache):
    """Recency order (reference: cache.h:133). O(1) updates via an
    ordered dict; optional capacity auto-evicts the stalest keys."""

    def __init__(self, capacity: int = 0):
        self.capacity = capacity
        self._od: "OrderedDict[int, int]" = OrderedDict()
        self._step = 0

    def add_to_cache(self, ids: torch.Tensor):
        self._step += 1
        for k in ids.reshape(-1).tolist():
            self._od[k] = self._step
            self._od.move_to_end(k)
        if self.capacity:
            while len(self._od) > self.capacity:
                self._od.popitem(last=False)

    def get_evict_ids(self, k: int) -> torch.Tensor:
        out = []
        for _ in range(min(k, len(self._od))):
            key, _ = self._od.popitem(last=False)
            out.append(key)
        return torch.tensor(out, dtype=torch.int64)

    def get_cached_ids(self, k: int) -> torch.Tensor:
        keys = list(self._od.keys())[-k:][::-1]
        return torch.tensor(keys, dtype=torch.int64)

    def size(self) -> int:
        return len(self._od)

    def __contains__(self, key: int) -> bool:
        return key in self._od


class LFUCache(BatchCache):
    """Frequency order (reference: cache.h:272); ties break by recency."""

    def __init__(self, capacity: int = 0):
        self.capacity = capacity
        self._freq: dict = {}
        self._last: dict = {}
        self._step = 0

    def add_to_cache(self, ids: torch.Tensor):
        self._step += 1
        flat = ids.reshape(-1)
        uniq, counts = torch.unique(flat, return_counts=True)
        for k, c in zip(uniq.tolist(), counts.tolist()):
            self._freq[k] = self._freq.get(k, 0) + c
            self._last[k] = self._step
        if self.capacity and len(self._freq) > self.capacity:
            self.get_evict_ids(len(self._freq) - self.capacity)

    def _order(self, reverse: bool):
        return sorted(self._freq,
                      key=lambda k: (self._freq[k], self._last[k]),
                      reverse=reverse)

    def get_evict_ids(self, k: int) -> torch.Tensor:
        victims = self._order(reverse=False)[:k]
        for v in victims:
            del self._freq[v]
            del self._last[v]
        return torch.tensor(victims, dtype=torch.int64)

    def get_cached_ids(self, k: int) -> torch.Tensor:
        return torch.tensor(self._order(reverse=True)[:k],
                            dtype=torch.int64)

    def size(self) -> int:
        return len(self._freq)

    def __contains__(self, key: int) -> bool:
        return key in self._freq


def make_cache(strategy) -> BatchCache:
    from deeprec_amd.embedding.options import CacheStrategy
    if strategy == CacheStrategy.LRU:
        return LRUCache()
    return LFUCache()
