"""EV-lookup-to-remote-KV conversion for serving.

Capability parity with the reference's RemoteSessionInstance
(serving/processor/serving/model_instance.h: sparse parameters live in
an external feature store; the serving graph's EV lookups are rewritten
into remote-KV reads — framework/graph_optimizer.cc). Here the rewrite
is a storage swap: each EmbeddingVariable/Collection keeps its API but
reads rows from the store per request; training paths raise.
"""
from __future__ import annotations

import torch

from deeprec_amd.serving.feature_store import (FeatureStore,
                                               FileFeatureStore,
                                               LocalFeatureStore)


class RemoteKvStorage:
    """Serving-only storage: rows come from a FeatureStore, addressed by
    the EV's (composite) keys; there are no local slots."""

    def __init__(self, table: str, store: FeatureStore, dim: int,
                 device="cpu", default: float = 0.0):
        self.table = table
        self.store = store
        self.dim = dim
        self.device = torch.device(device)
        self.default = default
        self._last_rank = None
        self.slabs = {}
        self.key_bits = 0
        self.dvd_per_table = 1

    # -- lookup surface (inference only) --
    def lookup(self, keys):
        return torch.arange(keys.numel(), dtype=torch.int32,
                            device=keys.device)

    def lookup_or_create(self, keys, counts, step, train=True):
        if train:
            raise RuntimeError(
                "RemoteKvStorage is serving-only: training lookups must "
                "run against a local EV storage")
        return self.lookup(keys)

    def gather(self, keys, slots, out_dtype=None):
        rows = self.store.get(self.table, keys.cpu(), self.dim,
                              self.default)
        rows = rows.to(self.device)
        return rows.to(out_dtype) if out_dtype else rows

    def prefers_dedup(self):
        return False

    def observe_uniq_ratio(self, m, nnz):
        pass

    def frequencies(self, keys):
        return torch.zeros(keys.numel(), dtype=torch.int64)

    def versions(self, keys):
        return torch.full((keys.numel(),), -1, dtype=torch.int64)

    def size(self):
        return 0

    def memory_usage(self):
        return {"total_bytes": 0, "remote": True}


def make_store(cfg: dict) -> FeatureStore:
    kind = cfg.get("kind", "local")
    if kind == "redis":
        from deeprec_amd.serving.redis_store import RedisFeatureStore
        return RedisFeatureStore(cfg.get("host", "127.0.0.1"),
                                 int(cfg["port"]))
    if kind == "file":
        return FileFeatureStore(cfg["root"])
    return LocalFeatureStore()


def attach_remote_store(model, fs_cfg: dict):
    """Swap every EV/collection storage for remote-KV reads (tables are
    named like the checkpoint EV bundles, so publish_checkpoint output
    matches directly)."""
    store = fs_cfg if isinstance(fs_cfg, FeatureStore) else \
        make_store(fs_cfg)
    for ev in model.embedding_variables():
        base = getattr(ev, "local", ev)
        table = ev.name.replace("/", "__")
        base.storage = RemoteKvStorage(table, store, base.dim,
                                       device=base.device)
    return store
