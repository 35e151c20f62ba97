"""Serving-time sparse feature store.

Capability parity with the reference's RemoteSessionInstance path
(serving/processor/storage/feature_store_mgr.cc + redis_feature_store.cc):
sparse embedding weights live in an external KV store instead of the
serving process, published from checkpoints and looked up per request.

The store API is pluggable; built-ins:
- LocalFeatureStore — in-process dict (the unit-test / single-process tier)
- FileFeatureStore — a safetensors-file-backed store sharing the
  checkpoint format, usable by many serving processes on one host.
A Redis client implements the same three methods against a live server.
"""
from __future__ import annotations

import glob
import os
from typing import Dict, Optional

import torch


class FeatureStore:
    """Abstract batch KV for embedding rows of one or more tables."""

    def put(self, table: str, keys: torch.Tensor, values: torch.Tensor):
        raise NotImplementedError

    def get(self, table: str, keys: torch.Tensor, dim: int,
            default: float = 0.0) -> torch.Tensor:
        """[n, dim] rows; missing keys get `default`."""
        raise NotImplementedError

    def tables(self):
        raise NotImplementedError


class LocalFeatureStore(FeatureStore):
    def __init__(self):
        self._t: Dict[str, Dict[int, torch.Tensor]] = {}

    def put(self, table, keys, values):
        d = self._t.setdefault(table, {})
        for k, v in zip(keys.tolist(), values.cpu()):
            d[k] = v.clone()

    def get(self, table, keys, dim, default=0.0):
        d = self._t.get(table, {})
        out = torch.full((keys.numel(), dim), float(default))
        for i, k in enumerate(keys.tolist()):
            row = d.get(k)
            if row is not None:
                out[i] = row
        return out

    def tables(self):
        return list(self._t)


class FileFeatureStore(LocalFeatureStore):
    """Persistent variant: load()/flush() round-trip through safetensors
    files (one per table) under `root`."""

    def __init__(self, root: str):
        super().__init__()
        self.root = root
        os.makedirs(root, exist_ok=True)
        self.load()

    def flush(self):
        import json
        from safetensors.torch import save_file
        index = {}
        for i, (table, d) in enumerate(self._t.items()):
            if not d:
                continue
            keys = torch.tensor(list(d.keys()), dtype=torch.int64)
            values = torch.stack([d[k] for k in keys.tolist()])
            fn = f"table-{i}.fs"
            save_file({"keys": keys, "values": values},
                      os.path.join(self.root, fn))
            index[fn] = table
        with open(os.path.join(self.root, "index.json"), "w") as f:
            json.dump(index, f)

    def load(self):
        import json
        from safetensors.torch import load_file
        idx_fn = os.path.join(self.root, "index.json")
        if not os.path.exists(idx_fn):
            return
        with open(idx_fn) as f:
            index = json.load(f)
        for fn, table in index.items():
            data = load_file(os.path.join(self.root, fn))
            self.put(table, data["keys"], data["values"])


def publish_checkpoint(store: FeatureStore, ckpt_path: str,
                       tables: Optional[list] = None) -> int:
    """Push every EV bundle of a checkpoint into the store (the
    full-model publish step of the online-serving pipeline). Returns the
    number of rows published."""
    from safetensors.torch import load_file
    n = 0
    for fn in sorted(glob.glob(os.path.join(ckpt_path,
                                            "ev-*-part*.safetensors"))):
        base = os.path.basename(fn)
        name = base[len("ev-"):base.rindex("-part")]
        if tables is not None and name not in tables:
            continue
        data = load_file(fn)
        store.put(name, data["keys"], data["values"])
        n += data["keys"].numel()
    return n


def store_backed_lookup(store: FeatureStore, table: str,
                        keys: torch.Tensor, dim: int,
                        default: float = 0.0) -> torch.Tensor:
    """Per-request embedding fetch (≙ FeatureStoreMgr::GetValues)."""
    return store.get(table, keys, dim, default)
