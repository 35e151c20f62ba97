"""Append-only SSD embedding store with compaction.

Capability parity with the reference SSD tier (ssd_hash_kv.h, emb_file.h,
emb_file_creator.h — append-only emb files, an in-memory key index, and
sync/async compaction; tested by embedding_variable_ops_test.cc's SSD KV
cases). Re-designed for this engine: records are fixed-width fp32 rows,
files are mmap-read / append-write, and the index is a plain dict
(the GPU hash table remains the primary index; this store holds the
coldest tier's bytes).

Layout: <path>/emb-<gen>-<seq>.dat, each up to file_capacity_rows rows.
A record is the raw row bytes; the index maps key -> (file_id, row_in_
file). Overwrites append a fresh record and dead-mark the old one;
compaction rewrites files whose live ratio drops below the threshold
into the current append head, then deletes them (sync or on a worker
thread).
"""
from __future__ import annotations

import json
import os
import threading
from typing import Dict, Optional, Tuple

import numpy as np
import torch


class _EmbFile:
    def __init__(self, path: str, dim: int, capacity: int, create: bool):
        self.path = path
        self.dim = dim
        self.capacity = capacity
        self.rows = 0
        self.dead = 0
        if create:
            with open(path, "wb"):
                pass
        else:
            self.rows = os.path.getsize(path) // (dim * 4)
        self._mm: Optional[np.memmap] = None

    def append(self, rows_np: np.ndarray) -> int:
        """Append [n, dim] fp32; returns first row index."""
        first = self.rows
        with open(self.path, "ab") as f:
            f.write(rows_np.tobytes())
        self.rows += rows_np.shape[0]
        self._mm = None  # size changed; remap lazily
        return first

    def read(self, row_idx: np.ndarray) -> np.ndarray:
        if self._mm is None or self._mm.shape[0] != self.rows:
            self._mm = np.memmap(self.path, dtype=np.float32, mode="r",
                                 shape=(self.rows, self.dim))
        return np.array(self._mm[row_idx])

    @property
    def full(self) -> bool:
        return self.rows >= self.capacity

    @property
    def live_ratio(self) -> float:
        return 1.0 - self.dead / max(self.rows, 1)

    def delete(self):
        self._mm = None
        try:
            os.remove(self.path)
        except OSError:
            pass


class SsdKv:
    """key(int64) -> fp32[dim] row store on disk."""

    def __init__(self, path: str, dim: int,
                 file_capacity_rows: int = 1 << 18,
                 compact_live_ratio: float = 0.5):
        self.path = path
        self.dim = dim
        self.file_capacity = file_capacity_rows
        self.compact_live_ratio = compact_live_ratio
        os.makedirs(path, exist_ok=True)
        self.index: Dict[int, Tuple[int, int]] = {}
        self.files: Dict[int, _EmbFile] = {}
        self._next_file = 0
        self._head: Optional[int] = None
        self._lock = threading.RLock()
        self._compact_thread: Optional[threading.Thread] = None
        self._load_manifest()

    # ------------- manifest (crash-safe reopen) -------------
    def _manifest_path(self):
        return os.path.join(self.path, "manifest.json")

    def _save_manifest(self):
        idx = {str(k): v for k, v in self.index.items()}
        tmp = self._manifest_path() + ".tmp"
        with open(tmp, "w") as f:
            json.dump({"dim": self.dim, "next_file": self._next_file,
                       "files": {str(fid): {"rows": fl.rows,
                                            "dead": fl.dead}
                                 for fid, fl in self.files.items()},
                       "index": idx}, f)
        os.replace(tmp, self._manifest_path())

    def _load_manifest(self):
        mp = self._manifest_path()
        if not os.path.exists(mp):
            return
        m = json.load(open(mp))
        assert m["dim"] == self.dim, "SSD store dim mismatch"
        self._next_file = m["next_file"]
        for fid_s, meta in m["files"].items():
            fid = int(fid_s)
            p = os.path.join(self.path, f"emb-{fid}.dat")
            if os.path.exists(p):
                fl = _EmbFile(p, self.dim, self.file_capacity, create=False)
                fl.dead = meta["dead"]
                self.files[fid] = fl
        self.index = {int(k): tuple(v) for k, v in m["index"].items()
                      if int(v[0]) in self.files}

    # ------------- write / read -------------
    def _head_file(self) -> Tuple[int, _EmbFile]:
        if self._head is not None:
            fl = self.files[self._head]
            if not fl.full:
                return self._head, fl
        fid = self._next_file
        self._next_file += 1
        fl = _EmbFile(os.path.join(self.path, f"emb-{fid}.dat"), self.dim,
                      self.file_capacity, create=True)
        self.files[fid] = fl
        self._head = fid
        return fid, fl

    def write(self, keys: torch.Tensor, values: torch.Tensor):
        """Append records (overwrite semantics via the index)."""
        with self._lock:
            ks = keys.cpu().numpy().astype(np.int64)
            vs = values.detach().cpu().float().numpy()
            n = 0
            while n < len(ks):
                fid, fl = self._head_file()
                room = fl.capacity - fl.rows
                take = min(room, len(ks) - n)
                first = fl.append(vs[n: n + take])
                for j in range(take):
                    k = int(ks[n + j])
                    old = self.index.get(k)
                    if old is not None:
                        of = self.files.get(old[0])
                        if of is not None:
                            of.dead += 1
                    self.index[k] = (fid, first + j)
                n += take
            self._save_manifest()

    def read(self, keys: torch.Tensor,
             default: float = 0.0) -> torch.Tensor:
        with self._lock:
            ks = keys.cpu().numpy().astype(np.int64)
            out = np.full((len(ks), self.dim), default, dtype=np.float32)
            by_file: Dict[int, list] = {}
            for i, k in enumerate(ks):
                loc = self.index.get(int(k))
                if loc is not None:
                    by_file.setdefault(loc[0], []).append((i, loc[1]))
            for fid, pairs in by_file.items():
                oi = np.array([p[0] for p in pairs])
                ri = np.array([p[1] for p in pairs])
                out[oi] = self.files[fid].read(ri)
            return torch.from_numpy(out)

    def contains(self, keys: torch.Tensor) -> torch.Tensor:
        with self._lock:
            return torch.tensor([int(k) in self.index
                                 for k in keys.cpu().numpy()],
                                dtype=torch.bool)

    def delete(self, keys: torch.Tensor):
        with self._lock:
            for k in keys.cpu().numpy():
                loc = self.index.pop(int(k), None)
                if loc is not None and loc[0] in self.files:
                    self.files[loc[0]].dead += 1
            self._save_manifest()

    def size(self) -> int:
        return len(self.index)

    # ------------- compaction -------------
    def _victims(self):
        head = self._head
        return [fid for fid, fl in self.files.items()
                if fid != head and fl.rows > 0
                and fl.live_ratio < self.compact_live_ratio]

    def compact(self, sync: bool = True) -> int:
        """Rewrite low-live-ratio files; returns #files compacted.
        sync=False runs on a worker thread (reference capability:
        async compaction, ssd_hash_kv.h)."""
        if not sync:
            if self._compact_thread and self._compact_thread.is_alive():
                return 0
            self._compact_thread = threading.Thread(
                target=self.compact, args=(True,), daemon=True)
            self._compact_thread.start()
            return 0
        with self._lock:
            victims = self._victims()
            for fid in victims:
                fl = self.files[fid]
                live = [(k, loc[1]) for k, loc in self.index.items()
                        if loc[0] == fid]
                if live:
                    rows = fl.read(np.array([r for _, r in live]))
                    # re-append (updates index to the new head)
                    self.write(torch.tensor([k for k, _ in live]),
                               torch.from_numpy(rows))
                fl.delete()
                del self.files[fid]
                if self._head == fid:
                    self._head = None
            self._save_manifest()
            return len(victims)

    def wait_compaction(self):
        t = self._compact_thread
        if t is not None:
            t.join()

    def file_count(self) -> int:
        return len([f for f in self.files.values() if f.rows > 0])
