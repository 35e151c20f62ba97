"""ParquetDataset + WorkQueue.

Capability parity with the reference's data ops:
- ParquetDataset (reference: kernels/data/parquet_dataset_ops.cc, python/
  data/experimental/ops/parquet_dataset_ops.py): arrow-backed columnar
  batch reader with column selection;
- WorkQueue (reference: python/ops/work_queue.py:113, ops/work_queue_ops.
  cc): a global work-item (file/shard) queue shared by workers for elastic
  data distribution, whose unprocessed remainder persists in checkpoints.
"""
from __future__ import annotations

import json
import os
import threading
from typing import List, Optional, Sequence

import torch


class ParquetDataset:
    """Iterates batches from parquet files as dict[column -> torch tensor].

    field_map optionally renames columns; drop_remainder mirrors the
    reference's batching flag.
    """

    def __init__(self, filenames, batch_size: int,
                 columns: Optional[Sequence[str]] = None,
                 drop_remainder: bool = False, num_epochs: int = 1):
        import pyarrow.parquet as pq  # noqa: F401
        self.filenames = ([filenames] if isinstance(filenames, str)
                          else list(filenames))
        self.batch_size = batch_size
        self.columns = list(columns) if columns else None
        self.drop_remainder = drop_remainder
        self.num_epochs = num_epochs

    def _to_tensor(self, arr):
        import numpy as np
        np_arr = arr.to_numpy(zero_copy_only=False)
        if np_arr.dtype.kind in "iu":
            return torch.from_numpy(np_arr.astype("int64"))
        if np_arr.dtype.kind == "f":
            return torch.from_numpy(np_arr.astype("float32"))
        return torch.from_numpy(np.asarray(np_arr))

    def __iter__(self):
        import pyarrow.parquet as pq
        for _ in range(self.num_epochs):
            for fn in self.filenames:
                pf = pq.ParquetFile(fn)
                for rb in pf.iter_batches(batch_size=self.batch_size,
                                          columns=self.columns):
                    if self.drop_remainder and rb.num_rows < self.batch_size:
                        continue
                    yield {name: self._to_tensor(rb.column(i))
                           for i, name in enumerate(rb.schema.names)}


class CsvDataset:
    """CSV batch reader with the reference modelzoo's Criteo column layout
    (reference: build_model_input/parse_csv, modelzoo/dlrm/train.py:291):
    label, I1..I13 continuous, C1..C26 categorical."""

    def __init__(self, filenames, batch_size: int, column_names=None,
                 num_epochs: int = 1, label_column: str = "clicked"):
        import pandas  # noqa: F401
        self.filenames = ([filenames] if isinstance(filenames, str)
                          else list(filenames))
        self.batch_size = batch_size
        self.column_names = column_names
        self.num_epochs = num_epochs
        self.label_column = label_column

    def __iter__(self):
        import pandas as pd
        for _ in range(self.num_epochs):
            for fn in self.filenames:
                for chunk in pd.read_csv(fn, names=self.column_names,
                                         chunksize=self.batch_size):
                    out = {}
                    for col in chunk.columns:
                        ser = chunk[col]
                        if ser.dtype.kind in "iu":
                            out[col] = torch.tensor(ser.to_numpy("int64"))
                        elif ser.dtype.kind == "f":
                            out[col] = torch.tensor(
                                ser.to_numpy("float32"))
                        else:  # hash string categoricals to int64 ids
                            out[col] = torch.tensor(
                                [hash(x) & ((1 << 48) - 1)
                                 for x in ser.astype(str)],
                                dtype=torch.int64)
                    yield out


class WorkQueue:
    """Elastic work distribution: a shared queue of work items (file paths
    / shard descriptors). Rank 0 owns the queue; take() hands the next item
    to the calling rank via a broadcast round. Unprocessed items persist in
    checkpoints so a restarted job resumes where it left off."""

    def __init__(self, works: Sequence[str], name: str = "work_queue",
                 shuffle: bool = False, seed: int = 0):
        self.name = name
        items = list(works)
        if shuffle:
            g = torch.Generator().manual_seed(seed)
            perm = torch.randperm(len(items), generator=g).tolist()
            items = [items[i] for i in perm]
        self._items: List[str] = items
        self._cursor = 0
        self._lock = threading.Lock()

    def _dist(self):
        import torch.distributed as dist
        return dist if dist.is_available() and dist.is_initialized() else None

    def take(self) -> Optional[str]:
        """Next work item, globally unique across ranks."""
        dist = self._dist()
        if dist is None:
            with self._lock:
                if self._cursor >= len(self._items):
                    return None
                item = self._items[self._cursor]
                self._cursor += 1
                return item
        # rank 0 assigns: every rank calls take() collectively. A rank
        # may only STOP calling take() when a round starts past the end
        # of the list (every rank sees the same idx, so all agree the
        # queue is drained in the same round). In a TAIL round — idx
        # still in range but this rank's slot past the end — the rank
        # must keep participating in broadcast rounds, or the peers that
        # DID draw items would block forever on their next take().
        while True:
            idx = torch.tensor([0], dtype=torch.int64)
            if dist.get_rank() == 0:
                with self._lock:
                    idx[0] = self._cursor
                    if self._cursor < len(self._items):
                        self._cursor += dist.get_world_size()
            dist.broadcast(idx, src=0)
            start = int(idx[0])
            if start >= len(self._items):
                return None  # unanimous: drained
            if dist.get_rank() != 0:
                self._cursor = min(start + dist.get_world_size(),
                                   len(self._items))
            my = start + dist.get_rank()
            if my < len(self._items):
                return self._items[my]
            # tail round: no item for this rank; rejoin the next round

    def remaining(self) -> List[str]:
        return self._items[self._cursor:]

    # ---- checkpoint integration ----
    def state_dict(self) -> dict:
        return {"items": self._items, "cursor": self._cursor,
                "name": self.name}

    def load_state_dict(self, sd: dict):
        self._items = list(sd["items"])
        self._cursor = int(sd["cursor"])

    def save(self, path: str):
        with open(path, "w") as f:
            json.dump(self.state_dict(), f)

    @classmethod
    def restore(cls, path: str) -> "WorkQueue":
        sd = json.load(open(path))
        wq = cls(sd["items"], sd.get("name", "work_queue"))
        wq.load_state_dict(sd)
        return wq
