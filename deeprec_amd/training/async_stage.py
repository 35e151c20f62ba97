"""Async embedding stage — overlap embedding lookups with dense compute.

Capability parity with the reference's Async-Embedding-Stage
(docs/Async-Embedding-Stage.md: the embedding-lookup subgraph executes
asynchronously from the dense net, trading one step of parameter
staleness for throughput). Here the stage is a one-deep pipeline:

    stage = AsyncEmbeddingStage(model.collection)
    stage.submit(ids_0)
    for ids_next, (dense, labels) in batches:
        emb = stage.take()          # batch N's pooled embeddings
        stage.submit(ids_next)      # batch N+1 looks up on a side stream
        loss = head(dense, emb); loss.backward(); opt.step()

The N+1 lookup runs on its own HIP stream concurrently with batch N's
dense forward/backward/optimizer. Embedding values it reads may miss the
in-flight step-N sparse update — exactly the staleness the reference
stage accepts. Gradients are NOT stale: backward scatters into current
state on the default stream.
"""
from __future__ import annotations

from collections import deque

import torch


class AsyncEmbeddingStage:
    """One-deep lookup pipeline over an EmbeddingCollection (matrix
    layout). On CPU (or when streams are unavailable) it degrades to a
    synchronous queue with identical semantics minus the overlap."""

    def __init__(self, collection, depth: int = 1):
        self.coll = collection
        self.depth = max(1, depth)
        self._q = deque()
        dev = collection.device
        self._stream = (torch.cuda.Stream(device=dev)
                        if dev.type == "cuda" else None)

    def submit(self, ids: torch.Tensor, train: bool = True):
        """Queue a lookup for `ids` [batch, n_tables]."""
        assert len(self._q) < self.depth + 1, "take() before submitting more"
        if self._stream is None:
            self._q.append((self.coll.lookup_matrix(ids, train=train),
                            None))
            return
        # the side stream must see prior default-stream updates (step N-1
        # optimizer applies); later ones may or may not be visible — the
        # staleness window the stage trades on
        self._stream.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(self._stream):
            out = self.coll.lookup_matrix(ids, train=train)
        ev = torch.cuda.Event()
        ev.record(self._stream)
        self._q.append((out, ev))

    def take(self) -> torch.Tensor:
        """Oldest completed lookup (blocks the stream, not the host)."""
        out, ev = self._q.popleft()
        if ev is not None:
            torch.cuda.current_stream().wait_event(ev)
            # the buffer was allocated on the side stream; tell the
            # caching allocator it is consumed on this stream too
            out.record_stream(torch.cuda.current_stream())
        return out

    def __len__(self):
        return len(self._q)
