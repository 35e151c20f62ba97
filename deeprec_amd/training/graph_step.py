"""Generic hipGraph-captured training step.

The steady-state fwd+bwd+optimizer step of any collection-backed model is
captured once and replayed per step (HIP graphs — the platform-idiomatic
replacement for the reference's CUDA-Graph support, GPU-Memory-
Optimization.md). Requirements handled here:

- the model's EmbeddingCollections are pre-sized (no growth inside a
  captured step) and switched to the sync-free capture dedup path;
- the optimizer must be graph-safe (AdamAsyncOptimizer(graph_safe=True):
  device beta powers + fused dense adam);
- inputs flow through static buffers copied into before each replay.

Falls back to eager stepping on any capture failure.
"""
from __future__ import annotations

import logging
from typing import Callable, Sequence

import torch

log = logging.getLogger("deeprec_amd")


class GraphedTrainStep:
    """step = GraphedTrainStep(model, opt, loss_fn, example_batch)
    then step(batch) per iteration. loss_fn(model, *batch) -> scalar."""

    def __init__(self, model, optimizer, loss_fn: Callable,
                 example_batch: Sequence[torch.Tensor],
                 warmup_steps: int = 2,
                 expected_entries: int = 1 << 23,
                 expected_slots: int = 1 << 23):
        self.model = model
        self.optimizer = optimizer
        self.loss_fn = loss_fn
        self.graph = None
        from deeprec_amd.ops.fused_mlp import enable_weight_cache
        if optimizer.post_step_hook is None:
            optimizer.post_step_hook = enable_weight_cache(model)
        self._eager(example_batch)  # ensure slabs exist before presizing

        try:
            colls = [ev for ev in model.embedding_variables()
                     if hasattr(ev, "graph_mode")]
            for coll in colls:
                st = coll.storage
                st.enable_graph_mode(expected_entries, expected_slots)
                coll.graph_mode = True
            for _ in range(max(warmup_steps - 1, 1)):
                self._eager(example_batch)
            torch.cuda.synchronize()
            self.static = [t.clone() for t in example_batch]
            g = torch.cuda.CUDAGraph()
            optimizer.zero_grad()
            with torch.cuda.graph(g):
                loss = self.loss_fn(self.model, *self.static)
                loss.backward()
                optimizer.step()
            torch.cuda.synchronize()
            self.graph = g
            self.loss = loss
        except Exception as e:  # noqa: BLE001
            log.warning("graph capture failed (%s); using eager steps", e)
            for ev in model.embedding_variables():
                if hasattr(ev, "graph_mode"):
                    ev.graph_mode = False
            self.graph = None

    def _eager(self, batch):
        loss = self.loss_fn(self.model, *batch)
        self.optimizer.zero_grad()
        loss.backward()
        self.optimizer.step()
        return loss

    def __call__(self, batch) -> torch.Tensor:
        if self.graph is None:
            return self._eager(batch)
        for s, t in zip(self.static, batch):
            s.copy_(t, non_blocking=True)
        self.graph.replay()
        return self.loss
