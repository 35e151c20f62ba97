"""Generic hipGraph-captured training step.

The steady-state fwd+bwd+optimizer step of any collection-backed model is
captured once and replayed per step (HIP graphs — the platform-idiomatic
replacement for the reference's CUDA-Graph support, GPU-Memory-
Optimization.md). Requirements handled here:

- the model's EmbeddingCollections are pre-sized (no growth inside a
  captured step) and switched to the sync-free capture dedup path;
- the optimizer must be graph-safe (AdamAsyncOptimizer(graph_safe=True):
  device beta powers + fused dense adam);
- inputs flow through static buffers copied into before each replay;
- GROWTH under capture: a long run crossing the pre-sized watermark
  cannot grow inside a replay — every ``growth_check_interval`` replays
  the (cheap) device counters are read back, and nearing capacity or a
  capacity error (engine codes 1/2) triggers graph INVALIDATION: the
  table/slabs grow 2x (entries preserved — ``_rehash``/``_grow_slots``)
  and the step re-captures. Admissions of the step that tripped the flag
  are dropped, not corrupted (the engine refuses inserts past the
  watermark); they re-admit on later batches. Codes 3/4 are real
  invariant violations and still raise.

Falls back to eager stepping on any capture failure.
"""
from __future__ import annotations

import gc
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
                 expected_slots: int = 1 << 23,
                 growth_check_interval: int = 64):
        self.model = model
        self.optimizer = optimizer
        self.loss_fn = loss_fn
        self.graph = None
        self._interval = growth_check_interval
        self._n = 0
        self.recaptures = 0
        self._expected = [expected_entries, expected_slots]
        from deeprec_amd.ops.fused_mlp import enable_weight_cache
        if optimizer.post_step_hook is None:
            optimizer.post_step_hook = enable_weight_cache(model)
        self._eager(example_batch)  # ensure slabs exist before presizing

        self._colls = [ev for ev in model.embedding_variables()
                       if hasattr(ev, "graph_mode")]
        try:
            for coll in self._colls:
                coll.storage.enable_graph_mode(*self._expected)
                coll.graph_mode = True
            self._capture(example_batch, max(warmup_steps - 1, 1))
        except Exception as e:  # noqa: BLE001
            log.warning("graph capture failed (%s); using eager steps", e)
            for ev in model.embedding_variables():
                if hasattr(ev, "graph_mode"):
                    ev.graph_mode = False
            self.graph = None

    def _capture(self, example_batch, warmup: int):
        for _ in range(warmup):
            self._eager(example_batch)
        torch.cuda.synchronize()
        self.static = [t.clone() for t in example_batch]
        g = torch.cuda.CUDAGraph()
        self.optimizer.zero_grad()
        with torch.cuda.graph(g):
            loss = self.loss_fn(self.model, *self.static)
            loss.backward()
            self.optimizer.step()
        torch.cuda.synchronize()
        self.graph = g
        self.loss = loss

    def _eager(self, batch):
        loss = self.loss_fn(self.model, *batch)
        self.optimizer.zero_grad()
        loss.backward()
        self.optimizer.step()
        return loss

    # ---------------- growth / invalidation ----------------
    def _watermark_check(self, batch):
        """One small D2H read per collection every _interval replays:
        invalidate + grow + re-capture BEFORE the engine hits the wall
        (>85% of either capacity), or right after a capacity error."""
        grow = False
        for coll in self._colls:
            st = coll.storage
            err = int(st.error_flag.cpu())
            if err in (3, 4):
                st._check_error()  # real invariant violation: raise
            if err in (1, 2):
                log.warning("capacity error %d under capture; growing", err)
                st.error_flag.zero_()
                grow = True
                continue
            from deeprec_amd.ops.hip_backend import _LOAD_FACTOR
            ents = int(st.entry_counter.cpu())
            slots = int(st.slot_counter.cpu())
            if (ents > 0.85 * st.capacity * _LOAD_FACTOR
                    or slots > 0.85 * st.max_slots):
                grow = True
        if not grow:
            return
        self.recaptures += 1
        self._expected = [2 * self._expected[0], 2 * self._expected[1]]
        log.warning("growth watermark under capture: invalidating graph, "
                    "growing to %d entries / %d slots and re-capturing",
                    *self._expected)
        # destroy the old graph BEFORE mutating engine buffers it refers to
        self.graph = None
        gc.collect()
        torch.cuda.synchronize()
        for coll in self._colls:
            coll.storage.enable_graph_mode(*self._expected)
        self._capture(batch, warmup=1)

    def __call__(self, batch) -> torch.Tensor:
        if self.graph is None:
            return self._eager(batch)
        for s, t in zip(self.static, batch):
            s.copy_(t, non_blocking=True)
        self.graph.replay()
        self._n += 1
        if self._n % self._interval == 0:
            self._watermark_check(batch)
        return self.loss
