"""Training session + hooks.

Capability parity with the reference's MonitoredTrainingSession / hook
surface (reference: python/training/monitored_session.py,
basic_session_run_hooks.py): hook lifecycle (begin / before_run /
after_run / end), logging, step-rate counting, checkpointing with an
incremental-save timer (reference: CheckpointSaverHook gains
incremental_save_secs, basic_session_run_hooks.py:525-591), profiler hook
emitting chrome traces (reference: tf.train.ProfilerHook), and
stop-condition plumbing.
"""
from __future__ import annotations

import logging
import os
import time
from typing import Callable, List, Optional

import torch

from deeprec_amd.embedding.variable import get_global_step

log = logging.getLogger("deeprec_amd")


class SessionRunHook:
    def begin(self, session):
        pass

    def before_run(self, session):
        pass

    def after_run(self, session, results):
        pass

    def end(self, session):
        pass


class LoggingTensorHook(SessionRunHook):
    """Log scalars returned by the step fn every N steps."""

    def __init__(self, every_n_steps: int = 100, formatter=None):
        self.every_n = every_n_steps
        self.formatter = formatter

    def after_run(self, session, results):
        step = get_global_step()
        if step % self.every_n == 0:
            if self.formatter:
                msg = self.formatter(step, results)
            else:
                if isinstance(results, torch.Tensor):
                    results = {"loss": float(results)}
                msg = f"step {step}: " + ", ".join(
                    f"{k}={float(v):.6g}" for k, v in (results or {}).items())
            log.info(msg)


class StepCounterHook(SessionRunHook):
    def __init__(self, every_n_steps: int = 100, batch_size: int = None):
        self.every_n = every_n_steps
        self.batch_size = batch_size
        self._t0 = None
        self._step0 = 0

    def begin(self, session):
        self._t0 = time.perf_counter()
        self._step0 = get_global_step()

    def after_run(self, session, results):
        step = get_global_step()
        if step > self._step0 and (step - self._step0) % self.every_n == 0:
            dt = time.perf_counter() - self._t0
            sps = (step - self._step0) / dt
            msg = f"steps/sec: {sps:.2f}"
            if self.batch_size:
                msg += f"  samples/sec: {sps * self.batch_size:.1f}"
            log.info(msg)
            self._t0 = time.perf_counter()
            self._step0 = step


class CheckpointSaverHook(SessionRunHook):
    """Periodic full saves + optional higher-frequency incremental saves
    (reference: Incremental-Checkpoint.md — save_incremental_checkpoint_secs)."""

    def __init__(self, checkpoint_dir: str, saver, save_steps: int = None,
                 save_secs: float = None,
                 incremental_save_secs: float = None):
        self.dir = checkpoint_dir
        self.saver = saver
        self.save_steps = save_steps
        self.save_secs = save_secs
        self.incr_secs = incremental_save_secs
        self._last_save = time.time()
        self._last_incr = time.time()
        self._saved_once = False

    def after_run(self, session, results):
        step = get_global_step()
        now = time.time()
        due = ((self.save_steps and step % self.save_steps == 0) or
               (self.save_secs and now - self._last_save >= self.save_secs))
        if due:
            self.saver.save(self.dir, step)
            self._last_save = now
            self._last_incr = now
            self._saved_once = True
        elif (self.incr_secs and self._saved_once
              and now - self._last_incr >= self.incr_secs):
            self.saver.incremental_save(self.dir, step)
            self._last_incr = now

    def end(self, session):
        self.saver.save(self.dir, get_global_step())


class ProfilerHook(SessionRunHook):
    """Emit a chrome trace around selected steps (rocTX/torch.profiler —
    the reference's tf.train.ProfilerHook analog)."""

    def __init__(self, save_steps: int, output_dir: str = "timeline",
                 num_steps: int = 1):
        self.save_steps = save_steps
        self.output_dir = output_dir
        self.num_steps = num_steps
        self._prof = None

    def before_run(self, session):
        step = get_global_step()
        if step > 0 and step % self.save_steps == 0 and self._prof is None:
            acts = [torch.profiler.ProfilerActivity.CPU]
            if torch.cuda.is_available():
                acts.append(torch.profiler.ProfilerActivity.CUDA)
            self._prof = torch.profiler.profile(activities=acts)
            self._prof.__enter__()
            self._prof_steps = 0

    def after_run(self, session, results):
        if self._prof is not None:
            self._prof_steps += 1
            if self._prof_steps >= self.num_steps:
                self._prof.__exit__(None, None, None)
                os.makedirs(self.output_dir, exist_ok=True)
                out = os.path.join(self.output_dir,
                                   f"timeline-{get_global_step()}.json")
                self._prof.export_chrome_trace(out)
                log.info("wrote %s", out)
                self._prof = None


class StopAtStepHook(SessionRunHook):
    def __init__(self, last_step: int):
        self.last_step = last_step

    def after_run(self, session, results):
        if get_global_step() >= self.last_step:
            session.request_stop()


class MonitoredTrainingSession:
    """Training loop shell with hooks and checkpoint auto-restore.

    with MonitoredTrainingSession(checkpoint_dir=..., saver=...,
                                  hooks=[...]) as sess:
        while not sess.should_stop():
            sess.run(step_fn)   # step_fn() -> loss / dict of scalars
    """

    def __init__(self, hooks: Optional[List[SessionRunHook]] = None,
                 checkpoint_dir: Optional[str] = None, saver=None,
                 save_checkpoint_steps: Optional[int] = None,
                 save_checkpoint_secs: Optional[float] = None,
                 save_incremental_checkpoint_secs: Optional[float] = None,
                 max_steps: Optional[int] = None):
        self.hooks = list(hooks or [])
        self.saver = saver
        self._stop = False
        if checkpoint_dir and saver:
            from deeprec_amd.checkpoint.saver import latest_checkpoint
            ck = latest_checkpoint(checkpoint_dir)
            if ck:
                step = saver.restore(ck)
                log.info("restored from %s (step %d)", ck, step)
            if save_checkpoint_steps or save_checkpoint_secs or \
                    save_incremental_checkpoint_secs:
                self.hooks.append(CheckpointSaverHook(
                    checkpoint_dir, saver, save_checkpoint_steps,
                    save_checkpoint_secs, save_incremental_checkpoint_secs))
        if max_steps:
            self.hooks.append(StopAtStepHook(max_steps))

    def __enter__(self):
        for h in self.hooks:
            h.begin(self)
        return self

    def __exit__(self, exc_type, *a):
        if exc_type is None:
            for h in self.hooks:
                h.end(self)
        return False

    def should_stop(self) -> bool:
        return self._stop

    def request_stop(self):
        self._stop = True

    def run(self, step_fn: Callable, *args, **kw):
        for h in self.hooks:
            h.before_run(self)
        results = step_fn(*args, **kw)
        for h in self.hooks:
            h.after_run(self, results)
        return results


class RebalanceHook(SessionRunHook):
    """Every `every_steps`, repack multi-tier EmbeddingVariables so the
    hottest keys (by engine frequency counters) sit in the HBM tier
    (reference: CacheStrategy-driven promotion in multi_tier_storage)."""

    def __init__(self, every_steps: int = 10000, variables=None):
        self.every = max(1, every_steps)
        self.variables = variables
        self._n = 0

    def after_run(self, results):
        self._n += 1
        if self._n % self.every:
            return
        from deeprec_amd.embedding.variable import all_embedding_variables
        evs = self.variables if self.variables is not None             else all_embedding_variables()
        moved = sum(ev.rebalance() for ev in evs
                    if hasattr(ev, "rebalance"))
        if moved:
            log.info("rebalance: %d rows changed tier", moved)


class MemoryStatsHook(SessionRunHook):
    """Log device memory accounting every N steps (the reference's
    GPU-memory-optimization observability: allocator stats vs peak)."""

    def __init__(self, every_n_steps: int = 500):
        self.every_n = every_n_steps

    def after_run(self, session, results):
        if not torch.cuda.is_available():
            return
        step = get_global_step()
        if step % self.every_n == 0:
            alloc = torch.cuda.memory_allocated() / (1 << 20)
            reserved = torch.cuda.memory_reserved() / (1 << 20)
            peak = torch.cuda.max_memory_allocated() / (1 << 20)
            log.info("memory MiB: allocated=%.1f reserved=%.1f peak=%.1f",
                     alloc, reserved, peak)
