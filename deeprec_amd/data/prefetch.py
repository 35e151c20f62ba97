"""Async input staging — the reference's SmartStage / tf.staged analog
(reference: core/graph/smart_stage_pass.cc, python/ops/prefetch.py):
a background thread produces batches and stages H2D copies on a side HIP
stream so input work overlaps the training step; the consumer just waits
on an event.
"""
from __future__ import annotations

import queue
import threading

import torch


class PrefetchIterator:
    def __init__(self, dataset, depth: int = 2):
        self.ds = dataset
        # datasets expose next_batch(); plain iterables are wrapped too
        self._next = (dataset.next_batch if hasattr(dataset, "next_batch")
                      else iter(dataset).__next__)
        self.q: queue.Queue = queue.Queue(maxsize=depth)
        self._stop = threading.Event()
        self.use_cuda = torch.cuda.is_available() and \
            getattr(dataset, "device", torch.device("cpu")).type == "cuda"
        self.stream = torch.cuda.Stream() if self.use_cuda else None
        self.thread = threading.Thread(target=self._worker, daemon=True)
        self.thread.start()

    def _produce(self):
        return self._next()

    def _worker(self):
        while not self._stop.is_set():
            try:
                if self.stream is not None:
                    with torch.cuda.stream(self.stream):
                        batch = self._produce()
                    ev = torch.cuda.Event()
                    ev.record(self.stream)
                else:
                    batch, ev = self._produce(), None
            except BaseException as e:  # noqa: BLE001
                # surface producer errors (incl. StopIteration) to the
                # consumer instead of deadlocking its q.get()
                self.q.put((e, "error"))
                return
            try:
                self.q.put((batch, ev), timeout=1.0)
            except queue.Full:
                if self._stop.is_set():
                    return
                self.q.put((batch, ev))

    def __iter__(self):
        return self

    def __next__(self):
        batch, ev = self.q.get()
        if ev == "error":
            raise batch
        if ev is not None:
            torch.cuda.current_stream().wait_event(ev)
            for t in batch:
                if torch.is_tensor(t):
                    t.record_stream(torch.cuda.current_stream())
                elif hasattr(t, "values"):
                    t.values.record_stream(torch.cuda.current_stream())
        return batch

    def close(self):
        self._stop.set()
        try:
            while True:
                self.q.get_nowait()
        except queue.Empty:
            pass


class ShuffleBuffer:
    """Streaming shuffle with a bounded buffer (reference:
    ``dataset.shuffle(buffer_size=20000)`` in the zoo input pipelines,
    modelzoo/dlrm/train.py:324): fills ``buffer_size`` items, then each
    pull swaps a uniformly random buffered item with the next source
    item. Every source item is yielded exactly once; composes under
    PrefetchIterator (shuffle first, then prefetch)."""

    def __init__(self, source, buffer_size: int, seed: int = 0):
        self.source = source
        self.buffer_size = int(buffer_size)
        self.seed = seed

    def __iter__(self):
        import random
        rng = random.Random(self.seed)
        buf = []
        it = iter(self.source)
        for item in it:
            if len(buf) < self.buffer_size:
                buf.append(item)
                continue
            j = rng.randrange(len(buf))
            out, buf[j] = buf[j], item
            yield out
        rng.shuffle(buf)
        yield from buf
