"""Serving: predictor with online model update.

Capability parity with the reference's serving processor (SURVEY.md §2.6,
serving/processor/):
- Predictor ≙ LocalSessionInstance: loads a SavedModel-equivalent
  (model object + checkpoint), runs inference sessions;
- ModelUpdater ≙ the background version-poll loop applying
  FullModelUpdate / DeltaModelUpdate from full + incremental checkpoints
  (second-level online model updates, model_instance.h:44-46);
- SessionGroup ≙ N concurrent inference sessions; on GPU each session owns
  a HIP stream (reference: direct_session_group.h:28 — per-session CUDA
  streams), round-robin dispatch;
- process()/batch_process() ≙ the C-ABI entry contract
  (serving/processor/serving/processor.cc:8-102) with dict/JSON payloads.
"""
from __future__ import annotations

import glob
import os
import threading
from typing import Callable, List, Optional

import torch

from deeprec_amd.checkpoint.saver import Saver, latest_checkpoint


class SessionGroup:
    """N logical inference sessions; on GPU each has its own HIP stream so
    concurrent requests overlap (multi-stream serving)."""

    def __init__(self, num_sessions: int = 2, device=None):
        self.device = torch.device(device or (
            "cuda" if torch.cuda.is_available() else "cpu"))
        self.n = max(1, num_sessions)
        self.streams = ([torch.cuda.Stream(device=self.device)
                         for _ in range(self.n)]
                        if self.device.type == "cuda" else [None] * self.n)
        self._rr = 0
        self._lock = threading.Lock()

    def run(self, fn: Callable, *args, **kw):
        with self._lock:
            i = self._rr
            self._rr = (self._rr + 1) % self.n
        stream = self.streams[i]
        if stream is None:
            return fn(*args, **kw)
        with torch.cuda.stream(stream):
            out = fn(*args, **kw)
        stream.synchronize()
        return out


class Predictor:
    """Loads model weights from a checkpoint dir and serves predictions,
    optionally hot-updating from new full/incremental checkpoints."""

    def __init__(self, model, checkpoint_dir: str,
                 num_sessions: int = 2, device=None,
                 remote_sparse: bool = False, fp8_mlp: bool = False):
        self.model = model
        self.dir = checkpoint_dir
        # fp8_mlp: serve the dense MLPs as OCP e4m3 (ops/fp8.py) — the
        # converter is reverted around full model updates so restores
        # land in the fp32 originals, then re-quantized
        self._fp8 = None
        if fp8_mlp:
            from deeprec_amd.ops.fp8 import Fp8MlpConverter
            self._fp8 = Fp8MlpConverter(model)
        # remote_sparse = RemoteSessionInstance mode: sparse weights live
        # in an external feature store (attach_remote_store), so only the
        # dense module restores from checkpoints here
        self.saver = Saver(module=model,
                           embedding_variables=[] if remote_sparse
                           else model.embedding_variables())
        self.group = SessionGroup(num_sessions, device)
        self._applied = set()
        self._loaded_full: Optional[str] = None
        self._lock = threading.Lock()
        self.reload()
        if self._fp8 is not None and not self._fp8._sites:
            self._fp8.convert()  # no checkpoint yet: quantize init weights

    # ------------- model update -------------
    def reload(self) -> bool:
        """Full model update from the latest full checkpoint."""
        ck = latest_checkpoint(self.dir)
        if ck is None or ck == self._loaded_full:
            return False
        with self._lock:
            if self._fp8 is not None:
                self._fp8.revert()
            self.saver.restore(ck)  # also replays newer incr deltas
            if self._fp8 is not None:
                self._fp8.convert()
            self._loaded_full = ck
            self._applied = {p for p in self._incr_paths()}
        return True

    def _incr_paths(self) -> List[str]:
        if self._loaded_full is None:
            return []
        base_step = int(os.path.basename(self._loaded_full).split("-")[1])
        return sorted(
            (p for p in glob.glob(os.path.join(self.dir, "ckpt-*.incr"))
             if int(os.path.basename(p)[5:-5]) > base_step),
            key=lambda p: int(os.path.basename(p)[5:-5]))

    def poll_updates(self) -> int:
        """Delta model update: apply any new incremental checkpoints
        (reference: DeltaModelUpdate); returns number applied. A newer full
        checkpoint triggers a full reload."""
        if self.reload():
            return 1
        n = 0
        for p in self._incr_paths():
            if p in self._applied:
                continue
            with self._lock:
                self.saver._restore_ev_files(p, replay=True)
                self._applied.add(p)
            n += 1
        return n

    def start_update_thread(self, poll_secs: float = 5.0):
        self._stop = threading.Event()

        def loop():
            while not self._stop.is_set():
                try:
                    self.poll_updates()
                except Exception:  # noqa: BLE001
                    pass
                self._stop.wait(poll_secs)

        self._thread = threading.Thread(target=loop, daemon=True)
        self._thread.start()

    def stop_update_thread(self):
        if hasattr(self, "_stop"):
            self._stop.set()
            self._thread.join(timeout=5)

    # ------------- inference -------------
    @torch.no_grad()
    def predict(self, *features) -> torch.Tensor:
        def run():
            with self._lock:
                logits = self.model(*features, train=False)
            if isinstance(logits, (list, tuple)):
                return [torch.sigmoid(lg) for lg in logits]
            return torch.sigmoid(logits)
        return self.group.run(run)

    def process(self, request: dict) -> dict:
        """JSON-ish request/response (processor.cc C-ABI contract shape):
        {"dense": [[...]], "sparse": [[...ids...]], "compress": bool}
        -> {"probabilities": [...]}. compress=true deduplicates repeated
        samples before the forward (sample-aware compression — ranking
        batches repeat user/context rows across candidates)."""
        dense = torch.tensor(request["dense"], dtype=torch.float32,
                             device=self.group.device)
        sparse = torch.tensor(request["sparse"], dtype=torch.int64,
                              device=self.group.device)
        if request.get("compress"):
            from deeprec_amd.data.compression import compressed_forward
            probs, _ = compressed_forward(self.predict, dense, sparse)
        else:
            probs = self.predict(dense, sparse)
        if isinstance(probs, list):
            return {"probabilities": [p.cpu().tolist() for p in probs]}
        return {"probabilities": probs.cpu().tolist()}

    def batch_process(self, requests: List[dict]) -> List[dict]:
        return [self.process(r) for r in requests]
