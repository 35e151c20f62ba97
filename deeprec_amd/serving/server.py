"""HTTP serving shell over Predictor.

The reference's serving processor exposes a C-ABI (`process(request) ->
response`, serving/processor/serving/processor.cc:8-102) consumed by an
external RPC frontend. Here the frontend is in-process: a FastAPI app over
Predictor.process with
- POST /v1/predict         one request  {"dense": [[..]], "sparse": [[..]]}
- POST /v1/predict_batch   list of requests
- GET  /health             liveness + loaded checkpoint version
- POST /v1/reload          explicit full-model reload

plus a DynamicBatcher that coalesces concurrent single-row requests into
one batched GPU inference call (per-request latency bound + max batch) —
the serving-side analog of the session-group multi-stream design.
"""
from __future__ import annotations

import threading
import time
from concurrent.futures import Future
from typing import List, Optional

import torch

from deeprec_amd.serving.predictor import Predictor


class DynamicBatcher:
    """Coalesce concurrent predict requests into batched model calls.

    submit() returns a Future; a collector thread drains the queue every
    time it wakes (new arrivals within max_latency_ms join the batch, up
    to max_batch rows) and runs ONE batched predict.
    """

    def __init__(self, predictor: Predictor, max_batch: int = 256,
                 max_latency_ms: float = 2.0):
        self.predictor = predictor
        self.max_batch = max_batch
        self.max_latency = max_latency_ms / 1000.0
        self._pending: List[tuple] = []
        self._cv = threading.Condition()
        self._stop = False
        self._thread = threading.Thread(target=self._loop, daemon=True)
        self._thread.start()

    def submit(self, dense: torch.Tensor, sparse: torch.Tensor) -> Future:
        """dense [b, D], sparse [b, N] (b rows from one request)."""
        fut: Future = Future()
        with self._cv:
            self._pending.append((dense, sparse, fut))
            self._cv.notify()
        return fut

    def close(self):
        with self._cv:
            self._stop = True
            self._cv.notify()
        self._thread.join(timeout=5)

    def _loop(self):
        while True:
            with self._cv:
                while not self._pending and not self._stop:
                    self._cv.wait()
                if self._stop and not self._pending:
                    return
                # wait out the coalescing window for more arrivals
                deadline = time.monotonic() + self.max_latency
                while (sum(d.shape[0] for d, _, _ in self._pending)
                       < self.max_batch):
                    left = deadline - time.monotonic()
                    if left <= 0 or self._stop:
                        break
                    self._cv.wait(timeout=left)
                batch, self._pending = self._pending, []
            self._run(batch)

    def _run(self, batch):
        try:
            dense = torch.cat([d for d, _, _ in batch], dim=0)
            sparse = torch.cat([s for _, s, _ in batch], dim=0)
            probs = self.predictor.predict(dense, sparse)
            head = probs[0] if isinstance(probs, list) else probs
            off = 0
            for d, _, fut in batch:
                b = d.shape[0]
                fut.set_result(head[off:off + b])
                off += b
        except Exception as e:  # noqa: BLE001
            for _, _, fut in batch:
                if not fut.done():
                    fut.set_exception(e)


class _Metrics:
    """Prometheus metrics for the serving shell (one registry per app so
    tests can build many apps in one process). The reference ships no
    Prometheus surface (SURVEY §5 'nothing Prometheus-like'); a
    production serving deployment needs one, so this exceeds parity.
    No-ops gracefully when prometheus_client is unavailable."""

    def __init__(self):
        try:
            from prometheus_client import (CollectorRegistry, Counter,
                                           Gauge, Histogram)
        except ImportError:  # pragma: no cover
            self.registry = None
            return
        self.registry = CollectorRegistry()
        self.requests = Counter(
            "deeprec_requests_total", "Predict requests",
            ["endpoint", "status"], registry=self.registry)
        self.latency = Histogram(
            "deeprec_request_seconds", "Predict latency", ["endpoint"],
            buckets=(.0005, .001, .0025, .005, .01, .025, .05, .1, .25,
                     .5, 1.0, 2.5), registry=self.registry)
        self.rows = Histogram(
            "deeprec_request_rows", "Rows per request",
            buckets=(1, 2, 4, 8, 16, 32, 64, 128, 256, 512, 1024),
            registry=self.registry)
        self.reloads = Counter(
            "deeprec_model_reloads_total", "Full model updates applied",
            registry=self.registry)

    def observe(self, endpoint: str, status: str, secs: float,
                rows: Optional[int] = None):
        if self.registry is None:
            return
        self.requests.labels(endpoint, status).inc()
        self.latency.labels(endpoint).observe(secs)
        if rows is not None:
            self.rows.observe(rows)

    def expose(self):
        from prometheus_client import generate_latest
        return generate_latest(self.registry)


def create_app(predictor: Predictor, batcher: Optional[DynamicBatcher] = None):
    """Build the FastAPI app. Endpoints are sync `def`s — FastAPI runs them
    on its threadpool, and the batcher coalesces across those threads."""
    from fastapi import Body, FastAPI, HTTPException, Response

    app = FastAPI(title="deeprec_amd serving")
    app.state.predictor = predictor
    app.state.batcher = batcher
    metrics = _Metrics()
    app.state.metrics = metrics

    @app.get("/health")
    def health():
        return {"status": "ok",
                "checkpoint": predictor._loaded_full,
                "device": str(predictor.group.device)}

    @app.get("/metrics")
    def prometheus_metrics():
        if metrics.registry is None:
            raise HTTPException(501, "prometheus_client not installed")
        return Response(metrics.expose(),
                        media_type="text/plain; version=0.0.4")

    @app.post("/v1/predict")
    def predict(request: dict = Body(...)):
        t0 = time.monotonic()
        try:
            if batcher is not None:
                dense = torch.tensor(request["dense"], dtype=torch.float32,
                                     device=predictor.group.device)
                sparse = torch.tensor(request["sparse"], dtype=torch.int64,
                                      device=predictor.group.device)
                probs = batcher.submit(dense, sparse).result(timeout=30)
                out = {"probabilities": probs.cpu().tolist()}
            else:
                out = predictor.process(request)
            metrics.observe("predict", "ok", time.monotonic() - t0,
                            len(out["probabilities"]))
            return out
        except KeyError as e:
            metrics.observe("predict", "422", time.monotonic() - t0)
            raise HTTPException(422, f"missing field {e}")
        except (ValueError, TypeError, RuntimeError) as e:
            # ragged lists, wrong dtypes, shape mismatches
            metrics.observe("predict", "422", time.monotonic() - t0)
            raise HTTPException(422, f"malformed request: {e}")

    @app.post("/v1/predict_batch")
    def predict_batch(requests: list = Body(...)):
        t0 = time.monotonic()
        out = predictor.batch_process(requests)
        metrics.observe("predict_batch", "ok", time.monotonic() - t0,
                        len(requests))
        return out

    @app.post("/v1/reload")
    def reload():
        ok = predictor.reload()
        if ok and metrics.registry is not None:
            metrics.reloads.inc()
        return {"reloaded": ok,
                "checkpoint": predictor._loaded_full}

    return app


def serve(model, checkpoint_dir: str, host: str = "0.0.0.0",
          port: int = 8500, num_sessions: int = 2, device=None,
          poll_secs: float = 5.0, batching: bool = True):
    """Run the HTTP server (blocking). Online updates poll in background."""
    import uvicorn
    predictor = Predictor(model, checkpoint_dir, num_sessions, device)
    predictor.start_update_thread(poll_secs)
    batcher = DynamicBatcher(predictor) if batching else None
    try:
        uvicorn.run(create_app(predictor, batcher), host=host, port=port)
    finally:
        if batcher is not None:
            batcher.close()
        predictor.stop_update_thread()
