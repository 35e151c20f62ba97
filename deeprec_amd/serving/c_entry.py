"""Python side of the serving C ABI (called by processor.cpp).

Contract (≙ reference processor.cc):
- initialize(model_entry, model_config_json) -> opaque handle
    model_entry: a zoo model name ("dlrm", "wdl", ...) or
    "module:callable" returning an nn.Module with embedding_variables().
    model_config JSON: {"checkpoint_dir": ..., "model_kwargs": {...},
    "num_sessions": 2, "device": "cpu"|"cuda", "poll_secs": 0,
    "feature_store": {"kind": "redis", "host": ..., "port": ...,
    "tables": [...]}}  (feature_store switches the EVs to remote-KV
    lookups — the RemoteSessionInstance mode).
- process(handle, payload bytes) -> response bytes (JSON in/out)
- batch_process(handle, [payload bytes]) -> one JSON-array response
- shutdown(handle)
"""
from __future__ import annotations

import importlib
import json

import torch

from deeprec_amd.serving.predictor import Predictor


def _build_model(model_entry: str, cfg: dict):
    kwargs = dict(cfg.get("model_kwargs", {}))
    kwargs.setdefault("device", cfg.get("device", "cpu"))
    if ":" in model_entry:
        mod_name, fn_name = model_entry.split(":", 1)
        factory = getattr(importlib.import_module(mod_name), fn_name)
        return factory(**kwargs)
    from deeprec_amd.models import MODEL_REGISTRY
    return MODEL_REGISTRY[model_entry](**kwargs)


def initialize(model_entry: str, model_config: str):
    cfg = json.loads(model_config or "{}")
    model = _build_model(model_entry, cfg)
    fs_cfg = cfg.get("feature_store")
    if fs_cfg:
        from deeprec_amd.serving.remote_kv import attach_remote_store
        attach_remote_store(model, fs_cfg)
    pred = Predictor(model, cfg["checkpoint_dir"],
                     num_sessions=int(cfg.get("num_sessions", 2)),
                     device=cfg.get("device", "cpu"),
                     remote_sparse=bool(fs_cfg))
    poll = float(cfg.get("poll_secs", 0))
    if poll > 0:
        pred.start_update_thread(poll)
    return pred


def process(pred: Predictor, payload: bytes) -> bytes:
    req = json.loads(payload.decode())
    with torch.no_grad():
        resp = pred.process(req)
    return json.dumps(resp).encode()


def batch_process(pred: Predictor, payloads) -> bytes:
    reqs = [json.loads(p.decode()) for p in payloads]
    with torch.no_grad():
        resps = pred.batch_process(reqs)
    return json.dumps(resps).encode()


def shutdown(pred: Predictor):
    try:
        pred.stop_update_thread()
    except Exception:  # noqa: BLE001
        pass
