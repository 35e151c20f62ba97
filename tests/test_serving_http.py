"""HTTP serving shell: FastAPI endpoints + dynamic batcher (CPU)."""
import threading

import pytest
import torch

from deeprec_amd.models.dlrm import DLRM
from deeprec_amd.optimizers import AdagradOptimizer
from deeprec_amd.serving.predictor import Predictor
from deeprec_amd.serving.server import DynamicBatcher, create_app

fastapi = pytest.importorskip("fastapi")
from fastapi.testclient import TestClient  # noqa: E402


def _make_ckpt(tmp_path):
    torch.manual_seed(0)
    model = DLRM(device="cpu", bf16=False, num_sparse=4,
                 mlp_bot=(32, 16), mlp_top=(32,))
    opt = AdagradOptimizer(params=model.parameters(),
                           embedding_variables=model.embedding_variables(),
                           learning_rate=0.01)
    dense = torch.randn(8, 13)
    sparse = torch.randint(0, 100, (8, 4))
    labels = torch.randint(0, 2, (8,)).float()
    loss = model.loss_fn(model(dense, sparse), labels)
    opt.zero_grad()
    loss.backward()
    opt.step()
    from deeprec_amd.checkpoint.saver import Saver
    Saver(module=model, embedding_variables=model.embedding_variables(),
          optimizer=opt).save(str(tmp_path), global_step=1)
    return model


def test_http_predict_and_health(tmp_path):
    model = _make_ckpt(tmp_path)
    pred = Predictor(model, str(tmp_path), num_sessions=1, device="cpu")
    client = TestClient(create_app(pred))

    r = client.get("/health")
    assert r.status_code == 200 and r.json()["status"] == "ok"
    assert "ckpt-1" in r.json()["checkpoint"]

    req = {"dense": torch.randn(3, 13).tolist(),
           "sparse": torch.randint(0, 100, (3, 4)).tolist()}
    r = client.post("/v1/predict", json=req)
    assert r.status_code == 200
    probs = r.json()["probabilities"]
    assert len(probs) == 3 and all(0.0 <= p <= 1.0 for p in probs)

    # batch endpoint
    r = client.post("/v1/predict_batch", json=[req, req])
    assert r.status_code == 200 and len(r.json()) == 2

    # missing field -> 422
    r = client.post("/v1/predict", json={"dense": [[0.0] * 13]})
    assert r.status_code == 422
    # ragged / malformed payload -> 422, not 500
    r = client.post("/v1/predict", json={"dense": [[0.0] * 13, [0.0]],
                                         "sparse": [[1] * 4]})
    assert r.status_code == 422

    r = client.post("/v1/reload")
    assert r.status_code == 200 and r.json()["reloaded"] is False


def test_http_predict_with_batcher(tmp_path):
    model = _make_ckpt(tmp_path)
    pred = Predictor(model, str(tmp_path), num_sessions=1, device="cpu")
    batcher = DynamicBatcher(pred, max_batch=64, max_latency_ms=5.0)
    client = TestClient(create_app(pred, batcher))
    try:
        req = {"dense": torch.randn(2, 13).tolist(),
               "sparse": torch.randint(0, 100, (2, 4)).tolist()}
        direct = client.post("/v1/predict_batch", json=[req]).json()[0]
        via_batcher = client.post("/v1/predict", json=req).json()
        assert via_batcher["probabilities"] == pytest.approx(
            direct["probabilities"], abs=1e-6)
    finally:
        batcher.close()


def test_batcher_coalesces_concurrent_requests(tmp_path):
    model = _make_ckpt(tmp_path)
    pred = Predictor(model, str(tmp_path), num_sessions=1, device="cpu")
    seen_batches = []
    orig = pred.predict

    def spy(dense, sparse):
        seen_batches.append(dense.shape[0])
        return orig(dense, sparse)

    pred.predict = spy
    batcher = DynamicBatcher(pred, max_batch=256, max_latency_ms=50.0)
    try:
        futs = []
        start = threading.Barrier(8 + 1)

        def worker():
            start.wait()
            futs.append(batcher.submit(torch.randn(1, 13),
                                       torch.randint(0, 100, (1, 4))))

        ts = [threading.Thread(target=worker) for _ in range(8)]
        for t in ts:
            t.start()
        start.wait()
        for t in ts:
            t.join()
        rows = sum(f.result(timeout=10).shape[0] for f in futs)
        assert rows == 8
        # coalescing happened: fewer model calls than requests
        assert len(seen_batches) < 8 and sum(seen_batches) == 8
    finally:
        batcher.close()


def test_batcher_propagates_errors():
    class Boom:
        def predict(self, d, s):
            raise RuntimeError("boom")

    batcher = DynamicBatcher(Boom(), max_latency_ms=1.0)
    try:
        fut = batcher.submit(torch.randn(1, 13),
                             torch.randint(0, 10, (1, 4)))
        with pytest.raises(RuntimeError, match="boom"):
            fut.result(timeout=10)
    finally:
        batcher.close()


def test_feature_store_publish_and_lookup(tmp_path):
    from deeprec_amd.serving.feature_store import (
        FileFeatureStore, LocalFeatureStore, publish_checkpoint,
        store_backed_lookup)

    model = _make_ckpt(tmp_path)
    # exported truth from the live model
    tables = model.collection.export_tables()
    store = LocalFeatureStore()
    ckpt = sorted(tmp_path.glob("ckpt-*"))[0]
    n = publish_checkpoint(store, str(ckpt))
    assert n > 0 and len(store.tables()) > 0

    name = store.tables()[0]
    # find the matching exported table rows
    for tname, (k, v, f, ver) in tables.items():
        if tname in name or name in tname:
            got = store_backed_lookup(store, name, k[:5], v.shape[1])
            torch.testing.assert_close(got, v[:5], rtol=1e-6, atol=1e-6)
            break

    # file-backed round trip
    fstore = FileFeatureStore(str(tmp_path / "fs"))
    publish_checkpoint(fstore, str(ckpt))
    fstore.flush()
    fstore2 = FileFeatureStore(str(tmp_path / "fs"))
    assert sorted(fstore2.tables()) == sorted(fstore.tables())
    missing = store_backed_lookup(fstore2, name,
                                  torch.tensor([10 ** 12]), 4, default=0.5)
    assert torch.allclose(missing, torch.full((1, 4), 0.5))


def test_prometheus_metrics_endpoint(tmp_path):
    """/metrics exposes request counters and latency histograms
    (observability beyond the reference's hook-based step metrics,
    SURVEY §5)."""
    pytest.importorskip("prometheus_client")
    model = _make_ckpt(tmp_path)
    pred = Predictor(model, str(tmp_path), num_sessions=1, device="cpu")
    client = TestClient(create_app(pred))

    req = {"dense": torch.randn(3, 13).tolist(),
           "sparse": torch.randint(0, 100, (3, 4)).tolist()}
    assert client.post("/v1/predict", json=req).status_code == 200
    assert client.post("/v1/predict", json={"dense": [[1.0]]}
                       ).status_code == 422
    body = client.get("/metrics").text
    assert 'deeprec_requests_total{endpoint="predict",status="ok"} 1.0' \
        in body
    assert 'deeprec_requests_total{endpoint="predict",status="422"} 1.0' \
        in body
    assert "deeprec_request_seconds_bucket" in body
    # two apps in one process: per-app registries must not collide
    client2 = TestClient(create_app(pred))
    assert client2.get("/metrics").status_code == 200
