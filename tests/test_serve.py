"""HTTP serving surface tests (FastAPI TestClient, CPU)."""
import time

import pytest
import torch

fastapi = pytest.importorskip("fastapi")
from fastapi.testclient import TestClient  # noqa: E402

from fmda_amd.models import BiGRU  # noqa: E402
from fmda_amd.runtime.streaming import StreamingPredictor  # noqa: E402
from fmda_amd.serve import create_app  # noqa: E402


@pytest.fixture()
def client():
    torch.manual_seed(0)
    model = BiGRU(16, 12, 4, n_layers=1, spatial_dropout=False, dropout=0.0)
    pred = StreamingPredictor(model, torch.zeros(12), torch.ones(12),
                              window=6, device="cpu", use_graph=False)
    return TestClient(create_app(pred))


def test_healthz(client):
    r = client.get("/healthz")
    assert r.status_code == 200
    body = r.json()
    assert body["ok"] and body["window"] == 6 and body["n_features"] == 12


def test_ingest_predict_flow(client):
    now = time.time()
    # window not full yet
    r = client.post("/timestamp", json={"Timestamp": now})
    assert r.json() == {"ok": False, "reason": "window_not_full"}
    for i in range(6):
        r = client.post("/ingest", json={"row": [0.1 * i] * 12,
                                         "Timestamp": now + i})
        assert r.json()["ok"]
    assert r.json()["window_full"]

    r = client.post("/timestamp", json={"Timestamp": time.time()})
    body = r.json()
    assert body["ok"]
    pred = body["prediction"]
    assert len(pred["probabilities"]) == 4
    assert set(pred["pred_labels"]) <= {"up1", "up2", "down1", "down2"}

    r = client.get("/prediction/latest")
    assert r.json()["prediction"]["probabilities"] == pred["probabilities"]

    # stale timestamps are dropped (predict.py:135 semantics)
    r = client.post("/timestamp", json={"Timestamp": time.time() - 3600})
    assert r.json() == {"ok": False, "reason": "stale"}


def test_metrics_exposition(client):
    now = time.time()
    for i in range(6):
        client.post("/ingest", json={"row": [0.0] * 12})
    client.post("/timestamp", json={"Timestamp": now})
    r = client.get("/metrics")
    assert r.status_code == 200
    text = r.text
    assert "fmda_rows_ingested_total" in text
    assert "fmda_predict_latency_seconds" in text


def test_concurrent_ingest_and_timestamp():
    """serve.py handlers run on a threadpool: hammer ingest and timestamp
    concurrently; the predictor lock must keep every response well-formed
    and the ring consistent."""
    import threading

    import torch

    from fmda_amd.models import BiGRU
    from fmda_amd.runtime import StreamingPredictor

    torch.manual_seed(0)
    m = BiGRU(8, 16, 4, spatial_dropout=False)
    p = StreamingPredictor(m, torch.zeros(16), torch.ones(16), window=4,
                           use_graph=False, stale_after=1e12)
    now = 1_000_000.0
    errs = []

    def ingest(k):
        try:
            for i in range(50):
                p.push_row(torch.rand(16), ts=now + k * 50 + i)
        except Exception as e:  # pragma: no cover
            errs.append(e)

    def predict():
        try:
            for i in range(50):
                p.handle_timestamp({"Timestamp": now}, now=now)
        except Exception as e:  # pragma: no cover
            errs.append(e)

    threads = [threading.Thread(target=ingest, args=(k,)) for k in range(2)]
    threads += [threading.Thread(target=predict) for _ in range(2)]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=60)
    assert not errs
    assert p.ring.count == 100
    out = p.predict_window()
    assert len(out["probabilities"]) == 4
