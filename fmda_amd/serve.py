"""HTTP serving surface for the streaming predictor.

The reference publishes predictions to a Kafka `prediction` topic for
downstream consumers (predict.py:193-197). This module puts the same
predictor behind a FastAPI app for deployments where consumers pull over
HTTP instead of subscribing to a topic:

- POST /ingest        {"row": [108 floats], "Timestamp": t}  -> ring push
- POST /timestamp     {"Timestamp": t}  -> run the predict step (stale
                      messages are dropped exactly like predict.py:135)
- GET  /prediction/latest               -> last prediction dict
- GET  /healthz                         -> liveness + model/config info
- GET  /metrics                         -> Prometheus exposition

Run: uvicorn "fmda_amd.serve:build_app_from_artifacts(...)" or see
`create_app` for wiring an existing StreamingPredictor. On an MI355X the
predictor's batch-1 step is the hipGraph-captured HIP engine path.
"""
import time
from typing import Optional

import torch

try:
    from prometheus_client import (CONTENT_TYPE_LATEST, CollectorRegistry,
                                   Counter, Histogram, generate_latest)
    _PROM = True
except ImportError:  # pragma: no cover
    _PROM = False

from .runtime.streaming import StreamingPredictor


def create_app(predictor: StreamingPredictor):
    from fastapi import FastAPI
    from fastapi.responses import Response

    app = FastAPI(title="fmda_amd streaming predictor")
    state = {"latest": None, "n_ingested": 0}

    if _PROM:
        # per-app registry: multiple app instances (tests, multi-model
        # deployments) must not collide in the global default registry
        reg = CollectorRegistry()
        c_pred = Counter("fmda_predictions_total",
                         "prediction steps served", registry=reg)
        c_stale = Counter("fmda_stale_dropped_total",
                          "timestamp messages dropped as stale", registry=reg)
        c_rows = Counter("fmda_rows_ingested_total", "feature rows pushed",
                         registry=reg)
        h_lat = Histogram(
            "fmda_predict_latency_seconds", "predict step wall latency",
            buckets=(1e-4, 2.5e-4, 5e-4, 1e-3, 2.5e-3, 5e-3, 1e-2, 5e-2),
            registry=reg)

    @app.post("/ingest")
    def ingest(msg: dict):
        row = torch.tensor(msg["row"], dtype=torch.float32)
        predictor.push_row(row)
        state["n_ingested"] += 1
        if _PROM:
            c_rows.inc()
        return {"ok": True, "rows": state["n_ingested"],
                "window_full": predictor.ring.full}

    @app.post("/timestamp")
    def timestamp(msg: dict):
        t0 = time.perf_counter()
        pred = predictor.handle_timestamp(msg)
        if pred is None:
            if _PROM and predictor.ring.full:
                c_stale.inc()
            return {"ok": False,
                    "reason": ("stale" if predictor.ring.full
                               else "window_not_full")}
        state["latest"] = pred
        if _PROM:
            c_pred.inc()
            h_lat.observe(time.perf_counter() - t0)
        return {"ok": True, "prediction": pred}

    @app.get("/prediction/latest")
    def latest():
        return {"prediction": state["latest"]}

    @app.get("/healthz")
    def healthz():
        return {"ok": True,
                "device": str(predictor.device),
                "window": predictor.window,
                "n_features": predictor.n_features,
                "hipgraph": predictor._use_graph,
                "n_predictions": predictor.n_predictions}

    @app.get("/metrics")
    def metrics():
        if not _PROM:  # pragma: no cover
            return Response("prometheus_client not installed",
                            media_type="text/plain")
        return Response(generate_latest(reg),
                        media_type=CONTENT_TYPE_LATEST)

    return app


def build_app_from_artifacts(checkpoint: str = "model_params.pt",
                             norm_params: str = "norm_params",
                             window: int = 30,
                             device: Optional[str] = None):
    """App factory from the reference-format artifacts (the same pair
    predict.py loads at :104 and :110-122)."""
    from .data.norm import load_norm_params
    from .models.checkpoint import load_checkpoint

    model = load_checkpoint(checkpoint)
    _, x_min, x_max = load_norm_params(norm_params)
    if device is None:
        device = "cuda:0" if torch.cuda.is_available() else "cpu"
    dtype = torch.bfloat16 if device.startswith("cuda") else torch.float32
    if dtype is torch.bfloat16:
        model = model.to(dtype)
    pred = StreamingPredictor(model, x_min, x_max, window, device=device,
                              dtype=dtype)
    return create_app(pred)
