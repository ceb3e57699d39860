"""Stage-level tests: hermetic 4-stage flow on a local store (CPU)."""
from datetime import date

import numpy as np
import pytest
import torch

from bodywork_mlops_demo_amd.serving.scorer import BatchedScorer
from bodywork_mlops_demo_amd.stages import datagen, loadtest, train
from bodywork_mlops_demo_amd.store import contract


@pytest.fixture()
def seeded_store(tmp_store):
    for day in (1, 2, 3):
        datagen.run(tmp_store, n=500, date=date(2026, 1, day), device="cpu",
                    seed=day)
    return tmp_store


def test_stage3_datagen_writes_contract_csv(tmp_store):
    y, X, d = datagen.run(tmp_store, n=400, date=date(2026, 2, 1), device="cpu")
    key = contract.dataset_key(d)
    assert tmp_store.exists(key)
    header = tmp_store.get_bytes(key).decode().splitlines()[0]
    assert header == "date,y,X"
    assert y.shape == X.shape and y.shape[0] <= 400


def test_stage1_train_reads_all_history(seeded_store):
    metrics = train.run(seeded_store, model_type="linear", device="cpu")
    assert set(metrics) == {"MAPE", "r_squared", "max_residual"}
    # model + metrics artefacts keyed by newest dataset date
    assert seeded_store.exists(contract.model_key(date(2026, 1, 3)))
    assert seeded_store.exists(contract.model_metrics_key(date(2026, 1, 3)))
    rec = seeded_store.get_metrics_csv(contract.model_metrics_key(date(2026, 1, 3)))
    assert list(rec) == ["date", "MAPE", "r_squared", "max_residual"]


def test_stage2_scorer_and_stage4_loadtest_inproc(seeded_store):
    train.run(seeded_store, model_type="linear", device="cpu")
    artefact, model_date = seeded_store.get_latest_model()
    from bodywork_mlops_demo_amd.models import regressor_from_artifact

    scorer = BatchedScorer(regressor_from_artifact(artefact), "cpu")
    preds = scorer.score([0.0, 50.0, 100.0])
    assert preds.shape == (3,)

    metrics = loadtest.run(seeded_store, device="cpu", scorer=scorer)
    assert seeded_store.exists(contract.test_metrics_key(date(2026, 1, 3)))
    rec = seeded_store.get_metrics_csv(contract.test_metrics_key(date(2026, 1, 3)))
    # stage_4:106-112 schema + the response_time_kind annotation column
    # (so the in-process amortised figure can't be read as a latency)
    assert list(rec) == [
        "date", "MAPE", "r_squared", "max_residual", "mean_response_time",
        "response_time_kind",
    ]
    assert rec["response_time_kind"] == "amortised-gpu-batch"
    assert 0 <= metrics["r_squared"] <= 1
    assert metrics["mean_response_time"] > 0


def test_serving_http_wire_format(seeded_store):
    """POST /score/v1 responds exactly like the reference (stage_2:8-22)."""
    from fastapi.testclient import TestClient

    from bodywork_mlops_demo_amd.serving.server import create_app

    train.run(seeded_store, model_type="linear", device="cpu")
    app = create_app(seeded_store, device="cpu")
    with TestClient(app) as client:
        r = client.post("/score/v1", json={"X": 50})
        assert r.status_code == 200
        body = r.json()
        assert set(body) == {"prediction", "model_info"}
        assert isinstance(body["prediction"], float)
        assert body["model_info"] == "LinearRegression()"

        # batch extension
        r = client.post("/score/v1/batch", json={"X": [1.0, 2.0, 3.0]})
        assert r.json()["n"] == 3

        # list input on the v1 endpoint
        r = client.post("/score/v1", json={"X": [10.0, 20.0]})
        assert len(r.json()["prediction"]) == 2

        # binary wire: raw float32 in/out
        X = np.array([10.0, 20.0, 30.0], dtype=np.float32)
        r = client.post("/score/v1/binary", content=X.tobytes(),
                        headers={"Content-Type": "application/octet-stream"})
        assert r.status_code == 200
        preds = np.frombuffer(r.content, dtype=np.float32)
        assert preds.shape == (3,)
        assert r.headers["X-N"] == "3"

        r = client.get("/healthz")
        assert r.json()["status"] == "ok"

        # tracing: latency histogram + prometheus exposition
        r = client.get("/stats")
        snap = r.json()
        assert snap["requests"] >= 3
        assert snap["rows_scored"] >= 8
        assert snap["mean_latency_s"] > 0
        r = client.get("/metrics")
        assert r.status_code == 200
        assert b"scoring_requests_total" in r.content or "requests" in r.text


def test_mlp_stage1(seeded_store):
    metrics = train.run(
        seeded_store, model_type="mlp", device="cpu", mlp_steps=5,
        mlp_batch_size=256,
    )
    artefact, _ = seeded_store.get_latest_model()
    assert type(artefact).__name__ == "MLPRegressor"
    assert np.isfinite(metrics["MAPE"])


def test_loadtest_serial_sentinels(monkeypatch, seeded_store):
    """Connection failures produce -1 sentinel scores (stage_4:82-85)."""
    train.run(seeded_store, model_type="linear", device="cpu")

    calls = {"n": 0}

    class FakeSession:
        def mount(self, *a, **k):
            pass

        def post(self, url, json=None, timeout=None):
            calls["n"] += 1

            class R:
                ok = calls["n"] % 2 == 0

                @staticmethod
                def json():
                    return {"prediction": 42.0}

            return R()

    import requests

    monkeypatch.setattr(requests, "Session", FakeSession)
    X = np.array([1.0, 2.0, 3.0, 4.0])
    scores, times = loadtest._score_serial("http://x/score/v1", X)
    assert (scores == np.array([-1, 42.0, -1, 42.0])).all()
    assert (times >= 0).all()
