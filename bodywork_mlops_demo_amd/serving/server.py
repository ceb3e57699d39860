"""HTTP scoring service — wire-compatible with the reference.

Endpoint parity (reference ``stage_2_serve_model.py:73-80`` and docstring
``:8-22``): ``POST /score/v1`` with ``{"X": <x>}`` returns
``{"prediction": <p>, "model_info": "<str(model)>"}``.

Extensions beyond the reference (batched GPU serving):
- ``X`` may be a list → ``prediction`` is a list (scored in one fused
  kernel launch instead of N requests);
- ``POST /score/v1/batch`` with ``{"X": [...]}`` returns
  ``{"predictions": [...], "n": n, "model_info": ...}``;
- ``GET /healthz`` for the pipeline runner's startup probe (replaces the
  k8s readiness mechanism implied by ``bodywork.yaml:39``).

Served by uvicorn (ASGI) instead of the reference's Flask dev server —
one replica process per GPU, model resident in HBM, hipGraph-captured
batch scoring.

Concurrency model (deliberate): handlers are async but call the scorer
synchronously, so scoring requests SERIALIZE within a replica — the
scorer's static graph buffers are single-stream and one GPU stream is
already saturated by a single large batch.  Throughput scales by replica
fan-out (one process per GPU, the reference's ``replicas`` semantics),
not by intra-process concurrency.
"""
from __future__ import annotations

from contextlib import asynccontextmanager

import numpy as np
from fastapi import FastAPI, Request
from fastapi.responses import JSONResponse, Response

from bodywork_mlops_demo_amd.models import regressor_from_artifact
from bodywork_mlops_demo_amd.monitoring.tracing import RequestTracer, timed
from bodywork_mlops_demo_amd.serving.scorer import BatchedScorer
from bodywork_mlops_demo_amd.store import ArtefactStore
from bodywork_mlops_demo_amd.utils.logging import configure_logger

log = configure_logger(__name__)


def create_app(store: ArtefactStore, device: str = "cpu",
               use_graphs: bool = True) -> FastAPI:
    state: dict = {}
    tracer = RequestTracer()

    @asynccontextmanager
    async def lifespan(app: FastAPI):
        artefact, model_date = store.get_latest_model()
        model = regressor_from_artifact(artefact, device=device)
        state["scorer"] = BatchedScorer(model, device, use_graphs=use_graphs)
        state["model_info"] = str(model)
        state["model_date"] = str(model_date)
        log.info(f"loaded model={state['model_info']} trained on {model_date} "
                 f"onto {device}")
        yield

    app = FastAPI(lifespan=lifespan)

    @app.post("/score/v1")
    async def score_data_instance(request: Request) -> Response:
        payload = await request.json()
        features = payload["X"]
        scalar = np.isscalar(features)
        with timed(tracer, rows=1 if scalar else len(features)):
            preds = state["scorer"].score(features)
        prediction = float(preds[0]) if scalar else [float(p) for p in preds]
        return JSONResponse(
            {"prediction": prediction, "model_info": state["model_info"]}
        )

    @app.post("/score/v1/batch")
    async def score_batch(request: Request) -> Response:
        payload = await request.json()
        with timed(tracer, rows=len(payload["X"])):
            preds = state["scorer"].score(payload["X"])
        return JSONResponse(
            {
                "predictions": [float(p) for p in preds],
                "n": int(preds.shape[0]),
                "model_info": state["model_info"],
            }
        )

    @app.post("/score/v1/binary")
    async def score_binary(request: Request) -> Response:
        """High-throughput wire: raw little-endian float32 X in the body,
        raw float32 predictions back (no JSON float formatting/parsing on
        the hot path — the batch-JSON endpoint spends more time in number
        parsing than the GPU does scoring)."""
        body = await request.body()
        X = np.frombuffer(body, dtype=np.float32)
        with timed(tracer, rows=X.shape[0]):
            preds = state["scorer"].score(X)
        return Response(content=preds.astype(np.float32).tobytes(),
                        media_type="application/octet-stream",
                        headers={"X-Model-Info": state["model_info"],
                                 "X-N": str(preds.shape[0])})

    @app.get("/stats")
    async def stats() -> Response:
        return JSONResponse(tracer.snapshot())

    @app.get("/metrics")
    async def metrics() -> Response:
        prom = RequestTracer.prometheus()
        if prom is None:
            return JSONResponse(tracer.snapshot())
        data, content_type = prom
        return Response(content=data, media_type=content_type)

    @app.post("/reload/v1")
    async def reload_model() -> Response:
        """Hot-redeploy: re-read the latest model artefact from the store
        and swap it into the resident scorer.  Weights are copied into the
        captured graphs' tensors when shapes match (no recapture); an
        incompatible model gets a fresh scorer.  The pipeline runner calls
        this on every replica after each retrain when the service is kept
        running across pipeline repeats (k8s rolling-redeploy parity)."""
        try:
            artefact, model_date = store.get_latest_model()
            model = regressor_from_artifact(artefact, device=device)
        except Exception as e:  # no/unreadable artefact
            return JSONResponse({"status": "error", "error": str(e)},
                                status_code=500)
        scorer = state.get("scorer")
        if scorer is None or not scorer.update_model(model):
            state["scorer"] = BatchedScorer(model, device,
                                            use_graphs=use_graphs)
        state["model_info"] = str(model)
        state["model_date"] = str(model_date)
        log.info(f"reloaded model={state['model_info']} trained on "
                 f"{model_date}")
        return JSONResponse({"status": "ok",
                             "model_date": state["model_date"],
                             "model_info": state["model_info"]})

    @app.get("/healthz")
    async def healthz() -> Response:
        ok = "scorer" in state
        return JSONResponse(
            {"status": "ok" if ok else "starting",
             "model_date": state.get("model_date"),
             "model_info": state.get("model_info")},
            status_code=200 if ok else 503,
        )

    return app
