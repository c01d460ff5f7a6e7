"""FastAPI serving — attaches ``/``, ``/predict``, ``/health`` routes
(contract of the reference's unionml/fastapi.py:15-70).

MI355X-native addition: with ``batch=True`` requests are funneled
through a dynamic micro-batcher whose fused forward is replayed from
bucketed hipGraph captures (unionml_amd/serving/batcher.py).
"""

import os
import time
from typing import Optional

from unionml_amd._logging import logger
from unionml_amd.artifact import ModelArtifact


def serving_app(
    model,
    app,
    *,
    remote: bool = False,
    app_version: Optional[str] = None,
    model_version: str = "latest",
    batch: bool = False,
    max_batch_size: int = 64,
    max_delay_ms: float = 0.0,
    metrics: bool = True,
):
    from fastapi import HTTPException, Request

    state = {"batcher": None, "metrics": None}
    if metrics:
        try:
            from unionml_amd.serving.metrics import ServingMetrics

            state["metrics"] = ServingMetrics(model.name)
        except ImportError:  # prometheus_client not installed
            pass

    async def load_model():
        # local: $UNIONML_MODEL_PATH; remote: latest successful training
        # run from the backend registry (reference: fastapi.py:22-34)
        model_path = os.environ.get("UNIONML_MODEL_PATH")
        if model_path:
            model.artifact = ModelArtifact(model.load(model_path))
        elif remote:
            model.artifact = model._backend().fetch_model_artifact(
                model, app_version=app_version, model_version=model_version
            )
        if model.artifact is not None and batch:
            from unionml_amd.serving.batcher import DynamicBatcher

            state["batcher"] = DynamicBatcher(
                model, max_batch_size=max_batch_size, max_delay_ms=max_delay_ms
            )
            state["batcher"].start()

    async def stop_batcher():
        if state["batcher"] is not None:
            state["batcher"].stop()

    # wrap (don't replace) any lifespan the user's app already has —
    # the modern replacement for the deprecated @app.on_event hooks
    from contextlib import asynccontextmanager

    existing_lifespan = app.router.lifespan_context

    @asynccontextmanager
    async def lifespan(app_):
        async with existing_lifespan(app_):
            await load_model()
            try:
                yield
            finally:
                await stop_batcher()

    app.router.lifespan_context = lifespan

    @app.get("/")
    async def root():
        return {
            "app": "unionml_amd",
            "model": model.name,
            "message": f"unionml_amd serving app for model '{model.name}'. "
            "POST /predict with {'features': [...]} or {'inputs': {...}}.",
        }

    @app.post("/predict")
    async def predict(request: Request):
        # parse the body once ourselves — the dual-Body(...) signature
        # costs a pydantic validation pass per request on the hot path
        try:
            body = await request.json()
        except Exception:
            raise HTTPException(status_code=400, detail="body must be JSON")
        if not isinstance(body, dict):
            raise HTTPException(
                status_code=400, detail="body must be a JSON object"
            )
        inputs = body.get("inputs")
        features = body.get("features")
        if model.artifact is None:
            raise HTTPException(status_code=500, detail="model artifact not loaded")
        if inputs is None and features is None:
            raise HTTPException(
                status_code=400, detail="provide one of 'inputs' or 'features'"
            )
        mx = state["metrics"]
        t0 = time.perf_counter() if mx else 0.0
        try:
            if features is not None:
                n_rows = len(features) if hasattr(features, "__len__") else 1
                if state["batcher"] is not None:
                    out = await state["batcher"].submit(features)
                else:
                    features = model._dataset.get_features(features)
                    wf = model.predict_from_features_workflow()
                    out = _jsonable(
                        wf(model_object=model.artifact.model_object, features=features)
                    )
            else:
                out = _jsonable(model.predict(**(inputs or {})))
                n_rows = len(out) if hasattr(out, "__len__") else 1
            if mx:
                mx.observe(n_rows, time.perf_counter() - t0)
            return out
        except HTTPException:
            raise
        except Exception as exc:
            if mx:
                mx.error()
            logger.exception("prediction failed")
            raise HTTPException(status_code=500, detail=str(exc))

    if state["metrics"] is not None:

        @app.get("/metrics")
        async def metrics_route():
            from fastapi import Response

            payload, content_type = state["metrics"].render()
            return Response(content=payload, media_type=content_type)

    @app.get("/health")
    async def health():
        if model.artifact is None:
            raise HTTPException(status_code=500, detail="model artifact not loaded")
        return {"status": "ok"}

    @app.post("/reload")
    async def reload_model(request: Request):
        """Hot-reload the artifact (after e.g. `unionml-amd fetch-model`)
        without restarting the server: re-read $UNIONML_MODEL_PATH (or
        the backend's latest), swap the artifact, and rebuild the
        batcher's hipGraphs against the new weights.

        Reloads force GPU/memory churn (full hipGraph re-capture), so
        when $UNIONML_RELOAD_TOKEN is set the request must carry it in
        the X-Reload-Token header; unauthenticated reloads are only
        allowed when no token was configured (trusted-network default).
        """
        token = os.environ.get("UNIONML_RELOAD_TOKEN")
        if token and request.headers.get("x-reload-token") != token:
            raise HTTPException(status_code=403, detail="invalid or missing X-Reload-Token")
        model_path = os.environ.get("UNIONML_MODEL_PATH")
        try:
            if model_path:
                model.artifact = ModelArtifact(model.load(model_path))
            elif remote:
                model.artifact = model._backend().fetch_model_artifact(
                    model, app_version=app_version, model_version=model_version
                )
            else:
                raise HTTPException(
                    status_code=400,
                    detail="no reload source: set UNIONML_MODEL_PATH or serve with remote=True",
                )
        except HTTPException:
            raise
        except Exception as exc:
            logger.exception("model reload failed")
            raise HTTPException(status_code=500, detail=str(exc))
        if batch:
            from unionml_amd.serving.batcher import DynamicBatcher

            old = state["batcher"]
            state["batcher"] = DynamicBatcher(
                model, max_batch_size=max_batch_size, max_delay_ms=max_delay_ms
            )
            state["batcher"].start()
            if old is not None:
                old.stop()
        return {"status": "reloaded", "source": model_path or "backend"}

    return app


def _jsonable(predictions):
    import numpy as np

    if hasattr(predictions, "detach"):  # torch tensor
        return predictions.detach().cpu().tolist()
    if isinstance(predictions, np.ndarray):
        return predictions.tolist()
    if isinstance(predictions, (list, tuple)):
        return [_jsonable(p) for p in predictions]
    if isinstance(predictions, (np.generic,)):
        return predictions.item()
    return predictions
