"""FastAPI serving tests.

Route contract mirrors the reference (unionml/fastapi.py:15-70, tested
in its tests/integration/test_fastapi.py): `GET /` banner, `POST
/predict` with either features or reader inputs, `GET /health` failing
without an artifact; plus the subprocess `serve` integration with
retry-based health polling (reference test_fastapi.py:14-49) and this
build's dynamic micro-batcher."""

import asyncio
import json
import socket
import subprocess
import sys
import time
from pathlib import Path

import pandas as pd
import pytest
from fastapi import FastAPI
from fastapi.testclient import TestClient

from model_fixtures import build_sklearn_app


@pytest.fixture
def trained_app():
    model = build_sklearn_app()
    model.train()
    app = FastAPI()
    model.serve(app)
    return model, app


def test_root_banner(trained_app):
    model, app = trained_app
    with TestClient(app) as client:
        resp = client.get("/")
        assert resp.status_code == 200
        assert model.name in resp.json()["model"]


def test_health_requires_artifact():
    model = build_sklearn_app()
    app = FastAPI()
    model.serve(app)
    with TestClient(app) as client:
        assert client.get("/health").status_code == 500


def test_health_ok(trained_app):
    _, app = trained_app
    with TestClient(app) as client:
        assert client.get("/health").json() == {"status": "ok"}


def test_predict_with_features(trained_app):
    model, app = trained_app
    feats = [{"x1": 0.5, "x2": 0.1, "x3": 0.9}, {"x1": 0.2, "x2": 0.8, "x3": 0.4}]
    with TestClient(app) as client:
        resp = client.post("/predict", json={"features": feats})
        assert resp.status_code == 200, resp.text
        preds = resp.json()
        assert len(preds) == 2
        direct = model.predict(features=model._dataset.get_features(feats))
        assert preds == direct


def test_predict_with_reader_inputs(trained_app):
    _, app = trained_app
    with TestClient(app) as client:
        resp = client.post("/predict", json={"inputs": {"n": 7}})
        assert resp.status_code == 200, resp.text
        assert len(resp.json()) == 7


def test_predict_requires_body(trained_app):
    _, app = trained_app
    with TestClient(app) as client:
        assert client.post("/predict", json={}).status_code == 400


def test_predict_no_artifact_500():
    model = build_sklearn_app()
    app = FastAPI()
    model.serve(app)
    with TestClient(app) as client:
        resp = client.post("/predict", json={"features": [{"x1": 1, "x2": 2, "x3": 3}]})
        assert resp.status_code == 500


# ----------------------------------------------------------------------
# dynamic batcher (CPU path of the hipGraph-bucketed serve step)
# ----------------------------------------------------------------------


def test_batcher_coalesces_and_matches_direct():
    from unionml_amd.serving.batcher import DynamicBatcher

    model = build_sklearn_app()
    model.train()

    calls = []
    orig = model._run_predictor

    def counting(model_obj, features):
        calls.append(len(features))
        return orig(model_obj, features)

    model._run_predictor = counting

    batcher = DynamicBatcher(model, max_batch_size=64, max_delay_ms=30.0)
    batcher.start()
    try:
        feats = [
            [{"x1": 0.1 * i, "x2": 0.2, "x3": 0.3}, {"x1": 0.9, "x2": 0.1 * i, "x3": 0.5}]
            for i in range(8)
        ]

        async def fire():
            return await asyncio.gather(*(batcher.submit(f) for f in feats))

        results = asyncio.run(fire())
    finally:
        batcher.stop()

    batched_calls = list(calls)
    assert len(results) == 8
    for f, r in zip(feats, results):
        direct = model.predict(features=model._dataset.get_features(f))
        assert r == direct
    # 16 rows total, arriving together within max_delay -> far fewer
    # forwards than requests
    assert len(batched_calls) < 8, f"batcher did not coalesce: {batched_calls}"


def test_batcher_bucketing():
    from unionml_amd.serving.batcher import bucket_for

    assert [bucket_for(n, 64) for n in (1, 2, 3, 5, 17, 64, 100)] == [1, 2, 4, 8, 32, 64, 64]


def test_serving_app_with_batcher_route():
    model = build_sklearn_app()
    model.train()
    app = FastAPI()
    model.serve(app, batch=True, max_delay_ms=1.0)
    feats = [{"x1": 0.5, "x2": 0.1, "x3": 0.9}]
    with TestClient(app) as client:
        resp = client.post("/predict", json={"features": feats})
        assert resp.status_code == 200, resp.text
        assert resp.json() == model.predict(features=model._dataset.get_features(feats))


# ----------------------------------------------------------------------
# subprocess integration: unionml-amd serve + real HTTP
# ----------------------------------------------------------------------

SERVE_APP = """
from typing import List

import pandas as pd
from fastapi import FastAPI
from sklearn.linear_model import LogisticRegression

from unionml_amd import Dataset, Model

dataset = Dataset(name="serve_ds", targets=["y"], test_size=0.2, shuffle=True, random_state=0)
model = Model(name="serve_model", init=LogisticRegression, dataset=dataset)


@dataset.reader
def reader(n: int = 50) -> pd.DataFrame:
    import numpy as np

    rng = np.random.RandomState(3)
    X = rng.rand(n, 3)
    return pd.DataFrame(
        {"x1": X[:, 0], "x2": X[:, 1], "x3": X[:, 2], "y": (X.sum(axis=1) > 1.5).astype(int)}
    )


@model.trainer
def trainer(
    estimator: LogisticRegression, features: pd.DataFrame, target: pd.DataFrame
) -> LogisticRegression:
    return estimator.fit(features, target.squeeze())


@model.predictor
def predictor(estimator: LogisticRegression, features: pd.DataFrame) -> List[float]:
    return [float(x) for x in estimator.predict(features)]


@model.evaluator
def evaluator(estimator: LogisticRegression, features: pd.DataFrame, target: pd.DataFrame) -> float:
    return float((estimator.predict(features) == target.squeeze().to_numpy()).mean())


app = FastAPI()
model.serve(app)

if __name__ == "__main__":
    model.train()
    model.save("model.joblib")
"""


def _free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


@pytest.mark.timeout(120)
def test_serve_subprocess_http(tmp_path):
    import os

    import httpx

    import unionml_amd

    env = dict(
        os.environ,
        PYTHONPATH=str(Path(unionml_amd.__file__).parent.parent)
        + os.pathsep
        + os.environ.get("PYTHONPATH", ""),
    )

    (tmp_path / "serve_app.py").write_text(SERVE_APP)
    # train + save the artifact in-process (reference trains via runpy:
    # test_fastapi.py:63-78)
    proc = subprocess.run(
        [sys.executable, "serve_app.py"], cwd=tmp_path, capture_output=True, text=True, env=env
    )
    assert proc.returncode == 0, proc.stderr
    assert (tmp_path / "model.joblib").exists()

    port = _free_port()
    server = subprocess.Popen(
        [
            sys.executable,
            "-m",
            "unionml_amd.cli",
            "serve",
            "serve_app:app",
            "--model-path",
            "model.joblib",
            "--port",
            str(port),
        ],
        cwd=tmp_path,
        stdout=subprocess.PIPE,
        stderr=subprocess.STDOUT,
        env=env,
    )
    try:
        # retry-based health polling (reference: test_fastapi.py:30-49)
        deadline = time.monotonic() + 60
        healthy = False
        while time.monotonic() < deadline:
            if server.poll() is not None:
                out = server.stdout.read().decode()
                raise AssertionError(f"server exited early:\n{out[-3000:]}")
            try:
                if httpx.get(f"http://127.0.0.1:{port}/health", timeout=1.0).status_code == 200:
                    healthy = True
                    break
            except httpx.HTTPError:
                time.sleep(0.5)
        assert healthy, "server never became healthy"

        feats = [{"x1": 0.9, "x2": 0.9, "x3": 0.9}, {"x1": 0.0, "x2": 0.1, "x3": 0.0}]
        resp = httpx.post(
            f"http://127.0.0.1:{port}/predict", json={"features": feats}, timeout=10.0
        )
        assert resp.status_code == 200, resp.text
        preds = resp.json()
        assert len(preds) == 2
        assert all(p in (0.0, 1.0) for p in preds)
    finally:
        server.terminate()
        try:
            server.wait(timeout=10)
        except subprocess.TimeoutExpired:
            server.kill()


def test_metrics_endpoint():
    """GET /metrics exposes Prometheus counters + latency histogram
    (net-new observability vs the reference's /health-only surface)."""
    model = build_sklearn_app()
    model.train()
    app = FastAPI()
    model.serve(app)
    feats = [{"x1": 0.5, "x2": 0.1, "x3": 0.9}, {"x1": 0.2, "x2": 0.8, "x3": 0.4}]
    with TestClient(app) as client:
        for _ in range(3):
            assert client.post("/predict", json={"features": feats}).status_code == 200
        resp = client.get("/metrics")
        assert resp.status_code == 200
        body = resp.text
        assert "unionml_predict_requests_total" in body
        assert "unionml_predict_rows_total" in body
        assert "unionml_predict_latency_seconds_bucket" in body
        import re

        m = re.search(r'unionml_predict_requests_total{[^}]*} (\d+\.\d+)', body)
        assert m and float(m.group(1)) == 3.0, body[:500]
        m = re.search(r'unionml_predict_rows_total{[^}]*} (\d+\.\d+)', body)
        assert m and float(m.group(1)) == 6.0


def test_reload_endpoint(tmp_path, monkeypatch):
    """POST /reload hot-swaps the artifact from $UNIONML_MODEL_PATH."""
    model = build_sklearn_app()
    model.train()
    path_a = tmp_path / "a.joblib"
    model.save(path_a)
    coef_a = model.artifact.model_object.coef_.copy()

    # a second, differently-trained artifact
    model2 = build_sklearn_app()
    model2.train(hyperparameters={"C": 0.01})
    path_b = tmp_path / "b.joblib"
    model2.save(path_b)

    app = FastAPI()
    model.serve(app)
    monkeypatch.setenv("UNIONML_MODEL_PATH", str(path_a))
    with TestClient(app) as client:
        assert client.get("/health").status_code == 200
        import numpy as np

        assert np.allclose(model.artifact.model_object.coef_, coef_a)

        monkeypatch.setenv("UNIONML_MODEL_PATH", str(path_b))
        resp = client.post("/reload")
        assert resp.status_code == 200, resp.text
        assert not np.allclose(model.artifact.model_object.coef_, coef_a)
        # serving still works on the new artifact
        feats = [{"x1": 0.5, "x2": 0.1, "x3": 0.9}]
        assert client.post("/predict", json={"features": feats}).status_code == 200

    monkeypatch.delenv("UNIONML_MODEL_PATH", raising=False)


def test_reload_without_source_400():
    model = build_sklearn_app()
    model.train()
    app = FastAPI()
    model.serve(app)
    import os

    os.environ.pop("UNIONML_MODEL_PATH", None)
    with TestClient(app) as client:
        assert client.post("/reload").status_code == 400


def test_predict_malformed_inputs_never_crash(trained_app):
    """Error paths: malformed bodies must map to 4xx/5xx JSON errors,
    never an unhandled exception."""
    _, app = trained_app
    with TestClient(app) as client:
        cases = [
            ("not json at all", None),                      # invalid JSON
            (None, [1, 2, 3]),                              # non-object body
            (None, {"features": "not-a-record-list"}),      # wrong features type
            (None, {"features": [{"wrong": 1.0}]}),         # missing columns
            (None, {"inputs": {"nonexistent_kwarg": 1}}),   # bad reader kwarg
            (None, {"features": []}),                       # empty list
        ]
        for raw, body in cases:
            if raw is not None:
                resp = client.post(
                    "/predict", content=raw, headers={"Content-Type": "application/json"}
                )
            else:
                resp = client.post("/predict", json=body)
            assert resp.status_code in (400, 422, 500), (raw, body, resp.status_code)
            assert resp.json().get("detail"), (raw, body)


def test_batcher_route_propagates_errors_cleanly():
    """A bad request through the batcher must 500 with detail and leave
    the batcher serving subsequent good requests."""
    model = build_sklearn_app()
    model.train()
    app = FastAPI()
    model.serve(app, batch=True)
    good = [{"x1": 0.5, "x2": 0.1, "x3": 0.9}]
    with TestClient(app) as client:
        r = client.post("/predict", json={"features": [{"bogus": 1}]})
        assert r.status_code == 500 and r.json().get("detail")
        r = client.post("/predict", json={"features": good})
        assert r.status_code == 200, r.text


def test_reload_token_gate(tmp_path, monkeypatch):
    """With $UNIONML_RELOAD_TOKEN set, /reload requires the matching
    X-Reload-Token header (advisor finding: unauthenticated reloads can
    force artifact swaps + hipGraph re-capture churn)."""
    model = build_sklearn_app()
    model.train()
    path = tmp_path / "m.joblib"
    model.save(path)

    app = FastAPI()
    model.serve(app)
    monkeypatch.setenv("UNIONML_MODEL_PATH", str(path))
    monkeypatch.setenv("UNIONML_RELOAD_TOKEN", "s3cret")
    with TestClient(app) as client:
        assert client.post("/reload").status_code == 403
        assert client.post("/reload", headers={"X-Reload-Token": "wrong"}).status_code == 403
        assert client.post("/reload", headers={"X-Reload-Token": "s3cret"}).status_code == 200


@pytest.mark.timeout(240)
def test_serve_multiworker_supervisor(tmp_path):
    """`unionml-amd serve --workers 2`: the SO_REUSEPORT supervisor
    boots N single-worker processes sharing the port; requests succeed
    (kernel load-balances across workers), and SIGTERM tears the whole
    tree down cleanly."""
    import os
    import signal

    import httpx

    import unionml_amd

    env = dict(
        os.environ,
        PYTHONPATH=str(Path(unionml_amd.__file__).parent.parent)
        + os.pathsep
        + os.environ.get("PYTHONPATH", ""),
    )
    (tmp_path / "serve_app.py").write_text(SERVE_APP)
    proc = subprocess.run(
        [sys.executable, "serve_app.py"], cwd=tmp_path, capture_output=True, text=True, env=env
    )
    assert proc.returncode == 0, proc.stderr

    port = _free_port()
    server = subprocess.Popen(
        [sys.executable, "-m", "unionml_amd.cli", "serve", "serve_app:app",
         "--model-path", "model.joblib", "--port", str(port), "--workers", "2"],
        cwd=tmp_path, stdout=subprocess.PIPE, stderr=subprocess.STDOUT, env=env,
    )
    try:
        deadline = time.monotonic() + 90
        streak = 0
        while time.monotonic() < deadline and streak < 8:  # hit both workers
            if server.poll() is not None:
                raise AssertionError(
                    f"supervisor exited early:\n{server.stdout.read().decode()[-3000:]}"
                )
            try:
                if httpx.get(f"http://127.0.0.1:{port}/health", timeout=1.0).status_code == 200:
                    streak += 1
                    continue
            except httpx.HTTPError:
                pass
            streak = 0
            time.sleep(0.5)
        assert streak >= 8, "workers never became (all) healthy"

        for _ in range(10):  # round-robins across both workers
            resp = httpx.post(
                f"http://127.0.0.1:{port}/predict",
                json={"features": [{"x1": 0.9, "x2": 0.9, "x3": 0.9}]},
                timeout=10.0,
            )
            assert resp.status_code == 200, resp.text

        server.send_signal(signal.SIGTERM)
        server.wait(timeout=20)   # supervisor reaps its workers
        assert server.returncode is not None
    finally:
        if server.poll() is None:
            server.kill()
