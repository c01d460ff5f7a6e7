"""The importable serving entry modules (`unionml_amd.models.mlp_serve`
and `.mnist_serve`) are what `bench.py --mode serve` and the multi-worker
supervisor point uvicorn at. Exercise them on CPU: a fresh import with
$UNIONML_MODEL_PATH set must come up serving the artifact."""

import importlib
import sys

import pytest
from fastapi.testclient import TestClient


def _fresh(module_name, monkeypatch, artifact_path):
    monkeypatch.setenv("UNIONML_MODEL_PATH", str(artifact_path))
    # drop the app modules so model.serve() re-runs against the new env
    for m in (module_name, module_name.rsplit("_", 1)[0]):
        sys.modules.pop(m, None)
    return importlib.import_module(module_name)


def test_mlp_serve_module_serves_artifact(tmp_path, monkeypatch):
    from unionml_amd.models import mlp

    importlib.reload(mlp)
    mlp.model.train(synthetic=True, n=256, trainer_kwargs={"epochs": 2})
    path = tmp_path / "digits.pt"
    mlp.model.save(path)

    mod = _fresh("unionml_amd.models.mlp_serve", monkeypatch, path)
    with TestClient(mod.app) as client:
        assert client.get("/health").status_code == 200
        feats = [{f"p{i}": 0.5 for i in range(64)}]
        r = client.post("/predict", json={"features": feats})
        assert r.status_code == 200, r.text
        assert len(r.json()) == 1


def test_mnist_serve_module_serves_artifact(tmp_path, monkeypatch):
    from unionml_amd.models import mnist

    importlib.reload(mnist)
    mnist.model.train(n=128, trainer_kwargs={"epochs": 1})
    path = tmp_path / "mnist.pt"
    mnist.model.save(path)

    mod = _fresh("unionml_amd.models.mnist_serve", monkeypatch, path)
    with TestClient(mod.app) as client:
        assert client.get("/health").status_code == 200
