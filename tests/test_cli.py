"""CLI tests — command set parity with the reference CLI (unionml/cli.py),
exercised against a real app module and the local process-cluster backend
(test design: reference tests drive a real sandbox; here the backend IS
local, so the full deploy→train→predict→fetch loop runs in-tree)."""

import json
import os
import subprocess
import sys
from pathlib import Path

import pytest
from typer.testing import CliRunner

from unionml_amd.cli import app

runner = CliRunner()

APP_SOURCE = '''
from typing import List

import pandas as pd
from sklearn.linear_model import LogisticRegression

from unionml_amd import Dataset, Model

dataset = Dataset(name="cli_ds", targets=["y"], test_size=0.2, shuffle=True, random_state=0)
model = Model(name="cli_model", init=LogisticRegression, dataset=dataset)
model.remote(backend_path=r"{backend_path}")


@dataset.reader
def reader(n: int = 60) -> pd.DataFrame:
    import numpy as np

    rng = np.random.RandomState(1)
    X = rng.rand(n, 3)
    return pd.DataFrame(
        {{"a": X[:, 0], "b": X[:, 1], "c": X[:, 2], "y": (X.sum(axis=1) > 1.5).astype(int)}}
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
'''


@pytest.fixture
def cli_app(tmp_path, monkeypatch):
    """Write a real app module into tmp_path and chdir there."""
    backend = tmp_path / "backend"
    (tmp_path / "cli_app.py").write_text(APP_SOURCE.format(backend_path=backend))
    monkeypatch.chdir(tmp_path)
    monkeypatch.syspath_prepend(str(tmp_path))
    yield tmp_path
    sys.modules.pop("cli_app", None)


def test_init_scaffolds_project(tmp_path, monkeypatch):
    monkeypatch.chdir(tmp_path)
    result = runner.invoke(app, ["init", "myapp", "--template", "basic"])
    assert result.exit_code == 0, result.output
    assert (tmp_path / "myapp" / "app.py").exists()
    text = (tmp_path / "myapp" / "app.py").read_text()
    assert "myapp" in text and "{{app_name}}" not in text
    # generated project is git-initialised (reference post-gen hook)
    assert (tmp_path / "myapp" / ".git").is_dir()


def test_init_unknown_template(tmp_path, monkeypatch):
    monkeypatch.chdir(tmp_path)
    result = runner.invoke(app, ["init", "x", "--template", "nope"])
    assert result.exit_code == 1
    assert "available" in result.output


def test_init_all_templates_render(tmp_path, monkeypatch):
    monkeypatch.chdir(tmp_path)
    from unionml_amd.cli import _available_templates

    templates = _available_templates()
    assert {"basic", "pytorch-mi355x", "basic-aws-lambda", "basic-bentoml"} <= set(templates)
    for t in templates:
        result = runner.invoke(app, ["init", f"app_{t.replace('-', '_')}", "--template", t])
        assert result.exit_code == 0, result.output


def test_train_local_and_save(cli_app):
    result = runner.invoke(
        app, ["train", "cli_app:model", "--local", "--output", "model.joblib"]
    )
    assert result.exit_code == 0, result.output
    assert "metrics" in result.output
    assert (cli_app / "model.joblib").exists()


def test_predict_local_with_features_file(cli_app):
    runner.invoke(app, ["train", "cli_app:model", "--local", "--output", "m.joblib"])
    feats = [{"a": 0.9, "b": 0.9, "c": 0.9}, {"a": 0.0, "b": 0.0, "c": 0.1}]
    (cli_app / "feats.json").write_text(json.dumps(feats))
    result = runner.invoke(
        app, ["predict", "cli_app:model", "--local", "--features", "feats.json"]
    )
    assert result.exit_code == 0, result.output
    preds = json.loads(result.output.strip().splitlines()[-1])
    assert len(preds) == 2


def test_deploy_train_predict_fetch_on_backend(cli_app):
    result = runner.invoke(app, ["deploy", "cli_app:model", "--allow-uncommitted"])
    assert result.exit_code == 0, result.output
    assert "deployed" in result.output

    result = runner.invoke(app, ["train", "cli_app:model"])
    assert result.exit_code == 0, result.output
    assert "SUCCEEDED" in result.output

    result = runner.invoke(app, ["list-model-versions", "cli_app:model"])
    assert result.exit_code == 0, result.output
    version = result.output.strip().splitlines()[0]
    assert version.startswith("train-")

    feats = [{"a": 0.9, "b": 0.9, "c": 0.9}]
    (cli_app / "feats.json").write_text(json.dumps(feats))
    result = runner.invoke(app, ["predict", "cli_app:model", "--features", "feats.json"])
    assert result.exit_code == 0, result.output
    preds = json.loads(result.output.strip().splitlines()[-1])
    assert len(preds) == 1

    result = runner.invoke(app, ["fetch-model", "cli_app:model", "fetched.joblib"])
    assert result.exit_code == 0, result.output
    assert (cli_app / "fetched.joblib").exists()

    result = runner.invoke(app, ["list-prediction-ids", "cli_app:model"])
    assert result.exit_code == 0, result.output
    pred_id = result.output.strip().splitlines()[0]
    result = runner.invoke(
        app, ["fetch-predictions", "cli_app:model", pred_id, "--output", "preds.json"]
    )
    assert result.exit_code == 0, result.output
    assert json.loads((cli_app / "preds.json").read_text()) == preds


def test_bad_model_spec(cli_app):
    result = runner.invoke(app, ["train", "nosuchmodule:model", "--local"])
    assert result.exit_code != 0


def test_serve_refuses_preset_env(cli_app, monkeypatch):
    monkeypatch.setenv("UNIONML_MODEL_PATH", "/tmp/x")
    result = runner.invoke(
        app, ["serve", "cli_app:fastapi_app", "--model-path", "m.joblib"]
    )
    assert result.exit_code == 1
    assert "already set" in result.output


def test_console_script_entrypoint():
    """pyproject [project.scripts] → unionml_amd.cli:main resolves."""
    from unionml_amd.cli import main  # noqa: F401

    proc = subprocess.run(
        [sys.executable, "-m", "unionml_amd.cli", "--help"],
        capture_output=True,
        text=True,
    )
    assert proc.returncode == 0
    assert "init" in proc.stdout and "serve" in proc.stdout


def test_bentoml_service_gated_import():
    """BentoMLService raises a helpful ImportError when bentoml is absent
    (reference gates the import: services/__init__.py:4-6)."""
    from unionml_amd.services.bentoml import BentoMLService, infer_io_descriptor
    from unionml_amd.utils.env import module_is_installed

    import numpy as np
    import pandas as pd

    assert infer_io_descriptor(np.ndarray) == "NumpyNdarray"
    assert infer_io_descriptor(pd.DataFrame) == "PandasDataFrame"
    assert infer_io_descriptor(list) == "JSON"

    if not module_is_installed("bentoml"):
        svc = BentoMLService.__new__(BentoMLService)
        with pytest.raises(RuntimeError):
            _ = BentoMLService(model=_DummyModel()).svc
        with pytest.raises(ImportError, match="bentoml"):
            BentoMLService(model=_DummyModel()).configure()


class _DummyModel:
    name = "dummy"
    artifact = None


APP_SCHED = APP_SOURCE.replace(
    'model.remote(backend_path=r"{backend_path}")',
    'model.remote(backend_path=r"{backend_path}")\n'
    "from datetime import timedelta\n"
    'model.schedule_training("retrain", fixed_rate=timedelta(minutes=5), inputs={{"n": 30}})\n'
    'model.schedule_prediction("batch_pred", expression="0 * * * *", inputs={{"n": 10}},'
    " activate_on_deploy=False)",
)


@pytest.fixture
def sched_cli_app(tmp_path, monkeypatch):
    backend = tmp_path / "backend"
    (tmp_path / "sched_cli_app.py").write_text(APP_SCHED.format(backend_path=backend))
    monkeypatch.chdir(tmp_path)
    monkeypatch.syspath_prepend(str(tmp_path))
    yield tmp_path
    sys.modules.pop("sched_cli_app", None)


def test_schedule_lifecycle_via_cli(sched_cli_app):
    """deploy -> activate/deactivate-schedules -> scheduled-run listings
    (the reference CLI's schedule surface, with its deactivate/list bugs
    fixed — SURVEY.md §8 quirks)."""
    import json as _json

    r = runner.invoke(app, ["deploy", "sched_cli_app:model", "--allow-uncommitted"])
    assert r.exit_code == 0, r.output

    import sched_cli_app as appmod

    backend = appmod.model._backend()
    manifest = backend._manifest(None)
    states = {lp["name"]: lp["active"] for lp in manifest["launchplans"]}
    assert states == {"retrain": True, "batch_pred": False}

    r = runner.invoke(app, ["activate-schedules", "sched_cli_app:model", "batch_pred"])
    assert r.exit_code == 0, r.output
    r = runner.invoke(app, ["deactivate-schedules", "sched_cli_app:model", "retrain"])
    assert r.exit_code == 0, r.output
    manifest = backend._manifest(None)
    states = {lp["name"]: lp["active"] for lp in manifest["launchplans"]}
    assert states == {"retrain": False, "batch_pred": True}

    # deactivate all
    r = runner.invoke(app, ["deactivate-schedules", "sched_cli_app:model"])
    assert r.exit_code == 0, r.output
    manifest = backend._manifest(None)
    assert all(not lp["active"] for lp in manifest["launchplans"])

    # run listings stay empty but exit cleanly
    for cmd in ("list-scheduled-training-runs", "list-scheduled-prediction-runs"):
        name = "retrain" if "training" in cmd else "batch_pred"
        r = runner.invoke(app, [cmd, "sched_cli_app:model", name])
        assert r.exit_code == 0, r.output


def test_run_scheduler_cli_bounded(sched_cli_app):
    r = runner.invoke(app, ["deploy", "sched_cli_app:model", "--allow-uncommitted"])
    assert r.exit_code == 0, r.output
    # bounded loop: primes next-fire times and exits cleanly without firing
    r = runner.invoke(
        app, ["run-scheduler", "sched_cli_app:model", "-n", "2", "--poll-s", "0.01"]
    )
    assert r.exit_code == 0, r.output
    assert "scheduler running" in r.output
