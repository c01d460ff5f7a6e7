"""Backend scheduler end-to-end: deploy an app with schedules, drive the
scheduler loop with a fake clock, and verify executions fire and are
listed (the reference delegates this to Flyte LaunchPlans; here the
in-process scheduler of unionml_amd/remote.py owns it)."""

import datetime
import sys
import time
from pathlib import Path

import pytest

APP_SOURCE = '''
from datetime import timedelta
from typing import List

import pandas as pd
from sklearn.linear_model import LogisticRegression

from unionml_amd import Dataset, Model

dataset = Dataset(name="sched_ds", targets=["y"], test_size=0.2, shuffle=True, random_state=0)
model = Model(name="sched_model", init=LogisticRegression, dataset=dataset)
model.remote(backend_path=r"{backend_path}")


@dataset.reader
def reader(n: int = 40) -> pd.DataFrame:
    import numpy as np

    rng = np.random.RandomState(5)
    X = rng.rand(n, 2)
    return pd.DataFrame({{"a": X[:, 0], "b": X[:, 1], "y": (X.sum(axis=1) > 1.0).astype(int)}})


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


model.schedule_training("every2min", fixed_rate=timedelta(minutes=2), inputs={{"n": 30}})
model.schedule_prediction("hourly_pred", expression="0 * * * *", inputs={{"n": 10}})
'''


@pytest.fixture
def sched_app(tmp_path, monkeypatch):
    backend = tmp_path / "backend"
    (tmp_path / "sched_app.py").write_text(APP_SOURCE.format(backend_path=backend))
    monkeypatch.chdir(tmp_path)
    monkeypatch.syspath_prepend(str(tmp_path))
    yield tmp_path
    sys.modules.pop("sched_app", None)


@pytest.mark.timeout(180)
def test_scheduler_fires_training_and_prediction(sched_app):
    import sched_app as appmod

    model = appmod.model
    backend = model._backend()
    backend.deploy(model, allow_uncommitted=True)

    # seed a model artifact for the prediction schedule to resolve
    execution = backend.execute(
        model,
        workflow="train",
        inputs=dict(
            hyperparameters=None, loader_kwargs=None, splitter_kwargs=None,
            parser_kwargs=None, trainer_kwargs=None, reader_kwargs={"n": 40},
        ),
    )
    backend.wait(execution, timeout=120)

    # fake clock: tick in 1-minute steps across an hour boundary so both
    # the 2-min fixed rate and the hourly cron fire
    t0 = datetime.datetime(2026, 9, 13, 11, 58, 30)
    ticks = [t0 + datetime.timedelta(minutes=i) for i in range(6)]
    it = iter(ticks)
    backend.run_scheduler(model, iterations=len(ticks), poll_s=0.0, now_fn=lambda: next(it))

    # give the fired worker subprocesses time to finish
    deadline = time.monotonic() + 120
    while time.monotonic() < deadline:
        train_runs = backend.list_scheduled_runs(model, "every2min", kind="training")
        pred_runs = backend.list_scheduled_runs(model, "hourly_pred", kind="prediction")
        if train_runs and pred_runs:
            statuses = {
                backend.get_execution(r).status for r in train_runs + pred_runs
            }
            if statuses <= {"SUCCEEDED", "FAILED"}:
                break
        time.sleep(0.5)

    assert train_runs, "fixed-rate training schedule never fired"
    assert pred_runs, "cron prediction schedule never fired"
    for run in train_runs + pred_runs:
        ex = backend.get_execution(run)
        assert ex.status == "SUCCEEDED", (
            run,
            ex.status,
            (Path(ex.path) / "worker.log").read_text()[-2000:],
        )


@pytest.mark.timeout(60)
def test_scheduler_routes_workflow_level_inputs(sched_app, monkeypatch):
    """Schedule inputs may mix workflow-level stage kwargs
    (hyperparameters, trainer_kwargs, ...) with reader kwargs; the
    scheduler must route each to its own workflow input instead of
    shoving everything into reader_kwargs."""
    import sched_app as appmod

    model = appmod.model
    backend = model._backend()
    model.schedule_training(
        "tuned",
        fixed_rate=datetime.timedelta(minutes=1),
        inputs={"hyperparameters": {"C": 0.5}, "trainer_kwargs": None, "n": 25},
    )
    backend.deploy(model, allow_uncommitted=True)

    fired = []

    def record(model, workflow, app_version=None, inputs=None, schedule_name=None, **kw):
        fired.append((workflow, schedule_name, inputs))

    monkeypatch.setattr(backend, "execute", record)
    # minute ticks away from an hour boundary: only the fixed-rate
    # schedules fire (the mocked execute would break the prediction
    # schedule's artifact lookup)
    t0 = datetime.datetime(2026, 9, 13, 11, 10, 30)
    ticks = [t0 + datetime.timedelta(minutes=i) for i in range(4)]
    it = iter(ticks)
    backend.run_scheduler(model, iterations=len(ticks), poll_s=0.0, now_fn=lambda: next(it))

    tuned = [f for f in fired if f[1] == "tuned"]
    assert tuned, f"tuned schedule never fired: {fired}"
    workflow, _, inputs = tuned[0]
    assert workflow == "train"
    assert inputs["hyperparameters"] == {"C": 0.5}
    assert inputs["trainer_kwargs"] is None
    assert inputs["reader_kwargs"] == {"n": 25}

    plain = [f for f in fired if f[1] == "every2min"]
    assert plain and plain[0][2]["reader_kwargs"] == {"n": 30}
    assert plain[0][2]["hyperparameters"] is None
