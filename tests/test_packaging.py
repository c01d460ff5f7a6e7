"""Packaging path tests: fast-registration code snapshots, patch
(code-only) re-registration, worker isolation from live edits, and
docker failure surfacing (reference semantics: unionml/remote.py
:104-122 docker build, :138-152 fast registration)."""

import sys

import pytest

APP_TEMPLATE = '''
from typing import List

import pandas as pd
from sklearn.linear_model import LogisticRegression

from unionml_amd import Dataset, Model

MARKER = "{marker}"

dataset = Dataset(name="pkg_ds", targets=["y"], test_size=0.25, random_state=0)
model = Model(name="pkg_model", init=LogisticRegression, dataset=dataset)
model.remote(backend_path=r"{backend_path}")


@dataset.reader
def reader(n: int = 60) -> pd.DataFrame:
    import numpy as np

    rng = np.random.RandomState(7)
    X = rng.rand(n, 2)
    return pd.DataFrame({{"a": X[:, 0], "b": X[:, 1], "y": (X.sum(axis=1) > 1.0).astype(int)}})


@model.trainer
def trainer(
    estimator: LogisticRegression, features: pd.DataFrame, target: pd.DataFrame
) -> LogisticRegression:
    estimator.fit(features, target.squeeze())
    estimator.marker_ = MARKER  # which CODE VERSION trained this model
    return estimator


@model.predictor
def predictor(estimator: LogisticRegression, features: pd.DataFrame) -> List[float]:
    return [float(x) for x in estimator.predict(features)]


@model.evaluator
def evaluator(estimator: LogisticRegression, features: pd.DataFrame, target: pd.DataFrame) -> float:
    return float((estimator.predict(features) == target.squeeze().to_numpy()).mean())
'''


@pytest.fixture
def pkg_app(tmp_path, monkeypatch):
    backend = tmp_path / "backend"

    def write(marker):
        (tmp_path / "pkg_app.py").write_text(
            APP_TEMPLATE.format(marker=marker, backend_path=backend)
        )

    write("v1")
    monkeypatch.chdir(tmp_path)
    monkeypatch.syspath_prepend(str(tmp_path))
    yield tmp_path, write
    sys.modules.pop("pkg_app", None)


@pytest.mark.timeout(240)
def test_deploy_snapshots_code_and_isolates_workers(pkg_app):
    """A deployed version keeps running its REGISTERED code even after
    the live file changes (the reference ships versioned source to the
    cluster; editing your checkout must not mutate a deployment)."""
    tmp_path, write = pkg_app
    import pkg_app as appmod

    backend = appmod.model._backend()
    version = backend.deploy(appmod.model, allow_uncommitted=True)

    snap = backend.root / "apps" / version / "code" / "pkg_app.py"
    assert snap.exists(), "deploy must snapshot the app source"
    assert 'MARKER = "v1"' in snap.read_text()

    # sabotage the live file: the worker must not see this
    write("live-edit-after-deploy")

    execution = backend.execute(appmod.model, "train", app_version=version)
    backend.wait(execution, timeout=120)
    artifact = backend.fetch_model_artifact(appmod.model, app_version=version)
    assert artifact.model_object.marker_ == "v1"


@pytest.mark.timeout(240)
def test_patch_redeploy_picks_up_code_without_image(pkg_app):
    """deploy -> edit app body -> deploy(patch=True) round-trips the
    NEW code with no image build (ref remote.py:138-152 semantics)."""
    tmp_path, write = pkg_app
    import pkg_app as appmod

    backend = appmod.model._backend()
    backend.deploy(appmod.model, allow_uncommitted=True, app_version="r1")

    write("v2")
    # patch: code-only re-registration under a new version
    backend.deploy(appmod.model, allow_uncommitted=True, patch=True, app_version="r2")
    snap = backend.root / "apps" / "r2" / "code" / "pkg_app.py"
    assert 'MARKER = "v2"' in snap.read_text()

    execution = backend.execute(appmod.model, "train", app_version="r2")
    backend.wait(execution, timeout=120)
    artifact = backend.fetch_model_artifact(appmod.model, app_version="r2")
    assert artifact.model_object.marker_ == "v2"

    # the original registration still runs v1 code
    execution = backend.execute(appmod.model, "train", app_version="r1")
    backend.wait(execution, timeout=120)
    arts = [
        e for e in backend._executions("train") if e.status == "SUCCEEDED"
    ]
    assert len(arts) >= 2


@pytest.mark.timeout(120)
def test_docker_failure_raises_when_registry_configured(pkg_app, monkeypatch):
    """An explicitly configured registry means a failed docker build is
    a deploy FAILURE, not a swallowed warning."""
    tmp_path, _ = pkg_app
    import pkg_app as appmod

    from unionml_amd.remote import Backend

    backend = Backend(
        project="pkg_model",
        backend_path=str(tmp_path / "backend2"),
        registry="localhost:1/unreachable",
    )
    # no Dockerfile in cwd + (likely) no docker daemon -> must raise
    with pytest.raises(RuntimeError):
        backend.deploy(appmod.model, allow_uncommitted=True)


@pytest.mark.timeout(120)
def test_framework_internal_apps_not_snapshotted(tmp_path):
    """unionml_amd.* apps are provided by the installed framework (the
    'image'); no code snapshot is taken for them."""
    import unionml_amd.models.mlp as mlp_mod
    from unionml_amd.remote import Backend

    backend = Backend(project="digits_mlp", backend_path=str(tmp_path / "b"))
    assert (
        backend._snapshot_code("unionml_amd.models.mlp", mlp_mod.__file__, tmp_path)
        is None
    )
    assert not (tmp_path / "code").exists()
