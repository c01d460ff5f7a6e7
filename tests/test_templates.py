"""Template integration: every scaffolded project must import, and the
basic app's __main__ path must actually train/predict/save
(the reference's cookiecutter templates are exercised the same way by
its docs CI)."""

import os
import subprocess
import sys
from pathlib import Path

import pytest
from typer.testing import CliRunner

import unionml_amd
from unionml_amd.cli import app

runner = CliRunner()

REPO = Path(unionml_amd.__file__).parent.parent


def _env():
    return dict(
        os.environ,
        PYTHONPATH=str(REPO) + os.pathsep + os.environ.get("PYTHONPATH", ""),
    )


@pytest.mark.timeout(300)
def test_basic_template_main_trains_and_saves(tmp_path, monkeypatch):
    monkeypatch.chdir(tmp_path)
    result = runner.invoke(app, ["init", "basicapp", "--template", "basic"])
    assert result.exit_code == 0, result.output
    proc = subprocess.run(
        [sys.executable, "app.py"],
        cwd=tmp_path / "basicapp",
        capture_output=True,
        text=True,
        timeout=240,
        env=_env(),
    )
    assert proc.returncode == 0, proc.stderr[-3000:]
    assert (tmp_path / "basicapp" / "model.joblib").exists()


@pytest.mark.timeout(300)
@pytest.mark.parametrize(
    "template,module_ok",
    [
        ("pytorch-mi355x", True),
        ("mnist-mi355x", True),
        ("basic-aws-lambda", True),
        ("basic-aws-lambda-s3", True),
        ("resnet-dp", True),
    ],
)
def test_templates_import_clean(tmp_path, monkeypatch, template, module_ok):
    monkeypatch.chdir(tmp_path)
    name = template.replace("-", "_")
    result = runner.invoke(app, ["init", name, "--template", template])
    assert result.exit_code == 0, result.output
    proc = subprocess.run(
        [sys.executable, "-c", "import app"],
        cwd=tmp_path / name,
        capture_output=True,
        text=True,
        timeout=240,
        env=_env(),
    )
    assert (proc.returncode == 0) == module_ok, proc.stderr[-3000:]


@pytest.mark.timeout(300)
def test_mnist_template_trains(tmp_path, monkeypatch):
    """The generalized-geometry template trains end to end on CPU."""
    monkeypatch.chdir(tmp_path)
    result = runner.invoke(app, ["init", "mnist_app", "--template", "mnist-mi355x"])
    assert result.exit_code == 0, result.output
    proc = subprocess.run(
        [sys.executable, "-c",
         "from app import model; obj, m = model.train(trainer_kwargs={'epochs': 3}, n=600); "
         "assert m['test'] > 0.8, m; assert obj.g.in_features == 784"],
        cwd=tmp_path / "mnist_app", capture_output=True, text=True, timeout=240,
        env=_env(),
    )
    assert proc.returncode == 0, proc.stderr[-3000:]
