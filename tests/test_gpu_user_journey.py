"""End-to-end user journey on an MI355X: scaffold the pytorch-mi355x
template with the CLI, train through the fused-kernel hot path, save
the artifact, serve it over HTTP, and predict — exactly the README
quickstart, run for real on device."""

import json
import os
import socket
import subprocess
import sys
import time
from pathlib import Path

import pytest
import torch
from typer.testing import CliRunner

import unionml_amd
from unionml_amd.cli import app

pytestmark = pytest.mark.gpu

REPO = Path(unionml_amd.__file__).parent.parent


def _env():
    return dict(
        os.environ,
        PYTHONPATH=str(REPO) + os.pathsep + os.environ.get("PYTHONPATH", ""),
    )


def _free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


@pytest.mark.timeout(420)
def test_quickstart_journey_on_gpu(tmp_path, monkeypatch):
    if not torch.cuda.is_available():
        pytest.skip("needs MI355X")
    import httpx

    monkeypatch.chdir(tmp_path)
    runner = CliRunner()
    result = runner.invoke(app, ["init", "journey", "--template", "pytorch-mi355x"])
    assert result.exit_code == 0, result.output
    appdir = tmp_path / "journey"

    # train through the CLI (in a subprocess, like a user would)
    proc = subprocess.run(
        [
            sys.executable, "-m", "unionml_amd.cli", "train", "app:model",
            "--local", "--output", "model.pt",
            "--inputs", json.dumps({"trainer_kwargs": {"epochs": 25, "lr": 2e-3}}),
        ],
        cwd=appdir, capture_output=True, text=True, timeout=300, env=_env(),
    )
    assert proc.returncode == 0, proc.stderr[-3000:]
    assert "metrics" in proc.stdout
    assert (appdir / "model.pt").exists()

    # serve it and predict over HTTP
    port = _free_port()
    server = subprocess.Popen(
        [
            sys.executable, "-m", "unionml_amd.cli", "serve", "app:fastapi_app",
            "--model-path", "model.pt", "--port", str(port),
        ],
        cwd=appdir, stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL, env=_env(),
    )
    try:
        url = f"http://127.0.0.1:{port}"
        deadline = time.monotonic() + 120
        while time.monotonic() < deadline:
            if server.poll() is not None:
                raise AssertionError("server exited early")
            try:
                if httpx.get(f"{url}/health", timeout=1.0).status_code == 200:
                    break
            except httpx.HTTPError:
                time.sleep(0.5)
        else:
            raise AssertionError("server never became healthy")

        feats = [{f"p{i}": float((i * 3) % 16) for i in range(64)}]
        r = httpx.post(f"{url}/predict", json={"features": feats}, timeout=15.0)
        assert r.status_code == 200, r.text
        preds = r.json()
        assert len(preds) == 1 and 0 <= preds[0] <= 9

        # /metrics shows the request; /reload hot-swaps the artifact
        m = httpx.get(f"{url}/metrics", timeout=10.0)
        assert m.status_code == 200 and "unionml_predict_requests_total" in m.text
        r = httpx.post(f"{url}/reload", timeout=60.0)
        assert r.status_code == 200, r.text
        r = httpx.post(f"{url}/predict", json={"features": feats}, timeout=15.0)
        assert r.status_code == 200, r.text
    finally:
        server.terminate()
        try:
            server.wait(timeout=10)
        except subprocess.TimeoutExpired:
            server.kill()
