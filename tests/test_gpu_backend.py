"""Backend-on-GPU test (BASELINE config 5 mechanics): a deployed app
whose tasks request MI355X devices runs in worker subprocesses with
HIP_VISIBLE_DEVICES assigned by the backend's allocator, and the
scheduled batch-predict path produces outputs."""

import json
import sys
from pathlib import Path

import pytest
import torch

pytestmark = pytest.mark.gpu

APP = '''
import os
from typing import List

import numpy as np
import pandas as pd
import torch

from unionml_amd import Dataset, Model
from unionml_amd.defaults import Resources
from unionml_amd.ops.tabular import TabularMLP

FEATURES = [f"p{i}" for i in range(64)]
dataset = Dataset(name="gpu_backend_ds", features=FEATURES, targets=["target"], test_size=0.2)
model = Model(name="gpu_backend_model", init=TabularMLP, dataset=dataset)
model.remote(backend_path=r"{backend_path}")


@dataset.reader
def reader(n: int = 512, seed: int = 0) -> pd.DataFrame:
    rng = np.random.RandomState(seed)
    frame = pd.DataFrame(rng.rand(n, 64).astype(np.float32) * 16.0, columns=FEATURES)
    frame["target"] = rng.randint(0, 10, size=n)
    return frame


@model.trainer(resources=Resources(cpu="1", mem="1Gi", gpu=1))
def trainer(clf: TabularMLP, features: pd.DataFrame, target: pd.DataFrame,
            *, epochs: int = 3, lr: float = 2e-3) -> TabularMLP:
    assert torch.cuda.is_available(), "worker must see a GPU"
    X = torch.tensor(features.to_numpy(), dtype=torch.float32, device=clf.device)
    y = torch.tensor(target.squeeze().to_numpy(), dtype=torch.int32, device=clf.device)
    clf.fit_standardizer(X)
    clf.train_epochs(clf.stage(X), y, epochs=epochs, lr=lr, use_graph=False)
    clf.visible_devices = os.environ.get("HIP_VISIBLE_DEVICES")
    return clf


@model.predictor
def predictor(clf: TabularMLP, features: pd.DataFrame) -> List[int]:
    X = torch.tensor(np.ascontiguousarray(features.to_numpy(), dtype=np.float32))
    return [int(i) for i in clf.predict(X).cpu()]


@model.evaluator
def evaluator(clf: TabularMLP, features: pd.DataFrame, target: pd.DataFrame) -> float:
    preds = predictor(clf, features)
    return float((np.asarray(preds) == target.squeeze().to_numpy()).mean())
'''


@pytest.mark.timeout(300)
def test_backend_gpu_execution_and_fanout(tmp_path, monkeypatch):
    if not torch.cuda.is_available():
        pytest.skip("needs MI355X")
    backend_path = tmp_path / "backend"
    (tmp_path / "gpu_backend_app.py").write_text(
        APP.replace("{backend_path}", str(backend_path))
    )
    monkeypatch.chdir(tmp_path)
    monkeypatch.syspath_prepend(str(tmp_path))
    try:
        import gpu_backend_app as appmod

        model = appmod.model
        backend = model._backend()
        backend.deploy(model, allow_uncommitted=True)
        assert backend.gpus.n_devices >= 1

        execution = model.remote_train(wait=True, trainer_kwargs={"epochs": 3})
        clf = model.artifact.model_object
        # the worker ran on device with an allocator-assigned GPU set
        assert getattr(clf, "visible_devices", None) is not None

        # scheduled-style batch predict fan-out: several predict
        # executions through the backend, each a separate worker process
        feats = appmod.reader(n=64).drop(columns=["target"])
        features = model._dataset.get_features(feats)
        executions = [
            backend.execute(
                model,
                workflow="predict_from_features",
                inputs=dict(model_object=clf, features=features),
            )
            for _ in range(3)
        ]
        for ex in executions:
            backend.wait(ex, timeout=180)
            preds = backend.fetch_output(ex)
            assert len(preds) == 64
        assert len(model.remote_list_prediction_ids()) >= 3
    finally:
        sys.modules.pop("gpu_backend_app", None)


@pytest.mark.gpu
def test_dp_oversubscription_raises_clearly():
    """dp > visible GPUs must fail fast with an actionable error, not
    die inside RCCL communicator init (only meaningful on boxes with
    fewer GPUs than the requested dp)."""
    import torch

    if not torch.cuda.is_available():
        pytest.skip("needs a GPU")
    n = torch.cuda.device_count()
    from unionml_amd.models.mlp import model

    with pytest.raises(ValueError, match="exceeds"):
        model.train(dp=n + 1, synthetic=True, n=256, trainer_kwargs={"epochs": 1})
