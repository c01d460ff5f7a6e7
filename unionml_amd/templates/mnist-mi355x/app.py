"""{{app_name}} — MNIST-shape (784x128x10) classifier on the GENERALIZED
CDNA4 tabular kernels.

Shows the any-geometry hot path: pick in_features/hidden/classes for
YOUR tabular problem (input width arbitrary, hidden <= 256, classes
<= 32; odd sizes zero-pad exactly) and the trainer runs the templated
gfx950 MFMA kernel family (unionml_amd/ops/hip/tabular_gen.hip). Falls
back to the fp32 torch reference on CPU. Walk-through:
docs/tutorials/mnist.md.

    unionml-amd train app:model --local --output model.pt
    unionml-amd serve app:fastapi_app --model-path model.pt --port 8000
"""

from typing import List, Optional

import numpy as np
import pandas as pd
import torch
from fastapi import FastAPI

from unionml_amd import Dataset, Model
from unionml_amd.ops.tabular import TabularMLP
from unionml_amd.parallel import get_world_size
from unionml_amd.serving.graph_runner import TabularGraphRunner, graphed
from unionml_amd.utils.staging import get_stager

N_PIXELS, HIDDEN, CLASSES = 784, 128, 10
FEATURES = [f"pixel{i + 1}" for i in range(N_PIXELS)]

dataset = Dataset(name="{{app_name}}_dataset", features=FEATURES, targets=["class"], test_size=0.2)


@dataset.reader(cache=True, cache_version="1")
def reader(n: int = 2000, path: Optional[str] = None, seed: int = 42) -> pd.DataFrame:
    """MNIST-shaped frame: a .npz with X [N,784]/y [N] when ``path`` is
    given, else synthetic per-class clusters."""
    if path is not None:
        blob = np.load(path)
        X, y = blob["X"].astype(np.float32), blob["y"].astype(np.int64)
        if n:
            X, y = X[:n], y[:n]
    else:
        rng = np.random.RandomState(seed)
        centers = rng.randn(CLASSES, N_PIXELS).astype(np.float32) * 2.0
        y = rng.randint(0, CLASSES, size=n)
        X = centers[y] + rng.randn(n, N_PIXELS).astype(np.float32) * 0.8
    frame = pd.DataFrame(X, columns=FEATURES)
    frame["class"] = y
    return frame


@dataset.feature_loader
def feature_loader(data) -> pd.DataFrame:
    """Accept a raw [N,784] (or flat) ndarray / nested list, or the
    default records/JSON forms."""
    if isinstance(data, (list, tuple)) and data and not isinstance(data[0], dict):
        data = np.asarray(data, dtype=np.float32)
    if isinstance(data, np.ndarray):
        return pd.DataFrame(data.reshape(-1, N_PIXELS).astype(np.float32), columns=FEATURES)
    return dataset._default_feature_loader(data)


model = Model(
    name="{{app_name}}",
    init=lambda hyperparameters=None: TabularMLP(
        in_features=N_PIXELS, hidden=HIDDEN, classes=CLASSES, **(hyperparameters or {})
    ),
    dataset=dataset,
)


@model.trainer(cache=True, cache_version="1")
def trainer(
    clf: TabularMLP,
    features: pd.DataFrame,
    target: pd.DataFrame,
    *,
    epochs: int = 20,
    batch_size: int = 512,
    lr: float = 3e-3,
) -> TabularMLP:
    stager = get_stager(clf.device)
    X = stager.to_device(features.to_numpy().astype(np.float32))
    y = stager.to_device(target.squeeze().to_numpy().astype(np.int32))
    clf.fit_standardizer(X)
    clf.train_epochs(
        clf.stage(X), y, epochs=epochs, batch_size=batch_size, lr=lr,
        world_size=get_world_size(),
    )
    return clf


@model.predictor
@graphed(lambda mlp, max_batch: TabularGraphRunner(mlp, max_batch))
def predictor(clf: TabularMLP, features: pd.DataFrame) -> List[int]:
    X = torch.from_numpy(np.ascontiguousarray(features.to_numpy(), dtype=np.float32))
    return [int(i) for i in clf.predict(X).cpu()]


@model.evaluator
def evaluator(clf: TabularMLP, features: pd.DataFrame, target: pd.DataFrame) -> float:
    preds = predictor(clf, features)
    return float((np.asarray(preds) == target.squeeze().to_numpy()).mean())


@model.saver
def saver(clf: TabularMLP, hyperparameters, file, **kwargs):
    torch.save({"state": clf.state_dict(), "hyperparameters": hyperparameters}, file)
    return file


@model.loader
def loader(file, **kwargs) -> TabularMLP:
    payload = torch.load(file, map_location="cpu", weights_only=False)
    clf = TabularMLP(in_features=N_PIXELS, hidden=HIDDEN, classes=CLASSES)
    clf.load_state_dict(payload["state"])
    return clf


fastapi_app = FastAPI()
model.serve(fastapi_app)
