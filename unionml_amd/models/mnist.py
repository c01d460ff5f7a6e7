"""MNIST-shape app: 784-feature / 10-class classifier on the
GENERALIZED CDNA4 tabular hot path.

Mirrors the reference's MNIST tutorial app structure
(/root/reference/docs/source/tutorials/mnist.md:106-202 — reader with
caching kwargs, trainer/predictor/evaluator decorators, an
ndarray->DataFrame feature_loader) with the MI355X-native difference
that the trainer runs on :class:`unionml_amd.ops.tabular.TabularMLP`'s
generalized MFMA kernels (784 padded to 800, hidden 128) instead of an
sklearn Pipeline. There is no network in this environment, so the
reader draws MNIST-shaped synthetic digits (clustered per class) rather
than fetching openml's mnist_784; pass a .npz path to train on real
data.

See docs/tutorials/mnist.md for the worked walk-through.
"""

from typing import List, Optional

import numpy as np
import pandas as pd
import torch

from unionml_amd import Dataset, Model
from unionml_amd.ops.tabular import TabularMLP
from unionml_amd.parallel import get_world_size
from unionml_amd.serving.graph_runner import TabularGraphRunner, graphed
from unionml_amd.utils.staging import get_stager

N_PIXELS = 784
FEATURES = [f"pixel{i + 1}" for i in range(N_PIXELS)]

dataset = Dataset(name="mnist", features=FEATURES, targets=["class"], test_size=0.2)


@dataset.reader(cache=True, cache_version="1")
def reader(n: int = 2000, path: Optional[str] = None, seed: int = 42) -> pd.DataFrame:
    """MNIST-shaped frame: a .npz with arrays X [N,784]/y [N] when
    ``path`` is given, else synthetic per-class clusters (no network for
    openml here — reference tutorial fetches mnist_784)."""
    if path is not None:
        blob = np.load(path)
        X, y = blob["X"].astype(np.float32), blob["y"].astype(np.int64)
        if n:
            X, y = X[:n], y[:n]
    else:
        rng = np.random.RandomState(seed)
        centers = rng.randn(10, N_PIXELS).astype(np.float32) * 2.0
        y = rng.randint(0, 10, size=n)
        X = centers[y] + rng.randn(n, N_PIXELS).astype(np.float32) * 0.8
        X = np.clip((X - X.min()) * 16.0, 0, 255.0)
    frame = pd.DataFrame(X, columns=FEATURES)
    frame["class"] = y
    return frame


@dataset.feature_loader
def feature_loader(data) -> pd.DataFrame:
    """Accept a raw [N,784] (or flat [784]) ndarray / nested list — the
    reference tutorial's gradio-style loader (mnist.md:196-202) — or the
    default records/JSON forms."""
    if isinstance(data, (list, tuple)) and data and not isinstance(data[0], dict):
        data = np.asarray(data, dtype=np.float32)
    if isinstance(data, np.ndarray):
        arr = data.reshape(-1, N_PIXELS).astype(np.float32)
        return pd.DataFrame(arr, columns=FEATURES)
    return dataset._default_feature_loader(data)


model = Model(name="mnist_mlp", init=lambda hyperparameters=None: TabularMLP(
    in_features=N_PIXELS, hidden=128, classes=10, **(hyperparameters or {})
), dataset=dataset)


@model.trainer(cache=True, cache_version="1")
def trainer(
    clf: TabularMLP,
    features: pd.DataFrame,
    target: pd.DataFrame,
    *,
    epochs: int = 20,
    batch_size: int = 512,
    lr: float = 3e-3,
    use_graph: bool = True,
) -> TabularMLP:
    """Stage features into HBM (pinned async H2D), fit the standardizer,
    then run the generalized fused-kernel minibatch loop (hipGraph-
    captured epochs). Under DP each rank gets a row shard and gradients
    all-reduce on RCCL inside the step."""
    stager = get_stager(clf.device)
    X = stager.to_device(features.to_numpy().astype(np.float32))
    y = stager.to_device(target.squeeze().to_numpy().astype(np.int32))
    clf.fit_standardizer(X)
    Xbf = clf.stage(X)
    clf.train_epochs(
        Xbf,
        y,
        epochs=epochs,
        batch_size=batch_size,
        lr=lr,
        use_graph=use_graph,
        world_size=get_world_size(),
    )
    return clf


@model.predictor
@graphed(lambda mlp, max_batch: TabularGraphRunner(mlp, max_batch))
def predictor(clf: TabularMLP, features: pd.DataFrame) -> List[int]:
    X = torch.from_numpy(np.ascontiguousarray(features.to_numpy(), dtype=np.float32))
    preds = clf.predict(X)
    return [int(i) for i in preds.cpu()]


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
    geo = payload["state"].get("geometry")
    inf, hid, cls = (int(x) for x in geo) if geo is not None else (N_PIXELS, 128, 10)
    clf = TabularMLP(in_features=inf, hidden=hid, classes=cls)
    clf.load_state_dict(payload["state"])
    return clf
