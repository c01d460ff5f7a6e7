"""Flagship app: digits MLP classifier on the CDNA4 tabular hot path.

The unionml_amd app for the 64-feature / 10-class digits shape
(BASELINE.md config 2): the same ``@dataset``/``@model`` decorator API
as any user app, with the trainer/predictor bodies running on
:class:`unionml_amd.ops.tabular.TabularMLP` — hand-written gfx950 MFMA
kernels on GPU, the torch reference on CPU. This module is importable
by the task resolver and the DP spawner (module-level ``dataset`` /
``model`` instances).
"""

from typing import List

import numpy as np
import pandas as pd
import torch

from unionml_amd import Dataset, Model
from unionml_amd.dataset import Dataset as _Dataset
from unionml_amd.ops.tabular import TabularMLP
from unionml_amd.parallel import get_world_size
from unionml_amd.serving.graph_runner import TabularGraphRunner, graphed
from unionml_amd.utils.staging import get_stager

FEATURES = [f"p{i}" for i in range(64)]

dataset = Dataset(name="digits", features=FEATURES, targets=["target"], test_size=0.2)


@dataset.reader
def reader(n: int = 0, synthetic: bool = False, seed: int = 17) -> pd.DataFrame:
    """sklearn's bundled digits set by default; ``synthetic=True`` draws
    n digits-shaped random rows (bench path — no network for datasets)."""
    if synthetic:
        rng = np.random.RandomState(seed)
        X = rng.rand(n or 1797, 64).astype(np.float32) * 16.0
        y = rng.randint(0, 10, size=len(X))
    else:
        from sklearn.datasets import load_digits

        digits = load_digits()
        X, y = digits.data, digits.target
        if n:
            X, y = X[:n], y[:n]
    frame = pd.DataFrame(X, columns=FEATURES)
    frame["target"] = y
    return frame


model = Model(name="digits_mlp", init=TabularMLP, dataset=dataset)


@model.trainer
def trainer(
    clf: TabularMLP,
    features: pd.DataFrame,
    target: pd.DataFrame,
    *,
    epochs: int = 30,
    batch_size: int = 512,
    lr: float = 2e-3,
    use_graph: bool = True,
) -> TabularMLP:
    """Stage features into HBM (pinned async H2D), fit the standardizer,
    then run the fused-kernel minibatch loop. Under DP each rank gets a
    row shard (sharded by Model.train(dp=N)) and gradients all-reduce on
    RCCL inside the step."""
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
    clf = TabularMLP()
    clf.load_state_dict(payload["state"])
    return clf
