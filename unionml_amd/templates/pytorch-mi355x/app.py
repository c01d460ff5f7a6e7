"""{{app_name}} — digits-shape MLP classifier on the CDNA4 tabular hot path.

MI355X-native counterpart of the reference's pytorch quickstart
(tests/integration/pytorch_app/quickstart.py:1-81): the trainer /
predictor bodies run on unionml_amd's hand-written gfx950 MFMA kernels
(fused fwd/bwd/Adam single-kernel step, hipGraph-captured training and
serving), and fall back to the fp32 torch reference on CPU.

    unionml-amd train app:model --local --output model.pt
    unionml-amd serve app:fastapi_app --model-path model.pt --port 8000

To train data-parallel across the 8 GPUs of one MI355X node:

    python -c "from app import model; model.train(dp=8)"
"""

from typing import List

import numpy as np
import pandas as pd
import torch
from fastapi import FastAPI

from unionml_amd import Dataset, Model
from unionml_amd.ops.tabular import TabularMLP
from unionml_amd.parallel import get_world_size
from unionml_amd.serving.graph_runner import TabularGraphRunner, graphed
from unionml_amd.utils.staging import get_stager

FEATURES = [f"p{i}" for i in range(64)]

dataset = Dataset(name="{{app_name}}_dataset", features=FEATURES, targets=["target"], test_size=0.2)
model = Model(name="{{app_name}}", init=TabularMLP, dataset=dataset)


@dataset.reader
def reader(n: int = 0) -> pd.DataFrame:
    from sklearn.datasets import load_digits

    digits = load_digits()
    X, y = digits.data, digits.target
    if n:
        X, y = X[:n], y[:n]
    frame = pd.DataFrame(X, columns=FEATURES)
    frame["target"] = y
    return frame


@model.trainer
def trainer(
    clf: TabularMLP,
    features: pd.DataFrame,
    target: pd.DataFrame,
    *,
    epochs: int = 30,
    batch_size: int = 512,
    lr: float = 2e-3,
) -> TabularMLP:
    stager = get_stager(clf.device)
    X = stager.to_device(features.to_numpy().astype(np.float32))
    y = stager.to_device(target.squeeze().to_numpy().astype(np.int32))
    clf.fit_standardizer(X)
    Xbf = clf.stage(X)
    clf.train_epochs(
        Xbf, y, epochs=epochs, batch_size=batch_size, lr=lr, world_size=get_world_size()
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
    clf = TabularMLP()
    clf.load_state_dict(payload["state"])
    return clf


fastapi_app = FastAPI()
model.serve(fastapi_app, batch=torch.cuda.is_available())


if __name__ == "__main__":
    model_object, metrics = model.train(trainer_kwargs={"epochs": 30})
    print(metrics)
    model.save("model.pt")
