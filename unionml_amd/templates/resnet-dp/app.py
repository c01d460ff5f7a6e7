"""{{app_name}} — image-CNN app trained data-parallel on one MI355X node.

A from-scratch ResNet-18 on synthetic images through the decorator API
(the reference's quickdraw template trains a from-scratch CNN —
templates/quickdraw/.../model.py:39-54; here the framework owns the
distributed loop): channels-last bf16 autocast, MIOpen conv autotuning,
and the framework's bucketed RCCL gradient all-reducer under
``model.train(dp=N)``. Swap the reader for your dataset.
"""

from typing import Dict, Tuple

import numpy as np
import torch
import torch.nn.functional as F

from unionml_amd import Dataset, Model
from unionml_amd.models.resnet import ResNet18
from unionml_amd.parallel import maybe_wrap

dataset = Dataset(name="{{app_name}}_dataset", test_size=0.1, shuffle=False)


@dataset.reader
def reader(n: int = 2048, image_size: int = 224, num_classes: int = 100, seed: int = 0) -> Dict[str, np.ndarray]:
    rng = np.random.RandomState(seed)
    return {
        "images": rng.rand(n, 3, image_size, image_size).astype(np.float32),
        "labels": rng.randint(0, num_classes, size=n).astype(np.int64),
    }


@dataset.loader
def loader(data: Dict[str, np.ndarray]) -> Dict[str, np.ndarray]:
    return data


@dataset.splitter
def splitter(
    data: Dict[str, np.ndarray], *, test_size: float, shuffle: bool, random_state: int
) -> Tuple[Dict[str, np.ndarray], Dict[str, np.ndarray]]:
    n = len(data["labels"])
    n_test = max(1, int(round(n * test_size)))
    pick = lambda sl: {k: v[sl] for k, v in data.items()}  # noqa: E731
    return pick(slice(n_test, n)), pick(slice(0, n_test))


@dataset.parser
def parser(data: Dict[str, np.ndarray], features, targets) -> Tuple[np.ndarray, np.ndarray]:
    return data["images"], data["labels"]


model = Model(name="{{app_name}}", dataset=dataset)


@model.init
def init(hyperparameters: dict) -> ResNet18:
    return ResNet18(**(hyperparameters or {"num_classes": 100}))


@model.trainer
def trainer(
    net: ResNet18, images: np.ndarray, labels: np.ndarray,
    *, epochs: int = 1, batch_size: int = 64, lr: float = 1e-3,
) -> ResNet18:
    device = torch.device("cuda" if torch.cuda.is_available() else "cpu")
    net = net.to(device)
    if device.type == "cuda":
        net = net.to(memory_format=torch.channels_last)
        torch.backends.cudnn.benchmark = True
    reducer = maybe_wrap(net)
    opt = torch.optim.Adam(net.parameters(), lr=lr, foreach=True)
    X, y = torch.from_numpy(images), torch.from_numpy(labels)
    net.train()
    for _ in range(epochs):
        for off in range(0, len(y), batch_size):
            xb = X[off : off + batch_size].to(device)
            yb = y[off : off + batch_size].to(device)
            if device.type == "cuda":
                xb = xb.to(memory_format=torch.channels_last)
            with torch.autocast(device.type, dtype=torch.bfloat16, enabled=device.type == "cuda"):
                loss = F.cross_entropy(net(xb), yb)
            loss.backward()
            if reducer is not None:
                reducer.finalize()
            opt.step()
            if reducer is not None:
                reducer.zero_grad()
            else:
                opt.zero_grad(set_to_none=True)
    if reducer is not None:
        reducer.detach()
    return net


@model.predictor
def predictor(net: ResNet18, images: np.ndarray) -> np.ndarray:
    device = next(net.parameters()).device
    with torch.no_grad():
        xb = torch.from_numpy(np.ascontiguousarray(images)).to(device)
        return net.eval()(xb).argmax(dim=1).cpu().numpy()


@model.evaluator
def evaluator(net: ResNet18, images: np.ndarray, labels: np.ndarray) -> float:
    return float((predictor(net, images) == labels).mean())


if __name__ == "__main__":
    import os

    dp = int(os.environ.get("DP", "1"))
    _, metrics = model.train(dp=dp, trainer_kwargs={"epochs": 1})
    print(metrics)
