"""Importable DP test app (spawned ranks re-import this module)."""

import os
import pickle
import tempfile
from pathlib import Path
from typing import List

import pandas as pd
import torch

from unionml_amd import Dataset, Model
from unionml_amd.parallel import get_world_size, maybe_wrap


class DPNet(torch.nn.Module):
    def __init__(self, in_dim: int = 3, hidden: int = 8, out_dim: int = 2):
        super().__init__()
        self.net = torch.nn.Sequential(
            torch.nn.Linear(in_dim, hidden), torch.nn.ReLU(), torch.nn.Linear(hidden, out_dim)
        )

    def forward(self, x):
        return self.net(x)


dataset = Dataset(name="dp_ds", targets=["y"], test_size=0.2, random_state=5)


@dataset.reader
def reader(n: int = 100) -> pd.DataFrame:
    import numpy as np

    rng = np.random.RandomState(3)
    return pd.DataFrame(
        {"x1": rng.randn(n), "x2": rng.randn(n), "x3": rng.randn(n), "y": rng.randint(0, 2, n)}
    )


model = Model(name="dp_model", init=DPNet, dataset=dataset)


@model.trainer
def trainer(
    net: DPNet, features: pd.DataFrame, target: pd.DataFrame, *, epochs: int = 2, lr: float = 0.05
) -> DPNet:
    torch.manual_seed(0)
    net = DPNet()  # identical init on every rank
    reducer = maybe_wrap(net)
    opt = torch.optim.SGD(net.parameters(), lr=lr)
    x = torch.tensor(features.to_numpy(), dtype=torch.float32)
    y = torch.tensor(target.squeeze().to_numpy(), dtype=torch.long)
    for _ in range(epochs):
        loss = torch.nn.functional.cross_entropy(net(x), y)
        loss.backward()
        if reducer:
            reducer.finalize()
        opt.step()
        if reducer:
            reducer.zero_grad()
        else:
            opt.zero_grad()
    return net


@model.predictor
def predictor(net: DPNet, features: pd.DataFrame) -> List[int]:
    x = torch.tensor(features.to_numpy(), dtype=torch.float32)
    with torch.no_grad():
        return [int(i) for i in net(x).argmax(dim=1)]


@model.evaluator
def evaluator(net: DPNet, features: pd.DataFrame, target: pd.DataFrame) -> float:
    preds = predictor(net, features)
    return float((pd.Series(preds).to_numpy() == target.squeeze().to_numpy()).mean())


# ----------------------------------------------------------------------
# TabularMLP DP equivalence harness
# ----------------------------------------------------------------------

N_ROWS = 256


def _make_data(shape=(64, 32, 10)):
    torch.manual_seed(7)
    inf, _, cls = shape
    X = torch.randn(N_ROWS, inf) * 1.5
    y = torch.randint(0, cls, (N_ROWS,), dtype=torch.int32)
    return X, y


def _make_clf(shape):
    from unionml_amd.ops.tabular import TabularMLP

    inf, hid, cls = shape
    return TabularMLP(in_features=inf, hidden=hid, classes=cls, device="cpu", seed=1)


def _tabular_worker(rank, world, port, out_dir, shape):
    import torch.distributed as dist

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from unionml_amd.parallel.ddp import shard

        X, y = _make_data(shape)
        Xr, yr = shard(X, rank, world), shard(y, rank, world)
        clf = _make_clf(shape)
        Xbf = clf.stage(Xr)  # identity standardizer (never fit): pads only
        clf.train_epochs(
            Xbf, yr, epochs=3, batch_size=len(Xr), lr=1e-3, world_size=world
        )
        if rank == 0:
            with open(Path(out_dir) / "dp_master.pkl", "wb") as f:
                pickle.dump(clf.master, f)
    finally:
        dist.destroy_process_group()


def run_tabular_dp(shape=(64, 32, 10)):
    import torch.multiprocessing as mp

    from unionml_amd.parallel.launch import _free_port

    # single-process oracle: full batch, invBtot = 1/N
    X, y = _make_data(shape)
    clf = _make_clf(shape)
    clf.train_epochs(clf.stage(X), y, epochs=3, batch_size=N_ROWS, lr=1e-3)
    single_master = clf.master.clone()

    with tempfile.TemporaryDirectory() as out_dir:
        port = _free_port()
        mp.start_processes(
            _tabular_worker, args=(2, port, out_dir, shape), nprocs=2, join=True,
            start_method="spawn",
        )
        with open(Path(out_dir) / "dp_master.pkl", "rb") as f:
            dp_master = pickle.load(f)
    return single_master, dp_master


def build_failing_model():
    """An app whose trainer raises — for DP failure-surfacing tests."""
    ds = Dataset(name="fail_ds", targets=["y"], test_size=0.2, random_state=5)

    @ds.reader
    def reader(n: int = 40) -> pd.DataFrame:
        import numpy as np

        rng = np.random.RandomState(0)
        X = rng.rand(n, 3)
        return pd.DataFrame(
            {"x1": X[:, 0], "x2": X[:, 1], "x3": X[:, 2], "y": (X.sum(axis=1) > 1.5).astype(int)}
        )

    m = Model(name="fail_model", init=DPNet, dataset=ds)

    @m.trainer
    def trainer(net: DPNet, features: pd.DataFrame, target: pd.DataFrame) -> DPNet:
        raise RuntimeError("injected trainer failure")

    @m.predictor
    def predictor(net: DPNet, features: pd.DataFrame) -> List[int]:
        return [0] * len(features)

    return m
