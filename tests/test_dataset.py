"""Dataset unit tests (coverage shape mirrors the reference's
tests/unit/test_dataset.py: registration, compiled task, defaults vs
custom pipeline, non-frame raw types, workflow composability)."""

import json

import numpy as np
import pandas as pd
import pytest

from unionml_amd import Dataset
from unionml_amd.task import Task, Workflow


def test_reader_registration(frame_dataset):
    assert frame_dataset._reader is not None
    assert frame_dataset.dataset_datatype["data"] is pd.DataFrame


def test_reader_requires_return_annotation():
    ds = Dataset(name="bad")
    from unionml_amd.type_guards import GuardError

    with pytest.raises(GuardError):

        @ds.reader
        def reader(n: int = 1):
            return None


def test_dataset_task_interface(frame_dataset):
    task = frame_dataset.dataset_task()
    assert isinstance(task, Task)
    assert task.name == "frame_ds.reader"
    out = task(n=10)
    assert isinstance(out, pd.DataFrame)
    assert len(out) == 10
    # cached builder returns the same object
    assert frame_dataset.dataset_task() is task


def test_get_data_default_pipeline(frame_dataset):
    raw = frame_dataset.dataset_task()(n=100)
    data = frame_dataset.get_data(raw)
    assert set(data) == {"train", "test"}
    features, target = data["train"]
    assert list(features.columns) == ["x1", "x2", "x3"]
    assert list(target.columns) == ["y"]
    assert len(features) == 80 and len(data["test"][0]) == 20


def test_get_data_kwargs_override(frame_dataset):
    raw = frame_dataset.dataset_task()(n=100)
    data = frame_dataset.get_data(raw, splitter_kwargs={"test_size": 0.5})
    assert len(data["train"][0]) == 50


def test_custom_feature_transformer():
    ds = Dataset(name="std_ds", targets=["y"])

    @ds.reader
    def reader(n: int = 50) -> pd.DataFrame:
        from dataset_fixtures import make_frame

        return make_frame(n)

    @ds.feature_transformer
    def standardize(features: pd.DataFrame) -> pd.DataFrame:
        return (features - features.mean()) / features.std()

    raw = reader(50)
    data = ds.get_data(raw)
    feats = data["train"][0]
    assert abs(feats.mean().mean()) < 0.5  # roughly centered


def test_custom_splitter_parser_over_list(list_dataset):
    from typing import Dict, List, Tuple

    @list_dataset.splitter
    def splitter(data: List[Dict], *, test_size: float, shuffle: bool, random_state: int):
        n_test = int(len(data) * test_size)
        return data[n_test:], data[:n_test]

    @list_dataset.parser
    def parser(data: List[Dict], features, targets) -> Tuple[List[List[float]], List[int]]:
        xs = [[row["x1"], row["x2"], row["x3"]] for row in data]
        ys = [row["y"] for row in data]
        return xs, ys

    raw = list_dataset.dataset_task()(n=20)
    data = list_dataset.get_data(raw)
    assert len(data["train"][0]) == 16
    assert len(data["test"][0]) == 4
    assert isinstance(data["train"][0][0], list)


def test_custom_loader_from_json(json_dataset):
    raw = json_dataset.dataset_task()(n=20)
    assert isinstance(raw, str)
    data = json_dataset.get_data(raw)
    assert isinstance(data["train"][0], pd.DataFrame)


def test_get_features_from_records(frame_dataset):
    features = frame_dataset.get_features([{"x1": 1.0, "x2": 2.0, "x3": 3.0}])
    assert isinstance(features, pd.DataFrame)
    assert list(features.columns) == ["x1", "x2", "x3"]


def test_get_features_from_json_file(tmp_path, frame_dataset):
    path = tmp_path / "features.json"
    path.write_text(json.dumps([{"x1": 0.1, "x2": 0.2, "x3": 0.3, "y": 1}]))
    features = frame_dataset.get_features(path)
    assert list(features.columns) == ["x1", "x2", "x3"]  # target dropped


def test_default_parser_feature_selection():
    """The corrected guard: features derived from non-target columns only
    when features= is not given (reference quirk fixed)."""
    ds = Dataset(name="sel", features=["x1"], targets=["y"])

    @ds.reader
    def reader() -> pd.DataFrame:
        return pd.DataFrame({"x1": [1.0, 2], "x2": [3.0, 4], "y": [0, 1]})

    parsed = ds._default_parser(reader(), **ds.parser_kwargs)
    assert list(parsed[0].columns) == ["x1"]  # explicit selection respected


def test_dataset_task_composable_in_workflow(frame_dataset):
    """unionml tasks embed in hand-written workflows (reference:
    test_dataset.py:129-145)."""
    task = frame_dataset.dataset_task()
    wf = Workflow(name="custom", inputs=["n"], outputs=[("rows", ("node", 1, None))])
    n0 = wf.add_node(task, bindings={"n": ("input", "n")})

    count_task = Task(lambda data: len(data), "count")
    wf.add_node(count_task, bindings={"data": ("node", n0, None)})
    assert wf(n=17) == 17


def test_sqlite_dataset(tmp_path):
    import sqlite3

    db = tmp_path / "test.db"
    with sqlite3.connect(db) as conn:
        conn.execute("CREATE TABLE points (x1 REAL, x2 REAL, y INTEGER)")
        conn.executemany(
            "INSERT INTO points VALUES (?, ?, ?)",
            [(float(i), float(-i), i % 2) for i in range(40)],
        )

    ds = Dataset.from_sqlite_task(
        "sql_ds", str(db), "SELECT * FROM points LIMIT {limit}", targets=["y"]
    )
    raw = ds.dataset_task()(limit=30)
    assert len(raw) == 30
    data = ds.get_data(raw)
    assert list(data["train"][0].columns) == ["x1", "x2"]


def test_stage_to_device_cpu(frame_dataset):
    import torch

    raw = frame_dataset.dataset_task()(n=10)
    data = frame_dataset.get_data(raw)
    staged = frame_dataset.stage_to_device(data["train"], device="cpu")
    assert torch.is_tensor(staged[0])
    assert staged[0].shape == (8, 3)


def test_dataset_from_task():
    """Dataset.from_task wraps an existing Task/callable as the reader
    (reference: Dataset._from_flytekit_task, dataset.py:426-440)."""
    import pandas as pd

    from unionml_amd.dataset import Dataset
    from unionml_amd.task import Task

    def produce(n: int = 10) -> pd.DataFrame:
        return pd.DataFrame({"a": range(n), "y": [i % 2 for i in range(n)]})

    task = Task(produce, "produce")
    ds = Dataset.from_task(task, targets=["y"], test_size=0.2, random_state=0)
    assert ds.name == "produce"
    data = ds.get_data(ds.dataset_task()(n=20))
    (Xtr, ytr), (Xte, yte) = data["train"], data["test"]
    assert len(Xtr) + len(Xte) == 20
    assert list(Xtr.columns) == ["a"]


def test_from_sqlalchemy_task_sqlite_fallback(tmp_path):
    """from_sqlalchemy_task works with or without sqlalchemy installed
    (falls back to sqlite3 for sqlite:/// URIs — reference capability:
    dataset.py:458-470)."""
    import sqlite3

    import pandas as pd

    from unionml_amd.dataset import Dataset

    db = tmp_path / "t.db"
    with sqlite3.connect(db) as conn:
        conn.execute("CREATE TABLE pts (a REAL, y INTEGER)")
        conn.executemany(
            "INSERT INTO pts VALUES (?, ?)", [(i * 0.1, i % 2) for i in range(30)]
        )
    ds = Dataset.from_sqlalchemy_task(
        "sql_ds", f"sqlite:///{db}", "SELECT * FROM pts WHERE a < {amax}",
        targets=["y"], test_size=0.2, random_state=0,
    )
    raw = ds.dataset_task()(amax=2.0)
    assert isinstance(raw, pd.DataFrame) and len(raw) == 20
    data = ds.get_data(raw)
    assert len(data["train"][0]) + len(data["test"][0]) == 20


def test_default_loader_rejects_loader_kwargs(frame_dataset):
    """loader_kwargs with the default loader is a GuardError, not a
    TypeError from the default loader's signature."""
    from unionml_amd.type_guards import GuardError

    raw = frame_dataset.dataset_task()(n=10)
    with pytest.raises(GuardError, match="no @dataset.loader"):
        frame_dataset.get_data(raw, loader_kwargs={"anything": 1})


def test_feature_loader_fast_path_preserves_dtypes():
    """The /predict records fast path must keep per-column dtypes (ints
    stay ints, bools stay bools) — not coerce everything to float64."""
    ds = Dataset(name="dtypes", features=["a", "b", "c"], targets=["t"])

    @ds.reader
    def reader() -> pd.DataFrame:
        return pd.DataFrame({"a": [1], "b": [0.5], "c": [True], "t": [0]})

    feats = ds.get_features([{"a": 1, "b": 0.25, "c": True}, {"a": 2, "b": 0.75, "c": False}])
    assert feats["a"].dtype.kind == "i"
    assert feats["b"].dtype.kind == "f"
    assert feats["c"].dtype.kind == "b"
    assert feats["a"].tolist() == [1, 2]


def test_feature_type_union_with_custom_loader():
    """A custom feature_loader with a different return type makes
    feature_type a FeatureTypeUnion[dataset_type, loaded_type]
    (reference dataset.py:405-424); bentoml IO inference unwraps the
    serve-time arm."""
    import typing

    from unionml_amd.dataset import FeatureTypeUnion

    ds = Dataset(name="ftu", features=["a"], targets=["t"])

    @ds.reader
    def reader() -> pd.DataFrame:
        return pd.DataFrame({"a": [1.0], "t": [0]})

    assert ds.feature_type in (pd.DataFrame, typing.Any)

    @ds.feature_loader
    def feature_loader(data) -> np.ndarray:
        return np.asarray(data)

    ft = ds.feature_type
    assert typing.get_origin(ft) is FeatureTypeUnion
    assert typing.get_args(ft)[1] is np.ndarray

    from unionml_amd.services.bentoml import infer_io_descriptor

    assert infer_io_descriptor(ft) == "NumpyNdarray"

    # predictor guards accept either arm of the union
    from unionml_amd import type_guards as tg

    assert tg._types_compatible(ft, np.ndarray)
    strict = FeatureTypeUnion[pd.DataFrame, np.ndarray]
    assert tg._types_compatible(strict, np.ndarray)
    assert tg._types_compatible(strict, pd.DataFrame)
    assert not tg._types_compatible(strict, int)
