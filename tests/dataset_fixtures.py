"""Dataset fixtures (design mirrors the reference's fixture strategy,
tests/unit/dataset_fixtures.py: simple readers over several raw types)."""

import json
from typing import Dict, List

import pandas as pd
import pytest

from unionml_amd import Dataset


def make_frame(n: int = 100) -> pd.DataFrame:
    import numpy as np

    rng = np.random.RandomState(7)
    return pd.DataFrame(
        {
            "x1": rng.randn(n),
            "x2": rng.randn(n),
            "x3": rng.randn(n),
            "y": rng.randint(0, 2, size=n),
        }
    )


@pytest.fixture
def frame_dataset() -> Dataset:
    ds = Dataset(name="frame_ds", targets=["y"], test_size=0.2, random_state=99)

    @ds.reader
    def reader(n: int = 100) -> pd.DataFrame:
        return make_frame(n)

    return ds


@pytest.fixture
def list_dataset() -> Dataset:
    ds = Dataset(name="list_ds", targets=["y"])

    @ds.reader
    def reader(n: int = 20) -> List[Dict]:
        frame = make_frame(n)
        return frame.to_dict(orient="records")

    return ds


@pytest.fixture
def json_dataset() -> Dataset:
    ds = Dataset(name="json_ds", targets=["y"])

    @ds.reader
    def reader(n: int = 20) -> str:
        return make_frame(n).to_json(orient="records")

    @ds.loader
    def loader(raw: str) -> pd.DataFrame:
        return pd.DataFrame(json.loads(raw))

    return ds
