"""Exhaustive positive/negative signature checks for every guard
(coverage shape of the reference's tests/unit/test_type_guards.py)."""

from typing import Dict, List, Optional, Tuple

import pandas as pd
import pytest

from unionml_amd import type_guards as tg


# ------------------------- reader ----------------------------------


def test_guard_reader_ok():
    def reader(n: int = 10) -> pd.DataFrame:
        ...

    tg.guard_reader(reader)


def test_guard_reader_missing_return():
    def reader(n: int = 10):
        ...

    with pytest.raises(tg.GuardError):
        tg.guard_reader(reader)


# ------------------------- loader ----------------------------------


def test_guard_loader_ok():
    def loader(raw: str) -> pd.DataFrame:
        ...

    tg.guard_loader(loader)


def test_guard_loader_wrong_arity():
    def loader(raw, extra):
        ...

    with pytest.raises(tg.GuardError):
        tg.guard_loader(loader)


# ------------------------- splitter --------------------------------


def test_guard_splitter_ok():
    def splitter(data, *, test_size: float, shuffle: bool, random_state: int):
        ...

    tg.guard_splitter(splitter)


def test_guard_splitter_var_kwargs_ok():
    def splitter(data, **kwargs):
        ...

    tg.guard_splitter(splitter)


def test_guard_splitter_missing_kwargs():
    def splitter(data, *, test_size: float):
        ...

    with pytest.raises(tg.GuardError):
        tg.guard_splitter(splitter)


def test_guard_splitter_positional_kwargs_rejected():
    def splitter(data, test_size, shuffle, random_state):
        ...

    with pytest.raises(tg.GuardError):
        tg.guard_splitter(splitter)


# ------------------------- parser ----------------------------------


def test_guard_parser_ok():
    def parser(data, features, targets) -> Tuple[pd.DataFrame, pd.DataFrame]:
        ...

    tg.guard_parser(parser)


def test_guard_parser_wrong_arity():
    def parser(data) -> Tuple[pd.DataFrame]:
        ...

    with pytest.raises(tg.GuardError):
        tg.guard_parser(parser)


def test_guard_parser_missing_return():
    def parser(data, features, targets):
        ...

    with pytest.raises(tg.GuardError):
        tg.guard_parser(parser)


# ------------------------- trainer ---------------------------------


def test_guard_trainer_ok():
    def trainer(model: dict, X: pd.DataFrame, y: pd.DataFrame) -> dict:
        ...

    tg.guard_trainer(trainer, dict, 2)


def test_guard_trainer_wrong_data_args():
    def trainer(model: dict, X: pd.DataFrame) -> dict:
        ...

    with pytest.raises(tg.GuardError):
        tg.guard_trainer(trainer, dict, 2)


def test_guard_trainer_wrong_model_type():
    def trainer(model: list, X: pd.DataFrame, y: pd.DataFrame) -> list:
        ...

    with pytest.raises(tg.GuardError):
        tg.guard_trainer(trainer, dict, 2)


def test_guard_trainer_keyword_only_args_ok():
    def trainer(model: dict, X: pd.DataFrame, y: pd.DataFrame, *, epochs: int = 5) -> dict:
        ...

    tg.guard_trainer(trainer, dict, 2)


# ------------------------- evaluator -------------------------------


def test_guard_evaluator_ok():
    def evaluator(model: dict, X: pd.DataFrame, y: pd.DataFrame) -> float:
        ...

    tg.guard_evaluator(evaluator, dict, 2)


def test_guard_evaluator_no_model_arg():
    def evaluator() -> float:
        ...

    with pytest.raises(tg.GuardError):
        tg.guard_evaluator(evaluator, dict, 2)


# ------------------------- predictor -------------------------------


def test_guard_predictor_ok():
    def predictor(model: dict, features: pd.DataFrame) -> List[float]:
        ...

    tg.guard_predictor(predictor, dict)


def test_guard_predictor_wrong_arity():
    def predictor(model: dict) -> List[float]:
        ...

    with pytest.raises(tg.GuardError):
        tg.guard_predictor(predictor, dict)


def test_guard_predictor_missing_return():
    def predictor(model: dict, features: pd.DataFrame):
        ...

    with pytest.raises(tg.GuardError):
        tg.guard_predictor(predictor, dict)


# ------------------------- callbacks -------------------------------


def test_guard_prediction_callback_ok():
    def cb(model, features, predictions) -> None:
        ...

    tg.guard_prediction_callback(cb)


def test_guard_prediction_callback_wrong_arity():
    def cb(model, predictions) -> None:
        ...

    with pytest.raises(tg.GuardError):
        tg.guard_prediction_callback(cb)


def test_guard_prediction_callback_nonnull_return():
    def cb(model, features, predictions) -> int:
        ...

    with pytest.raises(tg.GuardError):
        tg.guard_prediction_callback(cb)


# ------------------------- feature fns -----------------------------


def test_guard_feature_loader_ok():
    def fl(raw) -> pd.DataFrame:
        ...

    tg.guard_feature_loader(fl)


def test_guard_feature_transformer_wrong_arity():
    def ft(a, b):
        ...

    with pytest.raises(tg.GuardError):
        tg.guard_feature_transformer(ft)


# ------------------------- saver/loader ----------------------------


def test_guard_saver_ok():
    def saver(model, hyperparameters, file):
        ...

    tg.guard_saver(saver)


def test_guard_loader_fn_wrong_arity():
    def loader():
        ...

    with pytest.raises(tg.GuardError):
        tg.guard_loader_fn(loader)


# ------------------------- type compat -----------------------------


def test_types_compatible_union():
    from typing import Union

    assert tg._types_compatible(Union[str, bytes], str)
    assert tg._types_compatible(pd.DataFrame, pd.DataFrame)
    assert not tg._types_compatible(int, str)


def test_types_compatible_generics():
    assert tg._types_compatible(List[int], list)
    assert tg._types_compatible(Dict[str, int], dict)
    assert not tg._types_compatible(List[int], Dict[str, int])


# --------------- data-type compatibility matrix (r02) -----------------
# Mirrors the reference's compat checks (ref type_guards.py:28-69,
# 88-115): loader/splitter/parser input annotations vs upstream types,
# splitter split-container returns, trainer/evaluator data args vs
# parser return types, and model-type checks now WIRED through the
# Dataset/Model decorators (they were dead branches in r01).

from typing import Tuple as _Tuple


def test_guard_loader_input_compat():
    def good(data: pd.DataFrame):
        return data

    tg.guard_loader(good, pd.DataFrame)

    def bad(data: int):
        return data

    with pytest.raises(tg.GuardError, match="not compatible"):
        tg.guard_loader(bad, pd.DataFrame)


def test_guard_splitter_output_container():
    def good(data: pd.DataFrame, *, test_size: float, shuffle: bool, random_state: int) -> _Tuple[pd.DataFrame, pd.DataFrame]:
        return data, data

    tg.guard_splitter(good, pd.DataFrame, "reader")

    def bad_container(data: pd.DataFrame, *, test_size: float, shuffle: bool, random_state: int) -> pd.DataFrame:
        return data

    with pytest.raises(tg.GuardError, match="List, Tuple or NamedTuple"):
        tg.guard_splitter(bad_container, pd.DataFrame, "reader")

    def bad_elements(data: pd.DataFrame, *, test_size: float, shuffle: bool, random_state: int) -> _Tuple[int, int]:
        return 1, 2

    with pytest.raises(tg.GuardError, match="split elements"):
        tg.guard_splitter(bad_elements, pd.DataFrame, "reader")


def test_guard_trainer_model_and_data_types():
    class MyModel:
        pass

    def good(model: MyModel, features: pd.DataFrame, target: pd.DataFrame) -> MyModel:
        return model

    tg.guard_trainer(good, MyModel, 2, (pd.DataFrame, pd.DataFrame))

    def wrong_model(model: int, features: pd.DataFrame, target: pd.DataFrame) -> MyModel:
        return MyModel()

    with pytest.raises(tg.GuardError, match="not compatible"):
        tg.guard_trainer(wrong_model, MyModel, 2, (pd.DataFrame, pd.DataFrame))

    def wrong_return(model: MyModel, features: pd.DataFrame, target: pd.DataFrame) -> int:
        return 1

    with pytest.raises(tg.GuardError, match="must return the model type"):
        tg.guard_trainer(wrong_return, MyModel, 2, (pd.DataFrame, pd.DataFrame))

    def wrong_data(model: MyModel, features: int, target: pd.DataFrame) -> MyModel:
        return model

    with pytest.raises(tg.GuardError, match="parser output type"):
        tg.guard_trainer(wrong_data, MyModel, 2, (pd.DataFrame, pd.DataFrame))


def test_guard_predictor_feature_type():
    class MyModel:
        pass

    def good(model: MyModel, features: pd.DataFrame) -> List[int]:
        return []

    tg.guard_predictor(good, MyModel, pd.DataFrame)

    def bad(model: MyModel, features: int) -> List[int]:
        return []

    with pytest.raises(tg.GuardError, match="feature type"):
        tg.guard_predictor(bad, MyModel, pd.DataFrame)


def test_decorator_wiring_rejects_model_type_mismatch():
    """The Model decorators must actually FIRE the model-type checks
    (they passed model_type=None, dead code)."""
    from sklearn.linear_model import LogisticRegression

    from unionml_amd import Dataset, Model

    ds = Dataset(name="wired", features=["a"], targets=["t"])

    @ds.reader
    def reader() -> pd.DataFrame:
        return pd.DataFrame({"a": [1.0], "t": [0]})

    model = Model(name="wired", init=LogisticRegression, dataset=ds)

    with pytest.raises(tg.GuardError, match="not compatible"):

        @model.trainer
        def trainer(est: int, features: pd.DataFrame, target: pd.DataFrame) -> int:
            return est

    with pytest.raises(tg.GuardError, match="not compatible"):

        @model.predictor
        def predictor(est: dict, features: pd.DataFrame) -> List[float]:
            return []

    with pytest.raises(tg.GuardError, match="not compatible"):

        @model.evaluator
        def evaluator(est: str, features: pd.DataFrame, target: pd.DataFrame) -> float:
            return 0.0

    # correctly-annotated callables pass
    @model.trainer
    def trainer_ok(est: LogisticRegression, features: pd.DataFrame, target: pd.DataFrame) -> LogisticRegression:
        return est.fit(features, target.squeeze())


def test_decorator_wiring_rejects_dataset_type_mismatch():
    from unionml_amd import Dataset

    ds = Dataset(name="wired2")

    @ds.reader
    def reader() -> pd.DataFrame:
        return pd.DataFrame()

    with pytest.raises(tg.GuardError, match="not compatible"):

        @ds.loader
        def loader(data: int):
            return data

    with pytest.raises(tg.GuardError, match="split elements"):

        @ds.splitter
        def splitter(data: pd.DataFrame, *, test_size: float, shuffle: bool, random_state: int) -> _Tuple[int, int]:
            return 1, 2


def test_hyperparameter_config_branch():
    """Model(hyperparameter_config={'C': float}) synthesizes the dataclass
    (reference model.py:66,180-186 — used by its own test fixtures)."""
    import dataclasses

    from sklearn.linear_model import LogisticRegression

    from unionml_amd import Dataset, Model

    ds = Dataset(name="hpc", features=["x1"], targets=["t"])

    @ds.reader
    def reader(n: int = 20) -> pd.DataFrame:
        import numpy as np

        rng = np.random.RandomState(0)
        return pd.DataFrame({"x1": rng.rand(n), "t": rng.randint(0, 2, n)})

    model = Model(
        name="hpc",
        init=LogisticRegression,
        hyperparameter_config={"C": float, "max_iter": int},
        dataset=ds,
    )
    hp_type = model.hyperparameter_type
    assert dataclasses.is_dataclass(hp_type)
    assert {f.name: f.type for f in dataclasses.fields(hp_type)} == {"C": float, "max_iter": int}

    @model.trainer
    def trainer(est: LogisticRegression, features: pd.DataFrame, target: pd.DataFrame) -> LogisticRegression:
        return est.fit(features, target.squeeze())

    @model.predictor
    def predictor(est: LogisticRegression, features: pd.DataFrame) -> List[float]:
        return [float(x) for x in est.predict(features)]

    @model.evaluator
    def evaluator(est: LogisticRegression, features: pd.DataFrame, target: pd.DataFrame) -> float:
        return float((est.predict(features) == target.squeeze().to_numpy()).mean())

    obj, metrics = model.train(hyperparameters={"C": 0.5, "max_iter": 200})
    assert obj.C == 0.5 and obj.max_iter == 200
    assert set(metrics) == {"train", "test"}


def test_hyperparameter_type_and_config_mutually_exclusive():
    from unionml_amd import Dataset, Model

    ds = Dataset(name="x")
    with pytest.raises(ValueError, match="not both"):
        Model(name="x", hyperparameter_type=dict, hyperparameter_config={"C": float}, dataset=ds)
