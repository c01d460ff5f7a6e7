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
