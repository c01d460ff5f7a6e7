"""Dataset API — registers and composes data-pipeline callables.

Same capability surface as the reference Dataset
(unionml/dataset.py:44-527): ``@dataset.reader / .loader / .splitter /
.parser / .feature_loader / .feature_transformer`` decorators, type
derivation from signatures, ``get_data``/``get_features`` composition,
SQL constructors, and a compiled ``dataset_task``.

MI355X-native additions:

- :meth:`Dataset.stage_to_device` — stages a parsed split into a GPU's
  HBM via async H2D copies on a side stream
  (SURVEY.md §2c "pinned staging loader").
- The corrected default-parser column logic (the reference inverts its
  feature-selection guard — SURVEY.md §8 "Known reference quirks").
"""

import inspect
import json
import sqlite3
from inspect import signature
from pathlib import Path
from typing import Any, Callable, Dict, List, Optional, Tuple, Type

import numpy as np
import pandas as pd

from unionml_amd import type_guards
from unionml_amd.defaults import DEFAULT_RESOURCES, Resources
from unionml_amd.task import Task, inner_task
from unionml_amd.tracker import TrackedInstance


import typing as _typing

DT = _typing.TypeVar("DT")  # dataset (parser) feature type
FT = _typing.TypeVar("FT")  # serve-time loaded feature type


class FeatureTypeUnion(_typing.Generic[DT, FT]):
    """'raw-or-loaded features' union marker (reference dataset.py:30-31):
    when a custom feature_loader returns a different type than the
    parser's feature element, :attr:`Dataset.feature_type` is
    ``FeatureTypeUnion[dataset_type, loaded_type]`` — consumers that
    need the serve-time type (bentoml IO inference) unwrap the second
    argument."""


def _train_test_split(data: pd.DataFrame, test_size: float, shuffle: bool, random_state: int):
    """Deterministic train/test split on a DataFrame.

    Uses sklearn when importable (behavioral parity with the reference
    default splitter, dataset.py:478-487); otherwise falls back to a
    numpy permutation split with the same semantics.
    """
    try:
        from sklearn.model_selection import train_test_split

        return train_test_split(data, test_size=test_size, shuffle=shuffle, random_state=random_state)
    except ImportError:
        import numpy as np

        n = len(data)
        n_test = int(round(n * test_size))
        idx = np.arange(n)
        if shuffle:
            idx = np.random.RandomState(random_state).permutation(n)
        test_idx, train_idx = idx[:n_test], idx[n_test:]
        return data.iloc[train_idx], data.iloc[test_idx]


class Dataset(TrackedInstance):
    """Declarative data pipeline for a unionml_amd app."""

    def __init__(
        self,
        name: str = "dataset",
        *,
        features: Optional[List[str]] = None,
        targets: Optional[List[str]] = None,
        test_size: float = 0.2,
        shuffle: bool = True,
        random_state: int = 12345,
    ):
        self.name = name
        self._features = features or []
        self._targets = targets or []
        self._test_size = test_size
        self._shuffle = shuffle
        self._random_state = random_state

        self._reader: Optional[Callable] = None
        self._reader_task_kwargs: Dict[str, Any] = {}
        self._loader: Callable = self._default_loader
        self._splitter: Callable = self._default_splitter
        self._parser: Callable = self._default_parser
        self._parser_feature_key: int = 0
        self._feature_loader: Callable = self._default_feature_loader
        self._feature_transformer: Callable = self._default_feature_transformer

        self._dataset_task: Optional[Task] = None

    # ------------------------------------------------------------------
    # decorators
    # ------------------------------------------------------------------

    def reader(
        self,
        fn: Optional[Callable] = None,
        *,
        cache: bool = False,
        cache_version: str = "0",
        resources: Resources = DEFAULT_RESOURCES,
        **task_kwargs,
    ):
        """Register the function that produces raw data from the outside
        world (reference: dataset.py:103-116)."""

        def decorator(f: Callable) -> Callable:
            type_guards.guard_reader(f)
            self._reader = f
            self._reader_task_kwargs = dict(
                cache=cache, cache_version=cache_version, resources=resources, **task_kwargs
            )
            self._dataset_task = None
            return f

        return decorator(fn) if fn is not None else decorator

    def loader(self, fn: Callable) -> Callable:
        """Register the function converting reader output into the
        in-memory form the splitter/parser consume (reference:
        dataset.py:118-131)."""
        type_guards.guard_loader(fn, self._upstream_datatype(stage="reader"))
        self._loader = fn
        return fn

    def splitter(self, fn: Callable) -> Callable:
        """Register the train/test splitting function (reference:
        dataset.py:133-156)."""
        type_guards.guard_splitter(fn, *self._loaded_datatype())
        self._splitter = fn
        return fn

    def parser(self, fn: Callable, feature_key: int = 0) -> Callable:
        """Register the function parsing a split into (features, targets, ...)
        tuples; ``feature_key`` indexes the features element (reference:
        dataset.py:158-182)."""
        type_guards.guard_parser(fn, self._loaded_datatype()[0])
        self._parser = fn
        self._parser_feature_key = feature_key
        return fn

    def feature_loader(self, fn: Callable) -> Callable:
        """Register the function that converts raw serve-time feature
        payloads (JSON, file path, arrays) into the parser's feature type
        (reference: dataset.py:184-198)."""
        type_guards.guard_feature_loader(fn)
        self._feature_loader = fn
        return fn

    def feature_transformer(self, fn: Callable) -> Callable:
        """Register the transformation applied to features both at train
        and serve time (reference: dataset.py:200-212)."""
        type_guards.guard_feature_transformer(fn)
        self._feature_transformer = fn
        return fn

    # ------------------------------------------------------------------
    # derived types
    # ------------------------------------------------------------------

    def _upstream_datatype(self, stage: str = "reader"):
        """Type produced by the reader, for input-compat guards; None
        (skip the check) when no reader is registered yet."""
        if self._reader is None:
            return None
        return self.dataset_datatype["data"]

    def _loaded_datatype(self):
        """(type, source) of the data flowing out of the loader stage:
        a custom loader's return annotation, else the reader type."""
        if self._loader != self._default_loader:
            ret = signature(self._loader).return_annotation
            if ret is not inspect.Signature.empty:
                return ret, "loader"
            return None, "loader"
        return self._upstream_datatype(), "reader"


    @property
    def splitter_kwargs(self) -> Dict[str, Any]:
        return {
            "test_size": self._test_size,
            "shuffle": self._shuffle,
            "random_state": self._random_state,
        }

    @property
    def parser_kwargs(self) -> Dict[str, Any]:
        return {"features": self._features or None, "targets": self._targets}

    @property
    def dataset_datatype(self) -> Dict[str, Type]:
        """The reader's return type, keyed 'data' (reference: dataset.py:368-380)."""
        if self._reader is None:
            return {"data": pd.DataFrame}
        ret = signature(self._reader).return_annotation
        return {"data": ret}

    @property
    def parser_return_types(self) -> Tuple[Type, ...]:
        """Element types of the parser's return tuple — drives the
        trainer/evaluator data-argument count (reference: dataset.py:392-410)."""
        import typing

        ret = signature(self._parser).return_annotation
        if ret is signature(self._parser).empty:
            return (Any, Any)
        args = typing.get_args(ret)
        return args if args else (ret,)

    @property
    def n_parser_outputs(self) -> int:
        return len(self.parser_return_types)

    @property
    def feature_type(self) -> Type:
        """The type of one split's feature element; with a custom
        feature_loader producing a different type, the
        :class:`FeatureTypeUnion` of both (reference: dataset.py:405-424)."""
        dataset_type = self.parser_return_types[self._parser_feature_key]
        if self._feature_loader == self._default_feature_loader:
            return dataset_type
        loaded_type = (
            signature(self._feature_loader).return_annotation
            if self._feature_transformer == self._default_feature_transformer
            else signature(self._feature_transformer).return_annotation
        )
        if loaded_type is inspect.Signature.empty or loaded_type == dataset_type:
            return dataset_type
        return FeatureTypeUnion[dataset_type, loaded_type]  # type: ignore[misc]

    # ------------------------------------------------------------------
    # compiled task
    # ------------------------------------------------------------------

    def dataset_task(self) -> Task:
        """Compile the reader into a Task (reference: dataset.py:282-300)."""
        if self._dataset_task is not None:
            return self._dataset_task
        if self._reader is None:
            raise ValueError(f"dataset '{self.name}' has no @reader registered")

        reader = self._reader

        def dataset_task(**kwargs):
            return reader(**kwargs)

        self._dataset_task = inner_task(
            dataset_task,
            owner=self,
            name=reader.__name__,
            task_builder="dataset_task",
            signature=signature(reader),
            **self._reader_task_kwargs,
        )
        return self._dataset_task

    # ------------------------------------------------------------------
    # composition
    # ------------------------------------------------------------------

    def get_data(
        self,
        raw_data,
        loader_kwargs: Optional[Dict[str, Any]] = None,
        splitter_kwargs: Optional[Dict[str, Any]] = None,
        parser_kwargs: Optional[Dict[str, Any]] = None,
    ) -> Dict[str, List[Any]]:
        """raw reader output -> {"train": [...], "test": [...]} model-ready
        splits (reference: dataset.py:302-348): loader -> splitter ->
        parser per split -> feature_transformer on each split's feature
        element."""
        loader_kwargs = loader_kwargs or {}
        splitter_kwargs = {**self.splitter_kwargs, **(splitter_kwargs or {})}
        parser_kwargs = {**self.parser_kwargs, **(parser_kwargs or {})}

        if loader_kwargs and self._loader == self._default_loader:
            from unionml_amd.type_guards import GuardError

            raise GuardError(
                "loader_kwargs were provided but no @dataset.loader is registered: "
                "the default loader takes no keyword arguments"
            )
        data = self._loader(raw_data, **loader_kwargs)
        splits = self._splitter(data, **splitter_kwargs)

        out: Dict[str, List[Any]] = {}
        names = ("train",) if len(splits) == 1 else ("train", "test")
        for split_name, split in zip(names, splits):
            parsed = list(self._parser(split, **parser_kwargs))
            parsed[self._parser_feature_key] = self._feature_transformer(
                parsed[self._parser_feature_key]
            )
            out[split_name] = parsed
        return out

    def get_features(self, features):
        """Raw serve-time features -> model-ready features (reference:
        dataset.py:350-359)."""
        return self._feature_transformer(self._feature_loader(features))

    # ------------------------------------------------------------------
    # MI355X staging
    # ------------------------------------------------------------------

    def stage_to_device(self, split: List[Any], device=None, stream=None):
        """Stage a parsed split's arrays into GPU HBM.

        Converts DataFrame/ndarray elements to torch tensors, moves them
        through the pinned transfer pool and issues ``hipMemcpyAsync`` H2D on
        ``stream`` (or the current stream). Non-array elements pass through.
        """
        from unionml_amd.utils.staging import stage_split_to_device

        return stage_split_to_device(split, device=device, stream=stream)

    # ------------------------------------------------------------------
    # SQL constructors
    # ------------------------------------------------------------------

    @classmethod
    def from_task(cls, task, name: Optional[str] = None, **dataset_kwargs) -> "Dataset":
        """Build a Dataset whose reader is an existing :class:`Task` (or
        any annotated callable) — the analog of the reference's
        ``Dataset._from_flytekit_task`` (dataset.py:426-440), which the
        SQL constructors below are built on."""
        ds = cls(name=name or getattr(task, "name", "dataset"), **dataset_kwargs)
        fn = task.fn if hasattr(task, "fn") else task

        def task_reader(**kwargs):
            return task(**kwargs)

        task_reader.__annotations__ = dict(getattr(fn, "__annotations__", {}))
        task_reader.__signature__ = signature(fn)
        ds.reader(task_reader)
        return ds

    @classmethod
    def from_sqlite_task(
        cls,
        name: str,
        db_uri: str,
        query: str,
        **dataset_kwargs,
    ) -> "Dataset":
        """Build a Dataset whose reader executes ``query`` against a
        SQLite database (reference capability: dataset.py:442-456).

        ``query`` may contain ``{param}`` placeholders filled from reader
        kwargs at call time.
        """
        ds = cls(name=name, **dataset_kwargs)

        def sqlite_reader(**query_params) -> pd.DataFrame:
            q = query.format(**query_params) if query_params else query
            with sqlite3.connect(db_uri) as conn:
                return pd.read_sql_query(q, conn)

        sqlite_reader.__annotations__["return"] = pd.DataFrame
        ds.reader(sqlite_reader)
        return ds

    @classmethod
    def from_sqlalchemy_task(
        cls,
        name: str,
        db_uri: str,
        query: str,
        **dataset_kwargs,
    ) -> "Dataset":
        """Build a Dataset whose reader executes ``query`` through a
        SQLAlchemy engine (reference capability: dataset.py:458-470).
        Falls back to sqlite3 for ``sqlite://`` URIs when SQLAlchemy is
        not installed."""
        try:
            import sqlalchemy  # noqa: F401

            ds = cls(name=name, **dataset_kwargs)

            def sqlalchemy_reader(**query_params) -> pd.DataFrame:
                engine = sqlalchemy.create_engine(db_uri)
                q = query.format(**query_params) if query_params else query
                with engine.connect() as conn:
                    return pd.read_sql_query(sqlalchemy.text(q), conn)

            sqlalchemy_reader.__annotations__["return"] = pd.DataFrame
            ds.reader(sqlalchemy_reader)
            return ds
        except ImportError:
            if db_uri.startswith("sqlite:///"):
                return cls.from_sqlite_task(name, db_uri[len("sqlite:///") :], query, **dataset_kwargs)
            raise

    # ------------------------------------------------------------------
    # defaults
    # ------------------------------------------------------------------

    def _default_loader(self, data: Any) -> Any:
        """Coerce reader output to a DataFrame when the reader is
        DataFrame-typed (reference: dataset.py:472-476)."""
        (_, data_type), = self.dataset_datatype.items()
        if data_type is pd.DataFrame and not isinstance(data, pd.DataFrame):
            return pd.DataFrame(data)
        return data

    def _default_splitter(self, data: Any, *, test_size: float, shuffle: bool, random_state: int):
        if not isinstance(data, pd.DataFrame):
            return (data,)
        return _train_test_split(data, test_size=test_size, shuffle=shuffle, random_state=random_state)

    def _default_parser(
        self, data: Any, features: Optional[List[str]], targets: Optional[List[str]]
    ) -> Tuple[Any, Any]:
        if not isinstance(data, pd.DataFrame):
            return (data,)  # type: ignore[return-value]
        # corrected guard (the reference recomputes features when BOTH are
        # provided — dataset.py:498-499; here: only when features is missing)
        if not features and targets:
            features = [col for col in data.columns if col not in targets]
        try:
            target_data = data[targets] if targets else pd.DataFrame()
        except KeyError:
            target_data = pd.DataFrame()
        return data[features] if features else data, target_data

    def _default_feature_loader(self, features: Any) -> Any:
        """Accept a JSON file path, JSON-able records, arrays or a frame
        (reference: dataset.py:506-520)."""
        if isinstance(features, (str, Path)) and Path(features).exists():
            with open(features) as f:
                features = json.load(f)

        (_, data_type), = self.dataset_datatype.items()
        if data_type is pd.DataFrame:
            feature_names = self._features
            # serve-time fast path: list of records with declared feature
            # columns -> build one ndarray per column (an order of
            # magnitude cheaper than pandas' per-record inference; this
            # sits on the /predict hot path). Per-column arrays keep each
            # column's dtype (ints stay ints, bools stay bools) — a
            # single float64 matrix would silently change dtypes for
            # dtype-sensitive models.
            if (
                feature_names
                and isinstance(features, list)
                and features
                and isinstance(features[0], dict)
            ):
                try:
                    cols = {c: np.asarray([rec[c] for rec in features]) for c in feature_names}
                    if all(v.dtype != object for v in cols.values()):
                        return pd.DataFrame(cols, columns=feature_names)
                except (KeyError, TypeError, ValueError):
                    pass  # missing keys / ragged values: general path
            data = pd.DataFrame(features)
            if not feature_names and self._targets:
                feature_names = [col for col in data.columns if col not in self._targets]
            return data[feature_names] if feature_names else data
        return features

    def _default_feature_transformer(self, features: Any) -> Any:
        return features

    def __repr__(self):
        return f"Dataset(name={self.name!r}, features={self._features}, targets={self._targets})"
