"""Signature validation for every registered callable.

Validates user functions at decoration time so API misuse fails early
with an actionable message, mirroring the guard set of the reference
(unionml/type_guards.py:79-254) with an independent implementation.

Design notes (MI355X build): guards are pure ``inspect``/``typing``
checks — they never import torch or touch a device, so decoration stays
cheap and importable in CPU-only task-resolver containers.
"""

import inspect
import typing
from typing import Any, Callable, Dict, NamedTuple, Optional, Tuple, Type


class GuardError(TypeError):
    """Raised when a registered callable has an invalid signature."""


# keyword arguments the default splitter/parser understand; custom
# splitters/parsers must accept (data, **their own kwargs) and the kwargs
# must be keyword-friendly (reference: type_guards.py:12-21).
SPLITTER_KWTYPES: Dict[str, type] = {"test_size": float, "shuffle": bool, "random_state": int}
PARSER_KWTYPES: Dict[str, type] = {"features": Optional[typing.List[str]], "targets": typing.List[str]}


def _sig(fn: Callable) -> inspect.Signature:
    try:
        return inspect.signature(fn)
    except (TypeError, ValueError) as exc:  # builtins etc.
        raise GuardError(f"cannot inspect signature of {fn!r}: {exc}") from exc


def _name(fn: Callable) -> str:
    return getattr(fn, "__name__", repr(fn))


def _positional_params(sig: inspect.Signature):
    return [
        p
        for p in sig.parameters.values()
        if p.kind in (p.POSITIONAL_ONLY, p.POSITIONAL_OR_KEYWORD)
    ]


def _keyword_only_params(sig: inspect.Signature):
    return [p for p in sig.parameters.values() if p.kind == p.KEYWORD_ONLY]


def _types_compatible(expected: Any, actual: Any) -> bool:
    """True if ``actual`` annotation is acceptable where ``expected`` is required.

    Unannotated (`inspect.Parameter.empty`) or Any on either side is
    accepted; typing generics compare on origin (List[int] ~ list).
    """
    if expected is inspect.Parameter.empty or actual is inspect.Parameter.empty:
        return True
    if expected is Any or actual is Any:
        return True
    # a FeatureTypeUnion[dataset_type, loaded_type] accepts either arm
    if getattr(typing.get_origin(expected), "__name__", "") == "FeatureTypeUnion":
        return any(_types_compatible(arm, actual) for arm in typing.get_args(expected))
    exp_origin = typing.get_origin(expected) or expected
    act_origin = typing.get_origin(actual) or actual
    if exp_origin is typing.Union or act_origin is typing.Union:
        exp_set = set(typing.get_args(expected)) if exp_origin is typing.Union else {expected}
        act_set = set(typing.get_args(actual)) if act_origin is typing.Union else {actual}
        return bool(exp_set & act_set) or bool({Any} & (exp_set | act_set))
    if isinstance(exp_origin, type) and isinstance(act_origin, type):
        return issubclass(act_origin, exp_origin) or issubclass(exp_origin, act_origin)
    return exp_origin == act_origin


def _check_input_data_type(fn_name: str, actual: Any, expected: Any) -> None:
    """The first-argument annotation must be compatible with the type the
    upstream stage produces (reference: type_guards.py:28-40)."""
    if expected is None:
        return
    if not _types_compatible(expected, actual):
        raise GuardError(
            f"'{fn_name}' first argument annotated {actual} is not compatible with "
            f"the expected input type {expected} produced by the upstream stage"
        )


def _is_split_container(ann: Any) -> bool:
    origin = typing.get_origin(ann)
    if origin in (tuple, list):
        return True
    # NamedTuple classes subclass tuple
    return isinstance(ann, type) and issubclass(ann, tuple)


def _check_split_output(fn_name: str, ann: Any, expected: Any, source: str) -> None:
    """A splitter must return a List/Tuple/NamedTuple of the loaded data
    type (reference: type_guards.py:43-57)."""
    if ann is inspect.Signature.empty or ann is Any or expected is None:
        return
    if not _is_split_container(ann):
        raise GuardError(
            f"'{fn_name}' must return a List, Tuple or NamedTuple of data splits; "
            f"annotated {ann}"
        )
    for sub in typing.get_args(ann):
        if sub is Ellipsis:
            continue
        if not _types_compatible(expected, sub):
            raise GuardError(
                f"'{fn_name}' split elements annotated {sub} must match the "
                f"'{source}' output type {expected}"
            )


def _check_data_args(fn_name: str, params, expected_types) -> None:
    """Positional data arguments (after the model) must be compatible
    with the parser's per-split return types (reference:
    type_guards.py:118-132, 135-148)."""
    if not expected_types:
        return
    for p, expected in zip(params, expected_types):
        if not _types_compatible(expected, p.annotation):
            raise GuardError(
                f"'{fn_name}' data argument '{p.name}' annotated {p.annotation} is not "
                f"compatible with the parser output type {expected}"
            )


def guard_reader(fn: Callable) -> None:
    """A reader may take arbitrary kwargs but MUST annotate its return
    (the return type drives downstream type derivation —
    reference: type_guards.py:79-85)."""
    sig = _sig(fn)
    if sig.return_annotation is inspect.Signature.empty:
        raise GuardError(
            f"reader '{_name(fn)}' must have a return type annotation: the dataset "
            "derives its raw-data type from it"
        )


def guard_loader(fn: Callable, expected_data_type: Any = None) -> None:
    """loader(raw_data) -> loaded_data; exactly one positional argument
    whose annotation must accept the reader's return type."""
    sig = _sig(fn)
    pos = _positional_params(sig)
    if len(pos) != 1:
        raise GuardError(
            f"loader '{_name(fn)}' must take exactly one positional argument "
            f"(the reader output), got {len(pos)}"
        )
    _check_input_data_type(f"loader '{_name(fn)}'", pos[0].annotation, expected_data_type)


def guard_splitter(
    fn: Callable, expected_data_type: Any = None, expected_type_source: str = "loader"
) -> None:
    """splitter(data, *, test_size, shuffle, random_state) -> train/test splits."""
    sig = _sig(fn)
    pos = _positional_params(sig)
    if len(pos) != 1:
        raise GuardError(
            f"splitter '{_name(fn)}' must take exactly one positional argument "
            f"(the loaded data), got {len(pos)}"
        )
    _check_input_data_type(f"splitter '{_name(fn)}'", pos[0].annotation, expected_data_type)
    _check_split_output(
        f"splitter '{_name(fn)}'", sig.return_annotation, expected_data_type,
        expected_type_source,
    )
    kws = {p.name for p in _keyword_only_params(sig)}
    missing = set(SPLITTER_KWTYPES) - kws
    has_var_kw = any(p.kind == p.VAR_KEYWORD for p in sig.parameters.values())
    if missing and not has_var_kw:
        raise GuardError(
            f"splitter '{_name(fn)}' must accept keyword-only arguments "
            f"{sorted(SPLITTER_KWTYPES)}; missing {sorted(missing)}"
        )


def guard_parser(fn: Callable, expected_data_type: Any = None) -> None:
    """parser(data, features, targets) -> tuple of parsed outputs."""
    sig = _sig(fn)
    pos = _positional_params(sig)
    if len(pos) != 3:
        raise GuardError(
            f"parser '{_name(fn)}' must take three positional arguments "
            f"(data, features, targets), got {len(pos)}"
        )
    _check_input_data_type(f"parser '{_name(fn)}'", pos[0].annotation, expected_data_type)
    if sig.return_annotation is inspect.Signature.empty:
        raise GuardError(
            f"parser '{_name(fn)}' must annotate its return type as a tuple; the "
            "trainer/evaluator data argument types are derived from it"
        )


def guard_feature_loader(fn: Callable) -> None:
    sig = _sig(fn)
    pos = _positional_params(sig)
    if len(pos) != 1:
        raise GuardError(
            f"feature_loader '{_name(fn)}' must take exactly one positional argument, "
            f"got {len(pos)}"
        )


def guard_feature_transformer(fn: Callable) -> None:
    sig = _sig(fn)
    pos = _positional_params(sig)
    if len(pos) != 1:
        raise GuardError(
            f"feature_transformer '{_name(fn)}' must take exactly one positional "
            f"argument, got {len(pos)}"
        )


def guard_trainer(
    fn: Callable,
    model_type: Optional[type],
    expected_data_args: int,
    expected_data_types: Optional[Tuple[Any, ...]] = None,
) -> None:
    """trainer(model, *data_args, **hyperparam/kwargs) -> model.

    ``expected_data_args`` is the number of elements the parser returns
    for one split (reference: type_guards.py:118-132 checks arg0 and the
    return against the model type and counts the positional data args).
    """
    sig = _sig(fn)
    pos = _positional_params(sig)
    if len(pos) < 1:
        raise GuardError(f"trainer '{_name(fn)}' must take the model object as its first argument")
    n_data = len(pos) - 1
    if expected_data_args and n_data != expected_data_args:
        raise GuardError(
            f"trainer '{_name(fn)}' takes {n_data} positional data argument(s) after the "
            f"model, but the dataset parser provides {expected_data_args}"
        )
    _check_data_args(f"trainer '{_name(fn)}'", pos[1:], expected_data_types)
    if model_type is not None:
        a0 = pos[0].annotation
        if not _types_compatible(model_type, a0):
            raise GuardError(
                f"trainer '{_name(fn)}' first argument annotated {a0} is not compatible "
                f"with the model type {model_type}"
            )
        ra = sig.return_annotation
        if ra is not inspect.Signature.empty and not _types_compatible(model_type, ra):
            raise GuardError(
                f"trainer '{_name(fn)}' must return the model type {model_type}, "
                f"annotated {ra}"
            )


def guard_evaluator(
    fn: Callable,
    model_type: Optional[type],
    expected_data_args: int,
    expected_data_types: Optional[Tuple[Any, ...]] = None,
) -> None:
    """evaluator(model, *data_args) -> float."""
    sig = _sig(fn)
    pos = _positional_params(sig)
    if len(pos) < 1:
        raise GuardError(f"evaluator '{_name(fn)}' must take the model object as its first argument")
    n_data = len(pos) - 1
    if expected_data_args and n_data != expected_data_args:
        raise GuardError(
            f"evaluator '{_name(fn)}' takes {n_data} positional data argument(s) after the "
            f"model, but the dataset parser provides {expected_data_args}"
        )
    _check_data_args(f"evaluator '{_name(fn)}'", pos[1:], expected_data_types)
    if model_type is not None:
        a0 = pos[0].annotation
        if not _types_compatible(model_type, a0):
            raise GuardError(
                f"evaluator '{_name(fn)}' first argument annotated {a0} is not compatible "
                f"with the model type {model_type}"
            )


def guard_predictor(
    fn: Callable, model_type: Optional[type], feature_type: Any = None
) -> None:
    """predictor(model, features) -> predictions; return annotation required."""
    sig = _sig(fn)
    pos = _positional_params(sig)
    if len(pos) != 2:
        raise GuardError(
            f"predictor '{_name(fn)}' must take exactly two positional arguments "
            f"(model, features), got {len(pos)}"
        )
    if sig.return_annotation is inspect.Signature.empty:
        raise GuardError(
            f"predictor '{_name(fn)}' must annotate its return type: serving derives the "
            "response schema from it"
        )
    if feature_type is not None and not _types_compatible(feature_type, pos[1].annotation):
        raise GuardError(
            f"predictor '{_name(fn)}' features argument annotated {pos[1].annotation} is "
            f"not compatible with the dataset feature type {feature_type}"
        )
    if model_type is not None:
        a0 = pos[0].annotation
        if not _types_compatible(model_type, a0):
            raise GuardError(
                f"predictor '{_name(fn)}' first argument annotated {a0} is not compatible "
                f"with the model type {model_type}"
            )


def guard_prediction_callback(fn: Callable) -> None:
    """callback(model, features, predictions) -> None."""
    sig = _sig(fn)
    pos = _positional_params(sig)
    if len(pos) != 3:
        raise GuardError(
            f"prediction callback '{_name(fn)}' must take three positional arguments "
            f"(model, features, predictions), got {len(pos)}"
        )
    ra = sig.return_annotation
    if ra not in (inspect.Signature.empty, None, type(None)):
        raise GuardError(
            f"prediction callback '{_name(fn)}' must return None, annotated {ra}"
        )


def guard_saver(fn: Callable) -> None:
    """saver(model, hyperparameters, file, **kwargs)."""
    sig = _sig(fn)
    pos = _positional_params(sig)
    if len(pos) < 2:
        raise GuardError(
            f"saver '{_name(fn)}' must take at least (model, file) positionally, got {len(pos)}"
        )


def guard_loader_fn(fn: Callable) -> None:
    """loader(file, **kwargs) -> model."""
    sig = _sig(fn)
    pos = _positional_params(sig)
    if len(pos) < 1:
        raise GuardError(f"model loader '{_name(fn)}' must take the file/path as its first argument")
