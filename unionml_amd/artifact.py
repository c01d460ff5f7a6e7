"""Model-artifact persistence — default saver/loader per framework.

Format parity with the reference (unionml/model.py:1432-1519):

- sklearn estimators -> ``joblib.dump({"model_obj", "hyperparameters"})``
- torch modules      -> ``torch.save({"model_obj": state_dict, "hyperparameters"})``
  with load re-initializing the module from hyperparameters then
  ``load_state_dict``
- keras models       -> ``model.save`` / ``keras.models.load_model``
- anything else      -> joblib pickle fallback

MI355X note: torch checkpoints are saved from CPU tensors (state dicts
are moved off-device first) so artifacts restore on any machine, and
loads map to CPU then let the caller place on the ROCm device.
"""

from typing import Any, Callable, Dict, NamedTuple, Optional, Union

from unionml_amd._logging import logger


class ModelArtifact(NamedTuple):
    """A trained model plus its provenance (reference: model.py:46-56)."""

    model_object: Any
    hyperparameters: Optional[Union[dict, Any]] = None
    metrics: Optional[Dict[str, float]] = None


def is_torch_model(model_type: type) -> bool:
    """True when the class (or a base) comes from torch (reference:
    utils.py:63-64 checks module prefixes)."""
    return any(
        getattr(t, "__module__", "").split(".")[0] == "torch"
        for t in (model_type, *getattr(model_type, "__mro__", ()))
    )


def is_sklearn_model(model_type: type) -> bool:
    return any(
        getattr(t, "__module__", "").split(".")[0] == "sklearn"
        for t in (model_type, *getattr(model_type, "__mro__", ()))
    )


def is_keras_model(model_type: type) -> bool:
    return any(
        getattr(t, "__module__", "").split(".")[0] in ("keras", "tensorflow")
        for t in (model_type, *getattr(model_type, "__mro__", ()))
    )


def _hyperparameters_to_dict(hyperparameters) -> Any:
    import dataclasses

    if hyperparameters is None:
        return None
    if dataclasses.is_dataclass(hyperparameters):
        return dataclasses.asdict(hyperparameters)
    return hyperparameters


def default_saver(model_obj: Any, hyperparameters, file: Any, **kwargs) -> Any:
    """Serialize ``model_obj`` (+ hyperparameters) to ``file`` (path,
    Path, or binary file object)."""
    model_type = type(model_obj)
    hp = _hyperparameters_to_dict(hyperparameters)

    if is_keras_model(model_type):
        model_obj.save(file, **kwargs)
        return file

    if is_torch_model(model_type):
        import torch

        state_dict = model_obj.state_dict()
        cpu_state = {k: v.detach().cpu() if hasattr(v, "detach") else v for k, v in state_dict.items()}
        torch.save({"model_obj": cpu_state, "hyperparameters": hp}, file, **kwargs)
        return file

    import joblib

    joblib.dump({"model_obj": model_obj, "hyperparameters": hp}, file, **kwargs)
    return file


def default_loader(
    file: Any,
    model_type: Optional[type] = None,
    init: Optional[Callable] = None,
    hyperparameter_type: Optional[type] = None,
    **kwargs,
) -> Any:
    """Deserialize a model saved by :func:`default_saver`.

    For torch models, re-initializes the module via ``init`` (or the
    model type) from the stored hyperparameters then loads the state
    dict (reference: model.py:1501-1511).
    """
    if model_type is not None and is_keras_model(model_type):
        import keras

        return keras.models.load_model(file, **kwargs)

    if model_type is not None and is_torch_model(model_type):
        import torch

        payload = torch.load(file, map_location="cpu", weights_only=False, **kwargs)
        state_dict, hp = payload["model_obj"], payload.get("hyperparameters")
        hp = hp or {}
        if hyperparameter_type is not None and isinstance(hp, dict):
            try:
                hp_obj = hyperparameter_type(**hp)
            except TypeError:
                hp_obj = hp
        else:
            hp_obj = hp
        if init is not None:
            model_obj = init(hyperparameters=hp if isinstance(hp, dict) else vars(hp_obj))
        else:
            model_obj = model_type(**hp) if isinstance(hp, dict) else model_type()
        model_obj.load_state_dict(state_dict)
        return model_obj

    import joblib

    payload = joblib.load(file, **kwargs)
    if isinstance(payload, dict) and "model_obj" in payload:
        return payload["model_obj"]
    return payload
