"""BentoML serving integration — wraps a unionml_amd Model in a BentoML
Service/Runner with IO-descriptor inference.

Capability parity with the reference's unionml/services/bentoml.py:31-247:
``BentoMLService.configure/save_model/load_model``, a service factory
with sync/async predict APIs, and a runnable whose ``predict`` method is
``dataset.get_features`` → ``model.predict``. The runnable advertises
``amd.com/gpu`` (MI355X) instead of the reference's ``nvidia.com/gpu``
(services/bentoml.py:202).

BentoML itself is an optional dependency: constructing a
:class:`BentoMLService` without it installed raises ImportError with
install guidance, mirroring the reference's conditional import
(services/__init__.py:4-6).
"""

from typing import Any, Callable, Dict, Optional, Type

from unionml_amd._logging import logger


def _require_bentoml():
    try:
        import bentoml  # noqa: F401

        return bentoml
    except ImportError as exc:
        raise ImportError(
            "BentoML serving requires the 'bentoml' package: pip install bentoml>=1.0"
        ) from exc


#: framework name → bentoml sub-module used for save/load
FRAMEWORK_SAVERS = ("sklearn", "pytorch", "picklable_model")


def infer_framework(model_object: Any) -> str:
    """Pick the bentoml framework module for a model object (reference:
    services/bentoml.py:133-146 dispatches on the same three)."""
    mod = type(model_object).__module__.split(".")[0]
    if mod == "sklearn":
        return "sklearn"
    if mod in ("torch", "unionml_amd") or any(
        b.__module__.startswith("torch") for b in type(model_object).__mro__
    ):
        return "pytorch"
    return "picklable_model"


def infer_io_descriptor(type_: Type) -> Optional[str]:
    """Map a feature/prediction annotation to a bentoml IO descriptor
    name (reference IO_DESCRIPTOR_MAPPING: services/bentoml.py:33-38)."""
    import typing

    if getattr(typing.get_origin(type_), "__name__", "") == "FeatureTypeUnion":
        type_ = typing.get_args(type_)[1]  # serve-time (loaded) type
    origin = typing.get_origin(type_) or type_
    try:
        import numpy as np

        if origin is np.ndarray:
            return "NumpyNdarray"
    except ImportError:
        pass
    try:
        import pandas as pd

        if origin in (pd.DataFrame, pd.Series):
            return "PandasDataFrame"
    except ImportError:
        pass
    if origin in (list, dict, tuple):
        return "JSON"
    return None


class BentoMLService:
    """Configure, save, and serve a unionml_amd Model through BentoML."""

    def __init__(self, model, name: Optional[str] = None):
        self.model = model
        self.name = name or model.name
        self._svc = None
        self._enable_async: bool = False
        self._supported_resources: tuple = ("amd.com/gpu", "cpu")
        self._supports_multi_threading: bool = True

    @property
    def svc(self):
        if self._svc is None:
            raise RuntimeError("call .configure() first")
        return self._svc

    def configure(
        self,
        features: Optional[Type] = None,
        predictions: Optional[Type] = None,
        enable_async: bool = False,
        supported_resources: Optional[tuple] = None,
        supports_multi_threading: bool = True,
        runnable_method_kwargs: Optional[Dict[str, Any]] = None,
        service_kwargs: Optional[Dict[str, Any]] = None,
    ):
        """Build the Runner + Service (reference: services/bentoml.py:72-131)."""
        bentoml = _require_bentoml()

        self._enable_async = enable_async
        if supported_resources:
            self._supported_resources = supported_resources
        self._supports_multi_threading = supports_multi_threading

        runnable = create_runnable(
            self.model,
            supported_resources=self._supported_resources,
            supports_multi_threading=supports_multi_threading,
            method_kwargs=runnable_method_kwargs,
        )
        runner = bentoml.Runner(runnable, name=f"{self.name}_runner")
        self._svc = create_service(
            self,
            runner,
            features=features,
            predictions=predictions,
            enable_async=enable_async,
            **(service_kwargs or {}),
        )
        return self._svc

    def save_model(self, model_object: Any = None, framework: Optional[str] = None):
        """Persist the trained model to the local bento store
        (reference: services/bentoml.py:133-146)."""
        bentoml = _require_bentoml()
        if model_object is None:
            if self.model.artifact is None:
                raise ValueError("no trained artifact; train first or pass model_object")
            model_object = self.model.artifact.model_object
        framework = framework or infer_framework(model_object)
        saver = getattr(bentoml, framework)
        return saver.save_model(self.name, model_object)

    def load_model(self, tag: Optional[str] = None):
        """Load from the bento store and set ``model.artifact``
        (reference: services/bentoml.py:148-162)."""
        bentoml = _require_bentoml()
        from unionml_amd.artifact import ModelArtifact

        tag = tag or f"{self.name}:latest"
        bento_model = bentoml.models.get(tag)
        framework = bento_model.info.module.split(".")[-1]
        loader = getattr(bentoml, framework)
        model_object = loader.load_model(tag)
        self.model.artifact = ModelArtifact(model_object)
        return model_object


def create_runnable(
    model,
    supported_resources: tuple = ("amd.com/gpu", "cpu"),
    supports_multi_threading: bool = True,
    method_kwargs: Optional[Dict[str, Any]] = None,
):
    """A bentoml.Runnable whose predict = get_features → model.predict
    (reference: services/bentoml.py:190-213)."""
    bentoml = _require_bentoml()

    class UnionmlRunnable(bentoml.Runnable):
        SUPPORTED_RESOURCES = supported_resources
        SUPPORTS_CPU_MULTI_THREADING = supports_multi_threading

        @bentoml.Runnable.method(**(method_kwargs or {"batchable": False}))
        def predict(self, features):
            features = model._dataset.get_features(features)
            return model.predict(features=features)

    return UnionmlRunnable


def create_service(
    service: BentoMLService,
    runner,
    features: Optional[Type] = None,
    predictions: Optional[Type] = None,
    enable_async: bool = False,
    **service_kwargs,
):
    """Build the Service with inferred IO descriptors and a predict API
    (reference: services/bentoml.py:165-187)."""
    bentoml = _require_bentoml()
    import bentoml.io as bio

    if features is None:  # default to the dataset's declared feature type
        try:
            features = service.model._dataset.feature_type
        except Exception:
            features = None
    if predictions is None and service.model._predictor is not None:
        import inspect

        ra = inspect.signature(service.model._predictor).return_annotation
        predictions = None if ra is inspect.Signature.empty else ra

    in_name = (features is not None and infer_io_descriptor(features)) or "JSON"
    out_name = (predictions is not None and infer_io_descriptor(predictions)) or "JSON"
    input_io = getattr(bio, in_name)()
    output_io = getattr(bio, out_name)()

    svc = bentoml.Service(service.name, runners=[runner], **service_kwargs)

    if enable_async:

        @svc.api(input=input_io, output=output_io)
        async def predict(features):
            return await runner.predict.async_run(features)

    else:

        @svc.api(input=input_io, output=output_io)
        def predict(features):
            return runner.predict.run(features)

    logger.info("configured bentoml service '%s' (%s -> %s)", service.name, in_name, out_name)
    return svc
