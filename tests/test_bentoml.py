"""BentoML integration tests (coverage shape of the reference's
tests/unit/test_bentoml.py:17-60 — configure/save/load against a bento
store). When the real bentoml package is importable these run against
it; otherwise they run against tests/fake_bentoml.py, which mirrors the
API surface our integration touches, so the contract stays covered in
minimal CI environments."""

import sys

import numpy as np
import pandas as pd
import pytest

from tests.model_fixtures import build_sklearn_app


def _detect_real_bentoml() -> bool:
    try:
        import bentoml  # noqa: F401

        return True
    except ImportError:
        return False


# decide BEFORE any fixture installs the fake into sys.modules
_REAL_BENTOML = _detect_real_bentoml()


def _have_real_bentoml() -> bool:
    return _REAL_BENTOML


@pytest.fixture()
def bentoml_env(monkeypatch):
    if _have_real_bentoml():
        import bentoml

        yield bentoml
        return
    from tests import fake_bentoml

    yield fake_bentoml.install(monkeypatch)


@pytest.fixture()
def trained_model():
    model = build_sklearn_app()
    model.train()
    return model


def test_requires_bentoml_without_install(monkeypatch):
    if _have_real_bentoml():
        pytest.skip("bentoml installed")
    from unionml_amd.services.bentoml import BentoMLService

    svc = BentoMLService(build_sklearn_app())
    with pytest.raises(ImportError, match="bentoml"):
        svc.configure()


def test_configure_builds_runner_and_service(bentoml_env, trained_model):
    from unionml_amd.services.bentoml import BentoMLService

    service = BentoMLService(trained_model, name="digits_svc")
    with pytest.raises(RuntimeError, match="configure"):
        service.svc
    svc = service.configure(features=pd.DataFrame, predictions=list)
    assert service.svc is svc
    assert svc.name == "digits_svc"
    assert len(svc.runners) == 1


def test_runnable_predicts_through_feature_pipeline(bentoml_env, trained_model):
    from unionml_amd.services.bentoml import create_runnable

    runnable_cls = create_runnable(trained_model)
    assert "amd.com/gpu" in runnable_cls.SUPPORTED_RESOURCES  # MI355X, not nvidia
    instance = runnable_cls()
    feats = [{"x1": 0.5, "x2": 0.1, "x3": 0.9}]
    preds = instance.predict(feats)
    assert len(preds) == 1


def test_service_api_round_trip(bentoml_env, trained_model):
    """The registered predict API must run features through the runner
    (fake store executes synchronously; real bentoml runners need a
    server, so this leg is fake-only)."""
    if _have_real_bentoml():
        pytest.skip("real bentoml runners require an initialized server")
    from unionml_amd.services.bentoml import BentoMLService

    service = BentoMLService(trained_model)
    svc = service.configure()
    feats = [{"x1": 0.5, "x2": 0.1, "x3": 0.9}]
    out = svc.apis["predict"]["fn"](feats)
    assert len(out) == 1


def test_save_load_model_store(bentoml_env, trained_model):
    from unionml_amd.services.bentoml import BentoMLService

    service = BentoMLService(trained_model, name="digits_store")
    tag = service.save_model()
    assert str(tag).startswith("digits_store")

    # a fresh model object loads from the store and repopulates artifact
    fresh = build_sklearn_app()
    assert fresh.artifact is None
    service2 = BentoMLService(fresh, name="digits_store")
    obj = service2.load_model()
    assert fresh.artifact is not None
    np.testing.assert_allclose(obj.coef_, trained_model.artifact.model_object.coef_)


def test_save_model_requires_artifact(bentoml_env):
    from unionml_amd.services.bentoml import BentoMLService

    service = BentoMLService(build_sklearn_app())
    with pytest.raises(ValueError, match="train first"):
        service.save_model()


def test_infer_framework_and_io_descriptor(bentoml_env):
    from sklearn.linear_model import LogisticRegression
    from typing import Dict, List

    from unionml_amd.ops.tabular import TabularMLP
    from unionml_amd.services.bentoml import infer_framework, infer_io_descriptor

    assert infer_framework(LogisticRegression()) == "sklearn"
    assert infer_framework(TabularMLP(device="cpu")) == "pytorch"
    assert infer_framework(object()) == "picklable_model"

    assert infer_io_descriptor(np.ndarray) == "NumpyNdarray"
    assert infer_io_descriptor(pd.DataFrame) == "PandasDataFrame"
    assert infer_io_descriptor(List[float]) == "JSON"
    assert infer_io_descriptor(Dict[str, float]) == "JSON"
    assert infer_io_descriptor(int) is None


def test_configure_infers_io_from_app_types(bentoml_env):
    """configure() without explicit features/predictions infers the IO
    descriptors from the dataset's declared feature type (unwrapping
    FeatureTypeUnion to the serve-time side) and the predictor's return
    annotation (reference defaults its IO mapping the same way:
    services/bentoml.py:33-38, 238-247)."""
    from unionml_amd.services.bentoml import BentoMLService

    model = build_sklearn_app()

    @model._dataset.feature_loader
    def feature_loader(data) -> np.ndarray:
        return np.asarray(data, dtype=np.float64)

    model.train()
    svc = BentoMLService(model, name="io_infer_app").configure()
    api = svc.apis["predict"]
    # FeatureTypeUnion[dataset_type, np.ndarray] -> serve side -> NumpyNdarray
    assert type(api["input"]).__name__ == "NumpyNdarray", type(api["input"])
    # predictor returns List[float] -> JSON
    assert type(api["output"]).__name__ == "JSON", type(api["output"])
