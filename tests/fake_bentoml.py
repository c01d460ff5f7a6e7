"""A minimal in-memory stand-in for the bentoml API surface our
integration uses (Runnable/Runner/Service/io descriptors/per-framework
model store). Lets the BentoML contract tests run in environments
without bentoml installed; tests/test_bentoml.py exercises the real
package instead whenever it is importable."""

import sys
import types
from typing import Any, Dict


class _Tag(str):
    pass


class Runnable:
    SUPPORTED_RESOURCES = ("cpu",)
    SUPPORTS_CPU_MULTI_THREADING = True

    class method:
        def __init__(self, **kwargs):
            self.kwargs = kwargs

        def __call__(self, fn):
            fn.__bentoml_method__ = self.kwargs
            return fn


class _BoundMethod:
    def __init__(self, instance, fn):
        self._instance = instance
        self._fn = fn

    def run(self, *args, **kwargs):
        return self._fn(self._instance, *args, **kwargs)

    async def async_run(self, *args, **kwargs):
        return self._fn(self._instance, *args, **kwargs)


class Runner:
    def __init__(self, runnable_cls, name: str = ""):
        self.name = name
        self._instance = runnable_cls()
        for attr in dir(runnable_cls):
            fn = getattr(runnable_cls, attr)
            if callable(fn) and hasattr(fn, "__bentoml_method__"):
                setattr(self, attr, _BoundMethod(self._instance, fn))


class Service:
    def __init__(self, name, runners=None, **kwargs):
        self.name = name
        self.runners = runners or []
        self.apis: Dict[str, Any] = {}

    def api(self, input=None, output=None, **kwargs):
        def deco(fn):
            self.apis[fn.__name__] = {"fn": fn, "input": input, "output": output}
            return fn

        return deco


class _IoDescriptor:
    def __init__(self, *a, **k):
        pass


class _ModelInfo:
    def __init__(self, module):
        self.module = module


class _BentoModel:
    def __init__(self, obj, module):
        self.obj = obj
        self.info = _ModelInfo(module)


class _Store:
    def __init__(self):
        self._models: Dict[str, _BentoModel] = {}

    def get(self, tag: str) -> _BentoModel:
        name = str(tag).split(":")[0]
        if name not in self._models:
            raise KeyError(f"model {tag!r} not in store")
        return self._models[name]


class _FrameworkModule:
    def __init__(self, store: _Store, framework: str):
        self._store = store
        self._framework = framework

    def save_model(self, name: str, obj: Any, **kwargs) -> _Tag:
        self._store._models[name] = _BentoModel(obj, f"bentoml.{self._framework}")
        return _Tag(f"{name}:fake")

    def load_model(self, tag: str, **kwargs) -> Any:
        return self._store.get(tag).obj


def install(monkeypatch) -> types.ModuleType:
    """Install the fake as sys.modules['bentoml'] (+ bentoml.io)."""
    mod = types.ModuleType("bentoml")
    store = _Store()
    mod.Runnable = Runnable
    mod.Runner = Runner
    mod.Service = Service
    mod.models = store
    for framework in ("sklearn", "pytorch", "picklable_model"):
        setattr(mod, framework, _FrameworkModule(store, framework))
    io_mod = types.ModuleType("bentoml.io")
    for name in ("JSON", "NumpyNdarray", "PandasDataFrame", "PandasSeries"):
        setattr(io_mod, name, type(name, (_IoDescriptor,), {}))
    mod.io = io_mod
    monkeypatch.setitem(sys.modules, "bentoml", mod)
    monkeypatch.setitem(sys.modules, "bentoml.io", io_mod)
    return mod
