"""Instance tracking: record where each Dataset/Model was instantiated.

The remote task resolver rehydrates a compiled task inside a worker
process by importing the app module and looking the object up by its
left-hand-side variable name. This module records, at instantiation
time, which module (and file, for ``__main__``) each tracked instance
was created in (reference semantics: unionml/tracker.py:22-99).

Independent design: instead of walking the interpreter stack at
``find_lhs`` time, we capture the instantiation frame eagerly in the
metaclass ``__call__`` and resolve the variable name lazily by scanning
the module namespace — which also covers the ``__main__``-module edge
case by re-importing the main file under its stem name.
"""

import importlib
import importlib.util
import inspect
import sys
from pathlib import Path
from typing import Optional


def import_module_from_file(module_name: str, file_path: str):
    """Import ``file_path`` as ``module_name`` (reference: tracker.py:11-18)."""
    spec = importlib.util.spec_from_file_location(module_name, file_path)
    if spec is None or spec.loader is None:
        raise ImportError(f"cannot build import spec for {file_path}")
    module = importlib.util.module_from_spec(spec)
    sys.modules[module_name] = module
    spec.loader.exec_module(module)
    return module


class InstanceTrackingMeta(type):
    """Metaclass that records the instantiating module for each instance."""

    @staticmethod
    def _instantiating_module() -> Optional[str]:
        """Walk up the stack to the first frame outside this package."""
        pkg_dir = str(Path(__file__).parent)
        frame = inspect.currentframe()
        try:
            while frame is not None:
                fname = frame.f_code.co_filename
                if not fname.startswith(pkg_dir) and "importlib" not in fname:
                    mod = frame.f_globals.get("__name__")
                    return mod
                frame = frame.f_back
        finally:
            del frame
        return None

    @staticmethod
    def _module_file(module_name: str) -> Optional[str]:
        mod = sys.modules.get(module_name)
        return getattr(mod, "__file__", None) if mod is not None else None

    def __call__(cls, *args, **kwargs):
        instance = super().__call__(*args, **kwargs)
        module_name = InstanceTrackingMeta._instantiating_module()
        instance._instantiated_in = module_name
        instance._module_file = (
            InstanceTrackingMeta._module_file(module_name) if module_name else None
        )
        return instance


class TrackedInstance(metaclass=InstanceTrackingMeta):
    """Base for Dataset/Model: knows its own module and variable name."""

    _instantiated_in: Optional[str]
    _module_file: Optional[str]

    @property
    def instantiated_in(self) -> Optional[str]:
        return getattr(self, "_instantiated_in", None)

    def find_lhs(self) -> str:
        """Find the variable name this instance is bound to in its module.

        Scans the instantiating module's namespace for an attribute that
        *is* this instance; for ``__main__`` the module file is
        re-imported under its stem so worker processes can do the same
        (reference semantics: tracker.py:78-99).
        """
        module_name = self.instantiated_in
        if module_name is None:
            raise ValueError(f"cannot determine the instantiating module of {self!r}")

        module = sys.modules.get(module_name)
        if module is not None:
            for var_name, value in vars(module).items():
                if value is self:
                    return var_name

        # __main__ edge case: re-import the script by file under its stem
        # name and find an equivalent instance (same type + same app name).
        if self._module_file is not None:
            stem = Path(self._module_file).stem
            mod = sys.modules.get(stem)
            if mod is None:
                mod = import_module_from_file(stem, self._module_file)
            for var_name, value in vars(mod).items():
                if isinstance(value, type(self)) and getattr(value, "name", None) == getattr(
                    self, "name", None
                ):
                    return var_name

        raise ValueError(
            f"could not find a module-level variable bound to {self!r} in "
            f"module '{module_name}'. Assign the instance to a module-level name."
        )

    @property
    def app_module(self) -> str:
        """The import path a worker should use to rehydrate this instance."""
        module_name = self.instantiated_in
        if module_name == "__main__" and self._module_file is not None:
            return Path(self._module_file).stem
        if module_name is None:
            raise ValueError(f"cannot determine app module of {self!r}")
        return module_name
