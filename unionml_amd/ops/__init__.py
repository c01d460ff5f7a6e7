"""HIP extension loader for the CDNA4 hot-path kernels.

The extension is built IN-TREE (``python setup.py build_ext --inplace``)
so the ``.so`` ships with the repo snapshot. On a GPU machine a missing
or unloadable extension is a loud error (KernelExtensionNotBuilt) — the
HIP path must never silently fall back to eager PyTorch there. On
CPU-only machines the pure-torch reference implementations in
:mod:`unionml_amd.ops.reference` are used instead (they are also the
numerics oracle for the GPU tests).
"""

import importlib
from typing import Optional

from unionml_amd.exceptions import KernelExtensionNotBuilt

_ext = None
_load_error: Optional[BaseException] = None


def _try_load():
    global _ext, _load_error
    if _ext is not None:
        return _ext
    try:
        import torch  # noqa: F401  (loads libc10/libtorch the ext links against)

        _ext = importlib.import_module("unionml_amd.ops._tabular_hip")
    except ImportError as exc:
        _load_error = exc
        _ext = None
    return _ext


def hip_ext(required: Optional[bool] = None):
    """Return the loaded extension module.

    ``required=None`` resolves to "GPU present": with a visible GPU the
    extension MUST load; without one, returns None (CPU fallback).
    """
    ext = _try_load()
    if required is None:
        import torch

        required = torch.cuda.is_available()
    if required and ext is None:
        raise KernelExtensionNotBuilt(
            "unionml_amd.ops._tabular_hip is not built for this machine. Build it "
            "in-tree with: PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext "
            f"--inplace (import error: {_load_error})"
        )
    return ext


def hip_available() -> bool:
    import torch

    return torch.cuda.is_available() and _try_load() is not None
