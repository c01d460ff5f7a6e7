"""Environment probes."""

import importlib.util


def module_is_installed(name: str) -> bool:
    """True if ``name`` is importable (reference: unionml/utils.py:71-76)."""
    return importlib.util.find_spec(name) is not None


def gpu_available() -> bool:
    try:
        import torch

        return torch.cuda.is_available()
    except Exception:
        return False
