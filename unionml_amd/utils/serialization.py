"""CPU-portable payloads for process boundaries.

Plain pickling of CUDA-resident tensors/Modules does not round-trip
across processes (torch's non-``torch.save`` storage path), and a model
trained on one GPU must load on a machine with a different (or no)
device anyway. The backend therefore moves every tensor payload to CPU
at its execution boundaries: inputs before dispatch
(unionml_amd/remote.py) and outputs before they land in the execution
directory (unionml_amd/runner.py). Consumers re-place on device.
"""

import copy
from typing import Any


def tensors_to_cpu(obj: Any, _copy_modules: bool = True) -> Any:
    """Return a CPU-resident equivalent of ``obj``.

    Tensors are moved; ``nn.Module``s are deep-copied first so the
    caller's device-resident module is not mutated; containers recurse.
    Objects with their own ``__getstate__`` (e.g. TabularMLP) already
    pickle CPU-portably and pass through.
    """
    try:
        import torch
    except ImportError:
        return obj
    if isinstance(obj, torch.nn.Module):
        if any(p.is_cuda for p in obj.parameters()) or any(
            b.is_cuda for b in obj.buffers()
        ):
            module = copy.deepcopy(obj) if _copy_modules else obj
            return module.cpu()
        return obj
    if torch.is_tensor(obj):
        return obj.detach().cpu() if obj.is_cuda else obj
    if isinstance(obj, tuple):
        vals = [tensors_to_cpu(o, _copy_modules) for o in obj]
        # preserve NamedTuple subtypes (e.g. splitter outputs)
        return type(obj)(*vals) if hasattr(obj, "_fields") else tuple(vals)
    if isinstance(obj, list):
        return [tensors_to_cpu(o, _copy_modules) for o in obj]
    if isinstance(obj, dict):
        return {k: tensors_to_cpu(v, _copy_modules) for k, v in obj.items()}
    return obj
