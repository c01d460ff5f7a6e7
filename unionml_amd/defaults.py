"""Default resource requests for compiled tasks
(reference: unionml/defaults.py:5 — Resources(cpu="1", mem="1Gi")).

In the MI355X-native build a task's resources also carry a GPU count so
compiled tasks can request ``amd.com/gpu`` devices.
"""

from dataclasses import dataclass


@dataclass(frozen=True)
class Resources:
    cpu: str = "1"
    mem: str = "1Gi"
    gpu: int = 0  # number of amd.com/gpu (MI355X) devices


DEFAULT_RESOURCES = Resources(cpu="1", mem="1Gi", gpu=0)
GPU_RESOURCES = Resources(cpu="8", mem="32Gi", gpu=1)
