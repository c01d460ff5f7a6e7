"""unionml_amd — an MI355X-native ML-microservice framework.

Provides the ``Dataset``/``Model`` decorator API of unionai-oss/unionml
(reference: unionml/__init__.py:4-5) re-designed for AMD Instinct MI355X:
the default tabular hot path runs on hand-written CDNA4 HIP kernels
(MFMA + LDS tiling for gfx950), serving dynamic-batches into
hipGraph-captured inference, and ``Model.train`` scales data-parallel
across the 8 GPUs of a node with gradient all-reduce on RCCL over xGMI.
"""

from unionml_amd.dataset import Dataset
from unionml_amd.defaults import Resources
from unionml_amd.model import Model, ModelArtifact, BaseHyperparameters
from unionml_amd.schedule import Schedule

__all__ = ["Dataset", "Model", "ModelArtifact", "BaseHyperparameters", "Schedule", "Resources"]

__version__ = "0.1.0"
