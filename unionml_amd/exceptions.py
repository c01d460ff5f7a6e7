"""Framework exceptions (reference: unionml/exceptions.py:4-5)."""


class ModelArtifactNotFound(Exception):
    """Raised when a model artifact could not be found or resolved."""


class TrainingError(Exception):
    """Raised when a training run fails inside the execution engine."""


class KernelExtensionNotBuilt(RuntimeError):
    """Raised on a GPU machine when the gfx950 HIP extension is missing.

    The HIP hot path must never silently fall back to eager PyTorch on a
    GPU box; this error makes a missing/failed extension build loud.
    """


class VersionFetchError(Exception):
    """Raised when the app version (git sha) cannot be determined
    (reference: unionml/remote.py:26)."""
