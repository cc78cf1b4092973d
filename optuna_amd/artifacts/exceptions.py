"""Artifact exceptions (parity: reference ``optuna/artifacts/exceptions.py``)."""
from optuna_amd.exceptions import OptunaError


class ArtifactNotFound(OptunaError):
    """Raised when an artifact is not found in the store."""
