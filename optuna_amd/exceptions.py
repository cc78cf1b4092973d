"""Exception hierarchy.

Parity: reference ``optuna/exceptions.py`` (TrialPruned :22, OptunaError :8,
StorageInternalError :44, DuplicatedStudyError :56).
"""


class OptunaError(Exception):
    """Base class for all framework-specific exceptions."""


class TrialPruned(OptunaError):
    """Raised (by user code or ``Trial.should_prune`` users) to mark a trial pruned.

    Re-exported at package top level; catching it inside the optimize loop maps
    the trial to ``TrialState.PRUNED``.
    """


class CLIUsageError(OptunaError):
    """Command-line usage error."""


class StorageInternalError(OptunaError):
    """Low-level storage failure."""


class DuplicatedStudyError(OptunaError):
    """Raised when a study with the same name already exists."""


class UpdateFinishedTrialError(RuntimeError):
    """Raised on any mutation of a finished trial (storage contract)."""


class ExperimentalWarning(UserWarning):
    """Warning category for experimental API."""
