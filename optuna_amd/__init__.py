"""optuna_amd — an MI355X-native hyperparameter-optimization framework.

Drop-in surface of optuna (``create_study`` / ``Study.optimize`` / ``Trial.suggest_*``
/ storages / samplers / pruners), rebuilt MI355X-first: the sampler hot paths (TPE
Parzen-KDE + EI, GP posterior + acqf, CMA-ES eigendecomposition, NSGA-II
hypervolume) run as hand-written HIP/CDNA4 kernels (``optuna_amd._hip``); the
distributed-study path shares an in-memory trial table across one-process-per-GPU
workers over RCCL/xGMI (``optuna_amd.storages.RcclStorage``).

Parity: reference ``optuna/__init__.py`` re-export surface.
"""
from optuna_amd import distributions  # noqa: F401
from optuna_amd import exceptions  # noqa: F401
from optuna_amd import logging  # noqa: F401
from optuna_amd import pruners  # noqa: F401
from optuna_amd import samplers  # noqa: F401
from optuna_amd import search_space  # noqa: F401
from optuna_amd import storages  # noqa: F401
from optuna_amd import study  # noqa: F401
from optuna_amd import trial  # noqa: F401
from optuna_amd._callbacks import MaxTrialsCallback  # noqa: F401
from optuna_amd.exceptions import TrialPruned  # noqa: F401
from optuna_amd.study import (  # noqa: F401
    Study,
    StudyDirection,
    StudySummary,
    copy_study,
    create_study,
    delete_study,
    get_all_study_names,
    get_all_study_summaries,
    load_study,
)
from optuna_amd.trial import FixedTrial, FrozenTrial, Trial, TrialState, create_trial  # noqa: F401
from optuna_amd.version import __version__  # noqa: F401


__all__ = [
    "__version__",
    "FixedTrial",
    "FrozenTrial",
    "MaxTrialsCallback",
    "Study",
    "StudyDirection",
    "StudySummary",
    "Trial",
    "TrialPruned",
    "TrialState",
    "copy_study",
    "create_study",
    "create_trial",
    "delete_study",
    "distributions",
    "exceptions",
    "get_all_study_names",
    "get_all_study_summaries",
    "importance",
    "integration",
    "load_study",
    "logging",
    "pruners",
    "samplers",
    "search_space",
    "storages",
    "study",
    "trial",
    "version",
    "visualization",
    "artifacts",
    "terminator",
    "cli",
]


def __getattr__(name: str):
    """Lazy submodule access: ``optuna_amd.<submodule>`` works without an
    explicit import (matching the reference package, whose heavyweight
    subpackages are reachable as attributes)."""
    import importlib

    if name.startswith("__"):
        raise AttributeError(f"module {__name__!r} has no attribute {name!r}")
    try:
        return importlib.import_module(f"optuna_amd.{name}")
    except ModuleNotFoundError:
        raise AttributeError(f"module {__name__!r} has no attribute {name!r}") from None
