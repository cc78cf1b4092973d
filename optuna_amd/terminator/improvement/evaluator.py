"""Reference-compatible module path; implementation in ``_improvement_impl``."""
from optuna_amd.terminator._improvement_impl import *  # noqa: F401,F403
