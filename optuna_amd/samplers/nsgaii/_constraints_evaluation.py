"""Reference-compatible module path; implementation in ``_strategies.py``."""
from optuna_amd.samplers.nsgaii._strategies import (  # noqa: F401
    _constrained_dominates,
    _evaluate_penalty,
    _is_constrained_optimization,
    _validate_constraints,
)
