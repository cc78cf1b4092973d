"""Reference-compatible module path; implementation in ``_crossovers.py``."""
from optuna_amd.samplers.nsgaii._crossovers import (  # noqa: F401
    BaseCrossover,
    _inlined_categorical_uniform_crossover,
    perform_crossover,
)
