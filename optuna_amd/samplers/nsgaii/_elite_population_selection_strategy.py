"""Reference-compatible module path; implementation in ``_strategies.py``."""
from optuna_amd.samplers.nsgaii._strategies import (  # noqa: F401
    NSGAIIElitePopulationSelectionStrategy,
    _calc_crowding_distance,
    _constrained_dominates,
    _crowding_distance_sort,
    _rank_population,
    _validate_constraints,
)
