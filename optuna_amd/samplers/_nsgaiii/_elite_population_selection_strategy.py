"""Reference-compatible module path; implementation in ``_elite_selection.py``.

``_COEF`` is the reference's name for the infinite-objective clipping margin
multiplier (our ``_INF_CLIP_COEF``).
"""
from optuna_amd.samplers._nsgaiii._elite_selection import (  # noqa: F401
    _INF_CLIP_COEF as _COEF,
    NSGAIIIElitePopulationSelectionStrategy,
    _associate as _associate_individuals_with_reference_points,
    _filter_inf,
    _generate_default_reference_point,
    _normalize_objective_values,
    _preserve_niche_individuals,
)
