"""Constraint conventions: ``"constraints"`` trial system-attr and feasibility filters.

Parity: reference ``optuna/study/_constrained_optimization.py``
(_CONSTRAINTS_KEY, _get_feasible_trials :23).
"""
from __future__ import annotations

from typing import Sequence

from optuna_amd.trial import FrozenTrial


_CONSTRAINTS_KEY = "constraints"


def _get_constraints(trial: FrozenTrial) -> list[float] | None:
    constraints = trial.system_attrs.get(_CONSTRAINTS_KEY)
    if constraints is None:
        return None
    return list(constraints)


def _is_feasible(trial: FrozenTrial) -> bool:
    constraints = _get_constraints(trial)
    if constraints is None:
        return True
    return all(c <= 0.0 for c in constraints)


def _get_feasible_trials(trials: Sequence[FrozenTrial]) -> list[FrozenTrial]:
    return [t for t in trials if _is_feasible(t)]
