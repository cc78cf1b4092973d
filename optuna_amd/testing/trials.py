"""Trial construction helpers for tests.

Parity: reference ``optuna/testing/trials.py`` (_create_frozen_trial).
"""
from __future__ import annotations

from typing import Any, Sequence

from optuna_amd.distributions import BaseDistribution
from optuna_amd.study._constrained_optimization import _CONSTRAINTS_KEY
from optuna_amd.trial import FrozenTrial, TrialState, create_trial


def _create_frozen_trial(
    number: int = 0,
    values: Sequence[float] | None = None,
    constraints: Sequence[float] | None = None,
    params: dict[str, Any] | None = None,
    param_distributions: dict[str, BaseDistribution] | None = None,
    state: TrialState = TrialState.COMPLETE,
    *,
    distributions: dict[str, BaseDistribution] | None = None,
    intermediate_values: dict[int, float] | None = None,
    system_attrs: dict[str, Any] | None = None,
) -> FrozenTrial:
    """Bare FrozenTrial with the id set to ``number`` (no storage round trip).

    ``param_distributions`` is the reference's name; ``distributions`` is kept
    as an alias for this package's earlier tests.
    """
    dists = param_distributions or distributions or {}
    attrs = dict(system_attrs or {})
    if constraints is not None:
        attrs[_CONSTRAINTS_KEY] = list(constraints)
    if state == TrialState.COMPLETE and values is None:
        values = (1.0,)
    trial = create_trial(
        state=state,
        values=list(values) if values is not None else None,
        params=params or {},
        distributions=dists,
        intermediate_values=intermediate_values or {},
        system_attrs=attrs,
    )
    trial.number = number
    trial._trial_id = number
    return trial
