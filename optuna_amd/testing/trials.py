"""Frozen-trial factory for tests (parity: reference ``optuna/testing/trials.py`` :17-38)."""
from __future__ import annotations

from typing import Any, Sequence

from optuna_amd.distributions import BaseDistribution
from optuna_amd.trial import FrozenTrial, TrialState, create_trial


def _create_frozen_trial(
    number: int = 0,
    values: Sequence[float] = (1.0,),
    params: dict[str, Any] | None = None,
    distributions: dict[str, BaseDistribution] | None = None,
    state: TrialState = TrialState.COMPLETE,
    intermediate_values: dict[int, float] | None = None,
    system_attrs: dict[str, Any] | None = None,
) -> FrozenTrial:
    trial = create_trial(
        state=state,
        values=list(values) if state == TrialState.COMPLETE else None,
        params=params or {},
        distributions=distributions or {},
        intermediate_values=intermediate_values or {},
        system_attrs=system_attrs or {},
    )
    trial.number = number
    trial._trial_id = number
    return trial
