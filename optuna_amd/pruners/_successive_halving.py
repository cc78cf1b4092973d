"""Asynchronous Successive Halving (ASHA) pruner.

Rung r completes at ``min_resource * rf^(s+r)`` steps; a trial is promoted past a
rung only if its value is in the top 1/rf of all values recorded at that rung.
Rung membership is persisted in trial system attrs (``completed_rung_{r}``), which
is what makes the algorithm work asynchronously across distributed workers.

Parity: reference ``optuna/pruners/_successive_halving.py`` (SuccessiveHalvingPruner
:15, prune :167, _estimate_min_resource :219, rung bookkeeping :232-258).
"""
from __future__ import annotations

import math
from typing import TYPE_CHECKING

from optuna_amd.pruners._base import BasePruner
from optuna_amd.study._study_direction import StudyDirection
from optuna_amd.trial import FrozenTrial, TrialState


if TYPE_CHECKING:
    from optuna_amd.study import Study


def _completed_rung_key(rung: int) -> str:
    return f"completed_rung_{rung}"


def _current_rung(trial: FrozenTrial) -> int:
    rung = 0
    while _completed_rung_key(rung) in trial.system_attrs:
        rung += 1
    return rung


def _estimate_min_resource(trials: list[FrozenTrial]) -> int | None:
    """'auto' heuristic: 1/100th of the longest completed trial's last step."""
    last_steps = [
        t.last_step
        for t in trials
        if t.state == TrialState.COMPLETE and t.last_step is not None
    ]
    if not last_steps:
        return None
    return max(max(last_steps) // 100, 1)


def _values_at_rung(trials: list[FrozenTrial], rung_key: str, own_value: float) -> list[float]:
    values = [t.system_attrs[rung_key] for t in trials if rung_key in t.system_attrs]
    values.append(own_value)
    return values


def _promotable(
    value: float, competing: list[float], reduction_factor: int, direction: StudyDirection
) -> bool:
    top_k = len(competing) // reduction_factor - 1
    if top_k == -1:
        # Fewer than rf competitors: promote only the current best (trials cannot be
        # suspended/resumed, so the first rf-1 trials race on value alone).
        top_k = 0
    competing = sorted(competing)
    if direction == StudyDirection.MAXIMIZE:
        return value >= competing[-(top_k + 1)]
    return value <= competing[top_k]


class SuccessiveHalvingPruner(BasePruner):
    """Async SHA pruner (see module docstring)."""

    def __init__(
        self,
        min_resource: str | int = "auto",
        reduction_factor: int = 4,
        min_early_stopping_rate: int = 0,
        bootstrap_count: int = 0,
    ) -> None:
        if isinstance(min_resource, str) and min_resource != "auto":
            raise ValueError(
                f"The value of `min_resource` is {min_resource}, but must be either "
                "`min_resource >= 1` or 'auto'."
            )
        if isinstance(min_resource, int) and min_resource < 1:
            raise ValueError(
                f"The value of `min_resource` is {min_resource}, but must be either "
                "`min_resource >= 1` or 'auto'."
            )
        if reduction_factor < 2:
            raise ValueError(
                f"The value of `reduction_factor` is {reduction_factor}, but must be "
                ">= 2."
            )
        if min_early_stopping_rate < 0:
            raise ValueError(
                f"The value of `min_early_stopping_rate` is {min_early_stopping_rate}, "
                "but must be >= 0."
            )
        if bootstrap_count < 0:
            raise ValueError(
                f"The value of `bootstrap_count` is {bootstrap_count}, but must be >= 0."
            )
        if bootstrap_count > 0 and min_resource == "auto":
            raise ValueError(
                "bootstrap_count > 0 and min_resource == 'auto' are mutually incompatible."
            )

        self._min_resource: int | None = min_resource if isinstance(min_resource, int) else None
        self._reduction_factor = reduction_factor
        self._min_early_stopping_rate = min_early_stopping_rate
        self._bootstrap_count = bootstrap_count

    def prune(self, study: "Study", trial: FrozenTrial) -> bool:
        step = trial.last_step
        if step is None:
            return False

        rung = _current_rung(trial)
        value = trial.intermediate_values[step]
        trials: list[FrozenTrial] | None = None

        while True:
            if self._min_resource is None:
                trials = trials if trials is not None else study.get_trials(deepcopy=False)
                self._min_resource = _estimate_min_resource(trials)
                if self._min_resource is None:
                    return False

            rung_completion_step = self._min_resource * (
                self._reduction_factor ** (self._min_early_stopping_rate + rung)
            )
            if step < rung_completion_step:
                return False

            if math.isnan(value):
                return True

            trials = trials if trials is not None else study.get_trials(deepcopy=False)
            rung_key = _completed_rung_key(rung)
            study._storage.set_trial_system_attr(trial._trial_id, rung_key, value)
            competing = _values_at_rung(trials, rung_key, value)

            # `competing` already includes the current trial, hence `<=`.
            if len(competing) <= self._bootstrap_count:
                return True
            if not _promotable(value, competing, self._reduction_factor, study.direction):
                return True
            rung += 1
