"""Percentile pruner (and the building blocks shared with MedianPruner).

Prunes when the trial's best intermediate value so far is worse than the given
percentile of the intermediate values of completed trials at the same step.

Parity: reference ``optuna/pruners/_percentile.py`` (PercentilePruner :75,
_is_first_in_interval_step :57, percentile-over-trials :29-55).
"""
from __future__ import annotations

import functools
import math
from typing import TYPE_CHECKING, KeysView

import numpy as np

from optuna_amd.pruners._base import BasePruner
from optuna_amd.study._study_direction import StudyDirection
from optuna_amd.trial import FrozenTrial, TrialState


if TYPE_CHECKING:
    from optuna_amd.study import Study


def _get_best_intermediate_result_over_steps(
    trial: FrozenTrial, direction: StudyDirection
) -> float:
    values = np.asarray(list(trial.intermediate_values.values()), dtype=np.float64)
    if direction == StudyDirection.MAXIMIZE:
        return float(np.nanmax(values))
    return float(np.nanmin(values))


def _get_percentile_intermediate_result_over_trials(
    completed_trials: list[FrozenTrial],
    direction: StudyDirection,
    step: int,
    percentile: float,
    n_min_trials: int,
) -> float:
    if len(completed_trials) == 0:
        raise ValueError("No trials have been completed.")
    intermediate_values = [
        t.intermediate_values[step] for t in completed_trials if step in t.intermediate_values
    ]
    if len(intermediate_values) < n_min_trials:
        return math.nan
    if direction == StudyDirection.MAXIMIZE:
        percentile = 100 - percentile
    return float(
        np.nanpercentile(np.asarray(intermediate_values, dtype=np.float64), percentile)
    )


def _is_first_in_interval_step(
    step: int, intermediate_steps: KeysView[int], n_warmup_steps: int, interval_steps: int
) -> bool:
    nearest_lower_pruning_step = (
        (step - n_warmup_steps) // interval_steps * interval_steps + n_warmup_steps
    )
    assert nearest_lower_pruning_step >= 0
    # True iff this is the first reported step inside the current pruning interval.
    second_last_step = functools.reduce(
        lambda a, b: max(a, b) if b != step else a, intermediate_steps, -1
    )
    return second_last_step < nearest_lower_pruning_step


class PercentilePruner(BasePruner):
    """Keep the top-``percentile`` fraction of trials at each step; prune the rest."""

    def __init__(
        self,
        percentile: float,
        n_startup_trials: int = 5,
        n_warmup_steps: int = 0,
        interval_steps: int = 1,
        *,
        n_min_trials: int = 1,
    ) -> None:
        if not 0.0 <= percentile <= 100.0:
            raise ValueError(
                f"Percentile must be between 0 and 100 inclusive but got {percentile}."
            )
        if n_startup_trials < 0:
            raise ValueError(
                f"Number of startup trials cannot be negative but got {n_startup_trials}."
            )
        if n_warmup_steps < 0:
            raise ValueError(
                f"Number of warmup steps cannot be negative but got {n_warmup_steps}."
            )
        if interval_steps < 1:
            raise ValueError(
                f"Pruning interval steps must be at least 1 but got {interval_steps}."
            )
        if n_min_trials < 1:
            raise ValueError(
                f"Number of min trials for pruning must be at least 1 but got {n_min_trials}."
            )
        self._percentile = percentile
        self._n_startup_trials = n_startup_trials
        self._n_warmup_steps = n_warmup_steps
        self._interval_steps = interval_steps
        self._n_min_trials = n_min_trials

    def prune(self, study: "Study", trial: FrozenTrial) -> bool:
        step = trial.last_step
        if step is None:
            return False

        n_warmup_steps = self._n_warmup_steps
        if step < n_warmup_steps:
            return False

        if not _is_first_in_interval_step(
            step, trial.intermediate_values.keys(), n_warmup_steps, self._interval_steps
        ):
            return False

        completed_trials = study.get_trials(deepcopy=False, states=(TrialState.COMPLETE,))
        n_trials = len(completed_trials)
        if n_trials == 0:
            return False
        if n_trials < self._n_startup_trials:
            return False

        direction = study.direction
        best_intermediate_result = _get_best_intermediate_result_over_steps(trial, direction)
        if math.isnan(best_intermediate_result):
            return True

        p = _get_percentile_intermediate_result_over_trials(
            completed_trials, direction, step, self._percentile, self._n_min_trials
        )
        if math.isnan(p):
            return False

        if direction == StudyDirection.MAXIMIZE:
            return best_intermediate_result < p
        return best_intermediate_result > p
