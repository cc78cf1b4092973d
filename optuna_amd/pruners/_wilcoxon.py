"""Wilcoxon signed-rank pruner: statistical test against the best trial.

Compares the current trial's per-step values against the best trial's values at
the intersecting steps with a paired one-sided Wilcoxon signed-rank test; prunes
when p < ``p_threshold`` unless the current trial's average actually beats the
best trial's (safety valve).

Parity: reference ``optuna/pruners/_wilcoxon.py`` (WilcoxonPruner :27, test logic
:156-229).
"""
from __future__ import annotations

import warnings
from typing import TYPE_CHECKING

import numpy as np

from optuna_amd._experimental import experimental_class
from optuna_amd.pruners._base import BasePruner
from optuna_amd.study._study_direction import StudyDirection
from optuna_amd.trial import FrozenTrial


if TYPE_CHECKING:
    from optuna_amd.study import Study


@experimental_class("3.6.0")
class WilcoxonPruner(BasePruner):
    """Prune when a signed-rank test concludes the trial is worse than the best one."""

    def __init__(self, *, p_threshold: float = 0.1, n_startup_steps: int = 2) -> None:
        if n_startup_steps < 0:
            raise ValueError(f"n_startup_steps must be nonnegative but got {n_startup_steps}.")
        if not 0.0 <= p_threshold <= 1.0:
            raise ValueError(f"p_threshold must be between 0 and 1 but got {p_threshold}.")
        self._n_startup_steps = n_startup_steps
        self._p_threshold = p_threshold

    def prune(self, study: "Study", trial: FrozenTrial) -> bool:
        import scipy.stats as ss

        if len(trial.intermediate_values) == 0:
            return False

        steps, step_values = np.array(list(trial.intermediate_values.items())).T
        if np.any(~np.isfinite(step_values)):
            warnings.warn(
                f"The intermediate values of the current trial (trial {trial.number}) contain "
                "infinity/NaNs. WilcoxonPruner will not prune this trial."
            )
            return False

        try:
            best_trial = study.best_trial
        except ValueError:
            return False

        if len(best_trial.intermediate_values) == 0:
            warnings.warn(
                "The best trial has no intermediate values so WilcoxonPruner cannot prune "
                "trials. If you have added the best trial with Study.add_trial, consider "
                "setting the intermediate_values argument."
            )
            return False

        best_steps, best_step_values = np.array(list(best_trial.intermediate_values.items())).T
        if np.any(~np.isfinite(best_step_values)):
            warnings.warn(
                f"The intermediate values of the best trial (trial {best_trial.number}) "
                "contain infinity/NaNs. WilcoxonPruner will not prune the current trial."
            )
            return False

        _, idx1, idx2 = np.intersect1d(steps, best_steps, return_indices=True)
        if len(idx1) < len(step_values):
            warnings.warn(
                "WilcoxonPruner finds steps existing in the current trial but not in the "
                "best trial. Those values are ignored."
            )

        diff_values = step_values[idx1] - best_step_values[idx2]
        if len(diff_values) < max(2, self._n_startup_steps):
            return False

        if study.direction == StudyDirection.MAXIMIZE:
            alt = "less"
            average_is_best = best_step_values.mean() <= step_values.mean()
        else:
            alt = "greater"
            average_is_best = best_step_values.mean() >= step_values.mean()

        # zsplit keeps the test defined when all differences are zero.
        p = ss.wilcoxon(diff_values, alternative=alt, zero_method="zsplit").pvalue

        if p < self._p_threshold and average_is_best:
            # Test says "worse", averages say "better": be conservative, keep going.
            return False
        return bool(p < self._p_threshold)
