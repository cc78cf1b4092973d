"""Patient pruner: early-stopping wrapper with a patience window.

Parity: reference ``optuna/pruners/_patient.py`` (PatientPruner :17, prune :95).
"""
from __future__ import annotations

from typing import TYPE_CHECKING

import numpy as np

from optuna_amd._experimental import experimental_class
from optuna_amd.pruners._base import BasePruner
from optuna_amd.study._study_direction import StudyDirection
from optuna_amd.trial import FrozenTrial


if TYPE_CHECKING:
    from optuna_amd.study import Study


@experimental_class("2.8.0")
class PatientPruner(BasePruner):
    """Allow the wrapped pruner to act only after ``patience`` steps without improvement."""

    def __init__(
        self, wrapped_pruner: BasePruner | None, patience: int, min_delta: float = 0.0
    ) -> None:
        if patience < 0:
            raise ValueError(f"patience cannot be negative but got {patience}.")
        if min_delta < 0:
            raise ValueError(f"min_delta cannot be negative but got {min_delta}.")
        self._wrapped_pruner = wrapped_pruner
        self._patience = patience
        self._min_delta = min_delta

    def prune(self, study: "Study", trial: FrozenTrial) -> bool:
        step = trial.last_step
        if step is None:
            return False

        intermediate_values = trial.intermediate_values
        steps = np.asarray(list(intermediate_values.keys()))
        if steps.size <= self._patience + 1:
            return False
        steps.sort()

        steps_before = steps[: -self._patience - 1]
        scores_before = np.asarray([intermediate_values[s] for s in steps_before])
        steps_after = steps[-self._patience - 1 :]
        scores_after = np.asarray([intermediate_values[s] for s in steps_after])

        if study.direction == StudyDirection.MINIMIZE:
            maybe_prune = np.nanmin(scores_before) + self._min_delta < np.nanmin(scores_after)
        else:
            maybe_prune = np.nanmax(scores_before) - self._min_delta > np.nanmax(scores_after)

        if not maybe_prune:
            return False
        if self._wrapped_pruner is not None:
            return self._wrapped_pruner.prune(study, trial)
        return True
