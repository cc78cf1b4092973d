"""Statistical-error evaluators for the terminator.

Parity: reference ``optuna/terminator/erroreval.py`` (CrossValidationErrorEvaluator
— cv-score variance scaled by 1/k + 1/(k-1); StaticErrorEvaluator;
report_cross_validation_scores writing ``terminator:cv_scores``) and
``median_erroreval.py`` (MedianErrorEvaluator).
"""
from __future__ import annotations

import abc
import sys
from typing import TYPE_CHECKING

import numpy as np

from optuna_amd.study._study_direction import StudyDirection
from optuna_amd.trial import FrozenTrial, TrialState


if TYPE_CHECKING:
    from optuna_amd.terminator.improvement import BaseImprovementEvaluator
    from optuna_amd.trial import Trial

_CROSS_VALIDATION_SCORES_KEY = "terminator:cv_scores"


class BaseErrorEvaluator(abc.ABC):
    @abc.abstractmethod
    def evaluate(
        self, trials: list[FrozenTrial], study_direction: StudyDirection
    ) -> float:
        raise NotImplementedError


class CrossValidationErrorEvaluator(BaseErrorEvaluator):
    """Statistical error = scaled variance of the best trial's CV scores."""

    def evaluate(
        self, trials: list[FrozenTrial], study_direction: StudyDirection
    ) -> float:
        complete = [t for t in trials if t.state == TrialState.COMPLETE]
        assert len(complete) > 0
        if study_direction == StudyDirection.MAXIMIZE:
            best_trial = max(complete, key=lambda t: t.value)  # type: ignore[arg-type,return-value]
        else:
            best_trial = min(complete, key=lambda t: t.value)  # type: ignore[arg-type,return-value]
        attrs = best_trial.system_attrs
        if _CROSS_VALIDATION_SCORES_KEY not in attrs:
            raise ValueError(
                "Cross-validation scores have not been reported. Please call "
                "`report_cross_validation_scores(trial, scores)` during a trial and pass "
                "the list of scores as `scores`."
            )
        cv_scores = attrs[_CROSS_VALIDATION_SCORES_KEY]
        k = len(cv_scores)
        assert k > 1, "Should be guaranteed by `report_cross_validation_scores`."
        # Nadeau & Bengio's corrected variance of the CV mean estimate.
        scale = 1 / k + 1 / (k - 1)
        return scale * float(np.var(cv_scores, ddof=1))


class StaticErrorEvaluator(BaseErrorEvaluator):
    def __init__(self, constant: float) -> None:
        self._constant = constant

    def evaluate(
        self, trials: list[FrozenTrial], study_direction: StudyDirection
    ) -> float:
        return self._constant


class MedianErrorEvaluator(BaseErrorEvaluator):
    """Threshold = ratio × median of the paired improvement evaluator's values
    over the first ``n_initial_trials`` prefixes (after ``warm_up_trials``)."""

    def __init__(
        self,
        paired_improvement_evaluator: "BaseImprovementEvaluator",
        warm_up_trials: int = 10,
        n_initial_trials: int = 20,
        threshold_ratio: float = 0.01,
    ) -> None:
        if warm_up_trials < 0:
            raise ValueError("`warm_up_trials` is expected to be a non-negative integer.")
        if n_initial_trials <= 0:
            raise ValueError("`n_initial_trials` is expected to be a positive integer.")
        if threshold_ratio <= 0.0 or not np.isfinite(threshold_ratio):
            raise ValueError("`threshold_ratio` is expected to be a positive finite number.")
        self._paired_improvement_evaluator = paired_improvement_evaluator
        self._warm_up_trials = warm_up_trials
        self._n_initial_trials = n_initial_trials
        self._threshold_ratio = threshold_ratio
        self._threshold: float | None = None

    def evaluate(
        self, trials: list[FrozenTrial], study_direction: StudyDirection
    ) -> float:
        if self._threshold is not None:
            return self._threshold
        complete = [t for t in trials if t.state == TrialState.COMPLETE]
        if len(complete) < self._warm_up_trials + self._n_initial_trials:
            return -sys.float_info.min  # never terminate before warm-up completes
        complete.sort(key=lambda t: t.number)
        criteria = sorted(
            self._paired_improvement_evaluator.evaluate(
                complete[self._warm_up_trials : self._warm_up_trials + i], study_direction
            )
            for i in range(1, self._n_initial_trials + 1)
        )
        self._threshold = min(
            sys.float_info.max, criteria[len(criteria) // 2] * self._threshold_ratio
        )
        return self._threshold


def report_cross_validation_scores(trial: "Trial", scores: list[float]) -> None:
    """Store per-fold CV scores for CrossValidationErrorEvaluator."""
    if len(scores) <= 1:
        raise ValueError("The length of `scores` is expected to be greater than one.")
    trial.storage.set_trial_system_attr(
        trial._trial_id, _CROSS_VALIDATION_SCORES_KEY, list(scores)
    )
