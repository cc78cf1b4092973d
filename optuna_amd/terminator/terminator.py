"""Terminator: stop when remaining improvement < statistical error.

Parity: reference ``optuna/terminator/terminator.py`` (Terminator :25,
should_terminate :133-145).
"""
from __future__ import annotations

import abc
from typing import TYPE_CHECKING

from optuna_amd.terminator.erroreval import (
    BaseErrorEvaluator,
    CrossValidationErrorEvaluator,
    StaticErrorEvaluator,
)
from optuna_amd.terminator.improvement import (
    DEFAULT_MIN_N_TRIALS,
    BaseImprovementEvaluator,
    BestValueStagnationEvaluator,
    RegretBoundEvaluator,
)
from optuna_amd.trial import TrialState


if TYPE_CHECKING:
    from optuna_amd.study import Study


class BaseTerminator(abc.ABC):
    @abc.abstractmethod
    def should_terminate(self, study: "Study") -> bool:
        raise NotImplementedError


class Terminator(BaseTerminator):
    """Terminate when ``improvement_evaluator < error_evaluator``."""

    def __init__(
        self,
        improvement_evaluator: BaseImprovementEvaluator | None = None,
        error_evaluator: BaseErrorEvaluator | None = None,
        min_n_trials: int = DEFAULT_MIN_N_TRIALS,
    ) -> None:
        if min_n_trials <= 0:
            raise ValueError("`min_n_trials` is expected to be a positive integer.")
        self._improvement_evaluator = improvement_evaluator or RegretBoundEvaluator()
        self._error_evaluator = error_evaluator or self._default_error_evaluator()
        self._min_n_trials = min_n_trials

    def _default_error_evaluator(self) -> BaseErrorEvaluator:
        if isinstance(self._improvement_evaluator, BestValueStagnationEvaluator):
            return StaticErrorEvaluator(constant=0)
        return CrossValidationErrorEvaluator()

    def should_terminate(self, study: "Study") -> bool:
        trials = study.get_trials(states=[TrialState.COMPLETE])
        if len(trials) < self._min_n_trials:
            return False
        improvement = self._improvement_evaluator.evaluate(
            trials=study.trials, study_direction=study.direction
        )
        error = self._error_evaluator.evaluate(
            trials=study.trials, study_direction=study.direction
        )
        return improvement < error
