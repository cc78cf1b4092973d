from optuna_amd.terminator.callback import TerminatorCallback
from optuna_amd.terminator.erroreval import (
    BaseErrorEvaluator,
    CrossValidationErrorEvaluator,
    MedianErrorEvaluator,
    StaticErrorEvaluator,
    report_cross_validation_scores,
)
from optuna_amd.terminator.improvement import (
    BaseImprovementEvaluator,
    BestValueStagnationEvaluator,
    EMMREvaluator,
    RegretBoundEvaluator,
)
from optuna_amd.terminator.terminator import BaseTerminator, Terminator


__all__ = [
    "BaseErrorEvaluator",
    "BaseImprovementEvaluator",
    "BaseTerminator",
    "BestValueStagnationEvaluator",
    "CrossValidationErrorEvaluator",
    "EMMREvaluator",
    "MedianErrorEvaluator",
    "RegretBoundEvaluator",
    "StaticErrorEvaluator",
    "Terminator",
    "TerminatorCallback",
    "report_cross_validation_scores",
]
