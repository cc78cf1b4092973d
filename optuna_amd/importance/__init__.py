"""Hyperparameter-importance evaluation.

Parity: reference ``optuna/importance/__init__.py`` (get_param_importances :27-100,
default evaluator = PedAnova, normalize-to-1).
"""
from __future__ import annotations

from typing import TYPE_CHECKING, Callable

from optuna_amd.importance._base import BaseImportanceEvaluator
from optuna_amd.importance._fanova import FanovaImportanceEvaluator
from optuna_amd.importance._mean_decrease_impurity import (
    MeanDecreaseImpurityImportanceEvaluator,
)
from optuna_amd.importance._ped_anova import PedAnovaImportanceEvaluator


if TYPE_CHECKING:
    from optuna_amd.study import Study
    from optuna_amd.trial import FrozenTrial

__all__ = [
    "BaseImportanceEvaluator",
    "FanovaImportanceEvaluator",
    "MeanDecreaseImpurityImportanceEvaluator",
    "PedAnovaImportanceEvaluator",
    "get_param_importances",
]


def get_param_importances(
    study: "Study",
    *,
    evaluator: BaseImportanceEvaluator | None = None,
    params: list[str] | None = None,
    target: Callable[["FrozenTrial"], float] | None = None,
    normalize: bool = True,
) -> dict[str, float]:
    """Evaluate parameter importances from completed trials (higher = more important).

    Defaults to :class:`PedAnovaImportanceEvaluator`; the result is sorted
    descending and, with ``normalize=True``, sums to 1.
    """
    if not normalize:
        from optuna_amd._experimental import warn_experimental_argument

        warn_experimental_argument("normalize")
    if evaluator is None:
        evaluator = PedAnovaImportanceEvaluator()
    if not isinstance(evaluator, BaseImportanceEvaluator):
        raise TypeError("Evaluator must be a subclass of BaseImportanceEvaluator.")

    from optuna_amd.trial import TrialState

    if not study.get_trials(deepcopy=False, states=(TrialState.COMPLETE,)):
        # Nothing to attribute yet: an empty mapping, not an error (matching
        # the reference; evaluators may still raise for explicit `params`).
        return {}

    res = evaluator.evaluate(study, params=params, target=target)
    if normalize:
        s = sum(res.values())
        if s == 0.0:
            n_params = len(res)
            return {param: 1.0 / n_params for param in res} if n_params else {}
        return {param: value / s for param, value in res.items()}
    return res
