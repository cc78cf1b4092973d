"""Importance-evaluator base + shared helpers.

Parity: reference ``optuna/importance/_base.py``.
"""
from __future__ import annotations

import abc
import math
from typing import TYPE_CHECKING, Callable

from optuna_amd.distributions import BaseDistribution
from optuna_amd.search_space import intersection_search_space
from optuna_amd.trial import FrozenTrial, TrialState


if TYPE_CHECKING:
    from optuna_amd.study import Study


class BaseImportanceEvaluator(abc.ABC):
    @abc.abstractmethod
    def evaluate(
        self,
        study: "Study",
        params: list[str] | None = None,
        *,
        target: Callable[[FrozenTrial], float] | None = None,
    ) -> dict[str, float]:
        raise NotImplementedError


def _check_evaluate_args(completed_trials: list[FrozenTrial], params: list[str] | None) -> None:
    if len(completed_trials) == 0:
        raise ValueError("Cannot evaluate parameter importances without completed trials.")
    if params is not None:
        if not isinstance(params, (list, tuple)):
            raise TypeError(
                f"Parameters must be specified as a list. Actual parameters: {params}."
            )
        if any(not isinstance(p, str) for p in params):
            raise TypeError(
                f"Parameters must be specified by their names with strings. "
                f"Actual parameters: {params}."
            )
        if len(params) > 0:
            at_least_one_trial = any(
                all(p in t.params for p in params) for t in completed_trials
            )
            if not at_least_one_trial:
                raise ValueError(
                    f"Study must contain completed trials with all specified parameters. "
                    f"Specified parameters: {params}."
                )


def _get_distributions(study: "Study", params: list[str] | None) -> dict[str, BaseDistribution]:
    completed_trials = study.get_trials(deepcopy=False, states=(TrialState.COMPLETE,))
    _check_evaluate_args(completed_trials, params)
    if params is None:
        return intersection_search_space(study.get_trials(deepcopy=False))

    distributions: dict[str, BaseDistribution] | None = None
    for trial in completed_trials:
        trial_distributions = trial.distributions
        if not all(name in trial_distributions for name in params):
            continue
        if distributions is None:
            distributions = {
                name: dist for name, dist in trial_distributions.items() if name in params
            }
            continue
        if any(
            trial_distributions[name] != distribution
            for name, distribution in distributions.items()
        ):
            raise ValueError(
                "Parameters importances cannot be assessed with dynamic search spaces if "
                f"parameters are specified. Specified parameters: {params}."
            )
    assert distributions is not None
    return dict(sorted(distributions.items(), key=lambda kv: kv[0]))


def _get_filtered_trials(
    study: "Study", target: Callable[[FrozenTrial], float] | None
) -> list[FrozenTrial]:
    trials = study.get_trials(deepcopy=False, states=(TrialState.COMPLETE,))
    return [
        t
        for t in trials
        if (
            math.isfinite(target(t))
            if target is not None
            else t.values is not None and all(math.isfinite(v) for v in t.values)
        )
    ]


def _get_target_values(
    trials: list[FrozenTrial], target: Callable[[FrozenTrial], float] | None
) -> list[float]:
    if target is not None:
        return [target(t) for t in trials]
    out = []
    for t in trials:
        assert t.values is not None
        if len(t.values) > 1:
            raise ValueError(
                "If the `study` is being used for multi-objective optimization, "
                "please specify the `target`."
            )
        out.append(t.values[0])
    return out


def _sort_dict_by_importance(param_importances: dict[str, float]) -> dict[str, float]:
    return dict(sorted(param_importances.items(), key=lambda kv: kv[1], reverse=True))
