"""Shared helpers for the visualization info layer.

Parity: reference ``optuna/visualization/_utils.py``.
"""
from __future__ import annotations

import math
from typing import TYPE_CHECKING, Any, Callable, Sequence

from optuna_amd import logging as _logging
from optuna_amd.distributions import (
    BaseDistribution,
    CategoricalDistribution,
    FloatDistribution,
    IntDistribution,
)
from optuna_amd.trial import FrozenTrial, TrialState


if TYPE_CHECKING:
    from optuna_amd.study import Study

_logger = _logging.get_logger(__name__)


def _check_plot_args(
    study: "Study | Sequence[Study]",
    target: Callable[[FrozenTrial], float] | None,
    target_name: str,
) -> None:
    from optuna_amd.study import Study

    studies = [study] if isinstance(study, Study) else list(study)
    if target is None and any(s._is_multi_objective() for s in studies):
        raise ValueError(
            "If the `study` is being used for multi-objective optimization, "
            "please specify the `target`."
        )
    if target is not None and target_name == "Objective Value":
        _logger.warning(
            "`target` is specified, but `target_name` is the default. "
            "Consider specifying `target_name` as well."
        )


def _is_log_scale(trials: list[FrozenTrial], param: str) -> bool:
    for trial in trials:
        dist = trial.distributions.get(param)
        if dist is not None and isinstance(dist, (FloatDistribution, IntDistribution)):
            if dist.log:
                return True
    return False


def _is_categorical(trials: list[FrozenTrial], param: str) -> bool:
    return any(
        isinstance(t.distributions.get(param), CategoricalDistribution) for t in trials
    )


def _is_numerical(trials: list[FrozenTrial], param: str) -> bool:
    return all(
        (isinstance(t.params[param], (int, float)) and not isinstance(t.params[param], bool))
        for t in trials
        if param in t.params
    )


def _filter_nonfinite(
    trials: list[FrozenTrial],
    target: Callable[[FrozenTrial], float] | None = None,
    with_message: bool = True,
) -> list[FrozenTrial]:
    out = []
    has_nonfinite = False
    for trial in trials:
        value = target(trial) if target is not None else trial.value
        assert value is not None
        if math.isfinite(value):
            out.append(trial)
        else:
            has_nonfinite = True
    if has_nonfinite and with_message:
        _logger.warning("Trials with non-finite values were omitted from the plot.")
    return out


def _get_completed_trials(study: "Study") -> list[FrozenTrial]:
    return study.get_trials(deepcopy=False, states=(TrialState.COMPLETE,))


def _get_param_values(trials: list[FrozenTrial], p_name: str) -> list[Any]:
    values = [t.params[p_name] for t in trials if p_name in t.params]
    if _is_numerical(trials, p_name):
        return values
    return list(map(str, values))
