"""Terminator-improvement plot: improvement vs error as the study progresses.

Parity: reference ``optuna/visualization/_terminator_improvement.py``.
"""
from __future__ import annotations

from typing import TYPE_CHECKING, NamedTuple

from optuna_amd.trial import TrialState
from optuna_amd.visualization._plotly_imports import _imports


if TYPE_CHECKING:
    import plotly.graph_objects as go

    from optuna_amd.study import Study
    from optuna_amd.terminator import BaseErrorEvaluator, BaseImprovementEvaluator


class _ImprovementInfo(NamedTuple):
    trial_numbers: list[int]
    improvements: list[float]
    errors: list[float] | None


def _get_improvement_info(
    study: "Study",
    get_error: bool = False,
    improvement_evaluator: "BaseImprovementEvaluator | None" = None,
    error_evaluator: "BaseErrorEvaluator | None" = None,
) -> _ImprovementInfo:
    from optuna_amd.terminator import (
        CrossValidationErrorEvaluator,
        RegretBoundEvaluator,
    )

    improvement_evaluator = improvement_evaluator or RegretBoundEvaluator()
    error_evaluator = error_evaluator or CrossValidationErrorEvaluator()

    trial_numbers = []
    improvements = []
    errors = []
    completed: list = []
    for trial in study.trials:
        if trial.state != TrialState.COMPLETE:
            continue
        completed.append(trial)
        trial_numbers.append(trial.number)
        improvements.append(
            improvement_evaluator.evaluate(trials=completed, study_direction=study.direction)
        )
        if get_error:
            errors.append(
                error_evaluator.evaluate(trials=completed, study_direction=study.direction)
            )
    return _ImprovementInfo(
        trial_numbers=trial_numbers,
        improvements=improvements,
        errors=errors if get_error else None,
    )


def plot_terminator_improvement(
    study: "Study",
    plot_error: bool = False,
    improvement_evaluator: "BaseImprovementEvaluator | None" = None,
    error_evaluator: "BaseErrorEvaluator | None" = None,
    min_n_trials: int = 20,
) -> "go.Figure":
    _imports.check()
    import plotly.graph_objects as go

    info = _get_improvement_info(study, plot_error, improvement_evaluator, error_evaluator)
    fig = go.Figure()
    fig.add_trace(
        go.Scatter(
            x=info.trial_numbers, y=info.improvements, mode="markers+lines", name="Improvement"
        )
    )
    if info.errors is not None:
        fig.add_trace(
            go.Scatter(x=info.trial_numbers, y=info.errors, mode="markers+lines", name="Error")
        )
    fig.update_layout(
        title="Terminator Improvement Plot",
        xaxis_title="Trial",
        yaxis_title="Terminator Improvement",
    )
    return fig
