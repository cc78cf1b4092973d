"""Intermediate-values plot: learning curves of all trials.

Parity: reference ``optuna/visualization/_intermediate_values.py``.
"""
from __future__ import annotations

import math
from typing import TYPE_CHECKING, NamedTuple

from optuna_amd import logging as _logging
from optuna_amd.trial import TrialState
from optuna_amd.visualization._plotly_imports import _imports


if TYPE_CHECKING:
    import plotly.graph_objects as go

    from optuna_amd.study import Study

_logger = _logging.get_logger(__name__)


class _TrialInfo(NamedTuple):
    trial_number: int
    sorted_intermediate_values: list[tuple[int, float]]
    feasible: bool


class _IntermediatePlotInfo(NamedTuple):
    trial_infos: list[_TrialInfo]


def _get_intermediate_plot_info(study: "Study") -> _IntermediatePlotInfo:
    from optuna_amd.study._constrained_optimization import _is_feasible

    trials = study.get_trials(
        deepcopy=False, states=(TrialState.PRUNED, TrialState.COMPLETE, TrialState.RUNNING)
    )
    trial_infos = [
        _TrialInfo(
            trial.number,
            sorted(
                (step, value)
                for step, value in trial.intermediate_values.items()
                if not math.isnan(value)
            ),
            _is_feasible(trial),
        )
        for trial in trials
        if len(trial.intermediate_values) > 0
    ]
    if len(trials) == 0:
        _logger.warning("Study instance does not contain trials.")
    elif len(trial_infos) == 0:
        _logger.warning(
            "You need to set up the pruning feature to utilize `plot_intermediate_values()`."
        )
    return _IntermediatePlotInfo(trial_infos)


def plot_intermediate_values(study: "Study") -> "go.Figure":
    _imports.check()
    import plotly.graph_objects as go

    info = _get_intermediate_plot_info(study)
    fig = go.Figure()
    for trial_info in info.trial_infos:
        steps = [s for s, _ in trial_info.sorted_intermediate_values]
        values = [v for _, v in trial_info.sorted_intermediate_values]
        fig.add_trace(
            go.Scatter(
                x=steps,
                y=values,
                mode="lines+markers",
                marker={"maxdisplayed": 10},
                name=f"Trial{trial_info.trial_number}"
                + ("" if trial_info.feasible else " (Infeasible)"),
                line={"dash": None if trial_info.feasible else "dash"},
            )
        )
    fig.update_layout(
        title="Intermediate Values Plot",
        xaxis_title="Step",
        yaxis_title="Intermediate Value",
        showlegend=False,
    )
    return fig
