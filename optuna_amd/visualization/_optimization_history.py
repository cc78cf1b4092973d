"""Optimization history plot: per-trial values + running best.

Parity: reference ``optuna/visualization/_optimization_history.py``
(_OptimizationHistoryInfo data layer :30-60 consumed by plotly and matplotlib).
"""
from __future__ import annotations

import math
from enum import Enum
from typing import TYPE_CHECKING, Callable, NamedTuple, Sequence, cast

import numpy as np

from optuna_amd import logging as _logging
from optuna_amd.study._study_direction import StudyDirection
from optuna_amd.trial import FrozenTrial, TrialState
from optuna_amd.visualization._plotly_imports import _imports
from optuna_amd.visualization._utils import _check_plot_args


if TYPE_CHECKING:
    import plotly.graph_objects as go

    from optuna_amd.study import Study

_logger = _logging.get_logger(__name__)


class _ValueState(Enum):
    Feasible = 0
    Infeasible = 1
    Incomplete = 2


class _ValuesInfo(NamedTuple):
    values: list[float]
    stds: list[float] | None
    label_name: str
    states: list[_ValueState]


class _OptimizationHistoryInfo(NamedTuple):
    trial_numbers: list[int]
    values_info: _ValuesInfo
    best_values_info: _ValuesInfo | None


def _get_optimization_history_info_list(
    study: "Study | Sequence[Study]",
    target: Callable[[FrozenTrial], float] | None,
    target_name: str,
    error_bar: bool,
) -> list[_OptimizationHistoryInfo]:
    from optuna_amd.study import Study
    from optuna_amd.study._constrained_optimization import _is_feasible

    studies = [study] if isinstance(study, Study) else list(study)

    info_list: list[_OptimizationHistoryInfo] = []
    for s in studies:
        trials = s.get_trials(deepcopy=False)
        label_name = target_name if len(studies) == 1 else f"{target_name} of {s.study_name}"
        values = []
        value_states = []
        for trial in trials:
            if trial.state != TrialState.COMPLETE:
                values.append(float("nan"))
                value_states.append(_ValueState.Incomplete)
                continue
            value_states.append(
                _ValueState.Feasible if _is_feasible(trial) else _ValueState.Infeasible
            )
            if target is not None:
                values.append(float(target(trial)))
            else:
                values.append(cast(float, trial.value))

        if target is not None:
            best_values_info: _ValuesInfo | None = None
        else:
            feasible_best: list[float] = []
            best_so_far = float("inf") if s.direction == StudyDirection.MINIMIZE else -float("inf")
            op = min if s.direction == StudyDirection.MINIMIZE else max
            for v, st in zip(values, value_states):
                if st == _ValueState.Feasible and not math.isnan(v):
                    best_so_far = op(best_so_far, v)
                feasible_best.append(best_so_far)
            best_label = (
                "Best Value" if len(studies) == 1 else f"Best Value of {s.study_name}"
            )
            best_values_info = _ValuesInfo(
                feasible_best, None, best_label, [_ValueState.Feasible] * len(feasible_best)
            )
        info_list.append(
            _OptimizationHistoryInfo(
                [t.number for t in trials],
                _ValuesInfo(values, None, label_name, value_states),
                best_values_info,
            )
        )

    if len(info_list) == 0:
        _logger.warning("There are no studies.")

    if error_bar and len(info_list) > 1:
        # Aggregate across studies: mean ± std at each trial index.
        max_n = max(len(i.trial_numbers) for i in info_list)
        all_values = np.full((len(info_list), max_n), np.nan)
        all_best = np.full((len(info_list), max_n), np.nan)
        for r, info in enumerate(info_list):
            all_values[r, : len(info.values_info.values)] = info.values_info.values
            if info.best_values_info is not None:
                all_best[r, : len(info.best_values_info.values)] = info.best_values_info.values
        mean_v = np.nanmean(all_values, axis=0)
        std_v = np.nanstd(all_values, axis=0)
        mean_b = np.nanmean(all_best, axis=0)
        std_b = np.nanstd(all_best, axis=0)
        states = [_ValueState.Feasible] * max_n
        info_list = [
            _OptimizationHistoryInfo(
                list(range(max_n)),
                _ValuesInfo(mean_v.tolist(), std_v.tolist(), target_name, states),
                _ValuesInfo(mean_b.tolist(), std_b.tolist(), "Best Value", states),
            )
        ]
    return info_list


def plot_optimization_history(
    study: "Study | Sequence[Study]",
    *,
    target: Callable[[FrozenTrial], float] | None = None,
    target_name: str = "Objective Value",
    error_bar: bool = False,
) -> "go.Figure":
    """Objective values and the running best over trial numbers."""
    _imports.check()
    import plotly.graph_objects as go

    _check_plot_args(study, target, target_name)
    info_list = _get_optimization_history_info_list(study, target, target_name, error_bar)

    fig = go.Figure()
    for info in info_list:
        feasible_idx = [
            i for i, s in enumerate(info.values_info.states) if s == _ValueState.Feasible
        ]
        infeasible_idx = [
            i for i, s in enumerate(info.values_info.states) if s == _ValueState.Infeasible
        ]
        fig.add_trace(
            go.Scatter(
                x=[info.trial_numbers[i] for i in feasible_idx],
                y=[info.values_info.values[i] for i in feasible_idx],
                error_y=(
                    {"array": [info.values_info.stds[i] for i in feasible_idx]}
                    if info.values_info.stds is not None
                    else None
                ),
                mode="markers",
                name=info.values_info.label_name,
            )
        )
        if infeasible_idx:
            fig.add_trace(
                go.Scatter(
                    x=[info.trial_numbers[i] for i in infeasible_idx],
                    y=[info.values_info.values[i] for i in infeasible_idx],
                    mode="markers",
                    marker={"color": "#cccccc"},
                    name="Infeasible Trial",
                )
            )
        if info.best_values_info is not None:
            fig.add_trace(
                go.Scatter(
                    x=info.trial_numbers,
                    y=info.best_values_info.values,
                    error_y=(
                        {"array": info.best_values_info.stds}
                        if info.best_values_info.stds is not None
                        else None
                    ),
                    mode="lines",
                    name=info.best_values_info.label_name,
                )
            )
    fig.update_layout(
        title="Optimization History Plot",
        xaxis_title="Trial",
        yaxis_title=target_name,
    )
    return fig
