"""Empirical-distribution-function plot of objective values.

Parity: reference ``optuna/visualization/_edf.py``.
"""
from __future__ import annotations

from typing import TYPE_CHECKING, Callable, NamedTuple, Sequence, cast

import numpy as np

from optuna_amd.trial import FrozenTrial
from optuna_amd.visualization._plotly_imports import _imports
from optuna_amd.visualization._utils import (
    _check_plot_args,
    _filter_nonfinite,
    _get_completed_trials,
)


if TYPE_CHECKING:
    import plotly.graph_objects as go

    from optuna_amd.study import Study

NUM_SAMPLES_X_AXIS = 100


class _EDFLineInfo(NamedTuple):
    study_name: str
    y_values: np.ndarray


class _EDFInfo(NamedTuple):
    lines: list[_EDFLineInfo]
    x_values: np.ndarray


def _get_edf_info(
    study: "Study | Sequence[Study]",
    target: Callable[[FrozenTrial], float] | None = None,
    target_name: str = "Objective Value",
) -> _EDFInfo:
    from optuna_amd.study import Study

    studies = [study] if isinstance(study, Study) else list(study)
    if len(studies) == 0:
        return _EDFInfo(lines=[], x_values=np.array([]))

    all_values = []
    per_study_values = []
    for s in studies:
        trials = _filter_nonfinite(_get_completed_trials(s), target=target)
        values = np.array(
            [target(t) if target is not None else cast(float, t.value) for t in trials]
        )
        per_study_values.append((s.study_name, values))
        all_values.append(values)
    stacked = np.concatenate(all_values) if all_values else np.array([])
    if stacked.size == 0:
        return _EDFInfo(lines=[], x_values=np.array([]))

    x_values = np.linspace(stacked.min(), stacked.max(), NUM_SAMPLES_X_AXIS)
    lines = [
        _EDFLineInfo(
            study_name=name,
            y_values=np.sum(values[None, :] <= x_values[:, None], axis=1) / max(1, values.size),
        )
        for name, values in per_study_values
    ]
    return _EDFInfo(lines=lines, x_values=x_values)


def plot_edf(
    study: "Study | Sequence[Study]",
    *,
    target: Callable[[FrozenTrial], float] | None = None,
    target_name: str = "Objective Value",
) -> "go.Figure":
    _imports.check()
    import plotly.graph_objects as go

    _check_plot_args(study, target, target_name)
    info = _get_edf_info(study, target, target_name)
    fig = go.Figure()
    for line in info.lines:
        fig.add_trace(
            go.Scatter(x=info.x_values, y=line.y_values, mode="lines", name=line.study_name)
        )
    fig.update_layout(
        title="Empirical Distribution Function Plot",
        xaxis_title=target_name,
        yaxis_title="Cumulative Probability",
        yaxis_range=[0, 1],
    )
    return fig
