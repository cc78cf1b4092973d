"""Slice plot: objective value against each parameter.

Parity: reference ``optuna/visualization/_slice.py`` (info layer + plotly).
"""
from __future__ import annotations

from typing import TYPE_CHECKING, Any, Callable, NamedTuple, cast

from optuna_amd.trial import FrozenTrial
from optuna_amd.visualization._plotly_imports import _imports
from optuna_amd.visualization._utils import (
    _check_plot_args,
    _filter_nonfinite,
    _get_completed_trials,
    _get_param_values,
    _is_log_scale,
    _is_numerical,
)


if TYPE_CHECKING:
    import plotly.graph_objects as go

    from optuna_amd.study import Study


class _SlicePlotInfo(NamedTuple):
    param_name: str
    x: list[Any]
    y: list[float]
    trial_numbers: list[int]
    is_log: bool
    is_numerical: bool


class _SliceSubplotInfo(NamedTuple):
    subplots: list[_SlicePlotInfo]
    target_name: str


def _get_slice_plot_info(
    study: "Study",
    params: list[str] | None,
    target: Callable[[FrozenTrial], float] | None,
    target_name: str,
) -> _SliceSubplotInfo:
    trials = _filter_nonfinite(_get_completed_trials(study), target=target)
    all_params = {name for t in trials for name in t.params}
    if params is None:
        sorted_params = sorted(all_params)
    else:
        for name in params:
            if name not in all_params:
                raise ValueError(f"Parameter {name} does not exist in your study.")
        sorted_params = sorted(set(params))

    subplots = []
    for name in sorted_params:
        selected = [t for t in trials if name in t.params]
        subplots.append(
            _SlicePlotInfo(
                param_name=name,
                x=_get_param_values(selected, name),
                y=[
                    target(t) if target is not None else cast(float, t.value)
                    for t in selected
                ],
                trial_numbers=[t.number for t in selected],
                is_log=_is_log_scale(selected, name),
                is_numerical=_is_numerical(selected, name),
            )
        )
    return _SliceSubplotInfo(subplots, target_name)


def plot_slice(
    study: "Study",
    params: list[str] | None = None,
    *,
    target: Callable[[FrozenTrial], float] | None = None,
    target_name: str = "Objective Value",
) -> "go.Figure":
    _imports.check()
    from plotly.subplots import make_subplots

    import plotly.graph_objects as go

    _check_plot_args(study, target, target_name)
    info = _get_slice_plot_info(study, params, target, target_name)
    n = max(1, len(info.subplots))
    fig = make_subplots(
        rows=1, cols=n, subplot_titles=[s.param_name for s in info.subplots], shared_yaxes=True
    )
    for i, sub in enumerate(info.subplots, start=1):
        fig.add_trace(
            go.Scatter(
                x=sub.x,
                y=sub.y,
                mode="markers",
                marker={
                    "color": sub.trial_numbers,
                    "colorscale": "Blues",
                    "showscale": i == len(info.subplots),
                    "colorbar": {"title": "Trial"},
                },
                showlegend=False,
            ),
            row=1,
            col=i,
        )
        if sub.is_log:
            fig.update_xaxes(type="log", row=1, col=i)
    fig.update_layout(title="Slice Plot")
    fig.update_yaxes(title_text=info.target_name, row=1, col=1)
    return fig
