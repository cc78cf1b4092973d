"""Contour plot: pairwise parameter grids colored by objective.

Parity: reference ``optuna/visualization/_contour.py`` (info layer + plotly).
"""
from __future__ import annotations

from typing import TYPE_CHECKING, Any, Callable, NamedTuple, cast

from optuna_amd.trial import FrozenTrial
from optuna_amd.visualization._plotly_imports import _imports
from optuna_amd.visualization._utils import (
    _check_plot_args,
    _filter_nonfinite,
    _get_completed_trials,
    _is_log_scale,
    _is_numerical,
)


if TYPE_CHECKING:
    import plotly.graph_objects as go

    from optuna_amd.study import Study


class _AxisInfo(NamedTuple):
    name: str
    range: tuple[float, float]
    is_log: bool
    is_cat: bool
    indices: list[Any]
    values: list[Any]


class _SubContourInfo(NamedTuple):
    xaxis: _AxisInfo
    yaxis: _AxisInfo
    z_values: dict[tuple[int, int], float]
    constraints: list[bool]


class _ContourInfo(NamedTuple):
    sorted_params: list[str]
    sub_plot_infos: list[list[_SubContourInfo]]
    reverse_scale: bool
    target_name: str


def _get_axis_info(trials: list[FrozenTrial], param_name: str) -> _AxisInfo:
    values: list[Any]
    if _is_numerical(trials, param_name):
        values = [t.params.get(param_name) for t in trials]
    else:
        values = [
            str(t.params.get(param_name)) if param_name in t.params else None for t in trials
        ]
    present = [v for v in values if v is not None]
    is_cat = not _is_numerical(trials, param_name)
    is_log = _is_log_scale(trials, param_name)
    if is_cat:
        indices = sorted(set(present))
        r = (-0.05 * (len(indices) - 1), (len(indices) - 1) * 1.05)
    else:
        lo, hi = min(present), max(present)
        pad = 0.05 * (hi - lo) if hi > lo else 0.5
        r = (lo - pad, hi + pad)
        indices = sorted(set(present))
    return _AxisInfo(
        name=param_name, range=r, is_log=is_log, is_cat=is_cat, indices=indices, values=values
    )


def _get_contour_info(
    study: "Study",
    params: list[str] | None = None,
    target: Callable[[FrozenTrial], float] | None = None,
    target_name: str = "Objective Value",
) -> _ContourInfo:
    from optuna_amd.study._constrained_optimization import _is_feasible
    from optuna_amd.study._study_direction import StudyDirection

    trials = _filter_nonfinite(_get_completed_trials(study), target=target)
    all_params = {name for t in trials for name in t.params}
    if params is None:
        sorted_params = sorted(all_params)
    else:
        for name in params:
            if name not in all_params:
                raise ValueError(f"Parameter {name} does not exist in your study.")
        sorted_params = sorted(set(params))

    sub_plot_infos: list[list[_SubContourInfo]] = []
    for y_param in sorted_params:
        row = []
        for x_param in sorted_params:
            selected = [t for t in trials if x_param in t.params and y_param in t.params]
            xaxis = _get_axis_info(selected, x_param) if selected else _AxisInfo(
                x_param, (0, 1), False, False, [], []
            )
            yaxis = _get_axis_info(selected, y_param) if selected else _AxisInfo(
                y_param, (0, 1), False, False, [], []
            )
            z_values = {}
            constraints = []
            for i, t in enumerate(selected):
                value = target(t) if target is not None else cast(float, t.value)
                z_values[(i, i)] = value  # sparse store keyed per-trial
                constraints.append(_is_feasible(t))
            row.append(_SubContourInfo(xaxis, yaxis, z_values, constraints))
        sub_plot_infos.append(row)

    reverse_scale = (
        target is not None or study.direction == StudyDirection.MINIMIZE
    )
    return _ContourInfo(sorted_params, sub_plot_infos, reverse_scale, target_name)


def plot_contour(
    study: "Study",
    params: list[str] | None = None,
    *,
    target: Callable[[FrozenTrial], float] | None = None,
    target_name: str = "Objective Value",
) -> "go.Figure":
    _imports.check()
    import plotly.graph_objects as go
    from plotly.subplots import make_subplots

    _check_plot_args(study, target, target_name)
    info = _get_contour_info(study, params, target, target_name)
    n = len(info.sorted_params)
    if n < 2:
        return go.Figure(layout={"title": "Contour Plot"})

    trials = _filter_nonfinite(_get_completed_trials(study), target=target)
    values = [target(t) if target is not None else cast(float, t.value) for t in trials]

    fig = make_subplots(rows=n, cols=n, shared_xaxes=True, shared_yaxes=True)
    for yi, y_param in enumerate(info.sorted_params):
        for xi, x_param in enumerate(info.sorted_params):
            if x_param == y_param:
                continue
            xs, ys, zs = [], [], []
            for t, v in zip(trials, values):
                if x_param in t.params and y_param in t.params:
                    xs.append(t.params[x_param])
                    ys.append(t.params[y_param])
                    zs.append(v)
            if not xs:
                continue
            fig.add_trace(
                go.Contour(
                    x=xs,
                    y=ys,
                    z=zs,
                    colorscale="Blues",
                    reversescale=info.reverse_scale,
                    showscale=False,
                    connectgaps=True,
                    contours_coloring="heatmap",
                ),
                row=yi + 1,
                col=xi + 1,
            )
            fig.add_trace(
                go.Scatter(
                    x=xs, y=ys, mode="markers",
                    marker={"color": "black", "size": 3}, showlegend=False,
                ),
                row=yi + 1,
                col=xi + 1,
            )
            if _is_log_scale(trials, x_param):
                fig.update_xaxes(type="log", row=yi + 1, col=xi + 1)
            if _is_log_scale(trials, y_param):
                fig.update_yaxes(type="log", row=yi + 1, col=xi + 1)
    for i, name in enumerate(info.sorted_params):
        fig.update_xaxes(title_text=name, row=n, col=i + 1)
        fig.update_yaxes(title_text=name, row=i + 1, col=1)
    fig.update_layout(title="Contour Plot")
    return fig
