"""Parallel-coordinate plot over parameters + objective.

Parity: reference ``optuna/visualization/_parallel_coordinate.py``.
"""
from __future__ import annotations

import math
from typing import TYPE_CHECKING, Any, Callable, NamedTuple, cast

from optuna_amd.trial import FrozenTrial
from optuna_amd.visualization._plotly_imports import _imports
from optuna_amd.visualization._utils import (
    _check_plot_args,
    _filter_nonfinite,
    _get_completed_trials,
    _is_categorical,
    _is_log_scale,
)


if TYPE_CHECKING:
    import plotly.graph_objects as go

    from optuna_amd.study import Study


class _DimensionInfo(NamedTuple):
    label: str
    values: tuple[float, ...]
    range: tuple[float, float]
    is_log: bool
    is_cat: bool
    tickvals: list[int | float]
    ticktext: list[str]


class _ParallelCoordinateInfo(NamedTuple):
    dim_objective: _DimensionInfo
    dims_params: list[_DimensionInfo]
    reverse_scale: bool
    target_name: str


def _get_parallel_coordinate_info(
    study: "Study",
    params: list[str] | None = None,
    target: Callable[[FrozenTrial], float] | None = None,
    target_name: str = "Objective Value",
) -> _ParallelCoordinateInfo:
    from optuna_amd.study._study_direction import StudyDirection

    reverse_scale = target is not None or study.direction == StudyDirection.MINIMIZE

    trials = _filter_nonfinite(_get_completed_trials(study), target=target)
    all_params = {name for t in trials for name in t.params}
    if params is not None:
        for name in params:
            if name not in all_params:
                raise ValueError(f"Parameter {name} does not exist in your study.")
        all_params = set(params)
    sorted_params = sorted(all_params)

    # Only trials that have every shown parameter appear as lines.
    trials = [t for t in trials if all(p in t.params for p in sorted_params)]

    objective_values = tuple(
        target(t) if target is not None else cast(float, t.value) for t in trials
    )
    dim_objective = _DimensionInfo(
        label=target_name,
        values=objective_values,
        range=(min(objective_values, default=0), max(objective_values, default=1)),
        is_log=False,
        is_cat=False,
        tickvals=[],
        ticktext=[],
    )

    dims_params = []
    for p_name in sorted_params:
        if _is_categorical(trials, p_name):
            raw: list[Any] = [str(t.params[p_name]) for t in trials]
            categories = sorted(set(raw))
            mapping = {c: i for i, c in enumerate(categories)}
            values = tuple(float(mapping[v]) for v in raw)
            dims_params.append(
                _DimensionInfo(
                    label=p_name,
                    values=values,
                    range=(0, max(len(categories) - 1, 1)),
                    is_log=False,
                    is_cat=True,
                    tickvals=list(range(len(categories))),
                    ticktext=categories,
                )
            )
        else:
            is_log = _is_log_scale(trials, p_name)
            raw_num = [float(t.params[p_name]) for t in trials]
            values = tuple(math.log10(v) if is_log else v for v in raw_num)
            lo, hi = (min(values), max(values)) if values else (0.0, 1.0)
            if is_log:
                tickvals = list(range(math.ceil(lo), math.floor(hi) + 1))
                ticktext = [str(10**t) for t in tickvals]
            else:
                tickvals, ticktext = [], []
            dims_params.append(
                _DimensionInfo(
                    label=p_name,
                    values=values,
                    range=(lo, hi),
                    is_log=is_log,
                    is_cat=False,
                    tickvals=tickvals,
                    ticktext=ticktext,
                )
            )

    return _ParallelCoordinateInfo(dim_objective, dims_params, reverse_scale, target_name)


def plot_parallel_coordinate(
    study: "Study",
    params: list[str] | None = None,
    *,
    target: Callable[[FrozenTrial], float] | None = None,
    target_name: str = "Objective Value",
) -> "go.Figure":
    _imports.check()
    import plotly.graph_objects as go

    _check_plot_args(study, target, target_name)
    info = _get_parallel_coordinate_info(study, params, target, target_name)

    dims = [
        {
            "label": info.dim_objective.label,
            "values": info.dim_objective.values,
            "range": info.dim_objective.range,
        }
    ]
    for dim in info.dims_params:
        d: dict[str, Any] = {"label": dim.label, "values": dim.values, "range": dim.range}
        if dim.tickvals:
            d["tickvals"] = dim.tickvals
            d["ticktext"] = dim.ticktext
        dims.append(d)

    fig = go.Figure(
        go.Parcoords(
            dimensions=dims,
            line={
                "color": info.dim_objective.values,
                "colorscale": "Blues",
                "reversescale": info.reverse_scale,
                "showscale": True,
                "colorbar": {"title": info.target_name},
            },
            labelangle=30,
        )
    )
    fig.update_layout(title="Parallel Coordinate Plot")
    return fig
