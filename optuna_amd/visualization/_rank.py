"""Rank plot: param scatter colored by objective rank.

Parity: reference ``optuna/visualization/_rank.py``.
"""
from __future__ import annotations

from typing import TYPE_CHECKING, Any, Callable, NamedTuple, cast

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


class _RankSubplotInfo(NamedTuple):
    xaxis_name: str
    yaxis_name: str
    xs: list[Any]
    ys: list[Any]
    trial_numbers: list[int]
    zs: np.ndarray  # objective values
    colors: np.ndarray  # rank in [0, 1]


class _RankPlotInfo(NamedTuple):
    params: list[str]
    sub_plot_infos: list[list[_RankSubplotInfo]]
    target_name: str
    zs: np.ndarray
    colors: np.ndarray
    has_custom_target: bool


def _get_rank_info(
    study: "Study",
    params: list[str] | None,
    target: Callable[[FrozenTrial], float] | None,
    target_name: str,
) -> _RankPlotInfo:
    trials = _filter_nonfinite(_get_completed_trials(study), target=target)
    all_params = {name for t in trials for name in t.params}
    if params is None:
        sorted_params = sorted(all_params)[:4]  # cap dims like the reference default view
    else:
        for name in params:
            if name not in all_params:
                raise ValueError(f"Parameter {name} does not exist in your study.")
        sorted_params = sorted(set(params))

    zs = np.array(
        [target(t) if target is not None else cast(float, t.value) for t in trials]
    )
    ranks = np.argsort(np.argsort(zs))
    colors = ranks / max(1, len(zs) - 1)

    n = len(sorted_params)
    sub_plot_infos: list[list[_RankSubplotInfo]] = []
    for yi in range(n):
        row = []
        for xi in range(n):
            x_param, y_param = sorted_params[xi], sorted_params[yi]
            mask = [x_param in t.params and y_param in t.params for t in trials]
            sel = [t for t, m in zip(trials, mask) if m]
            row.append(
                _RankSubplotInfo(
                    xaxis_name=x_param,
                    yaxis_name=y_param,
                    xs=[t.params[x_param] for t in sel],
                    ys=[t.params[y_param] for t in sel],
                    trial_numbers=[t.number for t in sel],
                    zs=zs[mask],
                    colors=colors[mask],
                )
            )
        sub_plot_infos.append(row)
    return _RankPlotInfo(
        sorted_params, sub_plot_infos, target_name, zs, colors, target is not None
    )


def plot_rank(
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
    info = _get_rank_info(study, params, target, target_name)
    n = len(info.params)
    if n == 0:
        return go.Figure(layout={"title": "Rank Plot"})
    fig = make_subplots(rows=n, cols=n, shared_xaxes=False, shared_yaxes=False)
    for yi in range(n):
        for xi in range(n):
            sub = info.sub_plot_infos[yi][xi]
            fig.add_trace(
                go.Scatter(
                    x=sub.xs,
                    y=sub.ys,
                    mode="markers",
                    marker={
                        "color": sub.colors,
                        "colorscale": "RdYlBu_r",
                        "showscale": xi == 0 and yi == 0,
                        "colorbar": {"title": f"{info.target_name} Rank"},
                    },
                    showlegend=False,
                ),
                row=yi + 1,
                col=xi + 1,
            )
    for i, name in enumerate(info.params):
        fig.update_xaxes(title_text=name, row=n, col=i + 1)
        fig.update_yaxes(title_text=name, row=i + 1, col=1)
    fig.update_layout(title=f"Rank ({info.target_name})")
    return fig
