"""Timeline plot: per-trial start→complete bars colored by state.

Parity: reference ``optuna/visualization/_timeline.py``.
"""
from __future__ import annotations

import datetime
from typing import TYPE_CHECKING, NamedTuple

from optuna_amd.trial import TrialState
from optuna_amd.visualization._plotly_imports import _imports


if TYPE_CHECKING:
    import plotly.graph_objects as go

    from optuna_amd.study import Study


class _TimelineBarInfo(NamedTuple):
    number: int
    start: datetime.datetime
    complete: datetime.datetime
    state: TrialState
    hovertext: str
    infeasible: bool


class _TimelineInfo(NamedTuple):
    bars: list[_TimelineBarInfo]


def _get_timeline_info(study: "Study") -> _TimelineInfo:
    from optuna_amd.study._constrained_optimization import _is_feasible

    bars = []
    max_run_duration = max(
        (
            t.datetime_complete - t.datetime_start
            for t in study.get_trials(deepcopy=False)
            if t.datetime_start is not None and t.datetime_complete is not None
        ),
        default=datetime.timedelta(seconds=1),
    )
    now = datetime.datetime.now()
    for t in study.get_trials(deepcopy=False):
        if t.datetime_start is None:
            continue
        complete = t.datetime_complete or min(now, t.datetime_start + 5 * max_run_duration)
        text = f"Trial {t.number}<br>params: {t.params}"
        bars.append(
            _TimelineBarInfo(
                number=t.number,
                start=t.datetime_start,
                complete=complete,
                state=t.state,
                hovertext=text,
                infeasible=not _is_feasible(t),
            )
        )
    return _TimelineInfo(bars)


_STATE_COLORS = {
    TrialState.COMPLETE: "blue",
    TrialState.RUNNING: "green",
    TrialState.WAITING: "gray",
    TrialState.PRUNED: "orange",
    TrialState.FAIL: "red",
}


def plot_timeline(study: "Study") -> "go.Figure":
    _imports.check()
    import plotly.graph_objects as go

    info = _get_timeline_info(study)
    fig = go.Figure()
    for state in TrialState:
        bars = [b for b in info.bars if b.state == state]
        if not bars:
            continue
        fig.add_trace(
            go.Bar(
                base=[b.start.isoformat() for b in bars],
                x=[(b.complete - b.start).total_seconds() * 1000 for b in bars],
                y=[b.number for b in bars],
                text=[b.hovertext for b in bars],
                hovertemplate="%{text}",
                orientation="h",
                marker={"color": _STATE_COLORS[state]},
                name=state.name,
            )
        )
    fig.update_layout(
        title="Timeline Plot",
        xaxis_title="Datetime",
        yaxis_title="Trial",
        xaxis_type="date",
    )
    return fig
