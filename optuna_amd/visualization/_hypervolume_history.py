"""Hypervolume-history plot for multi-objective studies.

Parity: reference ``optuna/visualization/_hypervolume_history.py``.
"""
from __future__ import annotations

from typing import TYPE_CHECKING, NamedTuple, Sequence

import numpy as np

from optuna_amd._hypervolume import compute_hypervolume
from optuna_amd.study._multi_objective import _normalize_value
from optuna_amd.trial import TrialState
from optuna_amd.visualization._plotly_imports import _imports


if TYPE_CHECKING:
    import plotly.graph_objects as go

    from optuna_amd.study import Study


class _HypervolumeHistoryInfo(NamedTuple):
    trial_numbers: list[int]
    values: list[float]


def _get_hypervolume_history_info(
    study: "Study", reference_point: np.ndarray
) -> _HypervolumeHistoryInfo:
    completed = study.get_trials(deepcopy=False, states=(TrialState.COMPLETE,))
    trial_numbers = []
    values = []
    loss_so_far: list[list[float]] = []
    best_hv = 0.0
    for trial in completed:
        trial_numbers.append(trial.number)
        assert trial.values is not None
        loss = [_normalize_value(v, d) for v, d in zip(trial.values, study.directions)]
        loss_so_far.append(loss)
        arr = np.asarray(loss_so_far)
        finite = np.all(np.isfinite(arr), axis=1)
        if np.any(finite):
            best_hv = compute_hypervolume(arr[finite], reference_point)
        values.append(best_hv)
    return _HypervolumeHistoryInfo(trial_numbers, values)


def plot_hypervolume_history(
    study: "Study", reference_point: Sequence[float]
) -> "go.Figure":
    """Cumulative dominated hypervolume w.r.t. a (minimization-normalized) ref point."""
    _imports.check()
    import plotly.graph_objects as go

    if not study._is_multi_objective():
        raise ValueError(
            "Study must be multi-objective. For single-objective optimization, "
            "please use plot_optimization_history instead."
        )
    ref = np.asarray(reference_point, dtype=np.float64)
    if len(ref) != len(study.directions):
        raise ValueError(
            "The dimension of the reference point must be the same as the number of "
            "objectives."
        )
    # Normalize maximize axes onto the loss orientation.
    ref = np.array(
        [_normalize_value(v, d) for v, d in zip(ref, study.directions)]
    )
    info = _get_hypervolume_history_info(study, ref)
    fig = go.Figure(
        go.Scatter(x=info.trial_numbers, y=info.values, mode="lines+markers")
    )
    fig.update_layout(
        title="Hypervolume History Plot",
        xaxis_title="Trial",
        yaxis_title="Hypervolume",
    )
    return fig
