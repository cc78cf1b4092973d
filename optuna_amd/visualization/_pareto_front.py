"""Pareto-front scatter (2 or 3 objectives).

Parity: reference ``optuna/visualization/_pareto_front.py``.
"""
from __future__ import annotations

from typing import TYPE_CHECKING, Callable, NamedTuple, Sequence

from optuna_amd.study._multi_objective import _get_pareto_front_trials
from optuna_amd.trial import FrozenTrial, TrialState
from optuna_amd.visualization._plotly_imports import _imports


if TYPE_CHECKING:
    import plotly.graph_objects as go

    from optuna_amd.study import Study


class _ParetoFrontInfo(NamedTuple):
    n_targets: int
    target_names: list[str]
    best_trials_with_values: list[tuple[FrozenTrial, list[float]]]
    non_best_trials_with_values: list[tuple[FrozenTrial, list[float]]]
    infeasible_trials_with_values: list[tuple[FrozenTrial, list[float]]]
    axis_order: list[int]


def _get_pareto_front_info(
    study: "Study",
    target_names: list[str] | None = None,
    include_dominated_trials: bool = True,
    axis_order: list[int] | None = None,
    constraints_func: Callable[[FrozenTrial], Sequence[float]] | None = None,
    targets: Callable[[FrozenTrial], Sequence[float]] | None = None,
) -> _ParetoFrontInfo:
    from optuna_amd.study._constrained_optimization import _is_feasible

    if targets is not None and axis_order is not None:
        raise ValueError("Using both `targets` and `axis_order` is not supported.")

    def get_values(trial: FrozenTrial) -> list[float]:
        if targets is not None:
            return list(targets(trial))
        assert trial.values is not None
        return list(trial.values)

    completed = study.get_trials(deepcopy=False, states=(TrialState.COMPLETE,))
    has_constraints = constraints_func is not None or any(
        "constraints" in t.system_attrs for t in completed
    )
    if constraints_func is not None:
        feasible = [t for t in completed if all(c <= 0 for c in constraints_func(t))]
    elif has_constraints:
        feasible = [t for t in completed if _is_feasible(t)]
    else:
        feasible = completed
    infeasible = [t for t in completed if t not in feasible]

    best_trials = _get_pareto_front_trials(study, consider_constraint=has_constraints)
    best_ids = {t._trial_id for t in best_trials}
    non_best = [t for t in feasible if t._trial_id not in best_ids]

    n_targets = (
        len(get_values(completed[0])) if completed else len(study.directions)
    )
    if n_targets not in (2, 3):
        raise ValueError(
            "`plot_pareto_front` function only supports 2 or 3 targets. "
            f"you used {n_targets} targets now."
        )
    if target_names is None:
        target_names = [f"Objective {i}" for i in range(n_targets)]
    elif len(target_names) != n_targets:
        raise ValueError(f"The length of `target_names` is supposed to be {n_targets}.")

    if axis_order is None:
        axis_order = list(range(n_targets))
    else:
        if len(axis_order) != n_targets:
            raise ValueError(
                f"Size of `axis_order` {axis_order}. Expect: {n_targets}, "
                f"Actual: {len(axis_order)}."
            )
        if set(axis_order) != set(range(n_targets)):
            raise ValueError(f"Axis order {axis_order} is invalid.")

    return _ParetoFrontInfo(
        n_targets=n_targets,
        target_names=target_names,
        best_trials_with_values=[(t, get_values(t)) for t in best_trials],
        non_best_trials_with_values=[
            (t, get_values(t)) for t in (non_best if include_dominated_trials else [])
        ],
        infeasible_trials_with_values=[(t, get_values(t)) for t in infeasible],
        axis_order=axis_order,
    )


def plot_pareto_front(
    study: "Study",
    *,
    target_names: list[str] | None = None,
    include_dominated_trials: bool = True,
    axis_order: list[int] | None = None,
    constraints_func: Callable[[FrozenTrial], Sequence[float]] | None = None,
    targets: Callable[[FrozenTrial], Sequence[float]] | None = None,
) -> "go.Figure":
    _imports.check()
    import plotly.graph_objects as go

    info = _get_pareto_front_info(
        study, target_names, include_dominated_trials, axis_order, constraints_func, targets
    )
    order = info.axis_order

    def scatter(trials_with_values, name, color):  # type: ignore[no-untyped-def]
        if not trials_with_values:
            return None
        coords = [[vals[i] for _, vals in trials_with_values] for i in order]
        text = [f"Trial {t.number}" for t, _ in trials_with_values]
        if info.n_targets == 2:
            return go.Scatter(
                x=coords[0], y=coords[1], text=text, mode="markers",
                marker={"color": color}, name=name,
            )
        return go.Scatter3d(
            x=coords[0], y=coords[1], z=coords[2], text=text, mode="markers",
            marker={"color": color, "size": 4}, name=name,
        )

    traces = [
        scatter(info.infeasible_trials_with_values, "Infeasible Trial", "#cccccc"),
        scatter(info.non_best_trials_with_values, "Trial", "blue"),
        scatter(info.best_trials_with_values, "Best Trial", "red"),
    ]
    fig = go.Figure([t for t in traces if t is not None])
    names = [info.target_names[i] for i in order]
    if info.n_targets == 2:
        fig.update_layout(
            title="Pareto-front Plot", xaxis_title=names[0], yaxis_title=names[1]
        )
    else:
        fig.update_layout(
            title="Pareto-front Plot",
            scene={
                "xaxis_title": names[0],
                "yaxis_title": names[1],
                "zaxis_title": names[2],
            },
        )
    return fig
