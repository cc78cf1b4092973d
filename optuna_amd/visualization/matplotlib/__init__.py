"""Matplotlib mirrors of the plotly visualizations.

Every function consumes the same ``_get_*_info`` data layer as the plotly
renderers (parity: reference ``optuna/visualization/matplotlib/``).
"""
from __future__ import annotations

from typing import TYPE_CHECKING, Any, Callable, Sequence

from optuna_amd.trial import FrozenTrial
from optuna_amd.visualization.matplotlib._matplotlib_imports import _imports, is_available


if TYPE_CHECKING:
    from matplotlib.axes import Axes

    from optuna_amd.study import Study

__all__ = [
    "is_available",
    "plot_contour",
    "plot_edf",
    "plot_hypervolume_history",
    "plot_intermediate_values",
    "plot_optimization_history",
    "plot_parallel_coordinate",
    "plot_param_importances",
    "plot_pareto_front",
    "plot_rank",
    "plot_slice",
    "plot_terminator_improvement",
    "plot_timeline",
]


def plot_optimization_history(
    study: "Study | Sequence[Study]",
    *,
    target: Callable[[FrozenTrial], float] | None = None,
    target_name: str = "Objective Value",
    error_bar: bool = False,
) -> "Axes":
    _imports.check()
    from matplotlib import pyplot as plt

    from optuna_amd.visualization._optimization_history import (
        _ValueState,
        _get_optimization_history_info_list,
    )

    info_list = _get_optimization_history_info_list(study, target, target_name, error_bar)
    _, ax = plt.subplots()
    ax.set_title("Optimization History Plot")
    ax.set_xlabel("Trial")
    ax.set_ylabel(target_name)
    for info in info_list:
        feasible = [
            (n, v)
            for n, v, s in zip(
                info.trial_numbers, info.values_info.values, info.values_info.states
            )
            if s == _ValueState.Feasible
        ]
        if feasible:
            ax.scatter(*zip(*feasible), s=10, label=info.values_info.label_name)
        if info.best_values_info is not None:
            ax.plot(
                info.trial_numbers,
                info.best_values_info.values,
                color="tab:red",
                label=info.best_values_info.label_name,
            )
    ax.legend()
    return ax


def plot_slice(
    study: "Study",
    params: list[str] | None = None,
    *,
    target: Callable[[FrozenTrial], float] | None = None,
    target_name: str = "Objective Value",
) -> "Any":
    _imports.check()
    from matplotlib import pyplot as plt

    from optuna_amd.visualization._slice import _get_slice_plot_info

    info = _get_slice_plot_info(study, params, target, target_name)
    n = max(1, len(info.subplots))
    _, axes = plt.subplots(1, n, sharey=True, squeeze=False)
    for ax, sub in zip(axes[0], info.subplots):
        ax.scatter(sub.x, sub.y, c=sub.trial_numbers, cmap="Blues", s=10)
        ax.set_xlabel(sub.param_name)
        if sub.is_log:
            ax.set_xscale("log")
    axes[0][0].set_ylabel(info.target_name)
    plt.suptitle("Slice Plot")
    return axes[0]


def plot_edf(
    study: "Study | Sequence[Study]",
    *,
    target: Callable[[FrozenTrial], float] | None = None,
    target_name: str = "Objective Value",
) -> "Axes":
    _imports.check()
    from matplotlib import pyplot as plt

    from optuna_amd.visualization._edf import _get_edf_info

    info = _get_edf_info(study, target, target_name)
    _, ax = plt.subplots()
    ax.set_title("Empirical Distribution Function Plot")
    ax.set_xlabel(target_name)
    ax.set_ylabel("Cumulative Probability")
    for line in info.lines:
        ax.plot(info.x_values, line.y_values, label=line.study_name)
    if info.lines:
        ax.legend()
    return ax


def plot_intermediate_values(study: "Study") -> "Axes":
    _imports.check()
    from matplotlib import pyplot as plt

    from optuna_amd.visualization._intermediate_values import _get_intermediate_plot_info

    info = _get_intermediate_plot_info(study)
    _, ax = plt.subplots()
    ax.set_title("Intermediate Values Plot")
    ax.set_xlabel("Step")
    ax.set_ylabel("Intermediate Value")
    for trial_info in info.trial_infos:
        steps = [s for s, _ in trial_info.sorted_intermediate_values]
        values = [v for _, v in trial_info.sorted_intermediate_values]
        ax.plot(steps, values, marker="o", markersize=2, alpha=0.6)
    return ax


def plot_pareto_front(
    study: "Study",
    *,
    target_names: list[str] | None = None,
    include_dominated_trials: bool = True,
    axis_order: list[int] | None = None,
    constraints_func: Callable[[FrozenTrial], Sequence[float]] | None = None,
    targets: Callable[[FrozenTrial], Sequence[float]] | None = None,
) -> "Axes":
    _imports.check()
    from matplotlib import pyplot as plt

    from optuna_amd.visualization._pareto_front import _get_pareto_front_info

    info = _get_pareto_front_info(
        study, target_names, include_dominated_trials, axis_order, constraints_func, targets
    )
    if info.n_targets == 2:
        _, ax = plt.subplots()
    else:
        fig = plt.figure()
        ax = fig.add_subplot(projection="3d")
    ax.set_title("Pareto-front Plot")
    order = info.axis_order

    def scatter(twv, label, color):  # type: ignore[no-untyped-def]
        if not twv:
            return
        coords = [[vals[i] for _, vals in twv] for i in order]
        ax.scatter(*coords, label=label, color=color, s=12)

    scatter(info.non_best_trials_with_values, "Trial", "tab:blue")
    scatter(info.best_trials_with_values, "Best Trial", "tab:red")
    names = [info.target_names[i] for i in order]
    ax.set_xlabel(names[0])
    ax.set_ylabel(names[1])
    if info.n_targets == 3:
        ax.set_zlabel(names[2])
    ax.legend()
    return ax


def plot_param_importances(
    study: "Study",
    evaluator: Any = None,
    params: list[str] | None = None,
    *,
    target: Callable[[FrozenTrial], float] | None = None,
    target_name: str = "Objective Value",
) -> "Axes":
    _imports.check()
    from matplotlib import pyplot as plt

    from optuna_amd.visualization._param_importances import _get_importances_info

    info = _get_importances_info(study, evaluator, params, target, target_name)
    _, ax = plt.subplots()
    ax.barh(info.param_names, info.importance_values)
    ax.set_title(f"Hyperparameter Importances for {info.target_name}")
    ax.set_xlabel("Hyperparameter Importance")
    ax.set_ylabel("Hyperparameter")
    return ax


def plot_contour(
    study: "Study",
    params: list[str] | None = None,
    *,
    target: Callable[[FrozenTrial], float] | None = None,
    target_name: str = "Objective Value",
) -> "Any":
    _imports.check()
    from matplotlib import pyplot as plt

    from optuna_amd.visualization._contour import _get_contour_info
    from optuna_amd.visualization._utils import _filter_nonfinite, _get_completed_trials

    info = _get_contour_info(study, params, target, target_name)
    n = len(info.sorted_params)
    _, axes = plt.subplots(max(n, 1), max(n, 1), squeeze=False)
    trials = _filter_nonfinite(_get_completed_trials(study), target=target)
    values = [target(t) if target is not None else t.value for t in trials]
    for yi, y_param in enumerate(info.sorted_params):
        for xi, x_param in enumerate(info.sorted_params):
            ax = axes[yi][xi]
            if x_param == y_param:
                ax.set_visible(False)
                continue
            xs, ys, zs = [], [], []
            for t, v in zip(trials, values):
                if x_param in t.params and y_param in t.params:
                    xs.append(t.params[x_param])
                    ys.append(t.params[y_param])
                    zs.append(v)
            if xs:
                sc = ax.scatter(xs, ys, c=zs, cmap="Blues", s=10)
            ax.set_xlabel(x_param)
            ax.set_ylabel(y_param)
    plt.suptitle("Contour Plot")
    return axes


def plot_parallel_coordinate(
    study: "Study",
    params: list[str] | None = None,
    *,
    target: Callable[[FrozenTrial], float] | None = None,
    target_name: str = "Objective Value",
) -> "Axes":
    _imports.check()
    from matplotlib import pyplot as plt

    from optuna_amd.visualization._parallel_coordinate import (
        _get_parallel_coordinate_info,
    )

    info = _get_parallel_coordinate_info(study, params, target, target_name)
    _, ax = plt.subplots()
    dims = [info.dim_objective] + info.dims_params
    n_lines = len(info.dim_objective.values)
    for li in range(n_lines):
        ys = []
        for dim in dims:
            lo, hi = dim.range
            spread = hi - lo if hi > lo else 1.0
            ys.append((dim.values[li] - lo) / spread)
        ax.plot(range(len(dims)), ys, alpha=0.4)
    ax.set_xticks(range(len(dims)))
    ax.set_xticklabels([d.label for d in dims], rotation=30)
    ax.set_title("Parallel Coordinate Plot")
    return ax


def plot_rank(
    study: "Study",
    params: list[str] | None = None,
    *,
    target: Callable[[FrozenTrial], float] | None = None,
    target_name: str = "Objective Value",
) -> "Any":
    _imports.check()
    from matplotlib import pyplot as plt

    from optuna_amd.visualization._rank import _get_rank_info

    info = _get_rank_info(study, params, target, target_name)
    n = max(len(info.params), 1)
    _, axes = plt.subplots(n, n, squeeze=False)
    for yi in range(len(info.params)):
        for xi in range(len(info.params)):
            sub = info.sub_plot_infos[yi][xi]
            axes[yi][xi].scatter(sub.xs, sub.ys, c=sub.colors, cmap="RdYlBu_r", s=10)
    plt.suptitle(f"Rank ({info.target_name})")
    return axes


def plot_timeline(study: "Study") -> "Axes":
    _imports.check()
    from matplotlib import pyplot as plt

    from optuna_amd.visualization._timeline import _STATE_COLORS, _get_timeline_info

    info = _get_timeline_info(study)
    _, ax = plt.subplots()
    for bar in info.bars:
        ax.barh(
            bar.number,
            (bar.complete - bar.start).total_seconds(),
            left=bar.start.timestamp(),
            color=_STATE_COLORS[bar.state],
        )
    ax.set_title("Timeline Plot")
    ax.set_xlabel("Datetime")
    ax.set_ylabel("Trial")
    return ax


def plot_hypervolume_history(
    study: "Study", reference_point: Sequence[float]
) -> "Axes":
    _imports.check()
    import numpy as np
    from matplotlib import pyplot as plt

    from optuna_amd.study._multi_objective import _normalize_value
    from optuna_amd.visualization._hypervolume_history import _get_hypervolume_history_info

    if not study._is_multi_objective():
        raise ValueError("Study must be multi-objective.")
    ref = np.array(
        [_normalize_value(v, d) for v, d in zip(reference_point, study.directions)]
    )
    info = _get_hypervolume_history_info(study, ref)
    _, ax = plt.subplots()
    ax.plot(info.trial_numbers, info.values, marker="o")
    ax.set_title("Hypervolume History Plot")
    ax.set_xlabel("Trial")
    ax.set_ylabel("Hypervolume")
    return ax


def plot_terminator_improvement(
    study: "Study",
    plot_error: bool = False,
    improvement_evaluator: Any = None,
    error_evaluator: Any = None,
    min_n_trials: int = 20,
) -> "Axes":
    _imports.check()
    from matplotlib import pyplot as plt

    from optuna_amd.visualization._terminator_improvement import _get_improvement_info

    info = _get_improvement_info(study, plot_error, improvement_evaluator, error_evaluator)
    _, ax = plt.subplots()
    ax.plot(info.trial_numbers, info.improvements, marker="o", label="Improvement")
    if info.errors is not None:
        ax.plot(info.trial_numbers, info.errors, marker="o", label="Error")
    ax.set_title("Terminator Improvement Plot")
    ax.set_xlabel("Trial")
    ax.set_ylabel("Terminator Improvement")
    ax.legend()
    return ax
