"""Visualization: 13 plotly plot functions, each with a pure ``_get_*_info`` data
layer, mirrored under ``optuna_amd.visualization.matplotlib``.

Parity: reference ``optuna/visualization/__init__.py`` :1-33.
"""
from optuna_amd.visualization._contour import plot_contour
from optuna_amd.visualization._edf import plot_edf
from optuna_amd.visualization._hypervolume_history import plot_hypervolume_history
from optuna_amd.visualization._intermediate_values import plot_intermediate_values
from optuna_amd.visualization._optimization_history import plot_optimization_history
from optuna_amd.visualization._parallel_coordinate import plot_parallel_coordinate
from optuna_amd.visualization._param_importances import plot_param_importances
from optuna_amd.visualization._pareto_front import plot_pareto_front
from optuna_amd.visualization._plotly_imports import is_available
from optuna_amd.visualization._rank import plot_rank
from optuna_amd.visualization._slice import plot_slice
from optuna_amd.visualization._terminator_improvement import plot_terminator_improvement
from optuna_amd.visualization._timeline import plot_timeline


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
    "matplotlib",
]


def __getattr__(name: str):
    if name == "matplotlib":
        import optuna_amd.visualization.matplotlib as mod

        return mod
    raise AttributeError(f"module {__name__!r} has no attribute {name!r}")
