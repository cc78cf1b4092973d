"""Parameter-importance bar chart.

Parity: reference ``optuna/visualization/_param_importances.py``.
"""
from __future__ import annotations

from typing import TYPE_CHECKING, Callable, NamedTuple

from optuna_amd.importance import BaseImportanceEvaluator, get_param_importances
from optuna_amd.trial import FrozenTrial
from optuna_amd.visualization._plotly_imports import _imports


if TYPE_CHECKING:
    import plotly.graph_objects as go

    from optuna_amd.study import Study


class _ImportancesInfo(NamedTuple):
    importance_values: list[float]
    param_names: list[str]
    importance_labels: list[str]
    target_name: str


def _get_importances_info(
    study: "Study",
    evaluator: BaseImportanceEvaluator | None,
    params: list[str] | None,
    target: Callable[[FrozenTrial], float] | None,
    target_name: str,
) -> _ImportancesInfo:
    importances = get_param_importances(
        study, evaluator=evaluator, params=params, target=target
    )
    importances = dict(reversed(list(importances.items())))  # ascending for barh
    values = list(importances.values())
    return _ImportancesInfo(
        importance_values=values,
        param_names=list(importances.keys()),
        importance_labels=[f"{v:.2f}" if v > 0.01 else f"{v:.2e}" for v in values],
        target_name=target_name,
    )


def plot_param_importances(
    study: "Study",
    evaluator: BaseImportanceEvaluator | None = None,
    params: list[str] | None = None,
    *,
    target: Callable[[FrozenTrial], float] | None = None,
    target_name: str = "Objective Value",
) -> "go.Figure":
    _imports.check()
    import plotly.graph_objects as go

    info = _get_importances_info(study, evaluator, params, target, target_name)
    fig = go.Figure(
        go.Bar(
            x=info.importance_values,
            y=info.param_names,
            text=info.importance_labels,
            textposition="outside",
            orientation="h",
        )
    )
    fig.update_layout(
        title=f"Hyperparameter Importances for {info.target_name}",
        xaxis_title=f"Hyperparameter Importance",
        yaxis_title="Hyperparameter",
    )
    return fig
