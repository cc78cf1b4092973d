"""Threshold pruner: prune on an intermediate value crossing lower/upper/NaN.

Parity: reference ``optuna/pruners/_threshold.py`` (ThresholdPruner :29, prune :119).
"""
from __future__ import annotations

import math
from typing import TYPE_CHECKING, Any

from optuna_amd.pruners._base import BasePruner
from optuna_amd.pruners._percentile import _is_first_in_interval_step
from optuna_amd.trial import FrozenTrial


if TYPE_CHECKING:
    from optuna_amd.study import Study


def _check_value(value: Any) -> float:
    try:
        return float(value)
    except (TypeError, ValueError) as e:
        raise TypeError(
            f"The `value` argument is of type '{type(value).__name__}' but supposed to be a "
            "float."
        ) from e


class ThresholdPruner(BasePruner):
    def __init__(
        self,
        lower: float | None = None,
        upper: float | None = None,
        n_warmup_steps: int = 0,
        interval_steps: int = 1,
    ) -> None:
        if lower is None and upper is None:
            raise TypeError("Either lower or upper must be specified.")
        lower_f = _check_value(lower) if lower is not None else -math.inf
        upper_f = _check_value(upper) if upper is not None else math.inf
        if lower_f > upper_f:
            raise ValueError("lower should be smaller than upper.")
        if n_warmup_steps < 0:
            raise ValueError(
                f"Number of warmup steps cannot be negative but got {n_warmup_steps}."
            )
        if interval_steps < 1:
            raise ValueError(
                f"Pruning interval steps must be at least 1 but got {interval_steps}."
            )
        self._lower = lower_f
        self._upper = upper_f
        self._n_warmup_steps = n_warmup_steps
        self._interval_steps = interval_steps

    def prune(self, study: "Study", trial: FrozenTrial) -> bool:
        step = trial.last_step
        if step is None:
            return False
        if step < self._n_warmup_steps:
            return False
        if not _is_first_in_interval_step(
            step, trial.intermediate_values.keys(), self._n_warmup_steps, self._interval_steps
        ):
            return False
        latest_value = trial.intermediate_values[step]
        if math.isnan(latest_value):
            return True
        return latest_value < self._lower or latest_value > self._upper
