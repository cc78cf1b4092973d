"""Median pruner — PercentilePruner at the 50th percentile.

Parity: reference ``optuna/pruners/_median.py`` :4,77-87.
"""
from __future__ import annotations

from optuna_amd.pruners._percentile import PercentilePruner


class MedianPruner(PercentilePruner):
    """Prune if the trial's best intermediate result is worse than the median of
    intermediate results of previous trials at the same step."""

    def __init__(
        self,
        n_startup_trials: int = 5,
        n_warmup_steps: int = 0,
        interval_steps: int = 1,
        *,
        n_min_trials: int = 1,
    ) -> None:
        super().__init__(
            50.0,
            n_startup_trials,
            n_warmup_steps,
            interval_steps,
            n_min_trials=n_min_trials,
        )
