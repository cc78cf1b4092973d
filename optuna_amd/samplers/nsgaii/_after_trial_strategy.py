"""NSGA-II after-trial hook: persist constraint values.

Parity: reference ``optuna/samplers/nsgaii/_after_trial_strategy.py``.
``_process_constraints_after_trial`` is a module global so test doubles
patched on this module take effect.
"""
from __future__ import annotations

from typing import TYPE_CHECKING, Callable, Sequence

from optuna_amd.samplers._base import _process_constraints_after_trial
from optuna_amd.trial import FrozenTrial, TrialState


if TYPE_CHECKING:
    from optuna_amd.study import Study


class NSGAIIAfterTrialStrategy:
    def __init__(
        self, *, constraints_func: Callable[[FrozenTrial], Sequence[float]] | None = None
    ) -> None:
        self._constraints_func = constraints_func

    def __call__(
        self,
        study: "Study",
        trial: FrozenTrial,
        state: TrialState,
        values: Sequence[float] | None = None,
    ) -> None:
        if self._constraints_func is not None:
            _process_constraints_after_trial(self._constraints_func, study, trial, state)
