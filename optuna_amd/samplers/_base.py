"""BaseSampler — the three-phase sampling protocol.

Parity: reference ``optuna/samplers/_base.py`` (BaseSampler :33-230,
_process_constraints_after_trial :240, _INDEPENDENT_SAMPLING_WARNING_TEMPLATE).

Protocol per trial:
1. ``infer_relative_search_space(study, trial)`` — which params this sampler will
   sample jointly;
2. ``sample_relative(study, trial, search_space)`` — joint sample for those params,
   called lazily at the first ``suggest_*`` of the trial;
3. ``sample_independent(study, trial, name, dist)`` — fallback for params outside
   the relative space (dynamic / conditional search spaces).
"""
from __future__ import annotations

import abc
import warnings
from typing import TYPE_CHECKING, Any, Callable, Sequence

from optuna_amd.distributions import BaseDistribution
from optuna_amd.trial import FrozenTrial, TrialState


if TYPE_CHECKING:
    from optuna_amd.study import Study

_CONSTRAINTS_KEY = "constraints"


class BaseSampler(abc.ABC):
    """Base class for samplers."""

    def __str__(self) -> str:
        return self.__class__.__name__

    @abc.abstractmethod
    def infer_relative_search_space(
        self, study: "Study", trial: FrozenTrial
    ) -> dict[str, BaseDistribution]:
        raise NotImplementedError

    @abc.abstractmethod
    def sample_relative(
        self, study: "Study", trial: FrozenTrial, search_space: dict[str, BaseDistribution]
    ) -> dict[str, Any]:
        raise NotImplementedError

    @abc.abstractmethod
    def sample_independent(
        self,
        study: "Study",
        trial: FrozenTrial,
        param_name: str,
        param_distribution: BaseDistribution,
    ) -> Any:
        raise NotImplementedError

    def before_trial(self, study: "Study", trial: FrozenTrial) -> None:
        pass

    def after_trial(
        self,
        study: "Study",
        trial: FrozenTrial,
        state: TrialState,
        values: Sequence[float] | None,
    ) -> None:
        pass

    def reseed_rng(self) -> None:
        pass

    def _raise_error_if_multi_objective(self, study: "Study") -> None:
        if study._is_multi_objective():
            raise ValueError(
                f"If the study is being used for multi-objective optimization, "
                f"{self.__class__.__name__} cannot be used."
            )


def _process_constraints_after_trial(
    constraints_func: Callable[[FrozenTrial], Sequence[float]],
    study: "Study",
    trial: FrozenTrial,
    state: TrialState,
) -> None:
    """Evaluate constraints_func and persist the result under the constraints key.

    Parity: reference samplers/_base.py:240-266 (runs for COMPLETE/PRUNED trials;
    non-float or NaN constraint values are hard errors).
    """
    assert state in (TrialState.COMPLETE, TrialState.FAIL, TrialState.PRUNED)
    if state != TrialState.COMPLETE and state != TrialState.PRUNED:
        return
    constraints = None
    try:
        con = constraints_func(trial)
        # NaN must abort BEFORE anything is assigned — the finally block then
        # records None, leaving the trial with no constraint values at all.
        values = tuple(float(c) for c in con)
        if any(c != c for c in values):  # NaN
            raise ValueError("Constraint values cannot be NaN.")
        if not isinstance(con, (tuple, list)):
            warnings.warn(
                f"Constraints should be a sequence of floats but got {type(con).__name__}."
            )
        constraints = values
    finally:
        assert constraints is None or isinstance(constraints, tuple)
        study._storage.set_trial_system_attr(
            trial._trial_id, _CONSTRAINTS_KEY, constraints
        )


_INDEPENDENT_SAMPLING_WARNING_TEMPLATE = (
    "The parameter '{param_name}' in trial#{trial_number} is sampled independently "
    "instead of relatively by {sampler_name} (fallback: {fallback_name}). "
    "{reason}"
)
