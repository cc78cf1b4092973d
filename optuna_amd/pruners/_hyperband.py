"""Hyperband pruner: a portfolio of ASHA brackets with budget-weighted assignment.

Trials are assigned to a bracket by ``crc32(study_name_trial_number) % total_budget``
so assignment is stable across distributed workers with no extra storage traffic.
Each bracket runs its own SuccessiveHalvingPruner with
``min_early_stopping_rate = bracket_id``; samplers see only sibling trials via the
``_BracketStudy`` filtered view (hooked in ``pruners._filter_study``).

Parity: reference ``optuna/pruners/_hyperband.py`` (HyperbandPruner :21,
n_brackets formula :199-212, budget :216-227, bracket assignment :242-264,
_BracketStudy :266-326).
"""
from __future__ import annotations

import binascii
import math
from typing import TYPE_CHECKING, Container

from optuna_amd import logging as _logging
from optuna_amd.pruners._base import BasePruner
from optuna_amd.pruners._successive_halving import SuccessiveHalvingPruner
from optuna_amd.trial import FrozenTrial, TrialState


if TYPE_CHECKING:
    from optuna_amd.study import Study

_logger = _logging.get_logger(__name__)


class HyperbandPruner(BasePruner):
    """Hyperband = portfolio of ASHA brackets (see module docstring)."""

    def __init__(
        self,
        min_resource: int = 1,
        max_resource: str | int = "auto",
        reduction_factor: int = 3,
        bootstrap_count: int = 0,
    ) -> None:
        if not isinstance(max_resource, int) and max_resource != "auto":
            raise ValueError(
                f"The 'max_resource' should be integer or 'auto', but max_resource = "
                f"{max_resource}."
            )
        if bootstrap_count > 0 and max_resource == "auto":
            raise ValueError(
                "bootstrap_count > 0 and max_resource == 'auto' are mutually incompatible."
            )
        self._min_resource = min_resource
        self._max_resource = max_resource
        self._reduction_factor = reduction_factor
        self._bootstrap_count = bootstrap_count
        self._pruners: list[SuccessiveHalvingPruner] = []
        self._total_trial_allocation_budget = 0
        self._trial_allocation_budgets: list[int] = []
        self._n_brackets: int | None = None

    def prune(self, study: "Study", trial: FrozenTrial) -> bool:
        if len(self._pruners) == 0:
            self._try_initialization(study)
            if len(self._pruners) == 0:
                return False
        bracket_id = self._get_bracket_id(study, trial)
        _logger.debug(f"{bracket_id}th bracket is selected")
        bracket_study = self._create_bracket_study(study, bracket_id)
        return self._pruners[bracket_id].prune(bracket_study, trial)

    def _try_initialization(self, study: "Study") -> None:
        if self._max_resource == "auto":
            trials = study.get_trials(deepcopy=False, states=(TrialState.COMPLETE,))
            last_steps = [t.last_step for t in trials if t.last_step is not None]
            if not last_steps:
                return
            self._max_resource = max(last_steps) + 1
        assert isinstance(self._max_resource, int)

        if self._n_brackets is None:
            # n_brackets = floor(log_rf(max_resource / min_resource)) + 1
            self._n_brackets = (
                math.floor(
                    math.log(self._max_resource / self._min_resource, self._reduction_factor)
                )
                + 1
            )
        _logger.debug(f"Hyperband has {self._n_brackets} brackets")

        for bracket_id in range(self._n_brackets):
            budget = self._calculate_trial_allocation_budget(bracket_id)
            self._total_trial_allocation_budget += budget
            self._trial_allocation_budgets.append(budget)
            self._pruners.append(
                SuccessiveHalvingPruner(
                    min_resource=self._min_resource,
                    reduction_factor=self._reduction_factor,
                    min_early_stopping_rate=bracket_id,
                    bootstrap_count=self._bootstrap_count,
                )
            )

    def _calculate_trial_allocation_budget(self, bracket_id: int) -> int:
        assert self._n_brackets is not None
        s = self._n_brackets - 1 - bracket_id
        return math.ceil(self._n_brackets * (self._reduction_factor**s) / (s + 1))

    def _get_bracket_id(self, study: "Study", trial: FrozenTrial) -> int:
        if len(self._pruners) == 0:
            return 0
        assert self._n_brackets is not None
        n = (
            binascii.crc32(f"{study.study_name}_{trial.number}".encode())
            % self._total_trial_allocation_budget
        )
        for bracket_id in range(self._n_brackets):
            n -= self._trial_allocation_budgets[bracket_id]
            if n < 0:
                return bracket_id
        raise AssertionError("unreachable")

    def _create_bracket_study(self, study: "Study", bracket_id: int) -> "Study":
        from optuna_amd.study import Study as _Study

        pruner = self

        class _BracketStudy(_Study):
            # Whitelist: SHA only needs trial reads / direction / storage; anything
            # else is a programming error we want loud.
            _VALID_ATTRS = (
                "get_trials",
                "_get_trials",
                "directions",
                "direction",
                "_directions",
                "_storage",
                "_study_id",
                "pruner",
                "study_name",
                "_bracket_id",
                "sampler",
                "trials",
                "_is_multi_objective",
                "stop",
                "_study",
                "_thread_local",
            )

            def __init__(self, study: "Study", bracket_id: int) -> None:
                super().__init__(
                    study_name=study.study_name,
                    storage=study._storage,
                    sampler=study.sampler,
                    pruner=pruner,
                )
                self._study = study
                self._bracket_id = bracket_id

            def get_trials(
                self,
                deepcopy: bool = True,
                states: Container[TrialState] | None = None,
            ) -> list[FrozenTrial]:
                trials = super()._get_trials(deepcopy=deepcopy, states=states)
                return [
                    t for t in trials if pruner._get_bracket_id(self, t) == self._bracket_id
                ]

            def stop(self) -> None:
                self._study.stop()

            def __getattribute__(self, attr_name):  # type: ignore[no-untyped-def]
                if attr_name not in _BracketStudy._VALID_ATTRS:
                    raise AttributeError(
                        f"_BracketStudy does not have attribute of '{attr_name}'"
                    )
                return object.__getattribute__(self, attr_name)

        return _BracketStudy(study, bracket_id)
