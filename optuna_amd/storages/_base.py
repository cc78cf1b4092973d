"""BaseStorage — persistence AND the distributed coordination contract.

Parity: reference ``optuna/storages/_base.py`` (BaseStorage :21; thread-safety and
RUNNING-trial single-owner contract :29-47; default helpers get_best_trial :511,
get_trial_id_from_study_id_trial_number :280, check_trial_is_updatable :603).

Contract highlights every backend must honor (the conformance suite in
``optuna_amd/testing/pytest_storages.py`` enforces them):

* Thread-safe; one process may share a storage object across threads.
* A trial in a finished state is immutable; mutating it raises
  ``UpdateFinishedTrialError`` (RuntimeError).
* ``set_trial_state_values(trial_id, RUNNING)`` on a WAITING trial is a
  compare-and-swap: returns False if another worker claimed it first.
* Trial numbers are consecutive per study in creation order.
"""
from __future__ import annotations

import abc
from typing import Any, Container, Sequence

from optuna_amd.distributions import BaseDistribution
from optuna_amd.exceptions import UpdateFinishedTrialError
from optuna_amd.study._frozen import FrozenStudy
from optuna_amd.study._study_direction import StudyDirection
from optuna_amd.trial import FrozenTrial, TrialState


DEFAULT_STUDY_NAME_PREFIX = "no-name-"


class BaseStorage(abc.ABC):
    """Abstract storage: study/trial CRUD used by the whole runtime."""

    # ---- study CRUD -----------------------------------------------------------------

    @abc.abstractmethod
    def create_new_study(
        self, directions: Sequence[StudyDirection], study_name: str | None = None
    ) -> int:
        """Create a study; raise DuplicatedStudyError on a name collision."""
        raise NotImplementedError

    @abc.abstractmethod
    def delete_study(self, study_id: int) -> None:
        raise NotImplementedError

    @abc.abstractmethod
    def set_study_user_attr(self, study_id: int, key: str, value: Any) -> None:
        raise NotImplementedError

    @abc.abstractmethod
    def set_study_system_attr(self, study_id: int, key: str, value: Any) -> None:
        raise NotImplementedError

    @abc.abstractmethod
    def get_study_id_from_name(self, study_name: str) -> int:
        raise NotImplementedError

    @abc.abstractmethod
    def get_study_name_from_id(self, study_id: int) -> str:
        raise NotImplementedError

    @abc.abstractmethod
    def get_study_directions(self, study_id: int) -> list[StudyDirection]:
        raise NotImplementedError

    @abc.abstractmethod
    def get_study_user_attrs(self, study_id: int) -> dict[str, Any]:
        raise NotImplementedError

    @abc.abstractmethod
    def get_study_system_attrs(self, study_id: int) -> dict[str, Any]:
        raise NotImplementedError

    @abc.abstractmethod
    def get_all_studies(self) -> list[FrozenStudy]:
        raise NotImplementedError

    # ---- trial CRUD -----------------------------------------------------------------

    @abc.abstractmethod
    def create_new_trial(self, study_id: int, template_trial: FrozenTrial | None = None) -> int:
        """Create a trial (RUNNING, or a copy of ``template_trial``); returns trial_id.

        Trial numbers are dense 0..N-1 in creation order within the study, even
        under concurrent creation from many workers.
        """
        raise NotImplementedError

    @abc.abstractmethod
    def set_trial_param(
        self,
        trial_id: int,
        param_name: str,
        param_value_internal: float,
        distribution: BaseDistribution,
    ) -> None:
        raise NotImplementedError

    def get_trial_id_from_study_id_trial_number(self, study_id: int, trial_number: int) -> int:
        trials = self.get_all_trials(study_id, deepcopy=False)
        if len(trials) <= trial_number:
            raise KeyError(
                f"No trial with trial number {trial_number} exists in study with study_id "
                f"{study_id}."
            )
        return trials[trial_number]._trial_id

    def get_trial_number_from_id(self, trial_id: int) -> int:
        return self.get_trial(trial_id).number

    def get_trial_param(self, trial_id: int, param_name: str) -> float:
        trial = self.get_trial(trial_id)
        return trial.distributions[param_name].to_internal_repr(trial.params[param_name])

    @abc.abstractmethod
    def set_trial_state_values(
        self, trial_id: int, state: TrialState, values: Sequence[float] | None = None
    ) -> bool:
        """Update state (and optionally values). WAITING→RUNNING is a CAS; returns
        False when the claim lost a race; all other transitions return True."""
        raise NotImplementedError

    @abc.abstractmethod
    def set_trial_intermediate_value(
        self, trial_id: int, step: int, intermediate_value: float
    ) -> None:
        raise NotImplementedError

    @abc.abstractmethod
    def set_trial_user_attr(self, trial_id: int, key: str, value: Any) -> None:
        raise NotImplementedError

    @abc.abstractmethod
    def set_trial_system_attr(self, trial_id: int, key: str, value: Any) -> None:
        raise NotImplementedError

    # ---- reads ----------------------------------------------------------------------

    @abc.abstractmethod
    def get_trial(self, trial_id: int) -> FrozenTrial:
        raise NotImplementedError

    @abc.abstractmethod
    def get_all_trials(
        self,
        study_id: int,
        deepcopy: bool = True,
        states: Container[TrialState] | None = None,
    ) -> list[FrozenTrial]:
        raise NotImplementedError

    def get_n_trials(self, study_id: int, state: tuple[TrialState, ...] | TrialState | None = None) -> int:
        if isinstance(state, TrialState):
            state = (state,)
        return len(self.get_all_trials(study_id, deepcopy=False, states=state))

    def get_best_trial(self, study_id: int) -> FrozenTrial:
        """Best COMPLETE trial for a single-objective study (reference :511-560)."""
        all_trials = self.get_all_trials(study_id, deepcopy=False, states=(TrialState.COMPLETE,))
        if len(all_trials) == 0:
            raise ValueError("No trials are completed yet.")
        directions = self.get_study_directions(study_id)
        if len(directions) > 1:
            raise RuntimeError(
                "Best trial can be obtained only for single-objective optimization."
            )
        direction = directions[0]
        import math

        # NaN objective values rank worst in either direction.
        def _key_max(t: FrozenTrial) -> float:
            v = t.value
            assert v is not None
            return -math.inf if math.isnan(v) else v

        def _key_min(t: FrozenTrial) -> float:
            v = t.value
            assert v is not None
            return math.inf if math.isnan(v) else v

        if direction == StudyDirection.MAXIMIZE:
            best_trial = max(all_trials, key=_key_max)
        else:
            best_trial = min(all_trials, key=_key_min)
        # COMPLETE trials are immutable; the record from the no-copy listing IS
        # current, so no re-read (which would deepcopy in cached storages).
        return best_trial

    def get_trial_params(self, trial_id: int) -> dict[str, Any]:
        return self.get_trial(trial_id).params

    def get_trial_user_attrs(self, trial_id: int) -> dict[str, Any]:
        return self.get_trial(trial_id).user_attrs

    def get_trial_system_attrs(self, trial_id: int) -> dict[str, Any]:
        return self.get_trial(trial_id).system_attrs

    def remove_session(self) -> None:
        pass

    def check_trial_is_updatable(self, trial_id: int, trial_state: TrialState) -> None:
        if trial_state.is_finished():
            trial = self.get_trial(trial_id)
            raise UpdateFinishedTrialError(
                f"Trial#{trial.number} has already finished and can not be updated."
            )
