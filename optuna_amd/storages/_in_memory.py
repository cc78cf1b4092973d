"""In-process storage: dict + RLock, copy-on-write trial updates.

Parity: reference ``optuna/storages/_in_memory.py`` (InMemoryStorage :26, COW trial
updates :217-222, best-trial cache `_update_cache` :296, WAITING fast path :386-396).
"""
from __future__ import annotations

import copy
import math
import threading
import uuid
from datetime import datetime
from typing import Any, Container, Sequence

from optuna_amd.distributions import BaseDistribution, check_distribution_compatibility
from optuna_amd.exceptions import DuplicatedStudyError
from optuna_amd.storages._base import DEFAULT_STUDY_NAME_PREFIX, BaseStorage
from optuna_amd.study._frozen import FrozenStudy
from optuna_amd.study._study_direction import StudyDirection
from optuna_amd.trial import FrozenTrial, TrialState


class _StudyInfo:
    def __init__(self, name: str, directions: list[StudyDirection]) -> None:
        self.name = name
        self.directions = directions
        self.user_attrs: dict[str, Any] = {}
        self.system_attrs: dict[str, Any] = {}
        self.trials: list[FrozenTrial] = []
        self.param_distribution: dict[str, BaseDistribution] = {}
        self.best_trial_id: int | None = None
        # Per-state row index (trial numbers), so state-filtered reads cost
        # O(matching) instead of O(all trials). Finished states only grow; the
        # RUNNING/WAITING lists stay small, so removals are cheap.
        self.state_rows: dict[TrialState, list[int]] = {s: [] for s in TrialState}
        # COMPLETE/PRUNED trials in finish order: finished trials are immutable,
        # so samplers can read history deltas by log offset instead of
        # re-scanning all trials per suggest (see get_finished_trials_since).
        self.finished_log: list[FrozenTrial] = []


class InMemoryStorage(BaseStorage):
    """Single-process storage backend (the default for ``create_study(storage=None)``)."""

    def __init__(self) -> None:
        self._trial_id_offset = 0
        self._studies: dict[int, _StudyInfo] = {}
        self._max_study_id = -1
        self._study_name_to_id: dict[str, int] = {}
        self._trial_id_to_study_id_and_number: dict[int, tuple[int, int]] = {}
        self._lock = threading.RLock()

    def __getstate__(self) -> dict[str, Any]:
        state = self.__dict__.copy()
        del state["_lock"]
        return state

    def __setstate__(self, state: dict[str, Any]) -> None:
        self.__dict__.update(state)
        self._lock = threading.RLock()

    # ---- studies --------------------------------------------------------------------

    def create_new_study(
        self, directions: Sequence[StudyDirection], study_name: str | None = None
    ) -> int:
        with self._lock:
            study_id = self._max_study_id + 1
            if study_name is not None:
                if study_name in self._study_name_to_id:
                    raise DuplicatedStudyError(
                        f"Another study with name '{study_name}' already exists."
                    )
            else:
                study_name = DEFAULT_STUDY_NAME_PREFIX + str(uuid.uuid4())
            self._max_study_id = study_id
            self._studies[study_id] = _StudyInfo(study_name, list(directions))
            self._study_name_to_id[study_name] = study_id
            return study_id

    def delete_study(self, study_id: int) -> None:
        with self._lock:
            self._check_study_id(study_id)
            for trial in self._studies[study_id].trials:
                del self._trial_id_to_study_id_and_number[trial._trial_id]
            study_name = self._studies[study_id].name
            del self._study_name_to_id[study_name]
            del self._studies[study_id]

    def set_study_user_attr(self, study_id: int, key: str, value: Any) -> None:
        with self._lock:
            self._check_study_id(study_id)
            self._studies[study_id].user_attrs[key] = value

    def set_study_system_attr(self, study_id: int, key: str, value: Any) -> None:
        with self._lock:
            self._check_study_id(study_id)
            self._studies[study_id].system_attrs[key] = value

    def get_study_id_from_name(self, study_name: str) -> int:
        with self._lock:
            sid = self._study_name_to_id.get(study_name)
            if sid is None:
                raise KeyError(f"No such study {study_name}.")
            return sid

    def get_study_name_from_id(self, study_id: int) -> str:
        with self._lock:
            return self._study(study_id).name

    def get_study_directions(self, study_id: int) -> list[StudyDirection]:
        with self._lock:
            return self._study(study_id).directions

    def get_study_user_attrs(self, study_id: int) -> dict[str, Any]:
        with self._lock:
            return self._study(study_id).user_attrs

    def get_study_system_attrs(self, study_id: int) -> dict[str, Any]:
        with self._lock:
            return self._study(study_id).system_attrs

    def get_all_studies(self) -> list[FrozenStudy]:
        with self._lock:
            return [
                FrozenStudy(
                    study_name=info.name,
                    direction=None,
                    directions=info.directions,
                    user_attrs=copy.deepcopy(info.user_attrs),
                    system_attrs=copy.deepcopy(info.system_attrs),
                    study_id=study_id,
                )
                for study_id, info in self._studies.items()
            ]

    # ---- trials ---------------------------------------------------------------------

    def create_new_trial(self, study_id: int, template_trial: FrozenTrial | None = None) -> int:
        with self._lock:
            self._check_study_id(study_id)
            if template_trial is None:
                trial = self._create_running_trial()
            else:
                trial = copy.deepcopy(template_trial)
            trial_id = self._trial_id_offset + len(self._trial_id_to_study_id_and_number)
            trial.number = len(self._studies[study_id].trials)
            trial._trial_id = trial_id
            self._trial_id_to_study_id_and_number[trial_id] = (study_id, trial.number)
            self._studies[study_id].trials.append(trial)
            self._studies[study_id].state_rows[trial.state].append(trial.number)
            if trial.state in (TrialState.COMPLETE, TrialState.PRUNED):
                self._studies[study_id].finished_log.append(trial)
            self._update_cache(trial_id, study_id)
            return trial_id

    @staticmethod
    def _create_running_trial() -> FrozenTrial:
        # Fresh RUNNING record; id/number are assigned by create_new_trial.
        return FrozenTrial(
            state=TrialState.RUNNING,
            datetime_start=datetime.now(),
            datetime_complete=None,
            value=None,
            intermediate_values={},
            params={},
            distributions={},
            user_attrs={},
            system_attrs={},
            trial_id=-1,
            number=-1,
        )

    def set_trial_param(
        self,
        trial_id: int,
        param_name: str,
        param_value_internal: float,
        distribution: BaseDistribution,
    ) -> None:
        with self._lock:
            trial = self._get_trial(trial_id)
            self.check_trial_is_updatable(trial_id, trial.state)
            study_id = self._trial_id_to_study_id_and_number[trial_id][0]
            # Cross-trial compatibility check against the first recorded distribution.
            if param_name in self._studies[study_id].param_distribution:
                check_distribution_compatibility(
                    self._studies[study_id].param_distribution[param_name], distribution
                )
            self._studies[study_id].param_distribution[param_name] = distribution
            trial = copy.copy(trial)
            trial.params = {**trial.params, param_name: distribution.to_external_repr(param_value_internal)}
            trial.distributions = {**trial.distributions, param_name: distribution}
            self._set_trial(trial_id, trial)

    def get_trial_id_from_study_id_trial_number(self, study_id: int, trial_number: int) -> int:
        with self._lock:
            self._check_study_id(study_id)
            trials = self._studies[study_id].trials
            if len(trials) <= trial_number:
                raise KeyError(
                    f"No trial with trial number {trial_number} exists in study with "
                    f"study_id {study_id}."
                )
            return trials[trial_number]._trial_id

    def get_trial_number_from_id(self, trial_id: int) -> int:
        with self._lock:
            self._check_trial_id(trial_id)
            return self._trial_id_to_study_id_and_number[trial_id][1]

    def get_best_trial(self, study_id: int) -> FrozenTrial:
        with self._lock:
            self._check_study_id(study_id)
            if len(self._studies[study_id].directions) > 1:
                raise RuntimeError(
                    "Best trial can be obtained only for single-objective optimization."
                )
            best_trial_id = self._studies[study_id].best_trial_id
            if best_trial_id is None:
                raise ValueError("No trials are completed yet.")
            return self.get_trial(best_trial_id)

    def get_trial_param(self, trial_id: int, param_name: str) -> float:
        with self._lock:
            trial = self._get_trial(trial_id)
            distribution = trial.distributions[param_name]
            return distribution.to_internal_repr(trial.params[param_name])

    def set_trial_state_values(
        self, trial_id: int, state: TrialState, values: Sequence[float] | None = None
    ) -> bool:
        with self._lock:
            trial = self._get_trial(trial_id)
            self.check_trial_is_updatable(trial_id, trial.state)
            trial = copy.copy(trial)
            if state == TrialState.RUNNING and trial.state != TrialState.WAITING:
                return False  # WAITING→RUNNING claim lost (or invalid transition).
            study_id, number = self._trial_id_to_study_id_and_number[trial_id]
            if state != trial.state:
                state_rows = self._studies[study_id].state_rows
                state_rows[trial.state].remove(number)
                state_rows[state].append(number)
            trial.state = state
            if values is not None:
                trial.values = list(values)
            if state == TrialState.RUNNING:
                trial.datetime_start = datetime.now()
            if state.is_finished():
                trial.datetime_complete = datetime.now()
                self._set_trial(trial_id, trial)
                if state in (TrialState.COMPLETE, TrialState.PRUNED):
                    self._studies[study_id].finished_log.append(trial)
                self._update_cache(trial_id, study_id)
            else:
                self._set_trial(trial_id, trial)
            return True

    def _update_cache(self, trial_id: int, study_id: int) -> None:
        """Incremental best-trial cache: compare the newly finished trial only."""
        info = self._studies[study_id]
        if len(info.directions) > 1:
            return  # single-objective cache only
        trial = self._get_trial(trial_id)
        if trial.state != TrialState.COMPLETE:
            return
        if trial.value is None:
            # Legal via direct storage calls (study.tell() validates, the raw
            # set_trial_state_values() API does not require values).
            return
        if math.isnan(trial.value):
            return  # NaN never becomes the incumbent
        incumbent_id = info.best_trial_id
        if incumbent_id is None:
            info.best_trial_id = trial_id
            return
        incumbent_value = self._get_trial(incumbent_id).value
        assert incumbent_value is not None
        sign = -1.0 if info.directions[0] == StudyDirection.MINIMIZE else 1.0
        if math.isnan(incumbent_value) or sign * trial.value > sign * incumbent_value:
            info.best_trial_id = trial_id

    def set_trial_intermediate_value(
        self, trial_id: int, step: int, intermediate_value: float
    ) -> None:
        with self._lock:
            trial = self._get_trial(trial_id)
            self.check_trial_is_updatable(trial_id, trial.state)
            trial = copy.copy(trial)
            trial.intermediate_values = {**trial.intermediate_values, step: intermediate_value}
            self._set_trial(trial_id, trial)

    def set_trial_user_attr(self, trial_id: int, key: str, value: Any) -> None:
        with self._lock:
            trial = self._get_trial(trial_id)
            self.check_trial_is_updatable(trial_id, trial.state)
            trial = copy.copy(trial)
            trial.user_attrs = {**trial.user_attrs, key: value}
            self._set_trial(trial_id, trial)

    def set_trial_system_attr(self, trial_id: int, key: str, value: Any) -> None:
        with self._lock:
            trial = self._get_trial(trial_id)
            self.check_trial_is_updatable(trial_id, trial.state)
            trial = copy.copy(trial)
            trial.system_attrs = {**trial.system_attrs, key: value}
            self._set_trial(trial_id, trial)

    def get_trial(self, trial_id: int) -> FrozenTrial:
        with self._lock:
            return self._get_trial(trial_id)

    def _locate(self, trial_id: int) -> tuple[_StudyInfo, int]:
        """Resolve a trial id to its study record and row number."""
        where = self._trial_id_to_study_id_and_number.get(trial_id)
        if where is None:
            raise KeyError(f"No trial with trial_id {trial_id} exists.")
        return self._studies[where[0]], where[1]

    def _get_trial(self, trial_id: int) -> FrozenTrial:
        info, number = self._locate(trial_id)
        return info.trials[number]

    def _set_trial(self, trial_id: int, trial: FrozenTrial) -> None:
        info, number = self._locate(trial_id)
        info.trials[number] = trial

    def get_all_trials(
        self,
        study_id: int,
        deepcopy: bool = True,
        states: Container[TrialState] | None = None,
    ) -> list[FrozenTrial]:
        with self._lock:
            self._check_study_id(study_id)
            all_trials = self._studies[study_id].trials
            if states is not None:
                # Union of per-state row indices → O(matching log matching).
                state_rows = self._studies[study_id].state_rows
                rows: list[int] = []
                for s in TrialState:
                    if s in states:
                        rows.extend(state_rows[s])
                rows.sort()
                trials = [all_trials[i] for i in rows]
            else:
                trials = list(all_trials)
            if deepcopy:
                trials = copy.deepcopy(trials)
            return trials

    def get_finished_trials_since(self, study_id: int, start: int) -> list[FrozenTrial]:
        """Delta read of COMPLETE/PRUNED trials in finish order.

        Non-standard extension (absent from the reference's BaseStorage): finished
        trials are immutable, so a sampler holding an incremental history mirror can
        fetch only the trials finished since its last read — O(delta) instead of the
        O(all trials) ``get_all_trials`` scan per suggest. Callers discover it with
        ``getattr(storage, "get_finished_trials_since", None)``.
        """
        with self._lock:
            self._check_study_id(study_id)
            return list(self._studies[study_id].finished_log[start:])

    def get_n_trials(
        self, study_id: int, state: tuple[TrialState, ...] | TrialState | None = None
    ) -> int:
        if isinstance(state, TrialState):
            state = (state,)
        with self._lock:
            self._check_study_id(study_id)
            if state is None:
                return len(self._studies[study_id].trials)
            state_rows = self._studies[study_id].state_rows
            return sum(len(state_rows[s]) for s in TrialState if s in state)

    # ---- helpers --------------------------------------------------------------------

    def _study(self, study_id: int) -> _StudyInfo:
        """Checked lookup of the study record."""
        info = self._studies.get(study_id)
        if info is None:
            raise KeyError(f"No study with study_id {study_id} exists.")
        return info

    def _check_study_id(self, study_id: int) -> None:
        self._study(study_id)

    def _check_trial_id(self, trial_id: int) -> None:
        self._locate(trial_id)
