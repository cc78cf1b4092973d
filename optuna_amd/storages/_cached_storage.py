"""Write-through read cache for RDBStorage.

Finished trials are immutable, so they are cached forever by number; only
unfinished trials and trials newer than the last finished one are re-fetched
(delta query). Auto-applied by ``get_storage`` for RDB URLs.

Parity: reference ``optuna/storages/_cached_storage.py`` (_CachedStorage :36,
delta fetch :247-271).
"""
from __future__ import annotations

import copy
import threading
from typing import Any, Callable, Container, Sequence

from optuna_amd.distributions import BaseDistribution
from optuna_amd.storages._base import BaseStorage
from optuna_amd.storages._heartbeat import BaseHeartbeat
from optuna_amd.storages._rdb.storage import RDBStorage
from optuna_amd.study._frozen import FrozenStudy
from optuna_amd.study._study_direction import StudyDirection
from optuna_amd.trial import FrozenTrial, TrialState


class _StudyInfo:
    def __init__(self) -> None:
        # Cached finished trials, keyed by trial number.
        self.trials: dict[int, FrozenTrial] = {}
        self.unfinished_trial_ids: set[int] = set()
        # First-seen distribution per param name (compat check short-circuit).
        self.param_distribution: dict[str, Any] = {}
        self.last_finished_trial_id: int = -1
        self.directions: list[StudyDirection] | None = None
        self.name: str | None = None


class _CachedStorage(BaseStorage, BaseHeartbeat):
    def __init__(self, backend: RDBStorage) -> None:
        self._backend = backend
        self._studies: dict[int, _StudyInfo] = {}
        self._trial_id_to_study_id_and_number: dict[int, tuple[int, int]] = {}
        self._lock = threading.Lock()

    def __getstate__(self) -> dict[str, Any]:
        state = self.__dict__.copy()
        del state["_lock"]
        return state

    def __setstate__(self, state: dict[str, Any]) -> None:
        self.__dict__.update(state)
        self._lock = threading.Lock()

    # ---- study ops: pass-through (cheap / rare) -------------------------------------

    def create_new_study(
        self, directions: Sequence[StudyDirection], study_name: str | None = None
    ) -> int:
        study_id = self._backend.create_new_study(directions, study_name)
        with self._lock:
            study = _StudyInfo()
            study.name = self._backend.get_study_name_from_id(study_id)
            study.directions = list(directions)
            self._studies[study_id] = study
        return study_id

    def _get_cached_trial(self, trial_id: int) -> FrozenTrial | None:
        """The cached record iff it is finished (immutable); None otherwise."""
        mapped = self._trial_id_to_study_id_and_number.get(trial_id)
        if mapped is None:
            return None
        study_id, number = mapped
        trial = self._studies[study_id].trials.get(number)
        if trial is None or not trial.state.is_finished():
            return None
        return trial

    def delete_study(self, study_id: int) -> None:
        with self._lock:
            if study_id in self._studies:
                for number, trial in self._studies[study_id].trials.items():
                    self._trial_id_to_study_id_and_number.pop(trial._trial_id, None)
                del self._studies[study_id]
        self._backend.delete_study(study_id)

    def set_study_user_attr(self, study_id: int, key: str, value: Any) -> None:
        self._backend.set_study_user_attr(study_id, key, value)

    def set_study_system_attr(self, study_id: int, key: str, value: Any) -> None:
        self._backend.set_study_system_attr(study_id, key, value)

    def get_study_id_from_name(self, study_name: str) -> int:
        return self._backend.get_study_id_from_name(study_name)

    def get_study_name_from_id(self, study_id: int) -> str:
        with self._lock:
            info = self._studies.get(study_id)
            if info is not None and info.name is not None:
                return info.name
        return self._backend.get_study_name_from_id(study_id)

    def get_study_directions(self, study_id: int) -> list[StudyDirection]:
        with self._lock:
            info = self._studies.get(study_id)
            if info is not None and info.directions is not None:
                return list(info.directions)
        directions = self._backend.get_study_directions(study_id)
        with self._lock:
            self._studies.setdefault(study_id, _StudyInfo()).directions = list(directions)
        return directions

    def get_study_user_attrs(self, study_id: int) -> dict[str, Any]:
        return self._backend.get_study_user_attrs(study_id)

    def get_study_system_attrs(self, study_id: int) -> dict[str, Any]:
        return self._backend.get_study_system_attrs(study_id)

    def get_all_studies(self) -> list[FrozenStudy]:
        return self._backend.get_all_studies()

    # ---- trial ops ------------------------------------------------------------------

    def create_new_trial(self, study_id: int, template_trial: FrozenTrial | None = None) -> int:
        frozen_trial = self._backend._create_new_trial(study_id, template_trial)
        trial_id = frozen_trial._trial_id
        with self._lock:
            info = self._studies.setdefault(study_id, _StudyInfo())
            info.unfinished_trial_ids.add(trial_id)
            self._trial_id_to_study_id_and_number[trial_id] = (
                study_id,
                frozen_trial.number,
            )
            info.trials[frozen_trial.number] = frozen_trial
        return trial_id

    def set_trial_param(
        self,
        trial_id: int,
        param_name: str,
        param_value_internal: float,
        distribution: BaseDistribution,
    ) -> None:
        with self._lock:
            study_id, _ = self._trial_id_to_study_id_and_number[trial_id]
            info = self._studies.setdefault(study_id, _StudyInfo())
            cached_dist = info.param_distribution.get(param_name)
        # Write-through with the cached distribution so the backend can skip
        # its cross-trial compatibility query.
        self._backend._set_trial_param(
            trial_id, param_name, param_value_internal, distribution, cached_dist
        )
        if cached_dist is None:
            with self._lock:
                info.param_distribution[param_name] = distribution

    def get_trial_id_from_study_id_trial_number(self, study_id: int, trial_number: int) -> int:
        with self._lock:
            info = self._studies.get(study_id)
            if info is not None and trial_number in info.trials:
                return info.trials[trial_number]._trial_id
        return self._backend.get_trial_id_from_study_id_trial_number(study_id, trial_number)

    def get_trial_number_from_id(self, trial_id: int) -> int:
        with self._lock:
            mapped = self._trial_id_to_study_id_and_number.get(trial_id)
            if mapped is not None:
                return mapped[1]
        return self._backend.get_trial_number_from_id(trial_id)

    def get_trial_param(self, trial_id: int, param_name: str) -> float:
        return self._backend.get_trial_param(trial_id, param_name)

    def set_trial_state_values(
        self, trial_id: int, state: TrialState, values: Sequence[float] | None = None
    ) -> bool:
        return self._backend.set_trial_state_values(trial_id, state, values)

    def set_trial_intermediate_value(
        self, trial_id: int, step: int, intermediate_value: float
    ) -> None:
        self._backend.set_trial_intermediate_value(trial_id, step, intermediate_value)

    def set_trial_user_attr(self, trial_id: int, key: str, value: Any) -> None:
        self._backend.set_trial_user_attr(trial_id, key, value)

    def set_trial_system_attr(self, trial_id: int, key: str, value: Any) -> None:
        self._backend.set_trial_system_attr(trial_id, key, value)

    def get_trial(self, trial_id: int) -> FrozenTrial:
        with self._lock:
            mapped = self._trial_id_to_study_id_and_number.get(trial_id)
            if mapped is not None:
                study_id, number = mapped
                trial = self._studies[study_id].trials.get(number)
                # Only finished trials are immutable; an unfinished cached
                # snapshot can be stale (reference _cached_storage.py:207-223).
                if trial is not None and trial.state.is_finished():
                    return copy.deepcopy(trial)
        return self._backend.get_trial(trial_id)

    def get_n_trials(
        self, study_id: int, state: "tuple[TrialState, ...] | TrialState | None" = None
    ) -> int:
        # Counts come straight from the backend (one COUNT query on RDB) — no
        # cache materialization, always fresh.
        return self._backend.get_n_trials(study_id, state)

    def get_all_trials(
        self,
        study_id: int,
        deepcopy: bool = True,
        states: Container[TrialState] | None = None,
    ) -> list[FrozenTrial]:
        with self._lock:
            self._read_trials_from_remote_storage(study_id)
            info = self._studies[study_id]
            trials = list(info.trials.values())
            trials.sort(key=lambda t: t.number)
            if states is not None:
                trials = [t for t in trials if t.state in states]
            if deepcopy:
                trials = copy.deepcopy(trials)
            return trials

    def _read_trials_from_remote_storage(self, study_id: int) -> None:
        """Fetch only unfinished and newer-than-last-finished trials (delta)."""
        info = self._studies.setdefault(study_id, _StudyInfo())
        fetched = self._backend._get_trials(
            study_id,
            states=None,
            included_trial_ids=info.unfinished_trial_ids,
            trial_id_greater_than=info.last_finished_trial_id,
        )
        for trial in fetched:
            self._trial_id_to_study_id_and_number[trial._trial_id] = (study_id, trial.number)
            info.trials[trial.number] = trial
            if trial.state.is_finished():
                info.unfinished_trial_ids.discard(trial._trial_id)
                info.last_finished_trial_id = max(
                    info.last_finished_trial_id, trial._trial_id
                )
            else:
                info.unfinished_trial_ids.add(trial._trial_id)
        # Track monotone frontier: all trials with id <= last_finished and finished
        # are cached; ids above get re-fetched next round.

    # ---- heartbeat ------------------------------------------------------------------

    def record_heartbeat(self, trial_id: int) -> None:
        self._backend.record_heartbeat(trial_id)

    def _get_stale_trial_ids(self, study_id: int) -> list[int]:
        return self._backend._get_stale_trial_ids(study_id)

    def get_heartbeat_interval(self) -> int | None:
        return self._backend.get_heartbeat_interval()

    def get_failed_trial_callback(self) -> Callable[..., None] | None:
        return self._backend.get_failed_trial_callback()

    def remove_session(self) -> None:
        self._backend.remove_session()
