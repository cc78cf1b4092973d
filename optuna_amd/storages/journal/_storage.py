"""JournalStorage: append-only op log + in-memory replay.

Every write appends one JSON record then replays all unseen records under one
lock; the replayed state is the storage. The record format (op_code / worker_id /
per-op fields, ISO-8601 UTC datetimes) is compatible with the reference journal
log (reference ``optuna/storages/journal/_storage.py`` :41-52 opcodes, :145-151
write+sync, :104 worker-id ownership, :407-417 replay state), so log files
interoperate.

This op-log design is also the template for the distributed trial table
(``optuna_amd/storages/_rccl.py``): same opcodes, with the sequencing provided by
a TCPStore/RCCL tier instead of a file lock.
"""
from __future__ import annotations

import copy
import enum
import pickle
import threading
import uuid
from datetime import datetime, timezone
from typing import Any, Container, Iterable, Sequence

from optuna_amd import logging as _logging
from optuna_amd.distributions import (
    BaseDistribution,
    check_distribution_compatibility,
    distribution_to_json,
    json_to_distribution,
)
from optuna_amd.exceptions import DuplicatedStudyError, UpdateFinishedTrialError
from optuna_amd.storages._base import DEFAULT_STUDY_NAME_PREFIX, BaseStorage
from optuna_amd.storages.journal._base import BaseJournalBackend, BaseJournalSnapshot
from optuna_amd.study._frozen import FrozenStudy
from optuna_amd.study._study_direction import StudyDirection
from optuna_amd.trial import FrozenTrial, TrialState


_logger = _logging.get_logger(__name__)

NOT_FOUND_MSG = "Record does not exist."
UNUPDATABLE_MSG = "Trial#{trial_number} has already finished and can not be updated."
SNAPSHOT_INTERVAL = 100


class JournalOperation(enum.IntEnum):
    CREATE_STUDY = 0
    DELETE_STUDY = 1
    SET_STUDY_USER_ATTR = 2
    SET_STUDY_SYSTEM_ATTR = 3
    CREATE_TRIAL = 4
    SET_TRIAL_PARAM = 5
    SET_TRIAL_STATE_VALUES = 6
    SET_TRIAL_INTERMEDIATE_VALUE = 7
    SET_TRIAL_USER_ATTR = 8
    SET_TRIAL_SYSTEM_ATTR = 9
    DISCARD_TRIALS = 10  # written by other implementations; replay skips it
    # Private RcclStorage extension (never written to journal files; to_journal
    # expands it into standard SET_TRIAL_PARAM records): all of one trial's
    # buffered param writes as a single record, so N-rank replay parses one
    # JSON object instead of ~20 per finished trial.
    SET_TRIAL_PARAMS_BATCH = 100


def _utcnow_iso() -> str:
    return datetime.now(tz=timezone.utc).isoformat(timespec="microseconds")


def _iso_to_local_naive(s: str) -> datetime:
    return datetime.fromisoformat(s).astimezone().replace(tzinfo=None)


class JournalStorage(BaseStorage):
    """Storage over any append-only journal backend (file, Redis, ...)."""

    def __init__(self, log_storage: BaseJournalBackend) -> None:
        self._worker_id_prefix = str(uuid.uuid4()) + "-"
        self._backend = log_storage
        self._thread_lock = threading.Lock()
        self._replay_result = _ReplayState(self._worker_id_prefix)
        with self._thread_lock:
            if isinstance(self._backend, BaseJournalSnapshot):
                snapshot = self._backend.load_snapshot()
                if snapshot is not None:
                    self.restore_replay_result(snapshot)
            self._sync()

    def __getstate__(self) -> dict[str, Any]:
        state = self.__dict__.copy()
        del state["_worker_id_prefix"]
        del state["_replay_result"]
        del state["_thread_lock"]
        return state

    def __setstate__(self, state: dict[str, Any]) -> None:
        self.__dict__.update(state)
        self._worker_id_prefix = str(uuid.uuid4()) + "-"
        self._replay_result = _ReplayState(self._worker_id_prefix)
        self._thread_lock = threading.Lock()

    def restore_replay_result(self, snapshot: bytes) -> None:
        r: _ReplayState | None = pickle.loads(snapshot)
        if r is None or not isinstance(r, _ReplayState):
            raise RuntimeError("A snapshot is broken or uncompatible.")
        r._worker_id_prefix = self._worker_id_prefix
        r._worker_id_to_owned_trial_id = {}
        self._replay_result = r

    # ---- write plumbing -------------------------------------------------------------

    def _append(self, op: JournalOperation, fields: dict[str, Any]) -> None:
        record = {"op_code": int(op), "worker_id": self._replay_result.worker_id, **fields}
        self._backend.append_logs([record])

    def _sync(self) -> None:
        logs = self._backend.read_logs(self._replay_result.log_number_read)
        self._replay_result.apply_logs(logs)

    def _maybe_snapshot(self, counter: int) -> None:
        if (
            isinstance(self._backend, BaseJournalSnapshot)
            and counter != 0
            and counter % SNAPSHOT_INTERVAL == 0
        ):
            self._backend.save_snapshot(pickle.dumps(self._replay_result))

    # ---- studies --------------------------------------------------------------------

    def create_new_study(
        self, directions: Sequence[StudyDirection], study_name: str | None = None
    ) -> int:
        study_name = study_name or DEFAULT_STUDY_NAME_PREFIX + str(uuid.uuid4())
        with self._thread_lock:
            self._append(
                JournalOperation.CREATE_STUDY,
                {"study_name": study_name, "directions": [int(d) for d in directions]},
            )
            self._sync()
            for fs in self._replay_result.all_studies():
                if fs.study_name == study_name:
                    _logger.info(f"A new study created in Journal with name: {study_name}")
                    self._maybe_snapshot(fs._study_id)
                    return fs._study_id
            raise AssertionError("unreachable")

    def delete_study(self, study_id: int) -> None:
        with self._thread_lock:
            self._append(JournalOperation.DELETE_STUDY, {"study_id": study_id})
            self._sync()

    def set_study_user_attr(self, study_id: int, key: str, value: Any) -> None:
        with self._thread_lock:
            self._append(
                JournalOperation.SET_STUDY_USER_ATTR,
                {"study_id": study_id, "user_attr": {key: value}},
            )
            self._sync()

    def set_study_system_attr(self, study_id: int, key: str, value: Any) -> None:
        with self._thread_lock:
            self._append(
                JournalOperation.SET_STUDY_SYSTEM_ATTR,
                {"study_id": study_id, "system_attr": {key: value}},
            )
            self._sync()

    def get_study_id_from_name(self, study_name: str) -> int:
        with self._thread_lock:
            self._sync()
            for fs in self._replay_result.all_studies():
                if fs.study_name == study_name:
                    return fs._study_id
            raise KeyError(NOT_FOUND_MSG)

    def get_study_name_from_id(self, study_id: int) -> str:
        with self._thread_lock:
            self._sync()
            return self._replay_result.study(study_id).study_name

    def get_study_directions(self, study_id: int) -> list[StudyDirection]:
        with self._thread_lock:
            self._sync()
            return self._replay_result.study(study_id).directions

    def get_study_user_attrs(self, study_id: int) -> dict[str, Any]:
        with self._thread_lock:
            self._sync()
            return self._replay_result.study(study_id).user_attrs

    def get_study_system_attrs(self, study_id: int) -> dict[str, Any]:
        with self._thread_lock:
            self._sync()
            return self._replay_result.study(study_id).system_attrs

    def get_all_studies(self) -> list[FrozenStudy]:
        with self._thread_lock:
            self._sync()
            return copy.deepcopy(self._replay_result.all_studies())

    # ---- trials ---------------------------------------------------------------------

    def create_new_trial(self, study_id: int, template_trial: FrozenTrial | None = None) -> int:
        log: dict[str, Any] = {
            "study_id": study_id,
            "datetime_start": _utcnow_iso(),
        }
        if template_trial is not None:
            log["state"] = int(template_trial.state)
            if template_trial.values is not None and len(template_trial.values) > 1:
                log["value"] = None
                log["values"] = template_trial.values
            else:
                log["value"] = template_trial.value
                log["values"] = None
            if template_trial.datetime_start:
                log["datetime_start"] = (
                    template_trial.datetime_start.astimezone(timezone.utc).isoformat(
                        timespec="microseconds"
                    )
                )
            else:
                log["datetime_start"] = None
            if template_trial.datetime_complete:
                log["datetime_complete"] = (
                    template_trial.datetime_complete.astimezone(timezone.utc).isoformat(
                        timespec="microseconds"
                    )
                )
            log["distributions"] = {
                k: distribution_to_json(d) for k, d in template_trial.distributions.items()
            }
            log["params"] = {
                k: template_trial.distributions[k].to_internal_repr(v)
                for k, v in template_trial.params.items()
            }
            log["user_attrs"] = template_trial.user_attrs
            log["system_attrs"] = template_trial.system_attrs
            log["intermediate_values"] = template_trial.intermediate_values

        with self._thread_lock:
            self._append(JournalOperation.CREATE_TRIAL, log)
            self._sync()
            trial_id = self._replay_result.last_created_trial_id
            self._maybe_snapshot(trial_id)
            return trial_id

    def set_trial_param(
        self,
        trial_id: int,
        param_name: str,
        param_value_internal: float,
        distribution: BaseDistribution,
    ) -> None:
        with self._thread_lock:
            self._append(
                JournalOperation.SET_TRIAL_PARAM,
                {
                    "trial_id": trial_id,
                    "param_name": param_name,
                    "param_value_internal": param_value_internal,
                    "distribution": distribution_to_json(distribution),
                },
            )
            self._sync()

    def get_trial_id_from_study_id_trial_number(self, study_id: int, trial_number: int) -> int:
        with self._thread_lock:
            self._sync()
            trial_ids = self._replay_result._study_id_to_trial_ids.get(study_id)
            if trial_ids is None or len(trial_ids) <= trial_number:
                raise KeyError(
                    f"No trial with trial number {trial_number} exists in study with "
                    f"study_id {study_id}."
                )
            return trial_ids[trial_number]

    def set_trial_state_values(
        self, trial_id: int, state: TrialState, values: Sequence[float] | None = None
    ) -> bool:
        log: dict[str, Any] = {
            "trial_id": trial_id,
            "state": int(state),
            "values": list(values) if values is not None else None,
        }
        if state == TrialState.RUNNING:
            log["datetime_start"] = _utcnow_iso()
        elif state.is_finished():
            log["datetime_complete"] = _utcnow_iso()

        with self._thread_lock:
            if state == TrialState.RUNNING:
                # Claim pre-check: avoid a false-positive claim when this worker's
                # thread previously owned a trial popped by another process.
                self._sync()
                existing = self._replay_result._trials.get(trial_id)
                if existing is None:
                    raise KeyError(NOT_FOUND_MSG)
                if existing.state.is_finished():
                    raise UpdateFinishedTrialError(
                        UNUPDATABLE_MSG.format(trial_number=existing.number)
                    )
                if existing.state != TrialState.WAITING:
                    return False
            self._append(JournalOperation.SET_TRIAL_STATE_VALUES, log)
            self._sync()
            return state != TrialState.RUNNING or trial_id == self._replay_result.owned_trial_id

    def set_trial_intermediate_value(
        self, trial_id: int, step: int, intermediate_value: float
    ) -> None:
        with self._thread_lock:
            self._append(
                JournalOperation.SET_TRIAL_INTERMEDIATE_VALUE,
                {"trial_id": trial_id, "step": step, "intermediate_value": intermediate_value},
            )
            self._sync()

    def set_trial_user_attr(self, trial_id: int, key: str, value: Any) -> None:
        with self._thread_lock:
            self._append(
                JournalOperation.SET_TRIAL_USER_ATTR,
                {"trial_id": trial_id, "user_attr": {key: value}},
            )
            self._sync()

    def set_trial_system_attr(self, trial_id: int, key: str, value: Any) -> None:
        with self._thread_lock:
            self._append(
                JournalOperation.SET_TRIAL_SYSTEM_ATTR,
                {"trial_id": trial_id, "system_attr": {key: value}},
            )
            self._sync()

    def get_trial(self, trial_id: int) -> FrozenTrial:
        with self._thread_lock:
            self._sync()
            return self._replay_result.trial(trial_id)

    def get_all_trials(
        self,
        study_id: int,
        deepcopy: bool = True,
        states: Container[TrialState] | None = None,
    ) -> list[FrozenTrial]:
        with self._thread_lock:
            self._sync()
            trials = self._replay_result.all_trials(study_id, states)
            if deepcopy:
                trials = copy.deepcopy(trials)
            return trials

    def get_finished_trials_since(self, study_id: int, start: int) -> list[FrozenTrial]:
        """O(delta) read of COMPLETE/PRUNED trials in finish order (see
        InMemoryStorage.get_finished_trials_since for the contract)."""
        with self._thread_lock:
            self._sync()
            return self._replay_result.finished_trials_since(study_id, start)

    def get_n_trials(
        self, study_id: int, state: tuple[TrialState, ...] | TrialState | None = None
    ) -> int:
        if isinstance(state, tuple) and set(state) == {
            TrialState.COMPLETE,
            TrialState.PRUNED,
        }:
            with self._thread_lock:
                self._sync()
                return len(
                    self._replay_result._study_id_to_finished.get(study_id, [])
                )
        return super().get_n_trials(study_id, state)


class _ReplayState:
    """In-memory state reconstructed from the op log (one per storage object)."""

    def __init__(self, worker_id_prefix: str) -> None:
        self.log_number_read = 0
        self._worker_id_prefix = worker_id_prefix
        self._studies: dict[int, FrozenStudy] = {}
        self._trials: dict[int, FrozenTrial] = {}
        self._study_id_to_trial_ids: dict[int, list[int]] = {}
        # COMPLETE/PRUNED trials per study in finish order; finished trials are
        # immutable, so storages expose O(delta) history reads from this log.
        self._study_id_to_finished: dict[int, list[FrozenTrial]] = {}
        # Live-state index: RUNNING/WAITING trial ids per study, so the
        # per-suggest constant-liar fetch and the per-ask WAITING scan cost
        # O(matching) instead of O(all trials).
        self._study_id_to_live: dict[int, dict[TrialState, set[int]]] = {}
        self._trial_id_to_study_id: dict[int, int] = {}
        self._next_study_id = 0
        self._worker_id_to_owned_trial_id: dict[str, int] = {}
        self.last_created_trial_id = -1
        # Trial ids created by THIS worker, in creation order (bulk-create
        # bookkeeping for RcclStorage.bulk_create_trials).
        self.my_created_trial_ids: list[int] = []

    def __getstate__(self) -> dict[str, Any]:
        state = self.__dict__.copy()
        state.pop("last_created_trial_id", None)
        return state

    def __setstate__(self, state: dict[str, Any]) -> None:
        self.__dict__.update(state)
        self.last_created_trial_id = -1
        self.__dict__.setdefault("my_created_trial_ids", [])
        if "_study_id_to_live" not in self.__dict__:
            self._study_id_to_live = {}
            for sid, trial_ids in self._study_id_to_trial_ids.items():
                live: dict[TrialState, set[int]] = {}
                for tid in trial_ids:
                    st0 = self._trials[tid].state
                    if st0 in (TrialState.RUNNING, TrialState.WAITING):
                        live.setdefault(st0, set()).add(tid)
                self._study_id_to_live[sid] = live
        if "_study_id_to_finished" not in self.__dict__:
            # Snapshot from an older build: rebuild the finished log (trial-id
            # order; consumers dedupe by id, so order only affects row order).
            self._study_id_to_finished = {}
            for study_id, trial_ids in self._study_id_to_trial_ids.items():
                self._study_id_to_finished[study_id] = [
                    self._trials[tid]
                    for tid in trial_ids
                    if self._trials[tid].state
                    in (TrialState.COMPLETE, TrialState.PRUNED)
                ]

    def finished_trials_since(self, study_id: int, start: int) -> list[FrozenTrial]:
        return list(self._study_id_to_finished.get(study_id, [])[start:])

    # ---- queries --------------------------------------------------------------------

    @property
    def worker_id(self) -> str:
        return self._worker_id_prefix + str(threading.get_ident())

    @property
    def owned_trial_id(self) -> int | None:
        return self._worker_id_to_owned_trial_id.get(self.worker_id)

    def study(self, study_id: int) -> FrozenStudy:
        if study_id not in self._studies:
            raise KeyError(NOT_FOUND_MSG)
        return self._studies[study_id]

    def all_studies(self) -> list[FrozenStudy]:
        return list(self._studies.values())

    def trial(self, trial_id: int) -> FrozenTrial:
        if trial_id not in self._trials:
            raise KeyError(NOT_FOUND_MSG)
        return self._trials[trial_id]

    def all_trials(self, study_id: int, states: Container[TrialState] | None) -> list[FrozenTrial]:
        if study_id not in self._studies:
            raise KeyError(NOT_FOUND_MSG)
        try:
            state_set = set(states) if states is not None else None
        except TypeError:
            state_set = None
        if state_set is not None and state_set <= {
            TrialState.RUNNING,
            TrialState.WAITING,
        }:
            live = self._study_id_to_live.get(study_id, {})
            ids: list[int] = []
            for st0 in state_set:
                ids.extend(live.get(st0, ()))
            # trial ids are created in number order within a study
            return [self._trials[tid] for tid in sorted(ids)]
        out = []
        for trial_id in self._study_id_to_trial_ids[study_id]:
            t = self._trials[trial_id]
            if states is None or t.state in states:
                out.append(t)
        return out

    # ---- replay ---------------------------------------------------------------------

    def apply_logs(self, logs: Iterable[dict[str, Any]], advance: bool = True) -> None:
        """Apply records; advance=False previews records not yet in the log
        (RcclStorage write batching) — the log counter stays put so the same
        records re-apply idempotently once they arrive through the log."""
        handlers = {
            int(JournalOperation.CREATE_STUDY): self._on_create_study,
            int(JournalOperation.DELETE_STUDY): self._on_delete_study,
            int(JournalOperation.SET_STUDY_USER_ATTR): self._on_study_user_attr,
            int(JournalOperation.SET_STUDY_SYSTEM_ATTR): self._on_study_system_attr,
            int(JournalOperation.CREATE_TRIAL): self._on_create_trial,
            int(JournalOperation.SET_TRIAL_PARAM): self._on_trial_param,
            int(JournalOperation.SET_TRIAL_STATE_VALUES): self._on_trial_state_values,
            int(JournalOperation.SET_TRIAL_INTERMEDIATE_VALUE): self._on_trial_intermediate,
            int(JournalOperation.SET_TRIAL_USER_ATTR): self._on_trial_user_attr,
            int(JournalOperation.SET_TRIAL_SYSTEM_ATTR): self._on_trial_system_attr,
            int(JournalOperation.SET_TRIAL_PARAMS_BATCH): self._on_trial_params_batch,
        }
        for log in logs:
            if advance:
                self.log_number_read += 1
            op = log["op_code"]
            handler = handlers.get(op)
            if handler is None:
                if op == int(JournalOperation.DISCARD_TRIALS):
                    continue  # written by other implementations only
                raise AssertionError(f"unknown op_code {op}")
            handler(log)

    def _mine(self, log: dict[str, Any]) -> bool:
        return log["worker_id"] == self.worker_id

    def _study_exists(self, study_id: int, log: dict[str, Any]) -> bool:
        if study_id in self._studies:
            return True
        if self._mine(log):
            raise KeyError(NOT_FOUND_MSG)
        return False

    def _updatable_trial(self, trial_id: int, log: dict[str, Any]) -> bool:
        if trial_id not in self._trials:
            if self._mine(log):
                raise KeyError(NOT_FOUND_MSG)
            return False
        if self._trials[trial_id].state.is_finished():
            if self._mine(log):
                raise UpdateFinishedTrialError(
                    UNUPDATABLE_MSG.format(trial_number=self._trials[trial_id].number)
                )
            return False
        return True

    def _on_create_study(self, log: dict[str, Any]) -> None:
        study_name = log["study_name"]
        if any(s.study_name == study_name for s in self._studies.values()):
            if self._mine(log):
                raise DuplicatedStudyError
            return
        study_id = self._next_study_id
        self._next_study_id += 1
        self._studies[study_id] = FrozenStudy(
            study_name=study_name,
            direction=None,
            directions=[StudyDirection(d) for d in log["directions"]],
            user_attrs={},
            system_attrs={},
            study_id=study_id,
        )
        self._study_id_to_trial_ids[study_id] = []

    def _on_delete_study(self, log: dict[str, Any]) -> None:
        study_id = log["study_id"]
        if self._study_exists(study_id, log):
            self._studies.pop(study_id)

    def _on_study_user_attr(self, log: dict[str, Any]) -> None:
        study_id = log["study_id"]
        if self._study_exists(study_id, log):
            study = self._studies[study_id]
            study.user_attrs = {**study.user_attrs, **log["user_attr"]}

    def _on_study_system_attr(self, log: dict[str, Any]) -> None:
        study_id = log["study_id"]
        if self._study_exists(study_id, log):
            study = self._studies[study_id]
            study.system_attrs = {**study.system_attrs, **log["system_attr"]}

    def _on_create_trial(self, log: dict[str, Any]) -> None:
        study_id = log["study_id"]
        if not self._study_exists(study_id, log):
            return
        # Default ids are positional (global creation order, the journal
        # convention). RcclStorage's collective plane pre-assigns
        # rank-partitioned ids so a worker knows its trial id before the
        # records are merged — those records carry the id explicitly and
        # re-application is a no-op.
        trial_id = log.get("trial_id", len(self._trials))
        if trial_id in self._trials:
            return
        distributions = {
            k: json_to_distribution(v) for k, v in log.get("distributions", {}).items()
        }
        params = {
            k: distributions[k].to_external_repr(v) for k, v in log.get("params", {}).items()
        }
        datetime_start = (
            _iso_to_local_naive(log["datetime_start"])
            if log.get("datetime_start") is not None
            else None
        )
        datetime_complete = (
            _iso_to_local_naive(log["datetime_complete"])
            if "datetime_complete" in log
            else None
        )
        self._trials[trial_id] = FrozenTrial(
            trial_id=trial_id,
            number=len(self._study_id_to_trial_ids[study_id]),
            state=TrialState(log.get("state", int(TrialState.RUNNING))),
            params=params,
            distributions=distributions,
            user_attrs=log.get("user_attrs", {}),
            system_attrs=log.get("system_attrs", {}),
            value=log.get("value"),
            values=log.get("values"),
            intermediate_values={
                int(k): v for k, v in log.get("intermediate_values", {}).items()
            },
            datetime_start=datetime_start,
            datetime_complete=datetime_complete,
        )
        self._study_id_to_trial_ids[study_id].append(trial_id)
        self._trial_id_to_study_id[trial_id] = study_id
        st0 = self._trials[trial_id].state
        if st0 in (TrialState.RUNNING, TrialState.WAITING):
            self._study_id_to_live.setdefault(study_id, {}).setdefault(
                st0, set()
            ).add(trial_id)
        if self._trials[trial_id].state in (TrialState.COMPLETE, TrialState.PRUNED):
            self._study_id_to_finished.setdefault(study_id, []).append(
                self._trials[trial_id]
            )
        if self._mine(log):
            self.last_created_trial_id = trial_id
            self.my_created_trial_ids.append(trial_id)
            if self._trials[trial_id].state == TrialState.RUNNING:
                self._worker_id_to_owned_trial_id[self.worker_id] = trial_id

    def _on_trial_param(self, log: dict[str, Any]) -> None:
        trial_id = log["trial_id"]
        if not self._updatable_trial(trial_id, log):
            return
        param_name = log["param_name"]
        distribution = json_to_distribution(log["distribution"])
        study_id = self._trial_id_to_study_id[trial_id]
        for prev_id in self._study_id_to_trial_ids[study_id]:
            prev = self._trials[prev_id]
            if param_name in prev.params:
                try:
                    check_distribution_compatibility(
                        prev.distributions[param_name], distribution
                    )
                except Exception:
                    if self._mine(log):
                        raise
                    return
                break
        trial = copy.copy(self._trials[trial_id])
        trial.params = {
            **trial.params,
            param_name: distribution.to_external_repr(log["param_value_internal"]),
        }
        trial.distributions = {**trial.distributions, param_name: distribution}
        self._trials[trial_id] = trial

    def _on_trial_params_batch(self, log: dict[str, Any]) -> None:
        base = {"op_code": int(JournalOperation.SET_TRIAL_PARAM),
                "worker_id": log["worker_id"], "trial_id": log["trial_id"]}
        for name, fields in log["params"].items():
            self._on_trial_param(
                {
                    **base,
                    "param_name": name,
                    "param_value_internal": fields["value"],
                    "distribution": fields["distribution"],
                }
            )

    def _on_trial_state_values(self, log: dict[str, Any]) -> None:
        trial_id = log["trial_id"]
        if not self._updatable_trial(trial_id, log):
            return
        state = TrialState(log["state"])
        if state == self._trials[trial_id].state and state == TrialState.RUNNING:
            return  # claim lost: another worker raced us to RUNNING
        trial = copy.copy(self._trials[trial_id])
        if state == TrialState.RUNNING:
            trial.datetime_start = _iso_to_local_naive(log["datetime_start"])
            if self._mine(log):
                self._worker_id_to_owned_trial_id[self.worker_id] = trial_id
        if state.is_finished():
            trial.datetime_complete = _iso_to_local_naive(log["datetime_complete"])
        prev_state = self._trials[trial_id].state
        trial.state = state
        if log["values"] is not None:
            trial.values = log["values"]
        self._trials[trial_id] = trial
        sid = self._trial_id_to_study_id[trial_id]
        live = self._study_id_to_live.setdefault(sid, {})
        if prev_state in (TrialState.RUNNING, TrialState.WAITING):
            live.get(prev_state, set()).discard(trial_id)
        if state in (TrialState.RUNNING, TrialState.WAITING):
            live.setdefault(state, set()).add(trial_id)
        if state in (TrialState.COMPLETE, TrialState.PRUNED):
            self._study_id_to_finished.setdefault(
                self._trial_id_to_study_id[trial_id], []
            ).append(trial)

    def _on_trial_intermediate(self, log: dict[str, Any]) -> None:
        trial_id = log["trial_id"]
        if self._updatable_trial(trial_id, log):
            trial = copy.copy(self._trials[trial_id])
            trial.intermediate_values = {
                **trial.intermediate_values,
                log["step"]: log["intermediate_value"],
            }
            self._trials[trial_id] = trial

    def _on_trial_user_attr(self, log: dict[str, Any]) -> None:
        trial_id = log["trial_id"]
        if self._updatable_trial(trial_id, log):
            trial = copy.copy(self._trials[trial_id])
            trial.user_attrs = {**trial.user_attrs, **log["user_attr"]}
            self._trials[trial_id] = trial

    def _on_trial_system_attr(self, log: dict[str, Any]) -> None:
        trial_id = log["trial_id"]
        if self._updatable_trial(trial_id, log):
            trial = copy.copy(self._trials[trial_id])
            trial.system_attrs = {**trial.system_attrs, **log["system_attr"]}
            self._trials[trial_id] = trial
