"""Worker-failure detection via storage heartbeats.

A per-trial daemon thread records a heartbeat every ``heartbeat_interval`` seconds;
``fail_stale_trials`` marks RUNNING trials whose heartbeat is older than
``interval * grace_multiplier`` as FAIL, and runs the storage's
``failed_trial_callback`` (e.g. ``RetryFailedTrialCallback``).

Parity: reference ``optuna/storages/_heartbeat.py`` (BaseHeartbeat :18,
HeartbeatThread :117, fail_stale_trials :156).
"""
from __future__ import annotations

import abc
import copy
import threading
from threading import Event, Thread
from types import TracebackType
from typing import TYPE_CHECKING, Callable

from optuna_amd import logging as _logging
from optuna_amd.storages._base import BaseStorage
from optuna_amd.trial import FrozenTrial, TrialState


if TYPE_CHECKING:
    from optuna_amd.study import Study

_logger = _logging.get_logger(__name__)


class BaseHeartbeat(abc.ABC):
    """Mixin for storages that support heartbeat recording."""

    @abc.abstractmethod
    def record_heartbeat(self, trial_id: int) -> None:
        raise NotImplementedError

    @abc.abstractmethod
    def _get_stale_trial_ids(self, study_id: int) -> list[int]:
        raise NotImplementedError

    @abc.abstractmethod
    def get_heartbeat_interval(self) -> int | None:
        raise NotImplementedError

    def is_heartbeat_enabled(self) -> bool:
        return self.get_heartbeat_interval() is not None

    def get_heartbeat_stale_trial_callback(
        self,
    ) -> Callable[["Study", FrozenTrial], None] | None:
        """Current name for :meth:`get_failed_trial_callback`."""
        return self.get_failed_trial_callback()

    def get_failed_trial_callback(self) -> Callable[["Study", FrozenTrial], None] | None:
        return None


class BaseHeartbeatThread(abc.ABC):
    def __enter__(self) -> None:
        self.start()

    def __exit__(
        self,
        exc_type: type[Exception] | None,
        exc_value: Exception | None,
        traceback: TracebackType | None,
    ) -> None:
        self.join()

    @abc.abstractmethod
    def start(self) -> None:
        raise NotImplementedError

    @abc.abstractmethod
    def join(self) -> None:
        raise NotImplementedError


class NullHeartbeatThread(BaseHeartbeatThread):
    def start(self) -> None:
        pass

    def join(self) -> None:
        pass


class HeartbeatThread(BaseHeartbeatThread):
    def __init__(self, trial_id: int, heartbeat: BaseHeartbeat) -> None:
        self._trial_id = trial_id
        self._heartbeat = heartbeat
        self._thread: threading.Thread | None = None
        self._stop_event: threading.Event | None = None

    def start(self) -> None:
        self._stop_event = Event()
        # Module-level name + (target, args)-only construction so test doubles
        # can substitute the thread class.
        self._thread = Thread(target=self._record_periodically, args=())
        self._thread.start()

    def join(self) -> None:
        assert self._stop_event is not None and self._thread is not None
        self._stop_event.set()
        self._thread.join()

    def _record_periodically(self) -> None:
        assert self._stop_event is not None
        heartbeat_interval = self._heartbeat.get_heartbeat_interval()
        assert heartbeat_interval is not None
        while True:
            self._heartbeat.record_heartbeat(self._trial_id)
            if self._stop_event.wait(timeout=heartbeat_interval):
                break


def is_heartbeat_enabled(storage: BaseStorage) -> bool:
    return isinstance(storage, BaseHeartbeat) and storage.is_heartbeat_enabled()


def get_heartbeat_thread(trial_id: int, storage: BaseStorage) -> BaseHeartbeatThread:
    if is_heartbeat_enabled(storage):
        assert isinstance(storage, BaseHeartbeat)
        return HeartbeatThread(trial_id, storage)
    return NullHeartbeatThread()


def fail_stale_trials(study: "Study") -> None:
    """Mark RUNNING trials with stale heartbeats FAIL and fire the retry callback.

    Invoked at the top of every ``_run_trial`` (reference study/_optimize.py:191-195).
    """
    storage = study._storage
    if not isinstance(storage, BaseHeartbeat):
        return
    if not storage.is_heartbeat_enabled():
        return

    failed_trial_ids = []
    for trial_id in storage._get_stale_trial_ids(study._study_id):
        try:
            if storage.set_trial_state_values(trial_id, state=TrialState.FAIL):
                failed_trial_ids.append(trial_id)
        except RuntimeError:
            # Another worker raced us to fail (or finish) the trial.
            pass

    failed_trial_callback = storage.get_failed_trial_callback()
    if failed_trial_callback is not None:
        for trial_id in failed_trial_ids:
            failed_trial = copy.deepcopy(storage.get_trial(trial_id))
            failed_trial_callback(study, failed_trial)
