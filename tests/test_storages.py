"""Run the storage conformance suite over every CPU-available backend."""
from __future__ import annotations

import pytest

from optuna_amd.storages import BaseStorage
from optuna_amd.testing.pytest_storages import StorageTestCase
from optuna_amd.testing.storages import StorageSupplier


class TestInMemoryStorage(StorageTestCase):
    @pytest.fixture
    def storage(self) -> BaseStorage:
        from optuna_amd.storages import InMemoryStorage

        return InMemoryStorage()


class TestSqliteStorage(StorageTestCase):
    @pytest.fixture
    def storage(self):  # type: ignore[override]
        with StorageSupplier("sqlite") as s:
            yield s


class TestCachedSqliteStorage(StorageTestCase):
    @pytest.fixture
    def storage(self):  # type: ignore[override]
        with StorageSupplier("cached_sqlite") as s:
            yield s


class TestJournalStorage(StorageTestCase):
    @pytest.fixture
    def storage(self):  # type: ignore[override]
        with StorageSupplier("journal") as s:
            yield s


class TestGrpcSqliteStorage(StorageTestCase):
    @pytest.fixture
    def storage(self):  # type: ignore[override]
        with StorageSupplier("grpc_rdb") as s:
            yield s


class TestGrpcWireStorage(StorageTestCase):
    """Wire-protocol (reference api.proto) proxy over an in-memory backend."""

    @pytest.fixture
    def storage(self):  # type: ignore[override]
        with StorageSupplier("grpc_wire") as s:
            yield s


def test_finished_trials_since_inmemory_and_journal(tmp_path) -> None:
    import optuna_amd
    from optuna_amd.study import StudyDirection
    from optuna_amd.trial import TrialState

    """Delta-read contract: slice of the finish-order log, immutable entries."""
    from optuna_amd.storages import InMemoryStorage, JournalStorage
    from optuna_amd.storages.journal import JournalFileBackend

    for make in (
        lambda: InMemoryStorage(),
        lambda: JournalStorage(JournalFileBackend(str(tmp_path / "log.jsonl"))),
    ):
        storage = make()
        sid = storage.create_new_study([StudyDirection.MINIMIZE], study_name=None)
        # 3 running trials; finish them out of creation order.
        tids = [storage.create_new_trial(sid) for _ in range(3)]
        for tid in (tids[1], tids[0]):
            storage.set_trial_state_values(tid, TrialState.RUNNING)
            storage.set_trial_state_values(tid, TrialState.COMPLETE, [1.0])
        log0 = storage.get_finished_trials_since(sid, 0)
        assert [t._trial_id for t in log0] == [tids[1], tids[0]]
        assert storage.get_n_trials(sid, (TrialState.COMPLETE, TrialState.PRUNED)) == 2
        # Delta after cursor 2 is empty until another trial finishes.
        assert storage.get_finished_trials_since(sid, 2) == []
        storage.set_trial_state_values(tids[2], TrialState.RUNNING)
        storage.set_trial_state_values(tids[2], TrialState.FAIL)
        assert storage.get_finished_trials_since(sid, 2) == []  # FAIL not logged
        # Template-created finished trial lands in the log too.
        from optuna_amd.trial import FrozenTrial

        t = optuna_amd.create_trial(
            params={}, distributions={}, value=3.0
        )
        storage.create_new_trial(sid, template_trial=t)
        delta = storage.get_finished_trials_since(sid, 2)
        assert len(delta) == 1 and delta[0].value == 3.0


def test_heartbeat_fail_stale_and_retry(tmp_path) -> None:
    """Heartbeat-enabled RDB: a stale RUNNING trial is failed and re-enqueued by
    the retry callback (reference storages/_heartbeat.py:156 semantics)."""
    import time

    import optuna_amd
    from optuna_amd.storages import RDBStorage, RetryFailedTrialCallback, fail_stale_trials
    from optuna_amd.trial import TrialState

    import warnings

    with warnings.catch_warnings():
        warnings.simplefilter("ignore")
        storage = RDBStorage(
            f"sqlite:///{tmp_path}/hb.db",
            heartbeat_interval=1,
            grace_period=1,
            failed_trial_callback=RetryFailedTrialCallback(max_retry=2),
        )
    study = optuna_amd.create_study(storage=storage, study_name="hb")
    t = study.ask()
    t.suggest_float("x", 0, 1)
    storage.record_heartbeat(t._trial_id)
    time.sleep(2.1)  # heartbeat is now stale (grace 1s past 1s interval)
    fail_stale_trials(study)
    trials = study.get_trials(deepcopy=False)
    states = [tr.state for tr in trials]
    assert TrialState.FAIL in states
    # the retry callback enqueued a WAITING clone with the same params
    waiting = [tr for tr in trials if tr.state == TrialState.WAITING]
    assert len(waiting) == 1
    assert waiting[0].params == t.params
    assert RetryFailedTrialCallback.retried_trial_number(waiting[0]) == 0
