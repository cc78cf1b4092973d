"""JournalRedisBackend behavior against a minimal in-process fake redis
(the real ``redis`` package is not installed in this environment; the fake
implements exactly the five commands the backend uses)."""
from __future__ import annotations

import threading
from typing import Any

import pytest

import optuna_amd
from optuna_amd.storages.journal._redis import JournalRedisBackend


class _FakeRedis:
    def __init__(self) -> None:
        self._data: dict[str, bytes] = {}
        self._lock = threading.Lock()

    def get(self, key: str) -> bytes | None:
        with self._lock:
            return self._data.get(key)

    def set(self, key: str, value: Any) -> None:
        with self._lock:
            self._data[key] = value if isinstance(value, bytes) else str(value).encode()

    def incr(self, key: str, amount: int = 1) -> int:
        with self._lock:
            v = int(self._data.get(key, b"0")) + amount
            self._data[key] = str(v).encode()
            return v

    def close(self) -> None:
        pass


def _make_backend() -> JournalRedisBackend:
    backend = JournalRedisBackend.__new__(JournalRedisBackend)
    backend._url = "redis://fake"
    backend._redis = _FakeRedis()
    backend._prefix = "t"
    return backend


def test_append_and_read_logs_round_trip() -> None:
    backend = _make_backend()
    assert backend.read_logs(0) == []
    backend.append_logs([{"op_code": 0, "a": 1}, {"op_code": 5, "b": [1, 2]}])
    backend.append_logs([{"op_code": 7}])
    assert backend.read_logs(0) == [
        {"op_code": 0, "a": 1},
        {"op_code": 5, "b": [1, 2]},
        {"op_code": 7},
    ]
    assert backend.read_logs(2) == [{"op_code": 7}]


def test_snapshot_round_trip() -> None:
    backend = _make_backend()
    assert backend.load_snapshot() is None
    backend.save_snapshot(b"\x00\x01binary")
    assert backend.load_snapshot() == b"\x00\x01binary"


def test_full_journal_storage_over_fake_redis() -> None:
    backend = _make_backend()
    storage = optuna_amd.storages.JournalStorage(backend)
    study = optuna_amd.create_study(
        study_name="redis-study", storage=storage,
        sampler=optuna_amd.samplers.RandomSampler(seed=0),
    )
    study.optimize(lambda t: t.suggest_float("x", 0, 1) ** 2, n_trials=6)

    # A second storage over the same fake server replays the identical study.
    storage2 = optuna_amd.storages.JournalStorage(_clone(backend))
    study2 = optuna_amd.load_study(study_name="redis-study", storage=storage2)
    assert [t.value for t in study2.trials] == [t.value for t in study.trials]


def _clone(backend: JournalRedisBackend) -> JournalRedisBackend:
    other = JournalRedisBackend.__new__(JournalRedisBackend)
    other._url = backend._url
    other._redis = backend._redis  # same "server"
    other._prefix = backend._prefix
    return other
