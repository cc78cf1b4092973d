"""Redis journal backend: log records as sequential keys + snapshot.

Parity: reference ``optuna/storages/journal/_redis.py`` (JournalRedisBackend :20,
snapshot :91-96). Requires the ``redis`` package.
"""
from __future__ import annotations

import json
from typing import Any

from optuna_amd._imports import try_import
from optuna_amd.storages.journal._base import BaseJournalBackend, BaseJournalSnapshot


with try_import() as _imports:
    import redis


class JournalRedisBackend(BaseJournalBackend, BaseJournalSnapshot):
    def __init__(self, url: str, use_cluster: bool = False, prefix: str = "") -> None:
        _imports.check()
        self._url = url
        self._redis = (
            redis.cluster.RedisCluster.from_url(url)  # type: ignore[attr-defined]
            if use_cluster
            else redis.Redis.from_url(url)
        )
        self._prefix = prefix

    def __getstate__(self) -> dict[str, Any]:
        state = self.__dict__.copy()
        del state["_redis"]
        return state

    def __setstate__(self, state: dict[str, Any]) -> None:
        self.__dict__.update(state)
        self._redis = redis.Redis.from_url(self._url)

    def __del__(self) -> None:
        try:
            self._redis.close()
        except Exception:
            pass

    def read_logs(self, log_number_from: int) -> list[dict[str, Any]]:
        max_log_number_bytes = self._redis.get(f"{self._prefix}:log_number")
        if max_log_number_bytes is None:
            return []
        # The counter holds the number of logs written; keys are 0..counter-1.
        max_log_number = int(max_log_number_bytes)
        logs = []
        for log_number in range(log_number_from, max_log_number):
            payload = self._redis.get(self._key(log_number))
            if payload is None:
                continue
            logs.append(json.loads(payload))
        return logs

    def append_logs(self, logs: list[dict[str, Any]]) -> None:
        for log in logs:
            log_number = self._redis.incr(f"{self._prefix}:log_number", 1)
            self._redis.set(self._key(int(log_number) - 1), json.dumps(log))

    def save_snapshot(self, snapshot: bytes) -> None:
        self._redis.set(f"{self._prefix}:snapshot", snapshot)

    def load_snapshot(self) -> bytes | None:
        return self._redis.get(f"{self._prefix}:snapshot")

    def _key(self, log_number: int) -> str:
        return f"{self._prefix}:log:{log_number}"
