"""Journal backend ABC.

Parity: reference ``optuna/storages/journal/_base.py`` (BaseJournalBackend,
BaseJournalSnapshot).
"""
from __future__ import annotations

import abc
from typing import Any


class BaseJournalBackend(abc.ABC):
    """Append-only log of JSON-serializable operation records."""

    @abc.abstractmethod
    def read_logs(self, log_number_from: int) -> list[dict[str, Any]]:
        """Logs with number >= log_number_from (0-based, dense)."""
        raise NotImplementedError

    @abc.abstractmethod
    def append_logs(self, logs: list[dict[str, Any]]) -> None:
        raise NotImplementedError


class BaseJournalSnapshot(abc.ABC):
    """Optional snapshot support to bound replay time."""

    @abc.abstractmethod
    def save_snapshot(self, snapshot: bytes) -> None:
        raise NotImplementedError

    @abc.abstractmethod
    def load_snapshot(self) -> bytes | None:
        raise NotImplementedError
