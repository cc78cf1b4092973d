"""Exponential-backoff retry wrapper around any artifact store.

Parity: reference ``optuna/artifacts/_backoff.py``.
"""
from __future__ import annotations

import io
import time
from typing import BinaryIO

from optuna_amd.artifacts.exceptions import ArtifactNotFound


class Backoff:
    """Wraps another store and retries transient failures with exponential backoff."""

    def __init__(
        self,
        backend: "object",
        *,
        max_retries: int = 10,
        multiplier: float = 2,
        min_delay: float = 0.1,
        max_delay: float = 30,
    ) -> None:
        if max_retries <= 0:
            raise ValueError("max_retries should be a positive integer.")
        if multiplier <= 0:
            raise ValueError("multiplier should be a positive float.")
        if min_delay <= 0 or max_delay <= 0 or max_delay < min_delay:
            raise ValueError("delays should be positive with max_delay >= min_delay.")
        self._backend = backend
        self._max_retries = max_retries
        self._multiplier = multiplier
        self._min_delay = min_delay
        self._max_delay = max_delay

    def _sleep(self, attempt: int) -> None:
        time.sleep(min(self._min_delay * self._multiplier**attempt, self._max_delay))

    def _retry(self, func, *args):  # type: ignore[no-untyped-def]
        for attempt in range(self._max_retries):
            try:
                return func(*args)
            except ArtifactNotFound:
                raise
            except Exception:
                if attempt == self._max_retries - 1:
                    raise
                self._sleep(attempt)
        raise AssertionError("unreachable")

    def open_reader(self, artifact_id: str) -> BinaryIO:
        return self._retry(self._backend.open_reader, artifact_id)  # type: ignore[attr-defined]

    def write(self, artifact_id: str, content_body: BinaryIO) -> None:
        # Buffer so retries can re-seek the stream.
        data = content_body.read()
        self._retry(
            lambda aid, payload: self._backend.write(aid, io.BytesIO(payload)),  # type: ignore[attr-defined]
            artifact_id,
            data,
        )

    def remove(self, artifact_id: str) -> None:
        self._retry(self._backend.remove, artifact_id)  # type: ignore[attr-defined]
