"""Temporary-file pool whose files outlive the ``with`` block.

Storage tests hand a SQLite path to a storage object and keep using it after
the context manager exits; a plain ``NamedTemporaryFile(delete=True)`` would
unlink it too early (and on Windows could not be reopened at all). Files are
unlinked in one sweep at interpreter exit instead.

Parity: reference ``optuna/testing/tempfile_pool.py``.
"""
from __future__ import annotations

import atexit
import os
import tempfile
from typing import IO, Any


_pending_paths: list[str] = []
_sweep_registered = False


def _sweep() -> None:
    for path in _pending_paths:
        try:
            os.unlink(path)
        except OSError:
            pass
    _pending_paths.clear()


class NamedTemporaryFilePool:
    """``with NamedTemporaryFilePool() as f:`` — like NamedTemporaryFile, but
    deletion is deferred to interpreter exit."""

    def __init__(self, **kwargs: Any) -> None:
        self._kwargs = dict(kwargs)
        self._kwargs["delete"] = False
        self._file: IO[Any] | None = None

    def tempfile(self) -> IO[Any]:
        global _sweep_registered
        self._file = tempfile.NamedTemporaryFile(**self._kwargs)
        _pending_paths.append(self._file.name)
        if not _sweep_registered:
            atexit.register(_sweep)
            _sweep_registered = True
        return self._file

    def __enter__(self) -> IO[Any]:
        return self.tempfile()

    def __exit__(self, exc_type: Any, exc_value: Any, traceback: Any) -> None:
        assert self._file is not None
        self._file.close()
