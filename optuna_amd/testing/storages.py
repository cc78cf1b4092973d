"""StorageSupplier: materialize any storage mode for tests without a cluster.

Parity: reference ``optuna/testing/storages.py`` (StorageSupplier :34-207 —
sqlite in temp files, gRPC servers on free ports, etc.).
"""
from __future__ import annotations

import os
import socket
import tempfile
import threading
from types import TracebackType
from typing import Any

import optuna_amd
from optuna_amd.storages import BaseStorage


STORAGE_MODES: list[str] = [
    "inmemory",
    "sqlite",
    "cached_sqlite",
    "journal",
    # Named like the reference's gRPC modes so shared test suites apply the
    # same skips (the gRPC client doesn't use copy.deepcopy).
    "grpc_rdb",
    "grpc_journal_file",
    "grpc_wire",
]

STORAGE_MODES_HEARTBEAT = [
    "sqlite",
    "cached_sqlite",
]

SQLITE3_TIMEOUT = 300


def _find_free_port() -> int:
    with socket.socket(socket.AF_INET, socket.SOCK_STREAM) as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


class StorageSupplier:
    _ALIASES = {"grpc_sqlite": "grpc_rdb", "grpc_journal": "grpc_journal_file"}

    def __init__(self, storage_specifier: str, **kwargs: Any) -> None:
        storage_specifier = self._ALIASES.get(storage_specifier, storage_specifier)
        self.storage_specifier = storage_specifier
        self.extra_args = kwargs
        self.tempfile: Any = None
        self.server: Any = None
        self.proxy: Any = None
        self.thread: threading.Thread | None = None

    def __enter__(self) -> BaseStorage:
        if self.storage_specifier == "inmemory":
            if len(self.extra_args) > 0:
                raise ValueError("InMemoryStorage does not accept any arguments!")
            return optuna_amd.storages.InMemoryStorage()
        elif self.storage_specifier == "grpc_wire":
            return self._start_grpc(optuna_amd.storages.InMemoryStorage())
        elif (
            "sqlite" in self.storage_specifier
            or "rdb" in self.storage_specifier
            or "journal" in self.storage_specifier
        ):
            if self.storage_specifier in ("journal", "grpc_journal_file"):
                self.tempfile = tempfile.NamedTemporaryFile(suffix=".log", delete=False)
                from optuna_amd.storages.journal import JournalFileBackend, JournalStorage

                storage: BaseStorage = JournalStorage(JournalFileBackend(self.tempfile.name))
            else:
                self.tempfile = tempfile.NamedTemporaryFile(suffix=".db", delete=False)
                url = f"sqlite:///{self.tempfile.name}"
                from optuna_amd.storages._rdb.storage import RDBStorage

                rdb = RDBStorage(
                    url,
                    engine_kwargs={"connect_args": {"timeout": SQLITE3_TIMEOUT}},
                    **self.extra_args,
                )
                if self.storage_specifier == "cached_sqlite":
                    from optuna_amd.storages._cached_storage import _CachedStorage

                    storage = _CachedStorage(rdb)
                else:
                    storage = rdb
            if self.storage_specifier.startswith("grpc_"):
                return self._start_grpc(storage)
            return storage
        else:
            raise ValueError(f"Unknown storage specifier {self.storage_specifier}")

    def _start_grpc(self, backend: BaseStorage) -> BaseStorage:
        port = _find_free_port()
        if self.storage_specifier == "grpc_wire":
            from optuna_amd.storages._grpc.wire_client import GrpcWireStorageProxy
            from optuna_amd.storages._grpc.wire_server import make_wire_server

            self.server = make_wire_server(backend, "127.0.0.1", port)
            self.server.start()
            self.proxy = GrpcWireStorageProxy(host="127.0.0.1", port=port)
            return self.proxy
        from optuna_amd.storages._grpc.client import GrpcStorageProxy
        from optuna_amd.storages._grpc.server import make_server

        self.server = make_server(backend, "127.0.0.1", port)
        self.server.start()
        self.proxy = GrpcStorageProxy(host="127.0.0.1", port=port)
        return self.proxy

    def __exit__(
        self,
        exc_type: type[BaseException] | None,
        exc_val: BaseException | None,
        exc_tb: TracebackType | None,
    ) -> None:
        if self.proxy is not None:
            self.proxy.close()
            self.proxy = None
        if self.server is not None:
            self.server.stop(grace=None)
            self.server = None
        if self.tempfile is not None:
            name = self.tempfile.name
            self.tempfile.close()
            try:
                os.unlink(name)
            except OSError:
                pass
            self.tempfile = None
