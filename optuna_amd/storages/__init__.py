"""Storage registry and the ``get_storage`` resolver.

Parity: reference ``optuna/storages/__init__.py`` (get_storage :42-57:
None → InMemoryStorage; URL string → _CachedStorage(RDBStorage) or JournalStorage
for ``journal://`` paths; bare RDBStorage instances are auto-wrapped).
"""
from __future__ import annotations

from typing import TYPE_CHECKING, Union

from optuna_amd.storages._base import BaseStorage
from optuna_amd.storages._in_memory import InMemoryStorage


if TYPE_CHECKING:
    pass


def get_storage(storage: Union[None, str, BaseStorage]) -> BaseStorage:
    if storage is None:
        return InMemoryStorage()
    if isinstance(storage, str):
        from optuna_amd.storages._cached_storage import _CachedStorage
        from optuna_amd.storages._rdb.storage import RDBStorage

        if storage.startswith("redis"):
            raise ValueError(
                "Redis storage URLs are not supported directly; use JournalStorage with "
                "JournalRedisBackend."
            )
        return _CachedStorage(RDBStorage(storage))
    if type(storage).__name__ == "RDBStorage" and isinstance(storage, BaseStorage):
        from optuna_amd.storages._cached_storage import _CachedStorage

        return _CachedStorage(storage)  # type: ignore[arg-type]
    return storage


def __getattr__(name: str):  # lazy to avoid import cycles / optional deps
    if name == "RDBStorage":
        from optuna_amd.storages._rdb.storage import RDBStorage

        return RDBStorage
    if name == "_CachedStorage":
        from optuna_amd.storages._cached_storage import _CachedStorage

        return _CachedStorage
    if name == "JournalStorage":
        from optuna_amd.storages.journal import JournalStorage

        return JournalStorage
    if name == "JournalFileBackend":
        from optuna_amd.storages.journal import JournalFileBackend

        return JournalFileBackend
    if name == "RcclStorage":
        from optuna_amd.storages._rccl import RcclStorage

        return RcclStorage
    if name == "GrpcStorageProxy":
        from optuna_amd.storages._grpc.client import GrpcStorageProxy

        return GrpcStorageProxy
    if name == "run_grpc_proxy_server":
        from optuna_amd.storages._grpc.server import run_grpc_proxy_server

        return run_grpc_proxy_server
    if name == "GrpcWireStorageProxy":
        from optuna_amd.storages._grpc.wire_client import GrpcWireStorageProxy

        return GrpcWireStorageProxy
    if name == "run_grpc_wire_proxy_server":
        from optuna_amd.storages._grpc.wire_server import run_grpc_wire_proxy_server

        return run_grpc_wire_proxy_server
    if name == "RetryFailedTrialCallback":
        from optuna_amd._callbacks import RetryFailedTrialCallback

        return RetryFailedTrialCallback
    if name == "fail_stale_trials":
        from optuna_amd.storages._heartbeat import fail_stale_trials

        return fail_stale_trials
    if name == "BaseHeartbeat":
        from optuna_amd.storages._heartbeat import BaseHeartbeat

        return BaseHeartbeat
    if name == "BaseJournalBackend":
        from optuna_amd.storages.journal import BaseJournalBackend

        return BaseJournalBackend
    if name in ("JournalFileSymlinkLock", "JournalFileOpenLock"):
        from optuna_amd.storages.journal import _file

        return getattr(_file, name)
    # Deprecated aliases kept for drop-in parity with the reference
    # (storages/__init__.py): old names forward to the new classes.
    if name == "JournalFileStorage" or name == "BaseJournalLogStorage":
        import warnings

        from optuna_amd.storages.journal import BaseJournalBackend, JournalFileBackend

        warnings.warn(
            f"{name} is deprecated; use the *Backend classes instead.",
            FutureWarning,
        )
        return JournalFileBackend if name == "JournalFileStorage" else BaseJournalBackend
    if name == "JournalRedisStorage":
        import warnings

        from optuna_amd.storages.journal._redis import JournalRedisBackend

        warnings.warn(
            "JournalRedisStorage is deprecated; use JournalRedisBackend.",
            FutureWarning,
        )
        return JournalRedisBackend
    if name == "RetryHeartbeatStaleTrialCallback":
        import warnings

        from optuna_amd._callbacks import RetryFailedTrialCallback

        warnings.warn(
            "RetryHeartbeatStaleTrialCallback is deprecated; use "
            "RetryFailedTrialCallback.",
            FutureWarning,
        )
        return RetryFailedTrialCallback
    raise AttributeError(f"module {__name__!r} has no attribute {name!r}")


__all__ = [
    "BaseStorage",
    "_CachedStorage",
    "InMemoryStorage",
    "RDBStorage",
    "JournalStorage",
    "JournalFileBackend",
    "RcclStorage",
    "GrpcStorageProxy",
    "run_grpc_proxy_server",
    "GrpcWireStorageProxy",
    "run_grpc_wire_proxy_server",
    "RetryFailedTrialCallback",
    "fail_stale_trials",
    "BaseHeartbeat",
    "JournalFileSymlinkLock",
    "JournalFileOpenLock",
    "JournalFileStorage",
    "JournalRedisStorage",
    "BaseJournalLogStorage",
    "RetryHeartbeatStaleTrialCallback",
    "get_storage",
]
