from optuna_amd.storages.journal._base import BaseJournalBackend, BaseJournalSnapshot
from optuna_amd.storages.journal._file import (
    JournalFileBackend,
    JournalFileOpenLock,
    JournalFileSymlinkLock,
)
from optuna_amd.storages.journal._storage import JournalOperation, JournalStorage


__all__ = [
    "BaseJournalBackend",
    "BaseJournalSnapshot",
    "JournalFileBackend",
    "JournalFileOpenLock",
    "JournalFileSymlinkLock",
    "JournalOperation",
    "JournalStorage",
    "JournalRedisBackend",
]


def __getattr__(name: str):
    if name == "JournalRedisBackend":
        from optuna_amd.storages.journal._redis import JournalRedisBackend

        return JournalRedisBackend
    raise AttributeError(f"module {__name__!r} has no attribute {name!r}")
