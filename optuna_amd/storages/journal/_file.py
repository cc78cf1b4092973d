"""File-based journal backend: JSON-lines, fsync'd appends, inter-process locks.

Parity: reference ``optuna/storages/journal/_file.py`` (JournalFileBackend :24,
fsync append :104-111, offset index :65-102, JournalFileSymlinkLock /
JournalFileOpenLock :114+). The log format (one JSON object per line) is
compatible with the reference's file journal.
"""
from __future__ import annotations

import abc
import errno
import json
import os
import time
import uuid
from typing import Any

from optuna_amd.storages.journal._base import BaseJournalBackend


LOCK_FILE_SUFFIX = ".lock"
RENAME_FILE_SUFFIX = ".rename"


class BaseJournalFileLock(abc.ABC):
    @abc.abstractmethod
    def acquire(self) -> bool:
        raise NotImplementedError

    @abc.abstractmethod
    def release(self) -> None:
        raise NotImplementedError


class JournalFileSymlinkLock(BaseJournalFileLock):
    """Lock via atomic symlink creation (NFS-safe)."""

    def __init__(self, filepath: str) -> None:
        self._lock_target_file = filepath
        self._lock_file = filepath + LOCK_FILE_SUFFIX
        self._lock_rename_file = self._lock_file + str(uuid.uuid4()) + RENAME_FILE_SUFFIX

    def acquire(self) -> bool:
        sleep_secs = 0.001
        while True:
            try:
                os.symlink(self._lock_target_file, self._lock_file)
                return True
            except OSError as err:
                if err.errno == errno.EEXIST:
                    time.sleep(sleep_secs)
                    sleep_secs = min(sleep_secs * 2, 1)
                    continue
                raise
            except BaseException:
                self.release()
                raise

    def release(self) -> None:
        try:
            os.rename(self._lock_file, self._lock_rename_file)
            os.unlink(self._lock_rename_file)
        except OSError:
            raise RuntimeError("Error: did not possess lock")


class JournalFileOpenLock(BaseJournalFileLock):
    """Lock via O_CREAT|O_EXCL open (works where symlinks are unavailable)."""

    def __init__(self, filepath: str) -> None:
        self._lock_file = filepath + LOCK_FILE_SUFFIX

    def acquire(self) -> bool:
        sleep_secs = 0.001
        while True:
            try:
                open_flags = os.O_CREAT | os.O_EXCL | os.O_WRONLY
                os.close(os.open(self._lock_file, open_flags))
                return True
            except OSError as err:
                if err.errno == errno.EEXIST:
                    time.sleep(sleep_secs)
                    sleep_secs = min(sleep_secs * 2, 1)
                    continue
                raise
            except BaseException:
                self.release()
                raise

    def release(self) -> None:
        try:
            os.unlink(self._lock_file)
        except OSError:
            raise RuntimeError("Error: did not possess lock")


class JournalFileBackend(BaseJournalBackend):
    """JSON-lines journal file with a byte-offset index for incremental reads."""

    def __init__(self, file_path: str, lock_obj: BaseJournalFileLock | None = None) -> None:
        self._file_path = file_path
        self._lock = lock_obj or JournalFileSymlinkLock(self._file_path)
        if not os.path.exists(self._file_path):
            open(self._file_path, "ab").close()
        # log_number_offset[i] = byte offset where log i starts.
        self._log_number_offset: dict[int, int] = {0: 0}

    def read_logs(self, log_number_from: int) -> list[dict[str, Any]]:
        logs = []
        with open(self._file_path, "rb") as f:
            # Seek to the last known offset ≤ requested, then scan forward.
            log_number_start = 0
            if log_number_from in self._log_number_offset:
                f.seek(self._log_number_offset[log_number_from])
                log_number_start = log_number_from

            last_decode_error = None
            for log_number, line in enumerate(f, start=log_number_start):
                byte_len = len(line)
                if log_number + 1 not in self._log_number_offset:
                    self._log_number_offset[log_number + 1] = (
                        self._log_number_offset[log_number] + byte_len
                    )
                if log_number < log_number_from:
                    continue
                if last_decode_error is not None:
                    raise last_decode_error
                try:
                    logs.append(json.loads(line))
                except json.JSONDecodeError as err:
                    # A torn final line means a writer died mid-append; it will be
                    # retried/overwritten. Only raise if it is not the last line.
                    last_decode_error = err
                    del self._log_number_offset[log_number + 1]
            return logs

    def append_logs(self, logs: list[dict[str, Any]]) -> None:
        self._lock.acquire()
        try:
            what_to_write = (
                "".join(json.dumps(log, separators=(",", ":")) + "\n" for log in logs)
            )
            with open(self._file_path, "ab") as f:
                f.write(what_to_write.encode("utf-8"))
                f.flush()
                os.fsync(f.fileno())
        finally:
            self._lock.release()
