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
        with StorageSupplier("grpc_sqlite") as s:
            yield s
