"""Checkpoint-format interop with the reference implementation.

These tests import the read-only reference library (mounted at /root/reference)
and verify that studies written by optuna_amd through the RDB (schema v12) and
journal backends load back identically through the reference, and vice versa.
Skipped automatically when the reference tree is unavailable.
"""
from __future__ import annotations

import sys
import types

import pytest


REFERENCE_PATH = "/root/reference"


@pytest.fixture(scope="module")
def reference_optuna():
    import os

    if not os.path.isdir(REFERENCE_PATH):
        pytest.skip("reference tree not available")
    # The reference hard-requires colorlog; provide a minimal stand-in.
    if "colorlog" not in sys.modules:
        import logging as _logging

        stub = types.ModuleType("colorlog")

        class _Fmt(_logging.Formatter):
            def __init__(self, fmt=None, **kwargs):  # type: ignore[no-untyped-def]
                if fmt:
                    fmt = fmt.replace("%(log_color)s", "").replace("%(reset)s", "")
                super().__init__(fmt)

        class _TTYFmt(_Fmt):
            def __init__(self, *a, stream=None, **k):  # type: ignore[no-untyped-def]
                super().__init__(*a, **k)

        stub.ColoredFormatter = _Fmt  # type: ignore[attr-defined]
        stub.TTYColoredFormatter = _TTYFmt  # type: ignore[attr-defined]
        stub.StreamHandler = _logging.StreamHandler  # type: ignore[attr-defined]
        sys.modules["colorlog"] = stub
    sys.path.insert(0, REFERENCE_PATH)
    try:
        import optuna as reference

        yield reference
    finally:
        sys.path.remove(REFERENCE_PATH)


def _populate(module, storage_url: str) -> None:
    study = module.create_study(
        study_name="interop",
        storage=storage_url,
        direction="maximize",
        sampler=module.samplers.RandomSampler(seed=7),
    )
    study.set_user_attr("owner", "compat-test")

    def objective(trial):  # type: ignore[no-untyped-def]
        x = trial.suggest_float("x", -5.0, 5.0)
        lg = trial.suggest_float("lg", 1e-3, 1e2, log=True)
        i = trial.suggest_int("i", 0, 20, step=2)
        c = trial.suggest_categorical("c", ("red", "green", None))
        trial.report(x, 0)
        trial.report(x + 1, 1)
        trial.set_user_attr("tag", trial.number)
        return x + i * 0.1

    study.optimize(objective, n_trials=6)


def _check(module, storage_url: str) -> None:
    study = module.load_study(study_name="interop", storage=storage_url)
    assert study.user_attrs == {"owner": "compat-test"}
    trials = study.trials
    assert len(trials) == 6
    for t in trials:
        assert set(t.params) == {"x", "lg", "i", "c"}
        assert -5.0 <= t.params["x"] <= 5.0
        assert t.params["i"] % 2 == 0
        assert t.params["c"] in ("red", "green", None)
        assert t.intermediate_values[1] == pytest.approx(t.params["x"] + 1)
        assert t.user_attrs == {"tag": t.number}
        assert t.state.name == "COMPLETE"
    best = study.best_trial
    assert best.value == max(t.value for t in trials)


def test_rdb_schema_matches_reference_models(reference_optuna) -> None:
    """Table/column/type-level comparison against the reference ORM metadata.

    (The reference's RDBStorage itself needs alembic, unavailable here; its
    models module imports cleanly and defines the authoritative schema.)
    """
    import importlib

    ref_models = importlib.import_module("optuna.storages._rdb.models")
    import optuna_amd.storages._rdb.models as our_models

    assert our_models.SCHEMA_VERSION == ref_models.SCHEMA_VERSION == 12

    ref_tables = ref_models.BaseModel.metadata.tables
    our_tables = our_models.BaseModel.metadata.tables
    assert set(our_tables) == set(ref_tables)

    for name in sorted(ref_tables):
        ref_cols = {c.name: c for c in ref_tables[name].columns}
        our_cols = {c.name: c for c in our_tables[name].columns}
        assert set(our_cols) == set(ref_cols), f"column mismatch in {name}"
        for col_name, ref_col in ref_cols.items():
            our_col = our_cols[col_name]
            assert our_col.primary_key == ref_col.primary_key, (name, col_name)
            assert type(our_col.type).__name__ == type(ref_col.type).__name__, (
                name,
                col_name,
                our_col.type,
                ref_col.type,
            )
            # Enum columns must serialize the same labels.
            if hasattr(ref_col.type, "enums"):
                assert list(our_col.type.enums) == list(ref_col.type.enums), (
                    name,
                    col_name,
                )


def test_sqlite_roundtrip_through_raw_schema(reference_optuna, tmp_path) -> None:
    """Write with optuna_amd, then assemble FrozenTrials straight from the raw
    tables using the reference's value-type enum semantics."""
    import sqlite3

    import optuna_amd

    url = f"sqlite:///{tmp_path}/ours.db"
    _populate(optuna_amd, url)

    conn = sqlite3.connect(f"{tmp_path}/ours.db")
    cur = conn.cursor()
    cur.execute("SELECT schema_version FROM version_info")
    assert cur.fetchone()[0] == 12
    cur.execute("SELECT COUNT(*) FROM trials")
    assert cur.fetchone()[0] == 6
    cur.execute(
        "SELECT param_name, param_value, distribution_json FROM trial_params "
        "JOIN trials USING (trial_id) WHERE trials.number = 0"
    )
    rows = {name: (value, dist_json) for name, value, dist_json in cur.fetchall()}
    assert set(rows) == {"x", "lg", "i", "c"}
    # distribution_json decodes through the REFERENCE codec.
    ref_dist = reference_optuna.distributions.json_to_distribution(rows["x"][1])
    assert ref_dist.low == -5.0 and ref_dist.high == 5.0
    cur.execute("SELECT value_type FROM trial_values LIMIT 1")
    assert cur.fetchone()[0] == "FINITE"
    conn.close()


def test_reference_reads_our_journal(reference_optuna, tmp_path) -> None:
    import optuna_amd

    path = str(tmp_path / "ours.log")
    storage = optuna_amd.storages.JournalStorage(
        optuna_amd.storages.JournalFileBackend(path)
    )
    _populate(optuna_amd, storage)

    ref_storage = reference_optuna.storages.JournalStorage(
        reference_optuna.storages.journal.JournalFileBackend(path)
    )
    _check(reference_optuna, ref_storage)


def test_we_read_reference_journal(reference_optuna, tmp_path) -> None:
    import optuna_amd

    path = str(tmp_path / "theirs.log")
    ref_storage = reference_optuna.storages.JournalStorage(
        reference_optuna.storages.journal.JournalFileBackend(path)
    )
    _populate(reference_optuna, ref_storage)

    storage = optuna_amd.storages.JournalStorage(
        optuna_amd.storages.JournalFileBackend(path)
    )
    _check(optuna_amd, storage)
