"""Schema-upgrade tests: synthesized pre-v12 SQLite files → upgrade → read.

The old schemas are reconstructed from the reference's alembic chain
(v0.9.0.a base schema; v1.3.0.a `_number` attrs; v2.4.0.a single-direction
column and per-step trial_values; pre-v3 distribution JSON; float32-clamped
infinities) rather than copied binary assets.
"""
from __future__ import annotations

import json
import sqlite3

import pytest

import optuna_amd
from optuna_amd.storages._rdb.storage import RDBStorage


_F32_MAX = 3.4028234663852886e38


def _make_v09_db(path: str) -> None:
    conn = sqlite3.connect(path)
    c = conn.cursor()
    c.executescript(
        """
        CREATE TABLE studies (
            study_id INTEGER NOT NULL PRIMARY KEY,
            study_name VARCHAR(512) NOT NULL,
            direction VARCHAR(8) NOT NULL
        );
        CREATE UNIQUE INDEX ix_studies_study_name ON studies (study_name);
        CREATE TABLE version_info (
            version_info_id INTEGER NOT NULL PRIMARY KEY CHECK (version_info_id=1),
            schema_version INTEGER,
            library_version VARCHAR(256)
        );
        CREATE TABLE study_user_attributes (
            study_user_attribute_id INTEGER NOT NULL PRIMARY KEY,
            study_id INTEGER, key VARCHAR(512), value_json VARCHAR(2048),
            UNIQUE (study_id, key)
        );
        CREATE TABLE study_system_attributes (
            study_system_attribute_id INTEGER NOT NULL PRIMARY KEY,
            study_id INTEGER, key VARCHAR(512), value_json VARCHAR(2048),
            UNIQUE (study_id, key)
        );
        CREATE TABLE trials (
            trial_id INTEGER NOT NULL PRIMARY KEY,
            study_id INTEGER,
            state VARCHAR(8) NOT NULL,
            value FLOAT,
            datetime_start DATETIME,
            datetime_complete DATETIME
        );
        CREATE TABLE trial_params (
            param_id INTEGER NOT NULL PRIMARY KEY,
            trial_id INTEGER, param_name VARCHAR(512), param_value FLOAT,
            distribution_json VARCHAR(2048),
            UNIQUE (trial_id, param_name)
        );
        CREATE TABLE trial_user_attributes (
            trial_user_attribute_id INTEGER NOT NULL PRIMARY KEY,
            trial_id INTEGER, key VARCHAR(512), value_json VARCHAR(2048),
            UNIQUE (trial_id, key)
        );
        CREATE TABLE trial_system_attributes (
            trial_system_attribute_id INTEGER NOT NULL PRIMARY KEY,
            trial_id INTEGER, key VARCHAR(512), value_json VARCHAR(2048),
            UNIQUE (trial_id, key)
        );
        CREATE TABLE trial_values (
            trial_value_id INTEGER NOT NULL PRIMARY KEY,
            trial_id INTEGER, step INTEGER, value FLOAT,
            UNIQUE (trial_id, step)
        );
        """
    )
    c.execute("INSERT INTO version_info VALUES (1, 9, '0.9.0')")
    c.execute("INSERT INTO studies VALUES (1, 'old-study', 'MINIMIZE')")
    old_uniform = json.dumps(
        {"name": "UniformDistribution", "attributes": {"low": -5.0, "high": 5.0}}
    )
    old_intlog = json.dumps(
        {"name": "IntLogUniformDistribution", "attributes": {"low": 1, "high": 64, "step": 1}}
    )
    for i, (state, value) in enumerate(
        [("COMPLETE", 1.5), ("COMPLETE", _F32_MAX), ("PRUNED", None), ("COMPLETE", 0.25)]
    ):
        c.execute(
            "INSERT INTO trials (trial_id, study_id, state, value, datetime_start,"
            " datetime_complete) VALUES (?, 1, ?, ?, '2020-01-01 00:00:00',"
            " '2020-01-01 00:01:00')",
            (i + 1, state, value),
        )
        c.execute(
            "INSERT INTO trial_system_attributes (trial_id, key, value_json)"
            " VALUES (?, '_number', ?)",
            (i + 1, str(i)),
        )
        c.execute(
            "INSERT INTO trial_params (trial_id, param_name, param_value,"
            " distribution_json) VALUES (?, 'x', ?, ?)",
            (i + 1, float(i), old_uniform),
        )
        c.execute(
            "INSERT INTO trial_params (trial_id, param_name, param_value,"
            " distribution_json) VALUES (?, 'n', 3.0, ?)",
            (i + 1, old_intlog),
        )
    # pre-v2.4: trial_values rows are per-step intermediate values
    c.execute("INSERT INTO trial_values (trial_id, step, value) VALUES (1, 0, 0.5)")
    c.execute("INSERT INTO trial_values (trial_id, step, value) VALUES (1, 1, ?)", (_F32_MAX,))
    c.execute("INSERT INTO trial_values (trial_id, step, value) VALUES (3, 0, 0.9)")
    c.execute(
        "INSERT INTO trial_user_attributes (trial_id, key, value_json)"
        " VALUES (1, 'tag', '\"blue\"')"
    )
    c.execute(
        "INSERT INTO study_user_attributes (study_id, key, value_json)"
        " VALUES (1, 'owner', '\"me\"')"
    )
    conn.commit()
    conn.close()


def test_upgrade_v09_to_v12(tmp_path) -> None:
    db = str(tmp_path / "old.db")
    _make_v09_db(db)
    url = f"sqlite:///{db}"

    # Incompatible schema must be rejected until upgraded.
    with pytest.raises(RuntimeError):
        RDBStorage(url)

    storage = RDBStorage(url, skip_compatibility_check=True)
    storage.upgrade()

    # Now loads cleanly.
    storage = RDBStorage(url)
    study = optuna_amd.load_study(study_name="old-study", storage=storage)
    assert study.direction == optuna_amd.study.StudyDirection.MINIMIZE
    assert study.user_attrs["owner"] == "me"

    trials = study.trials
    assert [t.number for t in trials] == [0, 1, 2, 3]
    assert trials[0].value == 1.5
    assert trials[1].value == float("inf")  # float32 clamp → INF_POS
    assert trials[2].value is None and trials[2].state.name == "PRUNED"
    assert trials[3].value == 0.25
    assert trials[0].intermediate_values == {0: 0.5, 1: float("inf")}
    assert trials[2].intermediate_values == {0: 0.9}
    assert trials[0].params == {"x": 0.0, "n": 3}
    d = trials[0].distributions
    assert d["x"].low == -5.0 and d["x"].high == 5.0 and not d["x"].log
    assert d["n"].log and d["n"].low == 1 and d["n"].high == 64
    assert trials[0].user_attrs["tag"] == "blue"
    # `_number` bookkeeping attr is gone after migration.
    assert "_number" not in trials[0].system_attrs

    # Idempotent: a second upgrade applies nothing and changes nothing.
    storage2 = RDBStorage(url, skip_compatibility_check=True)
    storage2.upgrade()
    study2 = optuna_amd.load_study(study_name="old-study", storage=RDBStorage(url))
    assert [t.value for t in study2.trials] == [t.value for t in trials]

    # And the study keeps working: new trials append after old ones.
    study.optimize(lambda t: t.suggest_float("x", -5, 5) ** 2, n_trials=2)
    assert len(study.trials) == 6
    assert study.trials[-1].number == 5


def test_upgrade_intermediate_era_db(tmp_path) -> None:
    """A v3.0.0.b-era file: split tables exist but no value_type enums."""
    db = str(tmp_path / "mid.db")
    conn = sqlite3.connect(db)
    c = conn.cursor()
    c.executescript(
        """
        CREATE TABLE studies (
            study_id INTEGER NOT NULL PRIMARY KEY,
            study_name VARCHAR(512) NOT NULL
        );
        CREATE TABLE version_info (
            version_info_id INTEGER NOT NULL PRIMARY KEY CHECK (version_info_id=1),
            schema_version INTEGER, library_version VARCHAR(256)
        );
        CREATE TABLE study_directions (
            study_direction_id INTEGER NOT NULL PRIMARY KEY,
            direction VARCHAR(8) NOT NULL, study_id INTEGER NOT NULL,
            objective INTEGER NOT NULL, UNIQUE (study_id, objective)
        );
        CREATE TABLE study_user_attributes (
            study_user_attribute_id INTEGER NOT NULL PRIMARY KEY,
            study_id INTEGER, key VARCHAR(512), value_json TEXT,
            UNIQUE (study_id, key)
        );
        CREATE TABLE study_system_attributes (
            study_system_attribute_id INTEGER NOT NULL PRIMARY KEY,
            study_id INTEGER, key VARCHAR(512), value_json TEXT,
            UNIQUE (study_id, key)
        );
        CREATE TABLE trials (
            trial_id INTEGER NOT NULL PRIMARY KEY, number INTEGER,
            study_id INTEGER, state VARCHAR(8) NOT NULL,
            datetime_start DATETIME, datetime_complete DATETIME
        );
        CREATE TABLE trial_params (
            param_id INTEGER NOT NULL PRIMARY KEY,
            trial_id INTEGER, param_name VARCHAR(512), param_value FLOAT,
            distribution_json TEXT, UNIQUE (trial_id, param_name)
        );
        CREATE TABLE trial_user_attributes (
            trial_user_attribute_id INTEGER NOT NULL PRIMARY KEY,
            trial_id INTEGER, key VARCHAR(512), value_json TEXT,
            UNIQUE (trial_id, key)
        );
        CREATE TABLE trial_system_attributes (
            trial_system_attribute_id INTEGER NOT NULL PRIMARY KEY,
            trial_id INTEGER, key VARCHAR(512), value_json TEXT,
            UNIQUE (trial_id, key)
        );
        CREATE TABLE trial_values (
            trial_value_id INTEGER NOT NULL PRIMARY KEY,
            trial_id INTEGER NOT NULL, objective INTEGER NOT NULL, value FLOAT,
            UNIQUE (trial_id, objective)
        );
        CREATE TABLE trial_intermediate_values (
            trial_intermediate_value_id INTEGER NOT NULL PRIMARY KEY,
            trial_id INTEGER NOT NULL, step INTEGER NOT NULL,
            intermediate_value FLOAT, UNIQUE (trial_id, step)
        );
        """
    )
    c.execute("INSERT INTO version_info VALUES (1, 11, '2.10.0')")
    c.execute("INSERT INTO studies VALUES (1, 's')")
    c.execute("INSERT INTO study_directions VALUES (1, 'MAXIMIZE', 1, 0)")
    c.execute(
        "INSERT INTO trials VALUES (1, 0, 1, 'COMPLETE',"
        " '2022-01-01 00:00:00', '2022-01-01 00:01:00')"
    )
    c.execute("INSERT INTO trial_values VALUES (1, 1, 0, 7.0)")
    c.execute("INSERT INTO trial_intermediate_values VALUES (1, 1, 0, NULL)")  # NaN
    c.execute("INSERT INTO trial_intermediate_values VALUES (2, 1, 1, 2.0)")
    conn.commit()
    conn.close()

    url = f"sqlite:///{db}"
    storage = RDBStorage(url, skip_compatibility_check=True)
    storage.upgrade()
    study = optuna_amd.load_study(study_name="s", storage=RDBStorage(url))
    import math

    t = study.trials[0]
    assert t.value == 7.0
    assert math.isnan(t.intermediate_values[0])
    assert t.intermediate_values[1] == 2.0
    assert study.direction == optuna_amd.study.StudyDirection.MAXIMIZE


def test_reference_reads_our_upgraded_db(tmp_path) -> None:
    """After our in-place upgrade of a v0.9-era file, the REFERENCE library
    opens it and reads the study (the strongest upgrade-fidelity check)."""
    import logging as _logging
    import os
    import sys
    import types

    if not os.path.isdir("/root/reference"):
        pytest.skip("reference tree not available")
    db = str(tmp_path / "up.db")
    _make_v09_db(db)
    url = f"sqlite:///{db}"
    storage = RDBStorage(url, skip_compatibility_check=True)
    storage.upgrade()

    if "colorlog" not in sys.modules:  # the reference hard-requires colorlog
        stub = types.ModuleType("colorlog")

        class _Fmt(_logging.Formatter):
            def __init__(self, fmt=None, **kwargs):
                if fmt:
                    fmt = fmt.replace("%(log_color)s", "").replace("%(reset)s", "")
                super().__init__(fmt)

        stub.ColoredFormatter = _Fmt
        stub.TTYColoredFormatter = _Fmt
        stub.StreamHandler = _logging.StreamHandler
        sys.modules["colorlog"] = stub
    if "/root/reference" not in sys.path:
        sys.path.insert(0, "/root/reference")
    # alembic is not installed here and the reference's RDBStorage imports it;
    # read the upgraded file through the reference's ORM MODELS instead — the
    # schema contract is what the upgrade must satisfy.
    import sqlalchemy as sa
    import sqlalchemy.orm as sa_orm

    from optuna.storages._rdb import models as ref_models

    engine = sa.create_engine(url)
    with sa_orm.Session(engine) as session:
        trials = (
            session.query(ref_models.TrialModel)
            .order_by(ref_models.TrialModel.number)
            .all()
        )
        assert [t.number for t in trials] == [0, 1, 2, 3]
        values = {
            v.trial_id: ref_models.TrialValueModel.stored_repr_to_value(
                v.value, v.value_type
            )
            for v in session.query(ref_models.TrialValueModel).all()
        }
        assert values[trials[0].trial_id] == 1.5
        assert values[trials[1].trial_id] == float("inf")
        assert trials[2].trial_id not in values  # PRUNED without value
        params = (
            session.query(ref_models.TrialParamModel)
            .filter_by(trial_id=trials[0].trial_id)
            .all()
        )
        assert {p.param_name for p in params} == {"x", "n"}
        for p_ in params:
            assert "Distribution" in p_.distribution_json  # new-format JSON
            assert "Uniform" not in p_.distribution_json
        iv = (
            session.query(ref_models.TrialIntermediateValueModel)
            .filter_by(trial_id=trials[0].trial_id, step=0)
            .one()
        )
        got = ref_models.TrialIntermediateValueModel.stored_repr_to_intermediate_value(
            iv.intermediate_value, iv.intermediate_value_type
        )
        assert got == 0.5
        directions = session.query(ref_models.StudyDirectionModel).all()
        assert [d.direction.name for d in directions] == ["MINIMIZE"]
        vi = session.query(ref_models.VersionInfoModel).one()
        assert vi.schema_version == ref_models.SCHEMA_VERSION
