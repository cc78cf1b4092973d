"""Shape-driven RDB schema upgrade to SCHEMA_VERSION 12.

The reference evolves its schema through an alembic chain
(reference ``optuna/storages/_rdb/alembic/versions/v0.9.0.a.py`` …
``v3.2.0.a_.py``; ``_VersionManager.upgrade`` at storage.py:1096). alembic is
not available in this environment, so this module re-implements the chain as a
sequence of *shape-driven* steps: each step inspects the live schema and
applies its transform only when the old shape is present, which makes the
whole procedure idempotent and entry-point agnostic — a v0.9-era file walks
every step, a v3.0.0.c-era file only the tail.

Transforms covered (old shape → new shape):

* trials.number column added, backfilled from the ``_number`` trial system
  attribute (alembic v1.3.0.a).
* single ``studies.direction`` column → ``study_directions`` rows; old
  ``trial_values`` (per-step rows) → ``trial_intermediate_values``;
  ``trials.value`` → one ``trial_values`` row per objective (v2.4.0.a).
* old distribution JSON ({Log,Int,Discrete}Uniform…) rewritten to the
  Float/Int/Categorical format (v3.0.0.a) — the JSON codec in
  ``optuna_amd.distributions`` reads both, so this is a decode/encode pass.
* NULL-value rows dropped, error on COMPLETE trials with NULL values
  (v3.0.0.b).
* ``intermediate_value_type`` / ``value_type`` enum columns added with
  NaN/±inf (and legacy float32-clamp sentinel) re-encoding (v3.0.0.c/d).
* ``trials(study_id)`` index (v3.2.0.a) — created by ``metadata.create_all``.

SQLite cannot drop columns that participate in UNIQUE constraints, so tables
whose constraints change are rebuilt (create new → copy → drop old → rename),
the canonical SQLite migration idiom.
"""
from __future__ import annotations

import json
import math

import sqlalchemy
from sqlalchemy import inspect, text

from optuna_amd import version as _version
from optuna_amd.distributions import distribution_to_json, json_to_distribution
from optuna_amd.storages._rdb import models


# The float32 clamp old versions applied to ±inf before the enum encoding
# existed; treated as infinity on upgrade (alembic v3.0.0.c/d semantics).
_F32_MAX = 3.4028234663852886e38
_BIG = 1e16

_OLD_DIST_NAMES = (
    '"UniformDistribution"',
    '"LogUniformDistribution"',
    '"DiscreteUniformDistribution"',
    '"IntUniformDistribution"',
    '"IntLogUniformDistribution"',
)


def _columns(engine: sqlalchemy.engine.Engine, table: str) -> set[str]:
    return {c["name"] for c in inspect(engine).get_columns(table)}


def _tables(engine: sqlalchemy.engine.Engine) -> set[str]:
    return set(inspect(engine).get_table_names())


def _classify_inf(value: float | None) -> tuple[float | None, str]:
    """Re-encode a raw stored float into (value, value_type)."""
    if value is None or math.isnan(value):
        return None, "NAN"
    if value == float("inf") or (value > _BIG and math.isclose(value, _F32_MAX)):
        return None, "INF_POS"
    if value == float("-inf") or (value < -_BIG and math.isclose(value, -_F32_MAX)):
        return None, "INF_NEG"
    return value, "FINITE"


def _step_trial_numbers(engine: sqlalchemy.engine.Engine) -> bool:
    """v1.3.0.a: add trials.number, backfill from the '_number' system attr."""
    if "number" in _columns(engine, "trials"):
        return False
    with engine.begin() as conn:
        conn.execute(text("ALTER TABLE trials ADD COLUMN number INTEGER"))
        rows = conn.execute(
            text(
                "SELECT trial_id, value_json FROM trial_system_attributes "
                "WHERE key = '_number'"
            )
        ).fetchall()
        for trial_id, value_json in rows:
            conn.execute(
                text("UPDATE trials SET number = :n WHERE trial_id = :t"),
                {"n": int(json.loads(value_json)), "t": trial_id},
            )
        conn.execute(text("DELETE FROM trial_system_attributes WHERE key = '_number'"))
        # Trials predating the attr (rare): dense per-study order by id.
        missing = conn.execute(
            text("SELECT trial_id, study_id FROM trials WHERE number IS NULL ORDER BY trial_id")
        ).fetchall()
        counters: dict[int, int] = {}
        for trial_id, study_id in missing:
            taken = conn.execute(
                text("SELECT COUNT(*) FROM trials WHERE study_id = :s AND number IS NOT NULL"),
                {"s": study_id},
            ).scalar()
            n = counters.get(study_id, int(taken or 0))
            conn.execute(
                text("UPDATE trials SET number = :n WHERE trial_id = :t"),
                {"n": n, "t": trial_id},
            )
            counters[study_id] = n + 1
    return True


def _step_split_direction_and_values(engine: sqlalchemy.engine.Engine) -> bool:
    """v2.4.0.a (+ the v3.0.0.b/d trial_values shape, folded into the rebuild):
    studies.direction → study_directions rows; old per-step trial_values →
    trial_intermediate_values; trials.value → trial_values(objective=0)."""
    if "direction" not in _columns(engine, "studies"):
        return False
    old_tv_cols = _columns(engine, "trial_values")
    with engine.begin() as conn:
        conn.execute(
            text(
                "INSERT INTO study_directions (direction, study_id, objective) "
                "SELECT direction, study_id, 0 FROM studies"
            )
        )
        if "step" in old_tv_cols:
            # The pre-v2.4 trial_values table held intermediate values.
            for trial_id, step, value in conn.execute(
                text("SELECT trial_id, step, value FROM trial_values")
            ).fetchall():
                stored, vtype = _classify_inf(value)
                conn.execute(
                    text(
                        "INSERT INTO trial_intermediate_values "
                        "(trial_id, step, intermediate_value, intermediate_value_type) "
                        "VALUES (:t, :s, :v, :vt)"
                    ),
                    {"t": trial_id, "s": step, "v": stored, "vt": vtype},
                )
        # Rebuild trial_values with the final v12 shape (objective column +
        # value_type enum; the old UNIQUE(trial_id, step) blocks ALTERs on
        # SQLite).
        conn.execute(text("DROP TABLE trial_values"))
        conn.execute(
            text(
                "CREATE TABLE trial_values ("
                " trial_value_id INTEGER NOT NULL PRIMARY KEY,"
                " trial_id INTEGER NOT NULL,"
                " objective INTEGER NOT NULL,"
                " value FLOAT,"
                " value_type VARCHAR(7) NOT NULL,"
                " UNIQUE (trial_id, objective),"
                " FOREIGN KEY (trial_id) REFERENCES trials (trial_id)"
                ")"
            )
        )
        for trial_id, value, state in conn.execute(
            text("SELECT trial_id, value, state FROM trials")
        ).fetchall():
            if value is None:
                if str(state) == "COMPLETE":
                    raise ValueError(
                        "Found invalid trials records (value=None and state='COMPLETE')"
                    )
                continue
            stored, vtype = _classify_inf(value)
            if vtype == "NAN":
                continue  # objective values never store NaN in v12
            conn.execute(
                text(
                    "INSERT INTO trial_values (trial_id, objective, value, value_type) "
                    "VALUES (:t, 0, :v, :vt)"
                ),
                {"t": trial_id, "v": stored, "vt": vtype},
            )
        conn.execute(text("ALTER TABLE studies DROP COLUMN direction"))
        conn.execute(text("ALTER TABLE trials DROP COLUMN value"))
    return True


def _step_distribution_json(engine: sqlalchemy.engine.Engine) -> bool:
    """v3.0.0.a: decode/re-encode pre-v3 distribution JSON."""
    with engine.begin() as conn:
        rows = conn.execute(
            text("SELECT param_id, distribution_json FROM trial_params")
        ).fetchall()
        changed = False
        for param_id, dj in rows:
            if dj is None or not any(name in dj for name in _OLD_DIST_NAMES):
                continue
            new_dj = distribution_to_json(json_to_distribution(dj))
            conn.execute(
                text("UPDATE trial_params SET distribution_json = :d WHERE param_id = :p"),
                {"d": new_dj, "p": param_id},
            )
            changed = True
    return changed


def _step_value_types(engine: sqlalchemy.engine.Engine) -> bool:
    """v3.0.0.b/c/d for files that already had the split tables but predate
    the enum columns."""
    changed = False
    if "intermediate_value_type" not in _columns(engine, "trial_intermediate_values"):
        changed = True
        with engine.begin() as conn:
            conn.execute(
                text(
                    "ALTER TABLE trial_intermediate_values ADD COLUMN "
                    "intermediate_value_type VARCHAR(7) NOT NULL DEFAULT 'FINITE'"
                )
            )
            for row_id, value in conn.execute(
                text(
                    "SELECT trial_intermediate_value_id, intermediate_value "
                    "FROM trial_intermediate_values"
                )
            ).fetchall():
                stored, vtype = _classify_inf(value)
                if vtype != "FINITE":
                    conn.execute(
                        text(
                            "UPDATE trial_intermediate_values SET intermediate_value = :v,"
                            " intermediate_value_type = :vt"
                            " WHERE trial_intermediate_value_id = :i"
                        ),
                        {"v": stored, "vt": vtype, "i": row_id},
                    )
    if "value_type" not in _columns(engine, "trial_values"):
        changed = True
        with engine.begin() as conn:
            # v3.0.0.b: NULL values are only legal on unfinished trials.
            bad = conn.execute(
                text(
                    "SELECT COUNT(*) FROM trial_values tv JOIN trials t"
                    " ON tv.trial_id = t.trial_id"
                    " WHERE tv.value IS NULL AND t.state = 'COMPLETE'"
                )
            ).scalar()
            if bad:
                raise ValueError(
                    "Found invalid trial_values records (value=None and state='COMPLETE')"
                )
            conn.execute(text("DELETE FROM trial_values WHERE value IS NULL"))
            conn.execute(
                text(
                    "ALTER TABLE trial_values ADD COLUMN "
                    "value_type VARCHAR(7) NOT NULL DEFAULT 'FINITE'"
                )
            )
            for row_id, value in conn.execute(
                text("SELECT trial_value_id, value FROM trial_values")
            ).fetchall():
                stored, vtype = _classify_inf(value)
                if vtype not in ("FINITE", "NAN"):
                    conn.execute(
                        text(
                            "UPDATE trial_values SET value = :v, value_type = :vt"
                            " WHERE trial_value_id = :i"
                        ),
                        {"v": stored, "vt": vtype, "i": row_id},
                    )
    return changed


def upgrade_to_v12(engine: sqlalchemy.engine.Engine) -> list[str]:
    """Run every applicable step; returns the names of the steps applied.

    ``models.BaseModel.metadata.create_all`` must have run first (the storage
    constructor does) so the v12-only tables exist for the data moves.
    """
    applied = []
    for step in (
        _step_trial_numbers,
        _step_split_direction_and_values,
        _step_distribution_json,
        _step_value_types,
    ):
        if step(engine):
            applied.append(step.__name__.lstrip("_"))

    with engine.begin() as conn:
        if "alembic_version" not in _tables(engine):
            conn.execute(
                text("CREATE TABLE alembic_version (version_num VARCHAR(32) NOT NULL)")
            )
        n = conn.execute(text("SELECT COUNT(*) FROM alembic_version")).scalar()
        if n:
            conn.execute(text("UPDATE alembic_version SET version_num = 'v3.2.0.a'"))
        else:
            conn.execute(text("INSERT INTO alembic_version VALUES ('v3.2.0.a')"))
        conn.execute(
            text(
                "UPDATE version_info SET schema_version = :sv, library_version = :lv"
            ),
            {"sv": models.SCHEMA_VERSION, "lv": _version.__version__},
        )
    return applied
