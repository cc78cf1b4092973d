"""RDBStorage: SQLAlchemy-backed storage (SQLite / MySQL / PostgreSQL).

Parity: reference ``optuna/storages/_rdb/storage.py`` (RDBStorage :106, scoped
sessions + commit/rollback :73, row-lock trial numbering :456-594, heartbeat
:1041-1093, version manager :1096). The on-disk schema is v12-compatible
(see models.py), so databases interoperate with the reference; schema versioning
is handled by a lightweight version manager instead of alembic — pre-v12 files
upgrade in place through the shape-driven chain in ``upgrade.py``.
"""
from __future__ import annotations

import copy
import json
import time
from contextlib import contextmanager
from datetime import datetime, timedelta, timezone
from typing import Any, Callable, Container, Generator, Sequence

import sqlalchemy
from sqlalchemy import orm as sa_orm
from sqlalchemy.exc import IntegrityError, OperationalError

from optuna_amd import logging as _logging
from optuna_amd.distributions import (
    BaseDistribution,
    check_distribution_compatibility,
    distribution_to_json,
    json_to_distribution,
)
from optuna_amd.exceptions import DuplicatedStudyError, StorageInternalError
from optuna_amd.storages._base import DEFAULT_STUDY_NAME_PREFIX, BaseStorage
from optuna_amd.storages._heartbeat import BaseHeartbeat
from optuna_amd.storages._rdb import models
from optuna_amd.study._frozen import FrozenStudy
from optuna_amd.study._study_direction import StudyDirection
from optuna_amd.trial import FrozenTrial, TrialState
from optuna_amd.version import __version__


_logger = _logging.get_logger(__name__)


@contextmanager
def _create_scoped_session(
    scoped_session: sa_orm.scoped_session,
    ignore_integrity_error: bool = False,
) -> Generator[sa_orm.Session, None, None]:
    session = scoped_session()
    try:
        yield session
        session.commit()
    except IntegrityError as e:
        session.rollback()
        if ignore_integrity_error:
            _logger.debug(f"Ignoring {repr(e)}: probably caused by a benign race.")
        else:
            raise
    except sqlalchemy.exc.SQLAlchemyError as e:
        session.rollback()
        raise StorageInternalError(
            "An exception is raised during the commit. This typically happens due to invalid "
            f"data in the commit, e.g. exceeding max length. (The actual exception is: {e!r})"
        ) from e
    except Exception:
        session.rollback()
        raise
    finally:
        scoped_session.remove()


class RDBStorage(BaseStorage, BaseHeartbeat):
    """Storage backed by a relational database."""

    def __init__(
        self,
        url: str,
        engine_kwargs: dict[str, Any] | None = None,
        skip_compatibility_check: bool = False,
        *,
        heartbeat_interval: int | None = None,
        grace_period: int | None = None,
        heartbeat_stale_trial_callback: Callable[..., None] | None = None,
        failed_trial_callback: Callable[..., None] | None = None,
        skip_table_creation: bool = False,
    ) -> None:
        engine_kwargs = engine_kwargs or {}
        if heartbeat_interval is not None:
            from optuna_amd._experimental import warn_experimental_argument

            warn_experimental_argument("heartbeat_interval")
        if heartbeat_interval is not None and heartbeat_interval <= 0:
            raise ValueError("The value of `heartbeat_interval` should be a positive integer.")
        if grace_period is not None and grace_period <= 0:
            raise ValueError("The value of `grace_period` should be a positive integer.")
        if heartbeat_stale_trial_callback is not None and failed_trial_callback is not None:
            raise ValueError(
                "Specify only one of `heartbeat_stale_trial_callback` and "
                "`failed_trial_callback`."
            )
        if failed_trial_callback is not None:
            import warnings

            warnings.warn(
                "`failed_trial_callback` is deprecated; use "
                "`heartbeat_stale_trial_callback` instead.",
                FutureWarning,
            )
        self.url = url
        self.engine_kwargs = engine_kwargs
        self.skip_compatibility_check = skip_compatibility_check
        self.heartbeat_interval = heartbeat_interval
        self.grace_period = grace_period
        self.failed_trial_callback = heartbeat_stale_trial_callback or failed_trial_callback

        try:
            self.engine = sqlalchemy.engine.create_engine(url, **engine_kwargs)
        except ImportError as e:
            raise ImportError(
                f"Failed to import DB access module for the URL '{url}'. Install the driver "
                f"package for your database. Actual error: {e}."
            ) from e

        self.scoped_session = sa_orm.scoped_session(sa_orm.sessionmaker(bind=self.engine))
        if not skip_table_creation:
            models.BaseModel.metadata.create_all(self.engine)
        self._version_manager = _VersionManager(self.url, self.engine, self.scoped_session)
        if not skip_compatibility_check:
            self._version_manager.check_table_schema_compatibility()

    def __getstate__(self) -> dict[str, Any]:
        state = self.__dict__.copy()
        del state["scoped_session"]
        del state["engine"]
        del state["_version_manager"]
        return state

    def __setstate__(self, state: dict[str, Any]) -> None:
        self.__dict__.update(state)
        self.engine = sqlalchemy.engine.create_engine(self.url, **self.engine_kwargs)
        self.scoped_session = sa_orm.scoped_session(sa_orm.sessionmaker(bind=self.engine))
        models.BaseModel.metadata.create_all(self.engine)
        self._version_manager = _VersionManager(self.url, self.engine, self.scoped_session)

    # ---- studies --------------------------------------------------------------------

    def create_new_study(
        self, directions: Sequence[StudyDirection], study_name: str | None = None
    ) -> int:
        try:
            with _create_scoped_session(self.scoped_session) as session:
                if study_name is None:
                    study_name = self._create_unique_study_name(session)
                study = models.StudyModel(study_name=study_name)
                session.add(study)
                session.flush()
                for objective, d in enumerate(directions):
                    session.add(
                        models.StudyDirectionModel(
                            study_id=study.study_id, direction=d, objective=objective
                        )
                    )
                study_id = study.study_id
        except IntegrityError as e:
            raise DuplicatedStudyError(
                f"Another study with name '{study_name}' already exists. Please specify a "
                "different name, or reuse the existing one by setting `load_if_exists` to True."
            ) from e
        _logger.info(f"A new study created in RDB with name: {study_name}")
        return study_id

    @staticmethod
    def _create_unique_study_name(session: sa_orm.Session) -> str:
        import uuid

        while True:
            study_uuid = str(uuid.uuid4())
            study_name = DEFAULT_STUDY_NAME_PREFIX + study_uuid
            exists = (
                session.query(models.StudyModel)
                .filter(models.StudyModel.study_name == study_name)
                .one_or_none()
            )
            if exists is None:
                return study_name

    def delete_study(self, study_id: int) -> None:
        with _create_scoped_session(self.scoped_session, True) as session:
            study = self._get_study(session, study_id)
            session.delete(study)

    def set_study_user_attr(self, study_id: int, key: str, value: Any) -> None:
        with _create_scoped_session(self.scoped_session, True) as session:
            self._get_study(session, study_id)
            attr = (
                session.query(models.StudyUserAttributeModel)
                .filter_by(study_id=study_id, key=key)
                .one_or_none()
            )
            if attr is None:
                session.add(
                    models.StudyUserAttributeModel(
                        study_id=study_id, key=key, value_json=json.dumps(value)
                    )
                )
            else:
                attr.value_json = json.dumps(value)

    def set_study_system_attr(self, study_id: int, key: str, value: Any) -> None:
        with _create_scoped_session(self.scoped_session, True) as session:
            self._get_study(session, study_id)
            attr = (
                session.query(models.StudySystemAttributeModel)
                .filter_by(study_id=study_id, key=key)
                .one_or_none()
            )
            if attr is None:
                session.add(
                    models.StudySystemAttributeModel(
                        study_id=study_id, key=key, value_json=json.dumps(value)
                    )
                )
            else:
                attr.value_json = json.dumps(value)

    def get_study_id_from_name(self, study_name: str) -> int:
        with _create_scoped_session(self.scoped_session) as session:
            study = (
                session.query(models.StudyModel)
                .filter(models.StudyModel.study_name == study_name)
                .one_or_none()
            )
            if study is None:
                raise KeyError(models.NOT_FOUND_MSG)
            return study.study_id

    def get_study_name_from_id(self, study_id: int) -> str:
        with _create_scoped_session(self.scoped_session) as session:
            return self._get_study(session, study_id).study_name

    def get_study_directions(self, study_id: int) -> list[StudyDirection]:
        with _create_scoped_session(self.scoped_session) as session:
            self._get_study(session, study_id)
            rows = (
                session.query(models.StudyDirectionModel)
                .filter_by(study_id=study_id)
                .order_by(models.StudyDirectionModel.objective)
                .all()
            )
            return [StudyDirection(r.direction) for r in rows]

    def get_study_user_attrs(self, study_id: int) -> dict[str, Any]:
        with _create_scoped_session(self.scoped_session) as session:
            self._get_study(session, study_id)
            rows = (
                session.query(models.StudyUserAttributeModel)
                .filter_by(study_id=study_id)
                .all()
            )
            return {r.key: json.loads(r.value_json) for r in rows}

    def get_study_system_attrs(self, study_id: int) -> dict[str, Any]:
        with _create_scoped_session(self.scoped_session) as session:
            self._get_study(session, study_id)
            rows = (
                session.query(models.StudySystemAttributeModel)
                .filter_by(study_id=study_id)
                .all()
            )
            return {r.key: json.loads(r.value_json) for r in rows}

    def get_all_studies(self) -> list[FrozenStudy]:
        with _create_scoped_session(self.scoped_session) as session:
            studies = session.query(models.StudyModel).order_by(models.StudyModel.study_id).all()
            out = []
            for s in studies:
                directions = (
                    session.query(models.StudyDirectionModel)
                    .filter_by(study_id=s.study_id)
                    .order_by(models.StudyDirectionModel.objective)
                    .all()
                )
                uattrs = (
                    session.query(models.StudyUserAttributeModel)
                    .filter_by(study_id=s.study_id)
                    .all()
                )
                sattrs = (
                    session.query(models.StudySystemAttributeModel)
                    .filter_by(study_id=s.study_id)
                    .all()
                )
                out.append(
                    FrozenStudy(
                        study_name=s.study_name,
                        direction=None,
                        directions=[StudyDirection(d.direction) for d in directions],
                        user_attrs={a.key: json.loads(a.value_json) for a in uattrs},
                        system_attrs={a.key: json.loads(a.value_json) for a in sattrs},
                        study_id=s.study_id,
                    )
                )
            return out

    # ---- trials ---------------------------------------------------------------------

    def create_new_trial(self, study_id: int, template_trial: FrozenTrial | None = None) -> int:
        return self._create_new_trial(study_id, template_trial)._trial_id

    def _create_new_trial(
        self, study_id: int, template_trial: FrozenTrial | None = None
    ) -> FrozenTrial:
        """Insert a trial and return its full record (used by the caching
        wrapper to seed its cache without a second read)."""
        # Deadlocks on the study row lock get a small bounded retry.
        MAX_RETRIES = 5
        trial_id = -1
        for attempt in range(MAX_RETRIES):
            try:
                trial_id = self._insert_new_trial(study_id, template_trial)
                break
            except OperationalError:
                if attempt == MAX_RETRIES - 1:
                    raise
                time.sleep(0.05 * (attempt + 1))
        return self.get_trial(trial_id)

    def _insert_new_trial(self, study_id: int, template_trial: FrozenTrial | None) -> int:
        with _create_scoped_session(self.scoped_session) as session:
            # Serialize numbering on the study row.
            study = (
                session.query(models.StudyModel)
                .filter(models.StudyModel.study_id == study_id)
                .with_for_update()
                .one_or_none()
            )
            if study is None:
                raise KeyError(models.NOT_FOUND_MSG)

            if template_trial is None:
                trial = models.TrialModel(
                    study_id=study_id,
                    number=None,
                    state=TrialState.RUNNING,
                    datetime_start=_to_utc(datetime.now()),
                )
            else:
                # Insert in RUNNING, populate children, then flip to the real state
                # so "only RUNNING is mutable" stays invariant.
                trial = models.TrialModel(
                    study_id=study_id,
                    number=None,
                    state=TrialState.RUNNING,
                    datetime_start=_to_utc(template_trial.datetime_start),
                    datetime_complete=_to_utc(template_trial.datetime_complete),
                )
            session.add(trial)
            session.flush()
            trial.number = (
                session.query(sqlalchemy.func.count(models.TrialModel.trial_id))
                .filter(
                    models.TrialModel.study_id == study_id,
                    models.TrialModel.trial_id < trial.trial_id,
                )
                .scalar()
            )

            if template_trial is not None:
                if template_trial.values is not None:
                    for objective, value in enumerate(template_trial.values):
                        stored, vtype = models.TrialValueModel.value_to_stored_repr(value)
                        session.add(
                            models.TrialValueModel(
                                trial_id=trial.trial_id,
                                objective=objective,
                                value=stored,
                                value_type=vtype,
                            )
                        )
                for name, value in template_trial.params.items():
                    dist = template_trial.distributions[name]
                    session.add(
                        models.TrialParamModel(
                            trial_id=trial.trial_id,
                            param_name=name,
                            param_value=dist.to_internal_repr(value),
                            distribution_json=distribution_to_json(dist),
                        )
                    )
                for step, iv in template_trial.intermediate_values.items():
                    stored, ivtype = (
                        models.TrialIntermediateValueModel.intermediate_value_to_stored_repr(iv)
                    )
                    session.add(
                        models.TrialIntermediateValueModel(
                            trial_id=trial.trial_id,
                            step=step,
                            intermediate_value=stored,
                            intermediate_value_type=ivtype,
                        )
                    )
                for key, value in template_trial.user_attrs.items():
                    session.add(
                        models.TrialUserAttributeModel(
                            trial_id=trial.trial_id, key=key, value_json=json.dumps(value)
                        )
                    )
                for key, value in template_trial.system_attrs.items():
                    session.add(
                        models.TrialSystemAttributeModel(
                            trial_id=trial.trial_id, key=key, value_json=json.dumps(value)
                        )
                    )
                trial.state = template_trial.state
            return trial.trial_id

    def set_trial_param(
        self,
        trial_id: int,
        param_name: str,
        param_value_internal: float,
        distribution: BaseDistribution,
    ) -> None:
        self._set_trial_param(trial_id, param_name, param_value_internal, distribution, None)

    def _set_trial_param(
        self,
        trial_id: int,
        param_name: str,
        param_value_internal: float,
        distribution: BaseDistribution,
        previous_distribution: BaseDistribution | None,
    ) -> None:
        """Write one param; ``previous_distribution`` (when the caller already
        knows it, e.g. the caching wrapper) skips the cross-trial lookup."""
        with _create_scoped_session(self.scoped_session, True) as session:
            trial = self._get_trial_model(session, trial_id)
            self._check_trial_is_updatable_model(trial)
            if previous_distribution is None:
                # Cross-trial compatibility check against any prior use of the name.
                previous = (
                    session.query(models.TrialParamModel)
                    .join(models.TrialModel)
                    .filter(models.TrialModel.study_id == trial.study_id)
                    .filter(models.TrialParamModel.param_name == param_name)
                    .first()
                )
                if previous is not None:
                    check_distribution_compatibility(
                        json_to_distribution(previous.distribution_json), distribution
                    )
            else:
                check_distribution_compatibility(previous_distribution, distribution)
            session.add(
                models.TrialParamModel(
                    trial_id=trial_id,
                    param_name=param_name,
                    param_value=param_value_internal,
                    distribution_json=distribution_to_json(distribution),
                )
            )

    def get_trial_param(self, trial_id: int, param_name: str) -> float:
        with _create_scoped_session(self.scoped_session) as session:
            self._get_trial_model(session, trial_id)
            row = (
                session.query(models.TrialParamModel)
                .filter_by(trial_id=trial_id, param_name=param_name)
                .one_or_none()
            )
            if row is None:
                raise KeyError(models.NOT_FOUND_MSG)
            return row.param_value

    def set_trial_state_values(
        self, trial_id: int, state: TrialState, values: Sequence[float] | None = None
    ) -> bool:
        with _create_scoped_session(self.scoped_session) as session:
            trial = (
                session.query(models.TrialModel)
                .filter(models.TrialModel.trial_id == trial_id)
                .with_for_update()
                .one_or_none()
            )
            if trial is None:
                raise KeyError(models.NOT_FOUND_MSG)
            self._check_trial_is_updatable_model(trial)

            if state == TrialState.RUNNING and trial.state != TrialState.WAITING:
                return False

            trial.state = state
            if state == TrialState.RUNNING:
                trial.datetime_start = _to_utc(datetime.now())
            if state.is_finished():
                trial.datetime_complete = _to_utc(datetime.now())

            if values is not None:
                # Overwrite existing rows (tell on a RUNNING trial is one-shot, but
                # template-trial copies may re-set).
                session.query(models.TrialValueModel).filter_by(trial_id=trial_id).delete()
                for objective, value in enumerate(values):
                    stored, vtype = models.TrialValueModel.value_to_stored_repr(float(value))
                    session.add(
                        models.TrialValueModel(
                            trial_id=trial_id,
                            objective=objective,
                            value=stored,
                            value_type=vtype,
                        )
                    )
            return True

    def set_trial_intermediate_value(
        self, trial_id: int, step: int, intermediate_value: float
    ) -> None:
        with _create_scoped_session(self.scoped_session) as session:
            trial = self._get_trial_model(session, trial_id)
            self._check_trial_is_updatable_model(trial)
            stored, ivtype = models.TrialIntermediateValueModel.intermediate_value_to_stored_repr(
                intermediate_value
            )
            row = (
                session.query(models.TrialIntermediateValueModel)
                .filter_by(trial_id=trial_id, step=step)
                .one_or_none()
            )
            if row is None:
                session.add(
                    models.TrialIntermediateValueModel(
                        trial_id=trial_id,
                        step=step,
                        intermediate_value=stored,
                        intermediate_value_type=ivtype,
                    )
                )
            else:
                row.intermediate_value = stored
                row.intermediate_value_type = ivtype

    def set_trial_user_attr(self, trial_id: int, key: str, value: Any) -> None:
        with _create_scoped_session(self.scoped_session, True) as session:
            trial = self._get_trial_model(session, trial_id)
            self._check_trial_is_updatable_model(trial)
            row = (
                session.query(models.TrialUserAttributeModel)
                .filter_by(trial_id=trial_id, key=key)
                .one_or_none()
            )
            if row is None:
                session.add(
                    models.TrialUserAttributeModel(
                        trial_id=trial_id, key=key, value_json=json.dumps(value)
                    )
                )
            else:
                row.value_json = json.dumps(value)

    def set_trial_system_attr(self, trial_id: int, key: str, value: Any) -> None:
        with _create_scoped_session(self.scoped_session, True) as session:
            trial = self._get_trial_model(session, trial_id)
            self._check_trial_is_updatable_model(trial)
            row = (
                session.query(models.TrialSystemAttributeModel)
                .filter_by(trial_id=trial_id, key=key)
                .one_or_none()
            )
            if row is None:
                session.add(
                    models.TrialSystemAttributeModel(
                        trial_id=trial_id, key=key, value_json=json.dumps(value)
                    )
                )
            else:
                row.value_json = json.dumps(value)

    def get_trial_number_from_id(self, trial_id: int) -> int:
        with _create_scoped_session(self.scoped_session) as session:
            return self._get_trial_model(session, trial_id).number

    def get_trial(self, trial_id: int) -> FrozenTrial:
        with _create_scoped_session(self.scoped_session) as session:
            trial = self._get_trial_model(session, trial_id)
            return self._build_frozen_trial(session, trial)

    def get_n_trials(
        self, study_id: int, state: "tuple[TrialState, ...] | TrialState | None" = None
    ) -> int:
        """One COUNT query instead of materializing every trial row (the
        samplers poll this once per suggest as their startup/delta check)."""
        if isinstance(state, TrialState):
            state = (state,)
        with _create_scoped_session(self.scoped_session) as session:
            self._get_study(session, study_id)
            q = session.query(sqlalchemy.func.count(models.TrialModel.trial_id)).filter(
                models.TrialModel.study_id == study_id
            )
            if state is not None:
                q = q.filter(models.TrialModel.state.in_(list(state)))
            return int(q.scalar() or 0)

    def get_all_trials(
        self,
        study_id: int,
        deepcopy: bool = True,
        states: Container[TrialState] | None = None,
    ) -> list[FrozenTrial]:
        with _create_scoped_session(self.scoped_session) as session:
            self._get_study(session, study_id)
            query = session.query(models.TrialModel).filter(
                models.TrialModel.study_id == study_id
            )
            if states is not None:
                states_list = [s for s in TrialState if s in states]
                query = query.filter(models.TrialModel.state.in_(states_list))
            trials = query.order_by(models.TrialModel.trial_id).all()
            trial_ids = [t.trial_id for t in trials]

            # Bulk-load children per table (avoids N+1 queries).
            def _bulk(model: Any) -> dict[int, list[Any]]:
                out: dict[int, list[Any]] = {}
                if not trial_ids:
                    return out
                for row in (
                    session.query(model).filter(model.trial_id.in_(trial_ids)).all()
                ):
                    out.setdefault(row.trial_id, []).append(row)
                return out

            params = _bulk(models.TrialParamModel)
            values = _bulk(models.TrialValueModel)
            ivalues = _bulk(models.TrialIntermediateValueModel)
            uattrs = _bulk(models.TrialUserAttributeModel)
            sattrs = _bulk(models.TrialSystemAttributeModel)

            frozen = [
                self._assemble_frozen_trial(
                    t,
                    params.get(t.trial_id, []),
                    values.get(t.trial_id, []),
                    ivalues.get(t.trial_id, []),
                    uattrs.get(t.trial_id, []),
                    sattrs.get(t.trial_id, []),
                )
                for t in trials
            ]
            # Freshly assembled objects are already caller-private, but the
            # deepcopy=True contract is observable — honor it like the
            # reference does.
            return copy.deepcopy(frozen) if deepcopy else frozen

    def _get_trials(
        self,
        study_id: int,
        states: Container[TrialState] | None,
        included_trial_ids: Container[int],
        trial_id_greater_than: int,
    ) -> list[FrozenTrial]:
        """Trials with id > frontier, plus the explicitly-watched unfinished ids
        (the _CachedStorage delta protocol; reference storage.py:858-923)."""
        with _create_scoped_session(self.scoped_session) as session:
            self._get_study(session, study_id)
            included = [
                tid
                for tid in included_trial_ids  # type: ignore[union-attr]
                if tid <= trial_id_greater_than
            ]
            base_query = session.query(models.TrialModel).filter(
                models.TrialModel.study_id == study_id
            )
            if states is not None:
                base_query = base_query.filter(
                    models.TrialModel.state.in_([st for st in TrialState if st in states])
                )
            cond = models.TrialModel.trial_id > trial_id_greater_than
            if included:
                cond = sqlalchemy.or_(cond, models.TrialModel.trial_id.in_(included))
            try:
                trials = (
                    base_query.filter(cond).order_by(models.TrialModel.trial_id).all()
                )
            except OperationalError:
                # A huge `included` list can exceed the dialect's bound-variable
                # limit (SQLite: SQLITE_MAX_VARIABLE_NUMBER); fall back to
                # fetching the study's trials and filtering in Python.
                session.rollback()
                included_set = set(included)
                trials = [
                    t
                    for t in base_query.order_by(models.TrialModel.trial_id).all()
                    if t.trial_id > trial_id_greater_than or t.trial_id in included_set
                ]
            trial_ids = [t.trial_id for t in trials]

            def _bulk(model: Any) -> dict[int, list[Any]]:
                out: dict[int, list[Any]] = {}
                if not trial_ids:
                    return out
                for row in session.query(model).filter(model.trial_id.in_(trial_ids)).all():
                    out.setdefault(row.trial_id, []).append(row)
                return out

            params = _bulk(models.TrialParamModel)
            values = _bulk(models.TrialValueModel)
            ivalues = _bulk(models.TrialIntermediateValueModel)
            uattrs = _bulk(models.TrialUserAttributeModel)
            sattrs = _bulk(models.TrialSystemAttributeModel)
            return [
                self._assemble_frozen_trial(
                    t,
                    params.get(t.trial_id, []),
                    values.get(t.trial_id, []),
                    ivalues.get(t.trial_id, []),
                    uattrs.get(t.trial_id, []),
                    sattrs.get(t.trial_id, []),
                )
                for t in trials
            ]

    # ---- heartbeat ------------------------------------------------------------------

    def record_heartbeat(self, trial_id: int) -> None:
        with _create_scoped_session(self.scoped_session, True) as session:
            hb = (
                session.query(models.TrialHeartbeatModel)
                .filter_by(trial_id=trial_id)
                .with_for_update()
                .one_or_none()
            )
            if hb is None:
                session.add(models.TrialHeartbeatModel(trial_id=trial_id))
            else:
                hb.heartbeat = session.execute(
                    sqlalchemy.select(sqlalchemy.func.current_timestamp())
                ).scalar()

    def _get_stale_trial_ids(self, study_id: int) -> list[int]:
        assert self.heartbeat_interval is not None
        if self.grace_period is None:
            grace_period = timedelta(seconds=2 * self.heartbeat_interval)
        else:
            grace_period = timedelta(seconds=self.grace_period)
        stale: list[int] = []
        with _create_scoped_session(self.scoped_session, True) as session:
            current = session.execute(
                sqlalchemy.select(sqlalchemy.func.current_timestamp())
            ).scalar()
            assert current is not None
            running = (
                session.query(models.TrialModel)
                .filter(
                    models.TrialModel.study_id == study_id,
                    models.TrialModel.state == TrialState.RUNNING,
                )
                .all()
            )
            for trial in running:
                hb = (
                    session.query(models.TrialHeartbeatModel)
                    .filter_by(trial_id=trial.trial_id)
                    .one_or_none()
                )
                if hb is None:
                    continue
                if current - hb.heartbeat > grace_period:
                    stale.append(trial.trial_id)
        return stale

    def get_heartbeat_interval(self) -> int | None:
        return self.heartbeat_interval

    def get_failed_trial_callback(self) -> Callable[..., None] | None:
        return self.failed_trial_callback

    # ---- helpers --------------------------------------------------------------------

    @staticmethod
    def _get_study(session: sa_orm.Session, study_id: int) -> models.StudyModel:
        study = (
            session.query(models.StudyModel)
            .filter(models.StudyModel.study_id == study_id)
            .one_or_none()
        )
        if study is None:
            raise KeyError(models.NOT_FOUND_MSG)
        return study

    @staticmethod
    def _get_trial_model(session: sa_orm.Session, trial_id: int) -> models.TrialModel:
        trial = (
            session.query(models.TrialModel)
            .filter(models.TrialModel.trial_id == trial_id)
            .one_or_none()
        )
        if trial is None:
            raise KeyError(models.NOT_FOUND_MSG)
        return trial

    def _check_trial_is_updatable_model(self, trial: models.TrialModel) -> None:
        if trial.state.is_finished():
            from optuna_amd.exceptions import UpdateFinishedTrialError

            raise UpdateFinishedTrialError(
                f"Trial#{trial.number} has already finished and can not be updated."
            )

    def _build_frozen_trial(
        self, session: sa_orm.Session, trial: models.TrialModel
    ) -> FrozenTrial:
        params = (
            session.query(models.TrialParamModel)
            .filter_by(trial_id=trial.trial_id)
            .order_by(models.TrialParamModel.param_id)  # insertion order
            .all()
        )
        values = (
            session.query(models.TrialValueModel).filter_by(trial_id=trial.trial_id).all()
        )
        ivalues = (
            session.query(models.TrialIntermediateValueModel)
            .filter_by(trial_id=trial.trial_id)
            .all()
        )
        uattrs = (
            session.query(models.TrialUserAttributeModel)
            .filter_by(trial_id=trial.trial_id)
            .all()
        )
        sattrs = (
            session.query(models.TrialSystemAttributeModel)
            .filter_by(trial_id=trial.trial_id)
            .all()
        )
        return self._assemble_frozen_trial(trial, params, values, ivalues, uattrs, sattrs)

    @staticmethod
    def _assemble_frozen_trial(
        trial: models.TrialModel,
        params: list[Any],
        values: list[Any],
        ivalues: list[Any],
        uattrs: list[Any],
        sattrs: list[Any],
    ) -> FrozenTrial:
        param_dict = {}
        dist_dict = {}
        # param_id is monotonically assigned -> suggestion (insertion) order.
        for p in sorted(params, key=lambda p: p.param_id):
            dist = json_to_distribution(p.distribution_json)
            param_dict[p.param_name] = dist.to_external_repr(p.param_value)
            dist_dict[p.param_name] = dist
        values_list: list[float] | None = None
        if values:
            values_sorted = sorted(values, key=lambda v: v.objective)
            values_list = [
                models.TrialValueModel.stored_repr_to_value(v.value, v.value_type)
                for v in values_sorted
            ]
        return FrozenTrial(
            number=trial.number,
            state=TrialState(trial.state),
            value=None,
            values=values_list,
            datetime_start=_from_utc(trial.datetime_start),
            datetime_complete=_from_utc(trial.datetime_complete),
            params=param_dict,
            distributions=dist_dict,
            user_attrs={a.key: json.loads(a.value_json) for a in uattrs},
            system_attrs={a.key: json.loads(a.value_json) for a in sattrs},
            intermediate_values={
                iv.step: models.TrialIntermediateValueModel.stored_repr_to_intermediate_value(
                    iv.intermediate_value, iv.intermediate_value_type
                )
                for iv in ivalues
            },
            trial_id=trial.trial_id,
        )

    def get_head_version(self) -> str:
        return f"v{models.SCHEMA_VERSION}"

    def get_current_version(self) -> str:
        return self._version_manager.get_current_version()

    def get_all_versions(self) -> list[str]:
        return [f"v{models.SCHEMA_VERSION}"]

    def upgrade(self) -> None:
        self._version_manager.upgrade()

    def remove_session(self) -> None:
        self.scoped_session.remove()


def _to_utc(dt: datetime | None) -> datetime | None:
    """Naive local time → naive UTC (the v12 column convention)."""
    if dt is None:
        return None
    if dt.tzinfo is None:
        dt = dt.astimezone()
    return dt.astimezone(timezone.utc).replace(tzinfo=None)


def _from_utc(dt: datetime | None) -> datetime | None:
    if dt is None:
        return None
    return dt.replace(tzinfo=timezone.utc).astimezone().replace(tzinfo=None)


class _VersionManager:
    """Single-row ``version_info`` bookkeeping (alembic-free)."""

    def __init__(
        self,
        url: str,
        engine: sqlalchemy.engine.Engine,
        scoped_session: sa_orm.scoped_session,
    ) -> None:
        self.url = url
        self.engine = engine
        self.scoped_session = scoped_session
        self._init_version_info()

    def _init_version_info(self) -> None:
        with _create_scoped_session(self.scoped_session, True) as session:
            vi = session.query(models.VersionInfoModel).one_or_none()
            if vi is None:
                session.add(
                    models.VersionInfoModel(
                        schema_version=models.SCHEMA_VERSION,
                        library_version=__version__,
                    )
                )

    def get_current_version(self) -> str:
        with _create_scoped_session(self.scoped_session) as session:
            vi = session.query(models.VersionInfoModel).one()
            return f"v{vi.schema_version}"

    def check_table_schema_compatibility(self) -> None:
        with _create_scoped_session(self.scoped_session) as session:
            vi = session.query(models.VersionInfoModel).one_or_none()
            if vi is None:
                return
            if vi.schema_version != models.SCHEMA_VERSION:
                raise RuntimeError(
                    f"The runtime schema version {models.SCHEMA_VERSION} is no longer "
                    f"compatible with the table schema (set up by schema version "
                    f"{vi.schema_version}). Please run `optuna-amd storage upgrade`."
                )

    def upgrade(self) -> None:
        """Migrate an older-schema database in place to SCHEMA_VERSION 12.

        The alembic chain the reference ships (v0.9.0.a … v3.2.0.a) is
        re-implemented as shape-driven steps in ``upgrade.py`` — idempotent,
        and correct from any entry version.
        """
        from optuna_amd.storages._rdb.upgrade import upgrade_to_v12

        applied = upgrade_to_v12(self.engine)
        if applied:
            _logger.info(f"Applied schema migrations: {', '.join(applied)}")
        with _create_scoped_session(self.scoped_session, True) as session:
            vi = session.query(models.VersionInfoModel).one_or_none()
            if vi is not None:
                vi.schema_version = models.SCHEMA_VERSION
                vi.library_version = __version__
