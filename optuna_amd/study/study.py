"""Study — the user-facing optimization session facade.

Parity: reference ``optuna/study/study.py`` (Study :67, optimize :412, ask :526,
tell :612, enqueue_trial :869, add_trial :934, _pop_waiting_trial_id :1098,
create_study :1189, load_study :1330, delete_study :1409, copy_study :1461).
"""
from __future__ import annotations

import copy
import threading
import warnings
from typing import TYPE_CHECKING, Any, Callable, Container, Iterable, Sequence, Union

from optuna_amd import exceptions, logging as _logging
from optuna_amd import pruners as _pruners_mod
from optuna_amd import samplers as _samplers_mod
from optuna_amd import storages as _storages_mod
from optuna_amd.distributions import BaseDistribution
from optuna_amd.study._constrained_optimization import _CONSTRAINTS_KEY  # noqa: F401
from optuna_amd.study._frozen import FrozenStudy
from optuna_amd.study._multi_objective import _get_pareto_front_trials
from optuna_amd.study._study_direction import StudyDirection
from optuna_amd.study._study_summary import StudySummary
from optuna_amd.trial import FrozenTrial, Trial, TrialState, create_trial


if TYPE_CHECKING:
    import pandas as pd

    from optuna_amd.pruners import BasePruner
    from optuna_amd.samplers import BaseSampler
    from optuna_amd.storages import BaseStorage

_logger = _logging.get_logger(__name__)

ObjectiveFuncType = Callable[[Trial], Union[float, Sequence[float]]]

_SYSTEM_ATTR_METRIC_NAMES = "study:metric_names"


class _ThreadLocalStudyAttribute(threading.local):
    in_optimize_loop: bool = False


class Study:
    """A study: an optimization session over one objective (or several)."""

    def __init__(
        self,
        study_name: str,
        storage: Union[str, "BaseStorage"],
        sampler: "BaseSampler | None" = None,
        pruner: "BasePruner | None" = None,
    ) -> None:
        self.study_name = study_name
        storage = _storages_mod.get_storage(storage)
        study_id = storage.get_study_id_from_name(study_name)
        self._study_id = study_id
        self._storage = storage
        self._directions = storage.get_study_directions(study_id)

        self.sampler = sampler or _samplers_mod.TPESampler()
        self.pruner = pruner or _pruners_mod.MedianPruner()

        self._thread_local = _ThreadLocalStudyAttribute()
        self._stop_flag = False

    def __getstate__(self) -> dict[str, Any]:
        state = self.__dict__.copy()
        del state["_thread_local"]
        return state

    def __setstate__(self, state: dict[str, Any]) -> None:
        self.__dict__.update(state)
        self._thread_local = _ThreadLocalStudyAttribute()

    # ---- properties -----------------------------------------------------------------

    @property
    def best_params(self) -> dict[str, Any]:
        return self.best_trial.params

    @property
    def best_value(self) -> float:
        best_value = self.best_trial.value
        assert best_value is not None
        return best_value

    @property
    def best_trial(self) -> FrozenTrial:
        return self._get_best_trial(deepcopy=True)

    def _get_best_trial(self, deepcopy: bool) -> FrozenTrial:
        """Best COMPLETE trial; in constrained optimization the best among
        feasible trials (all constraint values ≤ 0)."""
        if self._is_multi_objective():
            raise RuntimeError(
                "A single best trial cannot be retrieved from a multi-objective study. "
                "Consider using Study.best_trials to retrieve a list containing the best trials."
            )
        best_trial = self._storage.get_best_trial(self._study_id)
        if any(v > 0.0 for v in best_trial.constraints.values()):
            from optuna_amd.study._constrained_optimization import _get_feasible_trials

            complete = self.get_trials(deepcopy=False, states=[TrialState.COMPLETE])
            feasible = _get_feasible_trials(complete)
            if not feasible:
                raise ValueError("No feasible trials are completed yet.")
            key = lambda t: t.value  # noqa: E731
            best_trial = (
                max(feasible, key=key)
                if self.direction == StudyDirection.MAXIMIZE
                else min(feasible, key=key)
            )
        return copy.deepcopy(best_trial) if deepcopy else best_trial

    @property
    def best_trials(self) -> list[FrozenTrial]:
        return _get_pareto_front_trials(self, consider_constraint=True)

    @property
    def direction(self) -> StudyDirection:
        if self._is_multi_objective():
            raise RuntimeError(
                "A single direction cannot be retrieved from a multi-objective study. "
                "Consider using Study.directions to retrieve a list containing all directions."
            )
        return self.directions[0]

    @property
    def directions(self) -> list[StudyDirection]:
        return self._directions

    @property
    def trials(self) -> list[FrozenTrial]:
        return self.get_trials(deepcopy=True, states=None)

    @property
    def user_attrs(self) -> dict[str, Any]:
        return copy.deepcopy(self._storage.get_study_user_attrs(self._study_id))

    @property
    def system_attrs(self) -> dict[str, Any]:
        return copy.deepcopy(self._storage.get_study_system_attrs(self._study_id))

    @property
    def metric_names(self) -> list[str] | None:
        return self._storage.get_study_system_attrs(self._study_id).get(
            _SYSTEM_ATTR_METRIC_NAMES
        )

    def _is_multi_objective(self) -> bool:
        return len(self.directions) > 1

    # ---- trials access --------------------------------------------------------------

    def get_trials(
        self,
        deepcopy: bool = True,
        states: Container[TrialState] | None = None,
    ) -> list[FrozenTrial]:
        return self._get_trials(deepcopy=deepcopy, states=states, use_cache=False)

    def _get_trials(
        self,
        deepcopy: bool = True,
        states: Container[TrialState] | None = None,
        use_cache: bool = False,
    ) -> list[FrozenTrial]:
        return self._storage.get_all_trials(self._study_id, deepcopy=deepcopy, states=states)

    # ---- optimize -------------------------------------------------------------------

    def optimize(
        self,
        func: ObjectiveFuncType,
        n_trials: int | None = None,
        timeout: float | None = None,
        n_jobs: int = 1,
        catch: Iterable[type[Exception]] | type[Exception] = (),
        callbacks: list[Callable[["Study", FrozenTrial], None]] | None = None,
        gc_after_trial: bool = False,
        show_progress_bar: bool = False,
    ) -> None:
        """Run the define-by-run optimization loop (reference study.py:412-514)."""
        from optuna_amd.study import _optimize

        _optimize._optimize(
            study=self,
            func=func,
            n_trials=n_trials,
            timeout=timeout,
            n_jobs=n_jobs,
            catch=tuple(catch) if isinstance(catch, Iterable) else (catch,),
            callbacks=callbacks,
            gc_after_trial=gc_after_trial,
            show_progress_bar=show_progress_bar,
        )

    # ---- ask / tell -----------------------------------------------------------------

    def ask(self, fixed_distributions: dict[str, BaseDistribution] | None = None) -> Trial:
        """Create a new trial manually (reference study.py:526-610)."""
        if not self._thread_local.in_optimize_loop and is_heartbeat_enabled(self._storage):
            warnings.warn("Heartbeat of storage is supposed to be used with Study.optimize.")

        fixed_distributions = fixed_distributions or {}
        from optuna_amd.distributions import _convert_old_distribution_to_new_distribution

        fixed_distributions = {
            name: _convert_old_distribution_to_new_distribution(dist)
            for name, dist in fixed_distributions.items()
        }

        trial_id = self._pop_waiting_trial_id()
        if trial_id is None:
            trial_id = self._storage.create_new_trial(self._study_id)
        trial = Trial(self, trial_id)

        for name, dist in fixed_distributions.items():
            trial._suggest(name, dist)
        return trial

    def tell(
        self,
        trial: Trial | int,
        values: float | Sequence[float] | None = None,
        state: TrialState | None = None,
        skip_if_finished: bool = False,
    ) -> FrozenTrial:
        """Finish a trial created with ask (reference study.py:612-741)."""
        from optuna_amd.study._tell import _get_frozen_trial, _tell_with_warning

        _tell_with_warning(
            study=self,
            trial=trial,
            value_or_values=values,
            state=state,
            skip_if_finished=skip_if_finished,
        )
        return copy.deepcopy(_get_frozen_trial(self, trial))

    # ---- attrs ----------------------------------------------------------------------

    def set_user_attr(self, key: str, value: Any) -> None:
        self._storage.set_study_user_attr(self._study_id, key, value)

    def set_system_attr(self, key: str, value: Any) -> None:
        warnings.warn("set_system_attr is deprecated.", FutureWarning)
        self._storage.set_study_system_attr(self._study_id, key, value)

    def set_metric_names(self, metric_names: list[str]) -> None:
        warnings.warn(
            "Study.set_metric_names is experimental (supported from v3.2.0). "
            "The interface can change in the future.",
            exceptions.ExperimentalWarning,
            stacklevel=2,
        )
        if len(self._directions) != len(metric_names):
            raise ValueError("The number of objectives must match the length of the metric names.")
        self._storage.set_study_system_attr(
            self._study_id, _SYSTEM_ATTR_METRIC_NAMES, metric_names
        )

    # ---- dataframe ------------------------------------------------------------------

    def trials_dataframe(
        self,
        attrs: tuple[str, ...] = (
            "number",
            "value",
            "datetime_start",
            "datetime_complete",
            "duration",
            "params",
            "user_attrs",
            "system_attrs",
            "state",
        ),
        multi_index: bool = False,
    ) -> "pd.DataFrame":
        from optuna_amd.study._dataframe import _trials_dataframe

        return _trials_dataframe(self, attrs, multi_index)

    # ---- stop / enqueue / add -------------------------------------------------------

    def stop(self) -> None:
        """Request the enclosing optimize loop to exit after the current trial."""
        if not self._thread_local.in_optimize_loop:
            raise RuntimeError(
                "`Study.stop` is supposed to be invoked inside an objective function or a "
                "callback."
            )
        self._stop_flag = True

    def enqueue_trial(
        self,
        params: dict[str, Any],
        user_attrs: dict[str, Any] | None = None,
        skip_if_exists: bool = False,
    ) -> None:
        """Queue a WAITING trial whose parameters are fixed (reference :869-932)."""
        if not isinstance(params, dict):
            raise TypeError(
                f"The params argument is of type '{type(params).__name__}' but must be a dict."
            )
        if skip_if_exists and self._should_skip_enqueue(params):
            _logger.info(f"Trial with params {params} already exists. Skipping enqueue.")
            return
        self.add_trial(
            create_trial(
                state=TrialState.WAITING,
                system_attrs={"fixed_params": params},
                user_attrs=user_attrs,
            )
        )

    def _should_skip_enqueue(self, params: dict[str, Any]) -> bool:
        import math

        for trial in self.get_trials(deepcopy=False):
            trial_params = trial.system_attrs.get("fixed_params", trial.params)
            if trial_params.keys() != params.keys():
                continue

            def _all_equal() -> bool:
                for k in params:
                    a, b = params[k], trial_params[k]
                    both_nan = (
                        isinstance(a, float)
                        and isinstance(b, float)
                        and math.isnan(a)
                        and math.isnan(b)
                    )
                    if a != b and not both_nan:
                        return False
                return True

            if _all_equal():
                return True
        return False

    def add_trial(self, trial: FrozenTrial) -> None:
        """Register an externally-built FrozenTrial (reference :934-1010)."""
        trial._validate()
        if trial.values is not None and len(self.directions) != len(trial.values):
            raise ValueError(
                f"The number of the values {len(trial.values)} did not match the number of "
                f"the objectives {len(self.directions)} in the study."
            )
        self._storage.create_new_trial(self._study_id, template_trial=trial)

    def add_trials(self, trials: Iterable[FrozenTrial]) -> None:
        trials = list(trials)
        bulk = getattr(self._storage, "bulk_create_trials", None)
        if bulk is not None and len(trials) > 1:
            for trial in trials:
                trial._validate()
            bulk(self._study_id, trials)
            return
        for trial in trials:
            self.add_trial(trial)

    def _pop_waiting_trial_id(self) -> int | None:
        for trial in self._storage.get_all_trials(
            self._study_id, deepcopy=False, states=(TrialState.WAITING,)
        ):
            # WAITING→RUNNING is a CAS: only one worker wins a given trial.
            # A racing worker may even have FINISHED the trial since the
            # listing — that surfaces as UpdateFinishedTrialError, not False.
            try:
                if not self._storage.set_trial_state_values(
                    trial._trial_id, state=TrialState.RUNNING
                ):
                    continue
            except exceptions.UpdateFinishedTrialError:
                continue
            _logger.debug(f"Trial#{trial.number} is popped from the trial queue.")
            return trial._trial_id
        return None

    # ---- logging helper -------------------------------------------------------------

    def _log_completed_trial(
        self, values: "list[float]", number: int, params: "dict[str, Any]"
    ) -> None:
        """Completion log line; touches storage (best-trial lookup) only when
        INFO is actually enabled, so quiet runs pay nothing per trial."""
        if not _logger.isEnabledFor(_logging.INFO):
            return
        metric_names = self.metric_names
        if len(values) > 1:
            shown: "list[float] | dict[str, float]" = (
                values
                if metric_names is None
                else {name: v for name, v in zip(metric_names, values)}
            )
            _logger.info(
                f"Trial {number} finished with values: {shown} and parameters: {params}."
            )
        elif len(values) == 1:
            shown_one: "float | dict[str, float]" = (
                values[0] if metric_names is None else {metric_names[0]: values[0]}
            )
            message = (
                f"Trial {number} finished with value: {shown_one} and parameters: {params}."
            )
            try:
                best = self.best_trial
                message += f" Best is trial {best.number} with value: {best.value}."
            except ValueError:
                pass
            _logger.info(message)


# ----------------------------------------------------------------------------------
# Module-level API
# ----------------------------------------------------------------------------------


def create_study(
    *,
    storage: Union[str, "BaseStorage", None] = None,
    sampler: "BaseSampler | None" = None,
    pruner: "BasePruner | None" = None,
    study_name: str | None = None,
    direction: str | StudyDirection | None = None,
    load_if_exists: bool = False,
    directions: Sequence[str | StudyDirection] | None = None,
) -> Study:
    """Create (or load, with load_if_exists) a study (reference study.py:1189-1328)."""
    if direction is None and directions is None:
        directions = ["minimize"]
    elif direction is not None and directions is not None:
        raise ValueError("Specify only one of `direction` and `directions`.")
    elif direction is not None:
        directions = [direction]
    assert directions is not None
    if len(directions) < 1:
        raise ValueError("The number of objectives must be greater than 0.")

    if not all(
        d in ("minimize", "maximize", StudyDirection.MINIMIZE, StudyDirection.MAXIMIZE)
        for d in directions
    ):
        raise ValueError(
            "Please set either 'minimize' or 'maximize' to direction. You can also set the "
            "corresponding `StudyDirection` member."
        )
    direction_objects = [
        d if isinstance(d, StudyDirection) else StudyDirection[d.upper()] for d in directions
    ]

    storage_obj = _storages_mod.get_storage(storage)
    try:
        study_id = storage_obj.create_new_study(direction_objects, study_name)
    except exceptions.DuplicatedStudyError:
        if load_if_exists:
            assert study_name is not None
            _logger.info(
                f"Using an existing study with name '{study_name}' instead of creating a new one."
            )
            study_id = storage_obj.get_study_id_from_name(study_name)
        else:
            raise

    study_name = storage_obj.get_study_name_from_id(study_id)
    return Study(study_name=study_name, storage=storage_obj, sampler=sampler, pruner=pruner)


def load_study(
    *,
    study_name: str | None,
    storage: Union[str, "BaseStorage"],
    sampler: "BaseSampler | None" = None,
    pruner: "BasePruner | None" = None,
) -> Study:
    """Load an existing study (reference study.py:1330-1406)."""
    storage_obj = _storages_mod.get_storage(storage)
    if study_name is None:
        studies = storage_obj.get_all_studies()
        if len(studies) != 1:
            raise ValueError(
                f"Could not determine the study name since the storage {storage} does not "
                "contain exactly 1 study. Specify `study_name`."
            )
        study_name = studies[0].study_name
    return Study(study_name=study_name, storage=storage_obj, sampler=sampler, pruner=pruner)


def delete_study(
    *,
    study_name: str,
    storage: Union[str, "BaseStorage"],
) -> None:
    storage_obj = _storages_mod.get_storage(storage)
    study_id = storage_obj.get_study_id_from_name(study_name)
    storage_obj.delete_study(study_id)


def copy_study(
    *,
    from_study_name: str,
    from_storage: Union[str, "BaseStorage"],
    to_storage: Union[str, "BaseStorage"],
    to_study_name: str | None = None,
) -> None:
    """Copy a study with trials and attributes (reference study.py:1461-1559)."""
    from_study = load_study(study_name=from_study_name, storage=from_storage)
    to_study = create_study(
        study_name=to_study_name or from_study_name,
        storage=to_storage,
        directions=from_study.directions,
        load_if_exists=False,
    )
    for key, value in from_study._storage.get_study_system_attrs(from_study._study_id).items():
        to_study._storage.set_study_system_attr(to_study._study_id, key, value)
    for key, value in from_study.user_attrs.items():
        to_study.set_user_attr(key, value)
    for trial in from_study.get_trials(deepcopy=False):
        to_study.add_trial(trial)


def get_all_study_summaries(
    storage: Union[str, "BaseStorage"], include_best_trial: bool = True
) -> list[StudySummary]:
    """Summaries of all studies in a storage (reference study.py:1562-1620)."""
    storage_obj = _storages_mod.get_storage(storage)
    frozen_studies = storage_obj.get_all_studies()
    summaries = []
    for fs in frozen_studies:
        best_trial: FrozenTrial | None = None
        n_trials = storage_obj.get_n_trials(fs._study_id)
        datetime_start = None
        all_trials = storage_obj.get_all_trials(fs._study_id, deepcopy=False)
        if len(all_trials) > 0:
            datetime_start = min(
                t.datetime_start for t in all_trials if t.datetime_start is not None
            ) if any(t.datetime_start is not None for t in all_trials) else None
        if include_best_trial and len(fs.directions) == 1:
            try:
                best_trial = storage_obj.get_best_trial(fs._study_id)
            except ValueError:
                best_trial = None
        summaries.append(
            StudySummary(
                study_name=fs.study_name,
                direction=None,
                directions=fs.directions,
                best_trial=best_trial,
                user_attrs=fs.user_attrs,
                system_attrs=fs.system_attrs,
                n_trials=n_trials,
                datetime_start=datetime_start,
                study_id=fs._study_id,
            )
        )
    return summaries


def get_all_study_names(storage: Union[str, "BaseStorage"]) -> list[str]:
    storage_obj = _storages_mod.get_storage(storage)
    return [s.study_name for s in storage_obj.get_all_studies()]


def is_heartbeat_enabled(storage: "BaseStorage") -> bool:
    from optuna_amd.storages._heartbeat import BaseHeartbeat

    return isinstance(storage, BaseHeartbeat) and storage.is_heartbeat_enabled()
