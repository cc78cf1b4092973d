"""Live Trial: the define-by-run parameter suggestion surface.

Parity: reference ``optuna/trial/_trial.py`` (Trial :57, suggest flow `_suggest`
:620 — precedence: storage-cached param → enqueued fixed param → single-valued
distribution → relative (joint) sample → independent sample; report :412,
should_prune :513, set_constraint (experimental) :778).
"""
from __future__ import annotations

import copy
import datetime
import warnings
from typing import TYPE_CHECKING, Any, Sequence

from optuna_amd import distributions as _distributions_mod
from optuna_amd import logging as _logging
from optuna_amd.distributions import (
    BaseDistribution,
    CategoricalChoiceType,
    CategoricalDistribution,
    FloatDistribution,
    IntDistribution,
    _get_single_value,
    check_distribution_compatibility,
)
from optuna_amd.trial._base import BaseTrial
from optuna_amd.trial._frozen import FrozenTrial, _checked_constraint_value
from optuna_amd.trial._state import TrialState


_UNSET = object()


if TYPE_CHECKING:
    from optuna_amd.study import Study

_logger = _logging.get_logger(__name__)

_FIXED_PARAMS_KEY = "fixed_params"


class _LazyTrialSystemAttrs(dict):
    """Trial system-attrs view that loads from storage on first read.

    Parity: reference trial/_trial.py:815-827 (_LazyTrialSystemAttrs).
    """

    def __init__(self, trial_id: int, storage: Any) -> None:
        super().__init__()
        object.__setattr__(self, "_trial_id_for_lazy", trial_id)
        object.__setattr__(self, "_storage_for_lazy", storage)
        object.__setattr__(self, "_lazy_initialized", False)

    def _fetch(self) -> None:
        if not object.__getattribute__(self, "_lazy_initialized"):
            object.__setattr__(self, "_lazy_initialized", True)
            storage = object.__getattribute__(self, "_storage_for_lazy")
            trial_id = object.__getattribute__(self, "_trial_id_for_lazy")
            super().update(storage.get_trial_system_attrs(trial_id))

    def __getitem__(self, key: Any) -> Any:
        self._fetch()
        return super().__getitem__(key)

    def __contains__(self, key: Any) -> bool:
        self._fetch()
        return super().__contains__(key)

    def __iter__(self) -> Any:
        self._fetch()
        return super().__iter__()

    def __len__(self) -> int:
        self._fetch()
        return super().__len__()

    def get(self, key: Any, default: Any = None) -> Any:
        self._fetch()
        return super().get(key, default)

    def items(self) -> Any:
        self._fetch()
        return super().items()

    def keys(self) -> Any:
        self._fetch()
        return super().keys()

    def values(self) -> Any:
        self._fetch()
        return super().values()

    def __eq__(self, other: Any) -> bool:
        self._fetch()
        return super().__eq__(other)

    def __ne__(self, other: Any) -> bool:
        return not self.__eq__(other)

    def __hash__(self) -> int:  # dicts are unhashable; keep that behavior
        raise TypeError("unhashable type: '_LazyTrialSystemAttrs'")


class Trial(BaseTrial):
    """A single in-progress evaluation of the objective function."""

    def __init__(self, study: "Study", trial_id: int) -> None:
        self.study = study
        self._trial_id = trial_id
        self.storage = study._storage
        self._cached_frozen_trial = self.storage.get_trial(trial_id)
        # Enqueued fixed params are snapshotted once at creation (reference :71).
        self._fixed_params = self._cached_frozen_trial.system_attrs.get("fixed_params", {})
        self._relative_params: dict[str, Any] | None = None
        study.sampler.before_trial(study, self._cached_frozen_trial)

    def _get_latest_trial(self) -> FrozenTrial:
        """Shallow copy of the cached trial whose system attrs re-read storage lazily.

        Samplers receive this view so attrs written after the snapshot (e.g.
        GridSampler's grid_id from before_trial) are visible without an eager
        storage round trip per suggest (reference trial/_trial.py:700-703).
        """
        latest = copy.copy(self._cached_frozen_trial)
        latest.system_attrs = _LazyTrialSystemAttrs(self._trial_id, self.storage)
        return latest

    # ---- relative (joint) sampling, computed lazily on first suggest ----------------

    @property
    def relative_params(self) -> dict[str, Any]:
        if self._relative_params is None:
            from optuna_amd.pruners import _filter_study

            trial = self._get_latest_trial()
            # Under HyperbandPruner the sampler must only observe sibling
            # trials of this trial's bracket.
            study = _filter_study(self.study, trial)
            self._relative_search_space = study.sampler.infer_relative_search_space(study, trial)
            self._relative_params = study.sampler.sample_relative(
                study, trial, self._relative_search_space
            )
        return self._relative_params

    # ---- suggest_* ------------------------------------------------------------------

    def suggest_float(
        self,
        name: str,
        low: float,
        high: float,
        *,
        step: float | None = None,
        log: bool = False,
    ) -> float:
        return self._suggest(name, FloatDistribution(low, high, log=log, step=step))

    def suggest_uniform(self, name: str, low: float, high: float) -> float:
        warnings.warn(
            "suggest_uniform has been deprecated; use suggest_float instead.", FutureWarning
        )
        return self.suggest_float(name, low, high)

    def suggest_loguniform(self, name: str, low: float, high: float) -> float:
        warnings.warn(
            "suggest_loguniform has been deprecated; use suggest_float(..., log=True) instead.",
            FutureWarning,
        )
        return self.suggest_float(name, low, high, log=True)

    def suggest_discrete_uniform(self, name: str, low: float, high: float, q: float) -> float:
        warnings.warn(
            "suggest_discrete_uniform has been deprecated; use suggest_float(..., step=...) "
            "instead.",
            FutureWarning,
        )
        return self.suggest_float(name, low, high, step=q)

    def suggest_int(self, name: str, low: int, high: int, *, step: int = 1, log: bool = False) -> int:
        return int(self._suggest(name, IntDistribution(low, high, log=log, step=step)))

    def suggest_categorical(
        self, name: str, choices: Sequence[CategoricalChoiceType]
    ) -> CategoricalChoiceType:
        return self._suggest(name, CategoricalDistribution(choices))

    # ---- report / prune -------------------------------------------------------------

    def report(self, value: float, step: int) -> None:
        """Record an intermediate objective value at ``step`` for pruning.

        Parity: reference trial/_trial.py:412-511 (multi-objective rejection,
        float-castability error, negative step error, duplicate-step warning).
        """
        if len(self.study.directions) > 1:
            raise NotImplementedError(
                "Trial.report is not supported for multi-objective optimization."
            )
        try:
            value = float(value)
        except (TypeError, ValueError) as e:
            raise TypeError(
                f"The `value` argument is of type '{type(value).__name__}' but supposed to be a "
                "float."
            ) from e
        try:
            step = int(step)
        except (TypeError, ValueError) as e:
            raise TypeError(
                f"The `step` argument is of type '{type(step).__name__}' but supposed "
                "to be an int."
            ) from e
        if step < 0:
            raise ValueError(f"The `step` argument is {step} but cannot be negative.")
        if step in self._cached_frozen_trial.intermediate_values:
            warnings.warn(
                f"The reported value is ignored because this `step` {step} is already reported."
            )
            return
        self.storage.set_trial_intermediate_value(self._trial_id, step, value)
        self._cached_frozen_trial.intermediate_values[step] = value

    def should_prune(self) -> bool:
        if len(self.study.directions) > 1:
            raise NotImplementedError(
                "Trial.should_prune is not supported for multi-objective optimization."
            )
        trial = self.study._storage.get_trial(self._trial_id)
        return self.study.pruner.prune(self.study, trial)

    # ---- attributes -----------------------------------------------------------------

    def set_user_attr(self, key: str, value: Any) -> None:
        self.storage.set_trial_user_attr(self._trial_id, key, value)
        self._cached_frozen_trial.user_attrs[key] = value

    def set_system_attr(self, key: str, value: Any) -> None:
        warnings.warn(
            "set_system_attr is deprecated; system attributes are internal.", FutureWarning
        )
        self.storage.set_trial_system_attr(self._trial_id, key, value)
        self._cached_frozen_trial.system_attrs[key] = value

    @property
    def constraints(self) -> dict[str, float]:
        """Constraint values as ``{key: value}``; feasible iff all ≤ 0."""
        from optuna_amd.study._constrained_optimization import (
            _get_constraints_from_system_attrs,
        )

        return _get_constraints_from_system_attrs(
            self.storage.get_trial_system_attrs(self._trial_id)
        )

    def set_constraint(self, key: Any, value: Any = _UNSET) -> None:
        """Record one named constraint value (feasible iff ≤ 0).

        ``set_constraint("cost", 1.5)`` writes the per-key system attr
        ``constraints:cost``; the single-argument legacy form
        ``set_constraint([c0, c1, ...])`` writes the whole list under
        ``constraints``. Both are visible through ``FrozenTrial.constraints``.

        Parity: reference trial/_trial.py:778-813.
        """
        from optuna_amd.study._constrained_optimization import _CONSTRAINTS_KEY

        if value is _UNSET:
            if isinstance(key, str) or not isinstance(key, Sequence):
                raise TypeError(
                    "set_constraint requires (key, value), or a single sequence of "
                    "constraint values (legacy form)."
                )
            self.storage.set_trial_system_attr(self._trial_id, _CONSTRAINTS_KEY, list(key))
            return

        value = _checked_constraint_value(key, value)
        constraint_key = f"{_CONSTRAINTS_KEY}:{key}"
        if constraint_key in self.storage.get_trial_system_attrs(self._trial_id):
            warnings.warn(
                f"The constraint value is ignored because this constraint `key={key!r}` "
                "is already set."
            )
            return
        self.storage.set_trial_system_attr(self._trial_id, constraint_key, value)
        self._cached_frozen_trial.system_attrs[constraint_key] = value

    # ---- the suggest core -----------------------------------------------------------

    def _suggest(self, name: str, distribution: BaseDistribution) -> Any:
        storage = self.storage
        trial_id = self._trial_id
        trial = self._cached_frozen_trial

        if name in trial.distributions:
            # Already suggested in this trial: verify compatibility, replay the
            # first call's value (warning if the spec quietly changed).
            check_distribution_compatibility(trial.distributions[name], distribution)
            self._check_distribution(name, distribution)
            return trial.distributions[name].to_external_repr(
                storage.get_trial_param(trial_id, name)
            )

        if self._is_fixed_param(name, distribution):
            param_value = self._fixed_params[name]
        elif distribution.single():
            # Via the module (not the local binding) so test doubles patched on
            # optuna_amd.distributions are honored.
            param_value = _distributions_mod._get_single_value(distribution)
        elif self._is_relative_param(name, distribution):
            param_value = self._relative_params[name]  # type: ignore[index]
        else:
            from optuna_amd.pruners import _filter_study

            latest = self._get_latest_trial()
            study = _filter_study(self.study, latest)
            param_value = study.sampler.sample_independent(
                study, latest, name, distribution
            )

        param_value_in_internal_repr = distribution.to_internal_repr(param_value)
        storage.set_trial_param(trial_id, name, param_value_in_internal_repr, distribution)
        trial.params[name] = param_value
        trial.distributions[name] = distribution
        return param_value

    def _check_distribution(self, name: str, distribution: BaseDistribution) -> None:
        old_distribution = self._cached_frozen_trial.distributions.get(name, distribution)
        if old_distribution != distribution:
            warnings.warn(
                f'Inconsistent parameter values for distribution with name "{name}"! '
                "The values of the first call are used and later calls are ignored. "
                f"Using these values: {old_distribution._asdict()}",
                RuntimeWarning,
            )

    def _is_fixed_param(self, name: str, distribution: BaseDistribution) -> bool:
        if name not in self._fixed_params:
            return False
        param_value = self._fixed_params[name]
        param_value_in_internal_repr = distribution.to_internal_repr(param_value)
        contained = distribution._contains(param_value_in_internal_repr)
        if not contained:
            warnings.warn(
                f"Fixed parameter '{name}' with value {param_value} is out of range "
                f"for distribution {distribution}."
            )
        # The fixed value is used even when out of range (matching the
        # reference: an enqueued/fixed param always wins, with a warning).
        return True

    def _is_relative_param(self, name: str, distribution: BaseDistribution) -> bool:
        if name not in self.relative_params:
            return False
        if name not in self._relative_search_space:
            raise ValueError(
                f"The parameter '{name}' was sampled by `sample_relative` method but it is not "
                "contained in the relative search space."
            )
        relative_distribution = self._relative_search_space[name]
        check_distribution_compatibility(relative_distribution, distribution)
        param_value = self._relative_params[name]  # type: ignore[index]
        return distribution._contains(distribution.to_internal_repr(param_value))

    # ---- properties -----------------------------------------------------------------

    @property
    def number(self) -> int:
        return self._cached_frozen_trial.number

    @property
    def params(self) -> dict[str, Any]:
        return copy.deepcopy(self._cached_frozen_trial.params)

    @property
    def distributions(self) -> dict[str, BaseDistribution]:
        return copy.deepcopy(self._cached_frozen_trial.distributions)

    @property
    def user_attrs(self) -> dict[str, Any]:
        return copy.deepcopy(self._cached_frozen_trial.user_attrs)

    @property
    def system_attrs(self) -> dict[str, Any]:
        return copy.deepcopy(self.storage.get_trial_system_attrs(self._trial_id))

    @property
    def datetime_start(self) -> datetime.datetime | None:
        return self._cached_frozen_trial.datetime_start
