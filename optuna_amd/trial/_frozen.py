"""FrozenTrial — the universal immutable trial record.

Parity: reference ``optuna/trial/_frozen.py`` (FrozenTrial :37, create_trial :524).
This is the value type returned by all storage reads; in the MI355X build each
storage also mirrors the numeric columns of finished trials into a device-resident
SoA table (see ``optuna_amd._hip.trial_table``) so sampler kernels never rebuild it.
"""
from __future__ import annotations

import copy
import datetime
import math
import warnings
from typing import Any, Sequence

from optuna_amd import logging as _logging
from optuna_amd.distributions import (
    BaseDistribution,
    CategoricalChoiceType,
    CategoricalDistribution,
    FloatDistribution,
    IntDistribution,
    check_distribution_compatibility,
)
from optuna_amd.trial._base import BaseTrial
from optuna_amd.trial._state import TrialState


_logger = _logging.get_logger(__name__)


def _checked_constraint_value(key: Any, value: Any) -> float:
    """Validate one constraint value: float-castable (TypeError otherwise),
    never NaN (ValueError)."""
    try:
        value = float(value)
    except (TypeError, ValueError):
        raise TypeError(
            f"The `value` argument is of type '{type(value)}' but supposed to be a float."
        ) from None
    if math.isnan(value):
        raise ValueError(
            f"Attempted to set a constraint for {key!r}, but NaN is not allowed."
        )
    return value

# Value types that deepcopy may share (immutable or treated as such everywhere).
_ATOMIC_TYPES = (int, float, str, bool, type(None), datetime.datetime, TrialState)


def _fast_deepcopy_value(v: Any, memo: dict) -> Any:
    """Deepcopy with fast paths for FrozenTrial's typical attribute shapes."""
    if isinstance(v, _ATOMIC_TYPES):
        return v
    tv = type(v)
    if tv is dict:
        return {k: _fast_deepcopy_value(x, memo) for k, x in v.items()}
    if tv is list:
        return [_fast_deepcopy_value(x, memo) for x in v]
    if tv is tuple:
        return tuple(_fast_deepcopy_value(x, memo) for x in v)
    if tv in (FloatDistribution, IntDistribution):
        # All fields are scalars; reconstruct without __init__ revalidation.
        d = object.__new__(tv)
        d.__dict__.update(v.__dict__)
        return d
    return copy.deepcopy(v, memo)


class FrozenTrial(BaseTrial):
    """A finished (or snapshotted) trial.

    Implements the ``suggest_*`` API by replaying stored parameters, so an objective
    function can be re-evaluated against a frozen trial.
    """

    def __init__(
        self,
        number: int,
        state: TrialState,
        value: float | None,
        datetime_start: datetime.datetime | None,
        datetime_complete: datetime.datetime | None,
        params: dict[str, Any],
        distributions: dict[str, BaseDistribution],
        user_attrs: dict[str, Any],
        system_attrs: dict[str, Any],
        intermediate_values: dict[int, float],
        trial_id: int,
        *,
        values: Sequence[float] | None = None,
    ) -> None:
        if value is not None and values is not None:
            raise ValueError("Specify only one of `value` and `values`.")
        self._number = number
        self.state = state
        if value is not None:
            self._values: list[float] | None = [value]
        elif values is not None:
            self._values = list(values)
        else:
            self._values = None
        self._datetime_start = datetime_start
        self.datetime_complete = datetime_complete
        self._params = params
        self._distributions = distributions
        self._user_attrs = user_attrs
        self._system_attrs = system_attrs
        self.intermediate_values = intermediate_values
        self._trial_id = trial_id

    def __eq__(self, other: Any) -> bool:
        if not isinstance(other, FrozenTrial):
            return NotImplemented
        return other.__dict__ == self.__dict__

    def __deepcopy__(self, memo: dict) -> "FrozenTrial":
        # Storages deepcopy trials on every read/tell to isolate their internal
        # state from callers; the generic copy.deepcopy walk dominates tell()
        # latency. Attribute values are overwhelmingly scalars, flat dicts of
        # scalars, and parameter distributions with scalar fields — copy those
        # directly and fall back to copy.deepcopy only for anything unusual
        # (e.g. categorical choices holding user objects).
        cls = self.__class__
        new = cls.__new__(cls)
        memo[id(self)] = new
        for key, v in self.__dict__.items():
            new.__dict__[key] = _fast_deepcopy_value(v, memo)
        return new

    def __lt__(self, other: Any) -> bool:
        if not isinstance(other, FrozenTrial):
            return NotImplemented
        return self.number < other.number

    def __le__(self, other: Any) -> bool:
        if not isinstance(other, FrozenTrial):
            return NotImplemented
        return self.number <= other.number

    def __hash__(self) -> int:
        return hash(tuple(getattr(self, field) for field in self.__dict__))

    def __repr__(self) -> str:
        return (
            f"FrozenTrial(number={self.number}, state={self.state!r}, values={self.values!r}, "
            f"params={self.params!r}, user_attrs={self.user_attrs!r})"
        )

    # ---- BaseTrial surface (replay) -------------------------------------------------

    def suggest_float(
        self, name: str, low: float, high: float, *, step: float | None = None, log: bool = False
    ) -> float:
        return self._suggest(name, FloatDistribution(low, high, log=log, step=step))

    def suggest_uniform(self, name: str, low: float, high: float) -> float:
        return self.suggest_float(name, low, high)

    def suggest_loguniform(self, name: str, low: float, high: float) -> float:
        return self.suggest_float(name, low, high, log=True)

    def suggest_discrete_uniform(self, name: str, low: float, high: float, q: float) -> float:
        return self.suggest_float(name, low, high, step=q)

    def suggest_int(self, name: str, low: int, high: int, step: int = 1, log: bool = False) -> int:
        return int(self._suggest(name, IntDistribution(low, high, log=log, step=step)))

    def suggest_categorical(
        self, name: str, choices: Sequence[CategoricalChoiceType]
    ) -> CategoricalChoiceType:
        return self._suggest(name, CategoricalDistribution(choices))

    def _suggest(self, name: str, distribution: BaseDistribution) -> Any:
        if name not in self._params:
            raise ValueError(
                f"The value of the parameter '{name}' is not found. Please set it at "
                "the construction of the FrozenTrial object."
            )
        value = self._params[name]
        param_value_in_internal_repr = distribution.to_internal_repr(value)
        if not distribution._contains(param_value_in_internal_repr):
            # The stored value is still replayed (matching the reference).
            warnings.warn(
                f"The value {value} of the parameter '{name}' is out of "
                f"the range of the distribution {distribution}."
            )
        if name in self._distributions:
            check_distribution_compatibility(self._distributions[name], distribution)
        self._distributions[name] = distribution
        return value

    def report(self, value: float, step: int) -> None:
        """No-op validation-only report (FrozenTrial is immutable); reference :372-407."""
        value = float(value)
        if step < 0:
            raise ValueError(f"The `step` argument is {step} but cannot be negative.")

    def should_prune(self) -> bool:
        return False

    def set_user_attr(self, key: str, value: Any) -> None:
        self._user_attrs[key] = value

    def set_system_attr(self, key: str, value: Any) -> None:
        self._system_attrs[key] = value

    # ---- Validation -----------------------------------------------------------------

    def _validate(self) -> None:
        if self.state != TrialState.WAITING and self.datetime_start is None:
            raise ValueError(
                "`datetime_start` is supposed to be set when the trial state is not waiting."
            )
        if self.state.is_finished():
            if self.datetime_complete is None:
                raise ValueError("`datetime_complete` is supposed to be set for a finished trial.")
        else:
            if self.datetime_complete is not None:
                raise ValueError(
                    "`datetime_complete` is supposed to be None for an unfinished trial."
                )
        if self.state == TrialState.FAIL and self._values is not None:
            raise ValueError(
                f"values should be None for a failed trial, but got {self._values}."
            )
        if self.state == TrialState.COMPLETE:
            if self._values is None:
                raise ValueError("`value` is supposed to be set for a complete trial.")
            if any(math.isnan(v) for v in self._values):
                raise ValueError("values should not contain NaN.")
        if set(self._params.keys()) != set(self._distributions.keys()):
            raise ValueError(
                "Inconsistent parameters and distributions: "
                f"{set(self._params.keys())} != {set(self._distributions.keys())}"
            )
        for param_name, param_value in self._params.items():
            distribution = self._distributions[param_name]
            param_value_in_internal_repr = distribution.to_internal_repr(param_value)
            if not distribution._contains(param_value_in_internal_repr):
                raise ValueError(
                    f"The value {param_value} of parameter '{param_name}' isn't contained in "
                    f"the distribution {distribution}."
                )

    # ---- Properties -----------------------------------------------------------------

    @property
    def number(self) -> int:
        return self._number

    @number.setter
    def number(self, value: int) -> None:
        self._number = value

    @property
    def value(self) -> float | None:
        if self._values is None:
            return None
        if len(self._values) > 1:
            raise RuntimeError(
                "This attribute is not available during multi-objective optimization."
            )
        return self._values[0]

    @value.setter
    def value(self, v: float | None) -> None:
        if self._values is not None and len(self._values) > 1:
            raise RuntimeError(
                "This attribute is not available during multi-objective optimization."
            )
        self._values = None if v is None else [v]

    @property
    def values(self) -> list[float] | None:
        return self._values

    @values.setter
    def values(self, v: Sequence[float] | None) -> None:
        self._values = None if v is None else list(v)

    # Plain pass-through fields of the record (readable AND assignable — the
    # storages' copy-on-write updates rebind them wholesale).
    def _record_field(slot: str) -> property:  # noqa: N805 — class-body helper
        return property(
            lambda self: getattr(self, slot),
            lambda self, value: setattr(self, slot, value),
        )

    datetime_start = _record_field("_datetime_start")
    params = _record_field("_params")
    distributions = _record_field("_distributions")
    user_attrs = _record_field("_user_attrs")
    system_attrs = _record_field("_system_attrs")
    del _record_field

    @property
    def constraints(self) -> dict[str, float]:
        """Constraint values as ``{key: value}``; feasible iff all ≤ 0.

        Read from both the legacy list attr and per-key ``constraints:<key>``
        attrs (see ``study/_constrained_optimization.py``).
        """
        from optuna_amd.study._constrained_optimization import (
            _get_constraints_from_system_attrs,
        )

        return _get_constraints_from_system_attrs(self._system_attrs)

    def set_constraint(self, key: str, value: Any) -> None:
        """Record one named constraint value (feasible iff ≤ 0); float-castable
        values accepted, NaN rejected, repeated keys ignored with a warning."""
        value = _checked_constraint_value(key, value)
        attr_key = f"constraints:{key}"
        if attr_key in self._system_attrs:
            warnings.warn(
                f"The constraint value is ignored because this constraint `key={key!r}` "
                "is already set."
            )
            return
        self._system_attrs[attr_key] = value

    @property
    def last_step(self) -> int | None:
        if len(self.intermediate_values) == 0:
            return None
        return max(self.intermediate_values.keys())

    @property
    def duration(self) -> datetime.timedelta | None:
        if self.datetime_start and self.datetime_complete:
            return self.datetime_complete - self.datetime_start
        return None

    @property
    def _lifecycle_duration(self) -> datetime.timedelta | None:
        return self.duration


def create_trial(
    *,
    state: TrialState | None = None,
    value: float | None = None,
    values: Sequence[float] | None = None,
    params: dict[str, Any] | None = None,
    distributions: dict[str, BaseDistribution] | None = None,
    user_attrs: dict[str, Any] | None = None,
    system_attrs: dict[str, Any] | None = None,
    intermediate_values: dict[int, float] | None = None,
    constraints: dict[str, float] | None = None,
) -> FrozenTrial:
    """Build a standalone FrozenTrial (e.g. for ``study.add_trial``).

    Parity: reference trial/_frozen.py:524-647.
    """
    from optuna_amd.distributions import _convert_old_distribution_to_new_distribution

    params = params or {}
    distributions = {
        name: _convert_old_distribution_to_new_distribution(dist)
        for name, dist in (distributions or {}).items()
    }
    user_attrs = user_attrs or {}
    system_attrs = system_attrs or {}
    intermediate_values = intermediate_values or {}
    if state is None:
        # NB: `state or COMPLETE` would be wrong — TrialState.RUNNING is enum
        # value 0 and therefore falsy.
        state = TrialState.COMPLETE

    # WAITING trials have not started; their start stamp is set at the
    # WAITING->RUNNING transition by the storage.
    datetime_start = None if state == TrialState.WAITING else datetime.datetime.now()
    datetime_complete = datetime_start if state.is_finished() else None

    trial = FrozenTrial(
        number=-1,
        trial_id=-1,
        state=state,
        value=value,
        values=values,
        datetime_start=datetime_start,
        datetime_complete=datetime_complete,
        params=copy.deepcopy(params),
        distributions=copy.deepcopy(distributions),
        user_attrs=copy.deepcopy(user_attrs),
        system_attrs=copy.deepcopy(system_attrs),
        intermediate_values=copy.deepcopy(intermediate_values),
    )
    if constraints is not None:
        for key, constraint_value in constraints.items():
            trial.set_constraint(key, constraint_value)
    trial._validate()
    return trial


def _check_float_finite_or_nan(value: float, name: str) -> None:
    if math.isinf(value):
        warnings.warn(f"{name} is infinite.")
