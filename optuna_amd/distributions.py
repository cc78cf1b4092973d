"""Parameter distributions and the internal/external representation codec.

A distribution describes one searchable parameter. Every parameter value has two
representations:

* **external repr** — what user code receives from ``trial.suggest_*`` (float, int,
  or an arbitrary categorical choice);
* **internal repr** — a single ``float`` used by samplers and storages (categorical
  values map to their choice index).

Parity: reference ``optuna/distributions.py`` (BaseDistribution :31,
FloatDistribution :109, IntDistribution :310, CategoricalDistribution :470,
json_to_distribution :565, distribution_to_json :609,
check_distribution_compatibility :623). The JSON codec is byte-compatible with the
reference (used by the RDB schema-v12 ``trial_params.distribution_json`` column and
the journal log format); legacy pre-v3 distribution names are decodable.
"""
from __future__ import annotations

import abc
import decimal
import json
import math
import warnings
from typing import Any, Sequence, Union


CategoricalChoiceType = Union[None, bool, int, float, str]

_FLOAT_CLS = "FloatDistribution"
_INT_CLS = "IntDistribution"
_CATEGORICAL_CLS = "CategoricalDistribution"


class BaseDistribution(abc.ABC):
    """Base class of parameter distributions."""

    def to_external_repr(self, param_value_in_internal_repr: float) -> Any:
        return param_value_in_internal_repr

    def to_internal_repr(self, param_value_in_external_repr: Any) -> float:
        return param_value_in_external_repr

    @abc.abstractmethod
    def single(self) -> bool:
        """Whether the domain contains exactly one value."""
        raise NotImplementedError

    @abc.abstractmethod
    def _contains(self, param_value_in_internal_repr: float) -> bool:
        raise NotImplementedError

    def _asdict(self) -> dict[str, Any]:
        return dict(self.__dict__)

    def __eq__(self, other: Any) -> bool:
        if not isinstance(other, BaseDistribution):
            return NotImplemented
        if type(self) is not type(other):
            return False
        return self.__dict__ == other.__dict__

    def __hash__(self) -> int:
        return hash((type(self),) + tuple(sorted(self.__dict__.items(), key=lambda x: x[0])))

    def __repr__(self) -> str:
        kwargs = ", ".join(f"{k}={v!r}" for k, v in sorted(self._asdict().items()))
        return f"{type(self).__name__}({kwargs})"


def _adjust_discrete_high(low: float, high: float, step: float) -> float:
    """Clip ``high`` onto the grid ``low + k*step`` (largest representable point)."""
    d_high = decimal.Decimal(str(high))
    d_low = decimal.Decimal(str(low))
    d_step = decimal.Decimal(str(step))
    d_r = d_high - d_low
    if d_r % d_step != 0:
        # Largest grid point low + k*step ≤ high, in exact decimal arithmetic
        # (reference distributions.py:660-675; a quantize to the step's decimal
        # places would corrupt non-decimal lows). Note the float() rounding of
        # the exact grid point means a re-ingested high may renormalize one
        # more grid point down for irrational steps — the reference behaves
        # identically, and stability there matters less than landing on the
        # true k on first construction.
        k = d_r // d_step
        adjusted = float(k * d_step + d_low)
        warnings.warn(
            f"The distribution is specified by [{low}, {high}] and step={step}, but the range "
            f"is not divisible by `step`. It will be replaced by [{low}, {adjusted}]."
        )
        return adjusted
    return high


class FloatDistribution(BaseDistribution):
    """A continuous (optionally log-scaled or step-discretized) float domain.

    Args mirror the reference (distributions.py:109-199): ``step`` and ``log=True``
    are mutually exclusive; ``log=True`` requires ``low > 0``.
    """

    def __init__(self, low: float, high: float, log: bool = False, step: float | None = None) -> None:
        if log and step is not None:
            raise ValueError("The parameter `step` is not supported when `log` is true.")
        if low > high:
            raise ValueError(
                f"The `low` value must be smaller than or equal to the `high` value "
                f"(low={low}, high={high})."
            )
        if log and low <= 0.0:
            raise ValueError(
                f"The `low` value must be larger than 0 for a log distribution (low={low})."
            )
        if step is not None and step <= 0:
            raise ValueError(f"The `step` value must be non-zero positive value, but step={step}.")
        if math.isnan(low) or math.isnan(high):
            raise ValueError("The `low` and `high` must not be NaN.")

        self.low = float(low)
        self.step = None if step is None else float(step)
        self.log = log
        if step is not None:
            self.high = float(_adjust_discrete_high(self.low, float(high), self.step))
        else:
            self.high = float(high)

    def single(self) -> bool:
        if self.step is None:
            return self.low == self.high
        if self.low == self.high:
            return True
        # Exact decimal arithmetic: float subtraction would call e.g.
        # [0.2, 0.3] step 0.1 single (0.3-0.2 = 0.0999...).
        return (
            decimal.Decimal(str(self.high)) - decimal.Decimal(str(self.low))
        ) < decimal.Decimal(str(self.step))

    def _contains(self, param_value_in_internal_repr: float) -> bool:
        value = param_value_in_internal_repr
        if self.step is None:
            return self.low <= value <= self.high
        k = (value - self.low) / self.step
        return self.low <= value <= self.high and abs(k - round(k)) < 1e-8

    def to_external_repr(self, param_value_in_internal_repr: float) -> float:
        return float(param_value_in_internal_repr)

    def to_internal_repr(self, param_value_in_external_repr: float) -> float:
        try:
            internal = float(param_value_in_external_repr)
        except (ValueError, TypeError) as e:
            raise ValueError(
                f"'{param_value_in_external_repr}' is not a valid type. "
                "float-castable value is expected."
            ) from e
        if math.isnan(internal):
            raise ValueError(f"`{param_value_in_external_repr}` is invalid value.")
        if self.log and internal <= 0.0:
            raise ValueError(
                f"`{param_value_in_external_repr}` is invalid value for the case log=True."
            )
        return internal


class IntDistribution(BaseDistribution):
    """An integer domain with optional log scale or step.

    Parity: reference distributions.py:310-454. ``log=True`` requires ``step == 1``
    and ``low >= 1``.
    """

    def __init__(self, low: int, high: int, log: bool = False, step: int = 1) -> None:
        if log and step != 1:
            raise ValueError("The parameter `step != 1` is not supported when `log` is true.")
        if low > high:
            raise ValueError(
                f"The `low` value must be smaller than or equal to the `high` value "
                f"(low={low}, high={high})."
            )
        if log and low < 1:
            raise ValueError(f"The `low` value must be equal to or greater than 1 (low={low}).")
        if step <= 0:
            raise ValueError(f"The `step` value must be non-zero positive value, but step={step}.")

        self.log = log
        self.low = int(low)
        self.step = int(step)
        if (high - low) % step != 0:
            adjusted = int(low + ((high - low) // step) * step)
            warnings.warn(
                f"The distribution is specified by [{low}, {high}] and step={step}, but the range "
                f"is not divisible by `step`. It will be replaced by [{low}, {adjusted}]."
            )
            self.high = adjusted
        else:
            self.high = int(high)

    def to_external_repr(self, param_value_in_internal_repr: float) -> int:
        return int(param_value_in_internal_repr)

    def to_internal_repr(self, param_value_in_external_repr: int) -> float:
        try:
            internal = float(param_value_in_external_repr)
        except (TypeError, ValueError) as e:
            raise ValueError(
                f"'{param_value_in_external_repr}' is not a valid type. "
                "float-castable value is expected."
            ) from e
        if math.isnan(internal):
            raise ValueError(f"`{param_value_in_external_repr}` is invalid value.")
        if self.log and internal <= 0.0:
            raise ValueError(
                f"`{param_value_in_external_repr}` is invalid value for the case log=True."
            )
        return internal

    def single(self) -> bool:
        if self.log:
            return self.low == self.high
        return self.high - self.low < self.step

    def _contains(self, param_value_in_internal_repr: float) -> bool:
        value = param_value_in_internal_repr
        return self.low <= value <= self.high and (value - self.low) % self.step == 0


class CategoricalDistribution(BaseDistribution):
    """A finite unordered set of choices; internal repr is the choice index.

    Parity: reference distributions.py:470-563 (NaN-aware index lookup :536-556).
    """

    def __init__(self, choices: Sequence[CategoricalChoiceType]) -> None:
        if len(choices) == 0:
            raise ValueError("The `choices` must contain one or more elements.")
        for choice in choices:
            if choice is not None and not isinstance(choice, (bool, int, float, str)):
                warnings.warn(
                    f"Choice {choice} is of type {type(choice).__name__}, which is not supported "
                    "by persistent storages. Use None, bool, int, float or str."
                )
        self.choices = tuple(choices)

    def to_external_repr(self, param_value_in_internal_repr: float) -> CategoricalChoiceType:
        return self.choices[int(param_value_in_internal_repr)]

    def to_internal_repr(self, param_value_in_external_repr: CategoricalChoiceType) -> float:
        try:
            return self.choices.index(param_value_in_external_repr)
        except ValueError:
            # `list.index` misses float('nan') because nan != nan; scan identity-or-isnan.
            v = param_value_in_external_repr
            if isinstance(v, float) and math.isnan(v):
                for i, c in enumerate(self.choices):
                    if isinstance(c, float) and math.isnan(c):
                        return i
            raise ValueError(f"'{v}' not in {self.choices}.") from None

    def single(self) -> bool:
        return len(self.choices) == 1

    def _contains(self, param_value_in_internal_repr: float) -> bool:
        index = int(param_value_in_internal_repr)
        return 0 <= index < len(self.choices)

    def __eq__(self, other: Any) -> bool:
        # Choice-wise comparison: tuple equality would miss NaN choices
        # (nan != nan), but two NaN choices ARE the same category.
        if not isinstance(other, BaseDistribution):
            return NotImplemented
        if type(self) is not type(other):
            return False
        if len(self.choices) != len(other.choices):
            return False
        return all(
            _categorical_choice_equal(a, b) for a, b in zip(self.choices, other.choices)
        )

    __hash__ = BaseDistribution.__hash__


# --------------------------------------------------------------------------------------
# JSON codec (RDB / journal compatibility format)
# --------------------------------------------------------------------------------------

# Legacy (pre-v3) class names that old storages may contain (reference
# distributions.py:201-456 keeps them as deprecated aliases; we only need decode).
_LEGACY_DECODERS = {
    "UniformDistribution": lambda a: FloatDistribution(a["low"], a["high"]),
    "LogUniformDistribution": lambda a: FloatDistribution(a["low"], a["high"], log=True),
    "DiscreteUniformDistribution": lambda a: FloatDistribution(a["low"], a["high"], step=a["q"]),
    "IntUniformDistribution": lambda a: IntDistribution(a["low"], a["high"], step=a.get("step", 1)),
    "IntLogUniformDistribution": lambda a: IntDistribution(a["low"], a["high"], log=True),
}


def json_to_distribution(json_str: str) -> BaseDistribution:
    parsed = json.loads(json_str)
    if "name" not in parsed:
        # Abbreviated format {"type": "float"|"int"|"categorical", ...}
        # (reference distributions.py:589-607 — written by some ecosystem
        # tools and older exports).
        kind = parsed.get("type")
        if kind == "categorical":
            return CategoricalDistribution(parsed["choices"])
        if kind in ("float", "int"):
            kwargs = {
                k: parsed[k] for k in ("low", "high", "log", "step") if k in parsed
            }
            if kind == "float":
                return FloatDistribution(**kwargs)
            return IntDistribution(**kwargs)
        raise ValueError(f"Invalid distribution JSON: {json_str}.")
    name = parsed["name"]
    attributes = parsed["attributes"]
    if name == _FLOAT_CLS:
        return FloatDistribution(**attributes)
    if name == _INT_CLS:
        return IntDistribution(**attributes)
    if name == _CATEGORICAL_CLS:
        return CategoricalDistribution(**attributes)
    if name in _LEGACY_DECODERS:
        return _LEGACY_DECODERS[name](attributes)
    raise ValueError(f"Unknown distribution class: {name}.")


def distribution_to_json(dist: BaseDistribution) -> str:
    if isinstance(dist, CategoricalDistribution):
        attributes: dict[str, Any] = {"choices": dist.choices}
    else:
        attributes = dist._asdict()
    return json.dumps({"name": type(dist).__name__, "attributes": attributes})


def _categorical_choice_equal(
    a: CategoricalChoiceType, b: CategoricalChoiceType
) -> bool:
    """Equality that treats two NaN choices as the same category."""
    both_nan = (
        isinstance(a, float)
        and isinstance(b, float)
        and math.isnan(a)
        and math.isnan(b)
    )
    return bool(a == b) or both_nan


def check_distribution_compatibility(
    dist_old: BaseDistribution, dist_new: BaseDistribution
) -> None:
    """Raise ValueError if the two distributions cannot share stored parameter values.

    Same-class requirement; log configuration must match for numerical
    distributions; categorical choices must match exactly
    (reference distributions.py:623-654).
    """
    if dist_old.__class__ is not dist_new.__class__:
        raise ValueError("Cannot set different distribution kind to the same parameter name.")
    if isinstance(dist_old, (FloatDistribution, IntDistribution)):
        assert isinstance(dist_new, (FloatDistribution, IntDistribution))
        if dist_old.log != dist_new.log:
            raise ValueError("Cannot set different log configuration to the same parameter name.")
    if isinstance(dist_old, CategoricalDistribution):
        assert isinstance(dist_new, CategoricalDistribution)
        if dist_old != dist_new:
            raise ValueError(
                CategoricalDistribution.__name__ + " does not support dynamic value space."
            )


class UniformDistribution(FloatDistribution):
    """Deprecated alias of ``FloatDistribution`` (pre-v3 API)."""

    def __init__(self, low: float, high: float) -> None:
        super().__init__(low=low, high=high)

    def _asdict(self) -> dict[str, Any]:
        d = dict(self.__dict__)
        d.pop("log", None)
        d.pop("step", None)
        return d


class LogUniformDistribution(FloatDistribution):
    """Deprecated alias of ``FloatDistribution(log=True)`` (pre-v3 API)."""

    def __init__(self, low: float, high: float) -> None:
        super().__init__(low=low, high=high, log=True)

    def _asdict(self) -> dict[str, Any]:
        d = dict(self.__dict__)
        d.pop("log", None)
        d.pop("step", None)
        return d


class DiscreteUniformDistribution(FloatDistribution):
    """Deprecated alias of ``FloatDistribution(step=q)`` (pre-v3 API)."""

    def __init__(self, low: float, high: float, q: float) -> None:
        super().__init__(low=low, high=high, step=q)

    @property
    def q(self) -> float:
        assert self.step is not None
        return self.step

    def _asdict(self) -> dict[str, Any]:
        d = dict(self.__dict__)
        d.pop("log", None)
        d["q"] = d.pop("step")
        return d


class IntUniformDistribution(IntDistribution):
    """Deprecated alias of ``IntDistribution`` (pre-v3 API)."""

    def __init__(self, low: int, high: int, step: int = 1) -> None:
        super().__init__(low=low, high=high, step=step)

    def _asdict(self) -> dict[str, Any]:
        d = dict(self.__dict__)
        d.pop("log", None)
        return d


class IntLogUniformDistribution(IntDistribution):
    """Deprecated alias of ``IntDistribution(log=True)`` (pre-v3 API)."""

    def __init__(self, low: int, high: int, step: int = 1) -> None:
        super().__init__(low=low, high=high, log=True)

    def _asdict(self) -> dict[str, Any]:
        d = dict(self.__dict__)
        d.pop("log", None)
        return d


def _convert_old_distribution_to_new_distribution(
    distribution: BaseDistribution,
    suppress_warning: bool = False,
) -> BaseDistribution:
    """Map a pre-v3 alias onto the current class (reference :709-790)."""
    new_distribution: BaseDistribution
    if isinstance(distribution, UniformDistribution):
        new_distribution = FloatDistribution(distribution.low, distribution.high)
    elif isinstance(distribution, LogUniformDistribution):
        new_distribution = FloatDistribution(
            distribution.low, distribution.high, log=True
        )
    elif isinstance(distribution, DiscreteUniformDistribution):
        new_distribution = FloatDistribution(
            distribution.low, distribution.high, step=distribution.q
        )
    elif isinstance(distribution, IntUniformDistribution):
        new_distribution = IntDistribution(
            distribution.low, distribution.high, step=distribution.step
        )
    elif isinstance(distribution, IntLogUniformDistribution):
        new_distribution = IntDistribution(
            distribution.low, distribution.high, log=True, step=distribution.step
        )
    else:
        new_distribution = distribution
    if new_distribution != distribution and not suppress_warning:
        warnings.warn(
            f"{distribution} is deprecated and internally converted to "
            f"{new_distribution}. See https://github.com/optuna/optuna/issues/2941.",
            FutureWarning,
        )
    return new_distribution


def _get_single_value(dist: BaseDistribution) -> Any:
    """External-repr value of a ``single()`` distribution (reference :691-703)."""
    assert dist.single()
    if isinstance(dist, CategoricalDistribution):
        return dist.choices[0]
    if isinstance(dist, (FloatDistribution, IntDistribution)):
        return dist.low
    raise AssertionError(f"unexpected distribution {dist}")


def _is_distribution_log(dist: BaseDistribution) -> bool:
    return bool(getattr(dist, "log", False))
