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
        # (reference distributions.py:661-675; a quantize to the step's decimal
        # places would corrupt non-decimal lows). float() rounds the exact grid
        # point, so iterate to a fixed point — otherwise a JSON round-trip of
        # the stored (rounded) high can renormalize to a smaller grid point.
        k = d_r // d_step
        adjusted = float(k * d_step + d_low)
        while (
            k > 0
            and (decimal.Decimal(str(adjusted)) - d_low) % d_step != 0
        ):
            k -= 1
            adjusted = float(k * d_step + d_low)
        if k == 0:
            adjusted = float(low)
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
        return self.high - self.low < self.step

    def _contains(self, param_value_in_internal_repr: float) -> bool:
        value = param_value_in_internal_repr
        if self.step is None:
            return self.low <= value <= self.high
        k = (value - self.low) / self.step
        return self.low <= value <= self.high and abs(k - round(k)) < 1e-8

    def to_external_repr(self, param_value_in_internal_repr: float) -> float:
        return float(param_value_in_internal_repr)


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
            return float(param_value_in_external_repr)
        except (TypeError, ValueError) as e:
            raise ValueError(
                f"'{param_value_in_external_repr}' is not a valid value for IntDistribution."
            ) from e

    def single(self) -> bool:
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


def check_distribution_compatibility(
    dist_old: BaseDistribution, dist_new: BaseDistribution
) -> None:
    """Raise ValueError if the two distributions cannot share stored parameter values.

    Same-class requirement; categorical choices must match exactly
    (reference distributions.py:623-654).
    """
    if dist_old.__class__ is not dist_new.__class__:
        raise ValueError(
            f"Cannot set different distribution kind to the same parameter name: "
            f"{dist_old} != {dist_new}."
        )
    if isinstance(dist_old, CategoricalDistribution):
        assert isinstance(dist_new, CategoricalDistribution)
        if dist_old.choices != dist_new.choices:
            raise ValueError(
                CategoricalDistribution.__name__ + " does not support dynamic value space."
            )


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
