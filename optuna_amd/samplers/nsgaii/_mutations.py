"""NSGA-II mutation operators.

Parity: reference ``optuna/samplers/nsgaii/_mutations/`` (BaseMutation,
PolynomialMutation following the original NSGA-II C implementation δq branches,
_polynomial.py:45-72) and ``_mutation.py`` (perform_mutation: numerical-only,
transform round trip + clip).
"""
from __future__ import annotations

import abc
from typing import TYPE_CHECKING, Any

import numpy as np

from optuna_amd._transform import _SearchSpaceTransform
from optuna_amd.distributions import (
    BaseDistribution,
    FloatDistribution,
    IntDistribution,
)


if TYPE_CHECKING:
    from optuna_amd.study import Study

_NUMERICAL_DISTRIBUTIONS = (FloatDistribution, IntDistribution)


class BaseMutation(abc.ABC):
    def __str__(self) -> str:
        return self.__class__.__name__

    @abc.abstractmethod
    def mutation(
        self,
        param: float,
        rng: np.random.RandomState,
        study: "Study",
        search_space_bounds: np.ndarray,
    ) -> float:
        raise NotImplementedError


class PolynomialMutation(BaseMutation):
    """Polynomial probability-distribution mutation (Deb & Agrawal)."""

    def __init__(self, eta: float = 20.0) -> None:
        if eta < 0:
            raise ValueError("`eta` must be a non-negative float value.")
        self._eta = eta

    def mutation(
        self,
        param: float,
        rng: np.random.RandomState,
        study: "Study",
        search_space_bounds: np.ndarray,
    ) -> float:
        lb, ub = search_space_bounds
        width = ub - lb
        if width <= 0.0:
            return param
        u = rng.rand()
        power = 1.0 / (self._eta + 1.0)
        if u <= 0.5:
            frac = 1.0 - (param - lb) / width
            value = 2.0 * u + (1.0 - 2.0 * u) * frac ** (self._eta + 1.0)
            delta_q = value**power - 1.0
        else:
            frac = 1.0 - (ub - param) / width
            value = 2.0 * (1.0 - u) + 2.0 * (u - 0.5) * frac ** (self._eta + 1.0)
            delta_q = 1.0 - value**power
        return param + delta_q * width


def perform_mutation(
    mutation: BaseMutation,
    rng: np.random.RandomState,
    study: "Study",
    distribution: BaseDistribution,
    value: Any,
) -> Any | None:
    """Mutate one numerical gene (None for categoricals → caller resamples)."""
    if not isinstance(distribution, _NUMERICAL_DISTRIBUTIONS):
        return None
    transform = _SearchSpaceTransform({"": distribution})
    trans_value = transform.transform({"": value})
    mutated = mutation.mutation(trans_value.item(), rng, study, transform.bounds[0])
    mutated = float(np.clip(mutated, transform.bounds[0, 0], transform.bounds[0, 1]))
    return transform.untransform(np.array([mutated]))[""]
