"""NSGA-II crossover operators over the transformed numerical space.

All six operators of the reference (``optuna/samplers/nsgaii/_crossovers/``:
uniform, BLX-α, SBX (_sbx.py:80-149), vSBX, SPX, UNDX) plus the orchestration
(`perform_crossover`, reference nsgaii/_crossover.py: binary-tournament parent
selection :123-154, retry-until-in-box :104-120, categorical uniform swap
:166-179). Numerical parameters go through ``_SearchSpaceTransform`` (log-space,
no one-hot); categoricals always use the inlined uniform swap.
"""
from __future__ import annotations

import abc
from typing import TYPE_CHECKING, Any, Callable, Sequence

import numpy as np

from optuna_amd._transform import _SearchSpaceTransform
from optuna_amd.distributions import (
    BaseDistribution,
    FloatDistribution,
    IntDistribution,
)


if TYPE_CHECKING:
    from optuna_amd.study import Study, StudyDirection
    from optuna_amd.trial import FrozenTrial

_NUMERICAL_DISTRIBUTIONS = (FloatDistribution, IntDistribution)


class BaseCrossover(abc.ABC):
    """Combine n_parents transformed parameter vectors into one child vector."""

    def __str__(self) -> str:
        return self.__class__.__name__

    @property
    @abc.abstractmethod
    def n_parents(self) -> int:
        raise NotImplementedError

    @abc.abstractmethod
    def crossover(
        self,
        parents_params: np.ndarray,
        rng: np.random.RandomState,
        study: "Study",
        search_space_bounds: np.ndarray,
    ) -> np.ndarray:
        raise NotImplementedError


class UniformCrossover(BaseCrossover):
    """Each gene picked from either parent with probability swapping_prob."""

    n_parents = 2

    def __init__(self, swapping_prob: float = 0.5) -> None:
        if not 0.0 <= swapping_prob <= 1.0:
            raise ValueError(
                "`swapping_prob` must be a float value within the range [0.0, 1.0]."
            )
        self._swapping_prob = swapping_prob

    def crossover(self, parents_params, rng, study, search_space_bounds):  # type: ignore[override]
        n_params = len(search_space_bounds)
        take_second = (rng.rand(n_params) >= self._swapping_prob).astype(int)
        return parents_params[take_second, np.arange(n_params)]


class BLXAlphaCrossover(BaseCrossover):
    """Uniform sample from the α-expanded hyper-rectangle of the two parents."""

    n_parents = 2

    def __init__(self, alpha: float = 0.5) -> None:
        self._alpha = alpha

    def crossover(self, parents_params, rng, study, search_space_bounds):  # type: ignore[override]
        lo = parents_params.min(axis=0)
        hi = parents_params.max(axis=0)
        spread = self._alpha * (hi - lo)
        low, high = lo - spread, hi + spread
        return low + (high - low) * rng.rand(len(search_space_bounds))


class SBXCrossover(BaseCrossover):
    """Bounded simulated binary crossover (Deb's β-spread construction)."""

    n_parents = 2

    def __init__(
        self,
        eta: float | None = None,
        uniform_crossover_prob: float = 0.5,
        use_child_gene_prob: float = 0.5,
    ) -> None:
        if eta is not None and eta < 0.0:
            raise ValueError("The value of `eta` must be greater than or equal to 0.0.")
        if not 0.0 <= uniform_crossover_prob <= 1.0:
            raise ValueError(
                "The value of `uniform_crossover_prob` must be in the range [0.0, 1.0]."
            )
        if not 0.0 < use_child_gene_prob <= 1.0:
            raise ValueError(
                "The value of `use_child_gene_prob` must be in the range (0.0, 1.0]."
            )
        self._eta = eta
        self._uniform_crossover_prob = uniform_crossover_prob
        self._use_child_gene_prob = use_child_gene_prob

    def crossover(self, parents_params, rng, study, search_space_bounds):  # type: ignore[override]
        xls = search_space_bounds[..., 0]
        xus = search_space_bounds[..., 1]
        x_min = np.min(parents_params, axis=0)
        x_max = np.max(parents_params, axis=0)
        eta = self._eta if self._eta is not None else (
            20.0 if study._is_multi_objective() else 2.0
        )

        x_diff = np.clip(x_max - x_min, 1e-10, None)
        # Bounded-SBX spread factors toward each box edge (Deb Eq. 3-5).
        beta1 = 1 + 2 * (x_min - xls) / x_diff
        beta2 = 1 + 2 * (xus - x_max) / x_diff
        alpha1 = 2 - np.power(beta1, -(eta + 1))
        alpha2 = 2 - np.power(beta2, -(eta + 1))
        us = rng.rand(len(search_space_bounds))

        betaq1 = np.power(us * alpha1, 1 / (eta + 1))
        over1 = us > 1 / alpha1
        betaq1[over1] = np.power(1 / (2 - us * alpha1), 1 / (eta + 1))[over1]
        betaq2 = np.power(us * alpha2, 1 / (eta + 1))
        over2 = us > 1 / alpha2
        betaq2[over2] = np.power(1 / (2 - us * alpha2), 1 / (eta + 1))[over2]

        c1 = 0.5 * ((x_min + x_max) - betaq1 * x_diff)
        c2 = 0.5 * ((x_min + x_max) + betaq2 * x_diff)
        return _collapse_two_children(
            c1,
            c2,
            parents_params[0],
            parents_params[1],
            rng,
            self._use_child_gene_prob,
            self._uniform_crossover_prob,
        )


class VSBXCrossover(BaseCrossover):
    """vSBX: SBX variant without the shared-midpoint restriction."""

    n_parents = 2

    def __init__(
        self,
        eta: float | None = None,
        uniform_crossover_prob: float = 0.5,
        use_child_gene_prob: float = 0.5,
    ) -> None:
        if eta is not None and eta < 0.0:
            raise ValueError("The value of `eta` must be greater than or equal to 0.0.")
        if not 0.0 <= uniform_crossover_prob <= 1.0:
            raise ValueError(
                "The value of `uniform_crossover_prob` must be in the range [0.0, 1.0]."
            )
        if not 0.0 < use_child_gene_prob <= 1.0:
            raise ValueError(
                "The value of `use_child_gene_prob` must be in the range (0.0, 1.0]."
            )
        self._eta = eta
        self._uniform_crossover_prob = uniform_crossover_prob
        self._use_child_gene_prob = use_child_gene_prob

    def crossover(self, parents_params, rng, study, search_space_bounds):  # type: ignore[override]
        eta = self._eta if self._eta is not None else (
            20.0 if study._is_multi_objective() else 2.0
        )
        eps = 1e-10
        us = rng.rand(len(search_space_bounds))
        beta_1 = np.power(1 / np.maximum(2 * us, eps), 1 / (eta + 1))
        beta_2 = np.power(1 / np.maximum(2 * (1 - us), eps), 1 / (eta + 1))

        p0, p1 = parents_params[0], parents_params[1]
        if rng.rand() <= 0.5:
            c1 = 0.5 * ((1 + beta_1) * p0 + (1 - beta_2) * p1)
        else:
            c1 = 0.5 * ((1 - beta_1) * p0 + (1 + beta_2) * p1)
        if rng.rand() <= 0.5:
            c2 = 0.5 * ((3 - beta_1) * p0 - (1 - beta_2) * p1)
        else:
            c2 = 0.5 * (-(1 - beta_1) * p0 + (3 - beta_2) * p1)
        return _collapse_two_children(
            c1, c2, p0, p1, rng, self._use_child_gene_prob, self._uniform_crossover_prob
        )


class SPXCrossover(BaseCrossover):
    """Simplex crossover: uniform sample in the ε-expanded parent simplex."""

    n_parents = 3

    def __init__(self, epsilon: float | None = None) -> None:
        self._epsilon = epsilon

    def crossover(self, parents_params, rng, study, search_space_bounds):  # type: ignore[override]
        n = self.n_parents - 1
        center = np.mean(parents_params, axis=0)
        rs = np.power(rng.rand(n), 1 / (np.arange(n) + 1))
        epsilon = (
            np.sqrt(len(search_space_bounds) + 2) if self._epsilon is None else self._epsilon
        )
        expanded = [center + epsilon * (p - center) for p in parents_params]
        ck = np.zeros_like(center)
        for k in range(1, self.n_parents):
            ck = rs[k - 1] * (expanded[k - 1] - expanded[k] + ck)
        return expanded[-1] + ck


class UNDXCrossover(BaseCrossover):
    """Unimodal normal distribution crossover around the parents' midline."""

    n_parents = 3

    def __init__(self, sigma_xi: float = 0.5, sigma_eta: float | None = None) -> None:
        self._sigma_xi = sigma_xi
        self._sigma_eta = sigma_eta

    @staticmethod
    def _unit_x1_to_x2(parents_params: np.ndarray) -> np.ndarray:
        v = parents_params[1] - parents_params[0]
        return v / np.clip(np.linalg.norm(v, ord=2), 1e-10, None)

    def _distance_to_primary_search_line(self, parents_params: np.ndarray) -> float:
        e_12 = self._unit_x1_to_x2(parents_params)
        v_13 = parents_params[2] - parents_params[0]
        orth = v_13 - np.dot(v_13, e_12) * e_12
        return float(np.linalg.norm(orth, ord=2))

    def _orthonormal_basis(self, parents_params: np.ndarray, n: int) -> np.ndarray:
        e_12 = self._unit_x1_to_x2(parents_params)
        basis = np.identity(n)
        if np.count_nonzero(e_12) != 0:
            basis[0] = e_12
        Q, _ = np.linalg.qr(basis.T)
        return Q.T[1:]

    def crossover(self, parents_params, rng, study, search_space_bounds):  # type: ignore[override]
        n = len(search_space_bounds)
        midpoint = (parents_params[0] + parents_params[1]) / 2
        d = parents_params[0] - parents_params[1]
        sigma_eta = self._sigma_eta if self._sigma_eta is not None else 0.35 / np.sqrt(n)
        etas = rng.normal(0, sigma_eta**2, size=n)
        xi = rng.normal(0, self._sigma_xi**2)
        child = midpoint + xi * d
        if n > 1:
            D = self._distance_to_primary_search_line(parents_params)
            basis = self._orthonormal_basis(parents_params, n)
            lateral = np.zeros(n)
            for i in range(n - 1):
                lateral += etas[i] * basis[i]
            child = child + D * lateral
        return child


def _collapse_two_children(
    c1: np.ndarray,
    c2: np.ndarray,
    x1: np.ndarray,
    x2: np.ndarray,
    rng: np.random.RandomState,
    use_child_gene_prob: float,
    uniform_crossover_prob: float,
) -> np.ndarray:
    """Per-gene coin flips fold the two SBX children into one returned child."""
    out1, out2 = [], []
    for c1_i, c2_i, x1_i, x2_i in zip(c1, c2, x1, x2):
        if rng.rand() < use_child_gene_prob:
            a, b = (c1_i, c2_i) if rng.rand() >= uniform_crossover_prob else (c2_i, c1_i)
        else:
            a, b = (x1_i, x2_i) if rng.rand() >= uniform_crossover_prob else (x2_i, x1_i)
        out1.append(a)
        out2.append(b)
    return np.array(out1 if rng.rand() < 0.5 else out2)


# ----------------------------------------------------------------------------------
# Orchestration
# ----------------------------------------------------------------------------------


def _inlined_categorical_uniform_crossover(
    parent_params: np.ndarray,
    rng: np.random.RandomState,
    swapping_prob: float,
    search_space: dict[str, BaseDistribution],
) -> np.ndarray:
    """Per-gene uniform pick between two parents' untransformed categorical rows
    (categoricals never go through the numerical transform)."""
    n = len(search_space)
    pick = (rng.rand(n) >= swapping_prob).astype(int)
    return parent_params[pick, np.arange(n)]


def perform_crossover(
    crossover: BaseCrossover,
    study: "Study",
    parent_population: Sequence["FrozenTrial"],
    search_space: dict[str, BaseDistribution],
    rng: np.random.RandomState,
    swapping_prob: float,
    dominates: Callable[["FrozenTrial", "FrozenTrial", Sequence["StudyDirection"]], bool],
) -> dict[str, Any]:
    numerical_space: dict[str, BaseDistribution] = {}
    categorical_space: dict[str, BaseDistribution] = {}
    for key, value in search_space.items():
        (numerical_space if isinstance(value, _NUMERICAL_DISTRIBUTIONS) else categorical_space)[
            key
        ] = value

    numerical_transform = (
        _SearchSpaceTransform(numerical_space) if numerical_space else None
    )

    while True:  # retry until the child lies inside the box
        parents = _select_parents(crossover, study, parent_population, rng, dominates)
        child_params = _combine(
            parents,
            crossover,
            study,
            rng,
            swapping_prob,
            categorical_space,
            numerical_space,
            numerical_transform,
        )
        if _is_contained(child_params, search_space):
            return child_params


def _combine(
    parents: list["FrozenTrial"],
    crossover: BaseCrossover,
    study: "Study",
    rng: np.random.RandomState,
    swapping_prob: float,
    categorical_space: dict[str, BaseDistribution],
    numerical_space: dict[str, BaseDistribution],
    numerical_transform: _SearchSpaceTransform | None,
) -> dict[str, Any]:
    child_params: dict[str, Any] = {}

    if categorical_space:
        # Categorical genes: uniform swap between the first and last parent.
        first, last = parents[0], parents[-1]
        take_last = rng.rand(len(categorical_space)) >= swapping_prob
        for i, name in enumerate(categorical_space):
            src = last if take_last[i] else first
            child_params[name] = src.params[name]

    if numerical_transform is None:
        return child_params

    parents_numerical = np.stack(
        [
            numerical_transform.transform({k: p.params[k] for k in numerical_space})
            for p in parents
        ]
    )
    child_array = crossover.crossover(
        parents_numerical, rng, study, numerical_transform.bounds
    )
    child_params.update(numerical_transform.untransform(child_array))
    return child_params


def _select_parents(
    crossover: BaseCrossover,
    study: "Study",
    parent_population: Sequence["FrozenTrial"],
    rng: np.random.RandomState,
    dominates: Callable[..., bool],
) -> list["FrozenTrial"]:
    parents: list["FrozenTrial"] = []
    for _ in range(crossover.n_parents):
        pool = [t for t in parent_population if t not in parents]
        parents.append(_binary_tournament(study, pool, rng, dominates))
    return parents


def _binary_tournament(
    study: "Study",
    pool: Sequence["FrozenTrial"],
    rng: np.random.RandomState,
    dominates: Callable[..., bool],
) -> "FrozenTrial":
    candidate0 = pool[rng.choice(len(pool))]
    candidate1 = pool[rng.choice(len(pool))]
    return candidate0 if dominates(candidate0, candidate1, study.directions) else candidate1


def _is_contained(params: dict[str, Any], search_space: dict[str, BaseDistribution]) -> bool:
    for name, value in params.items():
        dist = search_space[name]
        if not dist._contains(dist.to_internal_repr(value)):
            return False
    return True
