"""NSGA-III elite selection: Das-Dennis reference points + niching.

Parity: reference ``optuna/samplers/_nsgaiii/_elite_population_selection_strategy.py``
(:27-80 selection; _generate_default_reference_point; ASF-based normalization;
reference-line association; sparsity-driven niche preservation).
"""
from __future__ import annotations

from collections import defaultdict
from itertools import combinations_with_replacement
from typing import TYPE_CHECKING, Callable, Sequence

import numpy as np

from optuna_amd.samplers._lazy_random_state import LazyRandomState
from optuna_amd.samplers.nsgaii._strategies import (
    _is_constrained_optimization,
    _rank_population,
    _validate_constraints,
)
from optuna_amd.trial import FrozenTrial


if TYPE_CHECKING:
    from optuna_amd.study import Study

_INF_CLIP_COEF = 3


def _generate_default_reference_point(
    n_objectives: int, dividing_parameter: int = 3
) -> np.ndarray:
    """Das-Dennis simplex lattice points (each row sums to dividing_parameter)."""
    indices = np.array(
        list(combinations_with_replacement(range(n_objectives), dividing_parameter))
    )
    points = np.zeros((len(indices), n_objectives), dtype=float)
    rows = np.repeat(np.arange(len(indices)), dividing_parameter)
    np.add.at(points, (rows, indices.flatten()), 1.0)
    return points


def _filter_inf(population: list[FrozenTrial]) -> np.ndarray:
    """Clip infinite objectives into a margin around the finite range."""
    mat = np.asarray([t.values for t in population])
    finite = np.where(np.isfinite(mat), mat, np.nan)
    hi = np.nanmax(finite, axis=0)
    lo = np.nanmin(finite, axis=0)
    margins = _INF_CLIP_COEF * (hi - lo)
    return np.clip(mat, lo - margins, hi + margins)


def _normalize_objective_values(objective_matrix: np.ndarray) -> np.ndarray:
    """Subtract the ideal point; rescale by hyperplane intercepts through the
    extreme points (achievement-scalarizing-function selection)."""
    n_objectives = objective_matrix.shape[1]
    objective_matrix = objective_matrix - np.min(objective_matrix, axis=0)
    weights = np.eye(n_objectives)
    weights[weights == 0] = 1e6
    asf_value = np.max(np.einsum("nm,dm->dnm", objective_matrix, weights), axis=2)
    extreme_points = objective_matrix[np.argmin(asf_value, axis=1), :]
    if np.all(np.isfinite(extreme_points)) and np.linalg.matrix_rank(
        extreme_points
    ) == len(extreme_points):
        intercepts_inv = np.linalg.solve(extreme_points, np.ones(n_objectives))
    else:
        intercepts = np.max(objective_matrix, axis=0)
        intercepts_inv = 1 / np.where(intercepts == 0, 1, intercepts)
    objective_matrix = objective_matrix * np.where(
        np.isfinite(intercepts_inv), intercepts_inv, 1
    )
    return objective_matrix


def _associate(
    objective_matrix: np.ndarray, reference_points: np.ndarray
) -> tuple[np.ndarray, np.ndarray]:
    """Closest reference line per individual + perpendicular distance to it."""
    ref_norm_sq = np.linalg.norm(reference_points, axis=1) ** 2
    projections = np.einsum(
        "ni,pi,p,pm->npm",
        objective_matrix,
        reference_points,
        1 / ref_norm_sq,
        reference_points,
    )
    dist = np.linalg.norm(objective_matrix[:, np.newaxis, :] - projections, axis=2)
    return np.argmin(dist, axis=1), np.min(dist, axis=1)


def _preserve_niche_individuals(
    target_population_size: int,
    elite_population_num: int,
    population: list[FrozenTrial],
    closest_reference_points: np.ndarray,
    distance_reference_points: np.ndarray,
    rng: np.random.RandomState,
) -> list[FrozenTrial]:
    """Fill remaining slots from the borderline front, preferring the reference
    points with the fewest already-selected neighbors."""
    if len(population) < target_population_size:
        raise ValueError(
            "The population size must be greater than or equal to the target population size."
        )

    borderline_by_ref: defaultdict[int, list[tuple[float, int]]] = defaultdict(list)
    for i, ref_idx in enumerate(closest_reference_points[elite_population_num:]):
        pop_idx = i + elite_population_num
        borderline_by_ref[ref_idx].append((distance_reference_points[pop_idx], i))

    elite_count_by_ref: dict[int, int] = defaultdict(int)
    for ref_idx in closest_reference_points[:elite_population_num]:
        elite_count_by_ref[ref_idx] += 1

    refs_by_elite_count: defaultdict[int, list[int]] = defaultdict(list)
    for ref_idx in borderline_by_ref:
        refs_by_elite_count[elite_count_by_ref[ref_idx]].append(ref_idx)

    count = -1
    additional: list[FrozenTrial] = []
    is_shuffled: defaultdict[int, bool] = defaultdict(bool)
    while len(additional) < target_population_size:
        if len(refs_by_elite_count[count]) == 0:
            count += 1
            rng.shuffle(refs_by_elite_count[count])
            continue
        ref_idx = refs_by_elite_count[count].pop()
        if count > 0 and not is_shuffled[ref_idx]:
            rng.shuffle(borderline_by_ref[ref_idx])
            is_shuffled[ref_idx] = True
        elif count == 0:
            # Empty niche: take the closest individual first (sort desc, pop last).
            borderline_by_ref[ref_idx].sort(reverse=True)
        _, selected = borderline_by_ref[ref_idx].pop()
        additional.append(population[selected])
        if borderline_by_ref[ref_idx]:
            refs_by_elite_count[count + 1].append(ref_idx)
    return additional


class NSGAIIIElitePopulationSelectionStrategy:
    def __init__(
        self,
        *,
        population_size: int,
        constraints_func: Callable[[FrozenTrial], Sequence[float]] | None = None,
        reference_points: np.ndarray | None = None,
        dividing_parameter: int = 3,
        rng: LazyRandomState,
    ) -> None:
        if population_size < 2:
            raise ValueError("`population_size` must be greater than or equal to 2.")
        self._population_size = population_size
        self._constraints_func = constraints_func
        self._reference_points = reference_points
        self._dividing_parameter = dividing_parameter
        self._rng = rng

    def __call__(self, study: "Study", population: list[FrozenTrial]) -> list[FrozenTrial]:
        is_constrained = _is_constrained_optimization(population)
        _validate_constraints(population, is_constrained=is_constrained)
        per_rank = _rank_population(population, study.directions, is_constrained=is_constrained)
        elite: list[FrozenTrial] = []
        for front in per_rank:
            if len(elite) + len(front) < self._population_size:
                elite.extend(front)
                continue
            n_objectives = len(study.directions)
            if self._reference_points is None:
                self._reference_points = _generate_default_reference_point(
                    n_objectives, self._dividing_parameter
                )
            elif np.shape(self._reference_points)[1] != n_objectives:
                raise ValueError(
                    "The dimension of reference points vectors must be the same as the "
                    "number of objectives of the study."
                )
            objective_matrix = _normalize_objective_values(_filter_inf(elite + front))
            closest, distances = _associate(objective_matrix, self._reference_points)
            target = self._population_size - len(elite)
            elite.extend(
                _preserve_niche_individuals(
                    target, len(elite), front, closest, distances, self._rng.rng
                )
            )
            break
        return elite
