"""NSGA-II strategies: elite selection (rank + crowding), child generation,
after-trial constraints, and constrained domination.

Parity: reference ``optuna/samplers/nsgaii/``
(_elite_population_selection_strategy.py:23-118, _child_generation_strategy.py
:88-123, _after_trial_strategy.py, _constraints_evaluation.py:18-85).
"""
from __future__ import annotations

from collections import defaultdict
from typing import TYPE_CHECKING, Any, Callable, Sequence

import numpy as np

from optuna_amd.samplers._base import _process_constraints_after_trial
from optuna_amd.samplers._lazy_random_state import LazyRandomState
from optuna_amd.samplers.nsgaii._crossovers import BaseCrossover, perform_crossover
from optuna_amd.samplers.nsgaii._mutations import BaseMutation, perform_mutation  # noqa: F401 — re-export
from optuna_amd.study._multi_objective import _dominates, _fast_non_domination_rank
from optuna_amd.study._study_direction import StudyDirection
from optuna_amd.trial import FrozenTrial, TrialState


if TYPE_CHECKING:
    from optuna_amd.distributions import BaseDistribution
    from optuna_amd.study import Study


# ---- constrained domination ----------------------------------------------------------


def _is_constrained_optimization(population: Sequence[FrozenTrial]) -> bool:
    return any("constraints" in t.system_attrs for t in population)


def _constrained_dominates(
    trial0: FrozenTrial, trial1: FrozenTrial, directions: Sequence[StudyDirection]
) -> bool:
    """Feasible ≻ infeasible; both infeasible → smaller violation sum; both
    feasible → ordinary Pareto dominance."""
    constraints0 = trial0.constraints
    constraints1 = trial1.constraints
    if trial0.state != TrialState.COMPLETE:
        return False
    if trial1.state != TrialState.COMPLETE:
        return True
    feasible0 = all(v <= 0 for v in constraints0.values())
    feasible1 = all(v <= 0 for v in constraints1.values())
    if feasible0 and feasible1:
        return _dominates(trial0, trial1, directions)
    if feasible0:
        return True
    if feasible1:
        return False
    violation0 = sum(v for v in constraints0.values() if v > 0)
    violation1 = sum(v for v in constraints1.values() if v > 0)
    return violation0 < violation1


def _evaluate_penalty(population: Sequence[FrozenTrial]) -> np.ndarray:
    return np.array(
        [sum(v for v in t.constraints.values() if v > 0) for t in population]
    )


def _validate_constraints(
    population: Sequence[FrozenTrial], *, is_constrained: bool = False
) -> None:
    if not is_constrained:
        return
    for t in population:
        if np.any(np.isnan(list(t.constraints.values()))):
            raise ValueError("NaN is not acceptable as constraint value.")


# ---- elite selection -----------------------------------------------------------------


def _rank_population(
    population: list[FrozenTrial],
    directions: Sequence[StudyDirection],
    *,
    is_constrained: bool = False,
) -> list[list[FrozenTrial]]:
    if len(population) == 0:
        return []
    loss = np.array([t.values for t in population], dtype=np.float64)
    loss *= np.array([-1.0 if d == StudyDirection.MAXIMIZE else 1.0 for d in directions])
    penalty = _evaluate_penalty(population) if is_constrained else None
    ranks = _fast_non_domination_rank(loss, penalty=penalty)
    per_rank: list[list[FrozenTrial]] = [[] for _ in range(int(max(ranks)) + 1)]
    for trial, rank in zip(population, ranks):
        if rank == -1:
            continue
        per_rank[rank].append(trial)
    return per_rank


def _calc_crowding_distance(population: list[FrozenTrial]) -> defaultdict[int, float]:
    """Per-objective neighbor-gap sum, normalized by the finite value range."""
    distances: defaultdict[int, float] = defaultdict(float)
    if len(population) == 0:
        return distances
    for i in range(len(population[0].values)):  # type: ignore[arg-type]
        population.sort(key=lambda t: t.values[i])  # type: ignore[index]
        if population[0].values[i] == population[-1].values[i]:  # type: ignore[index]
            continue
        vs = [-float("inf")] + [t.values[i] for t in population] + [float("inf")]  # type: ignore[index]
        v_min = next(x for x in vs if x != -float("inf"))
        v_max = next(x for x in reversed(vs) if x != float("inf"))
        width = v_max - v_min
        if width <= 0:
            width = 1.0
        for j in range(len(population)):
            gap = 0.0 if vs[j] == vs[j + 2] else vs[j + 2] - vs[j]
            distances[population[j].number] += gap / width
    return distances


def _crowding_distance_sort(population: list[FrozenTrial]) -> None:
    distances = _calc_crowding_distance(population)
    population.sort(key=lambda t: distances[t.number], reverse=True)


class NSGAIIElitePopulationSelectionStrategy:
    def __init__(
        self,
        *,
        population_size: int,
        constraints_func: Callable[[FrozenTrial], Sequence[float]] | None = None,
    ) -> None:
        if population_size < 2:
            raise ValueError("`population_size` must be greater than or equal to 2.")
        self._population_size = population_size
        self._constraints_func = constraints_func

    def __call__(self, study: "Study", population: list[FrozenTrial]) -> list[FrozenTrial]:
        is_constrained = _is_constrained_optimization(population)
        _validate_constraints(population, is_constrained=is_constrained)
        per_rank = _rank_population(population, study.directions, is_constrained=is_constrained)
        elite: list[FrozenTrial] = []
        for front in per_rank:
            if len(elite) + len(front) < self._population_size:
                elite.extend(front)
            else:
                n = self._population_size - len(elite)
                _crowding_distance_sort(front)
                elite.extend(front[:n])
                break
        return elite


# ---- child generation ----------------------------------------------------------------


# The child-generation and after-trial strategies live in their own modules
# (reference layout + patchable call targets); re-exported here for callers
# of the earlier single-module layout.
from optuna_amd.samplers.nsgaii._after_trial_strategy import (  # noqa: E402,F401
    NSGAIIAfterTrialStrategy,
)
from optuna_amd.samplers.nsgaii._child_generation_strategy import (  # noqa: E402,F401
    NSGAIIChildGenerationStrategy,
)
