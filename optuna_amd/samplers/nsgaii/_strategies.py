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
from optuna_amd.samplers.nsgaii._mutations import BaseMutation, perform_mutation
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


class NSGAIIChildGenerationStrategy:
    def __init__(
        self,
        *,
        mutation: BaseMutation | None = None,
        mutation_prob: float | None = None,
        crossover: BaseCrossover,
        crossover_prob: float,
        swapping_prob: float,
        constraints_func: Callable[[FrozenTrial], Sequence[float]] | None = None,
        rng: LazyRandomState,
    ) -> None:
        if not (mutation_prob is None or 0.0 <= mutation_prob <= 1.0):
            raise ValueError(
                "`mutation_prob` must be None or a float value within the range [0.0, 1.0]."
            )
        if not 0.0 <= crossover_prob <= 1.0:
            raise ValueError("`crossover_prob` must be a float value within the range [0.0, 1.0].")
        if not 0.0 <= swapping_prob <= 1.0:
            raise ValueError("`swapping_prob` must be a float value within the range [0.0, 1.0].")
        if mutation is not None and not isinstance(mutation, BaseMutation):
            raise ValueError(f"'{mutation}' is not a valid mutation.")
        if not isinstance(crossover, BaseCrossover):
            raise ValueError(f"'{crossover}' is not a valid crossover.")
        self._mutation = mutation
        self._mutation_prob = mutation_prob
        self._crossover = crossover
        self._crossover_prob = crossover_prob
        self._swapping_prob = swapping_prob
        self._constraints_func = constraints_func
        self._rng = rng

    def __call__(
        self,
        study: "Study",
        search_space: dict[str, "BaseDistribution"],
        parent_population: list[FrozenTrial],
    ) -> dict[str, Any]:
        dominates = (
            _constrained_dominates
            if _is_constrained_optimization(parent_population)
            else _dominates
        )
        if self._rng.rng.rand() < self._crossover_prob:
            child_params = perform_crossover(
                self._crossover,
                study,
                parent_population,
                search_space,
                self._rng.rng,
                self._swapping_prob,
                dominates,
            )
        else:
            parent = parent_population[self._rng.rng.choice(len(parent_population))]
            child_params = {name: parent.params[name] for name in search_space}

        n_params = len(child_params)
        mutation_prob = (
            self._mutation_prob
            if self._mutation_prob is not None
            else 1.0 / max(1.0, n_params)
        )
        params = {}
        for name in child_params:
            if self._rng.rng.rand() >= mutation_prob:
                params[name] = child_params[name]
            elif self._mutation is not None:
                mutated = perform_mutation(
                    self._mutation, self._rng.rng, study, search_space[name], child_params[name]
                )
                if mutated is not None:
                    params[name] = mutated
            # else: drop the gene → Trial._suggest resamples it independently.
        return params


class NSGAIIAfterTrialStrategy:
    def __init__(
        self, *, constraints_func: Callable[[FrozenTrial], Sequence[float]] | None = None
    ) -> None:
        self._constraints_func = constraints_func

    def __call__(
        self,
        study: "Study",
        trial: FrozenTrial,
        state: TrialState,
        values: Sequence[float] | None = None,
    ) -> None:
        if self._constraints_func is not None:
            _process_constraints_after_trial(self._constraints_func, study, trial, state)
