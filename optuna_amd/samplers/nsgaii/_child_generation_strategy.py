"""NSGA-II child generation: crossover -> mutation -> gene dropout.

Parity: reference ``optuna/samplers/nsgaii/_child_generation_strategy.py``.
``perform_crossover``/``perform_mutation`` are module globals so test doubles
patched on this module take effect.
"""
from __future__ import annotations

from typing import TYPE_CHECKING, Any, Callable, Sequence

from optuna_amd.samplers._lazy_random_state import LazyRandomState
from optuna_amd.samplers.nsgaii._crossovers import BaseCrossover, perform_crossover
from optuna_amd.samplers.nsgaii._mutations import BaseMutation, perform_mutation
from optuna_amd.samplers.nsgaii._constraints_evaluation import (
    _constrained_dominates,
    _is_constrained_optimization,
)
from optuna_amd.study._multi_objective import _dominates
from optuna_amd.trial import FrozenTrial


if TYPE_CHECKING:
    from optuna_amd.distributions import BaseDistribution
    from optuna_amd.study import Study


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
            child_params = {name: parent.params[name] for name in search_space.keys()}

        n_params = len(child_params)
        mutation_prob = (
            self._mutation_prob
            if self._mutation_prob is not None
            else 1.0 / max(1.0, n_params)
        )
        params = {}
        for name in child_params.keys():
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
