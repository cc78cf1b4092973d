"""NSGAIIISampler — NSGA-II with reference-point niching for many objectives.

Parity: reference ``optuna/samplers/_nsgaiii/_sampler.py`` (NSGAIIISampler :33).
Shares the GA base, crossovers, mutations and child-generation strategy with
NSGA-II; only elite selection differs.
"""
from __future__ import annotations

from typing import TYPE_CHECKING, Any, Callable, Sequence

import numpy as np

from optuna_amd.distributions import BaseDistribution
from optuna_amd.samplers._ga import BaseGASampler
from optuna_amd.samplers._lazy_random_state import LazyRandomState
from optuna_amd.samplers._nsgaiii._elite_selection import (
    NSGAIIIElitePopulationSelectionStrategy,
)
from optuna_amd.samplers._random import RandomSampler
from optuna_amd.samplers.nsgaii._crossovers import BaseCrossover, UniformCrossover
from optuna_amd.samplers.nsgaii._mutations import BaseMutation
from optuna_amd.samplers.nsgaii._strategies import (
    NSGAIIAfterTrialStrategy,
    NSGAIIChildGenerationStrategy,
)
from optuna_amd.search_space import IntersectionSearchSpace
from optuna_amd.trial import FrozenTrial, TrialState


if TYPE_CHECKING:
    from optuna_amd.study import Study


class NSGAIIISampler(BaseGASampler):
    def __init__(
        self,
        *,
        population_size: int = 50,
        mutation: BaseMutation | None = None,
        mutation_prob: float | None = None,
        crossover: BaseCrossover | None = None,
        crossover_prob: float = 0.9,
        swapping_prob: float = 0.5,
        seed: int | None = None,
        constraints_func: Callable[[FrozenTrial], Sequence[float]] | None = None,
        reference_points: np.ndarray | None = None,
        dividing_parameter: int = 3,
        elite_population_selection_strategy: (
            Callable[["Study", list[FrozenTrial]], list[FrozenTrial]] | None
        ) = None,
        child_generation_strategy: (
            Callable[["Study", dict[str, BaseDistribution], list[FrozenTrial]], dict[str, Any]]
            | None
        ) = None,
        after_trial_strategy: (
            Callable[["Study", FrozenTrial, TrialState, Sequence[float] | None], None] | None
        ) = None,
    ) -> None:
        from optuna_amd._experimental import warn_experimental_argument

        if constraints_func is not None:
            import warnings

            warnings.warn(
                "`constraints_func` is deprecated; set constraints via "
                "Trial.set_constraint instead.",
                FutureWarning,
            )
        if after_trial_strategy is not None:
            warn_experimental_argument("after_trial_strategy")
        if child_generation_strategy is not None:
            warn_experimental_argument("child_generation_strategy")
        if elite_population_selection_strategy is not None:
            warn_experimental_argument("elite_population_selection_strategy")
        if population_size < 2:
            raise ValueError("`population_size` must be greater than or equal to 2.")
        if crossover is None:
            crossover = UniformCrossover(swapping_prob)
        if not isinstance(crossover, BaseCrossover):
            raise ValueError(f"'{crossover}' is not a valid crossover.")
        if population_size < crossover.n_parents:
            raise ValueError(
                f"Using {crossover}, the population size should be greater than or equal to "
                f"{crossover.n_parents}. The specified `population_size` is {population_size}."
            )
        super().__init__(population_size=population_size)
        self._random_sampler = RandomSampler(seed=seed)
        self._rng = LazyRandomState(seed)
        self._constraints_func = constraints_func
        self._search_space = IntersectionSearchSpace()
        if elite_population_selection_strategy is None:
            elite_population_selection_strategy = NSGAIIIElitePopulationSelectionStrategy(
                rng=self._rng,
                population_size=population_size,
                reference_points=reference_points,
                dividing_parameter=dividing_parameter,
                constraints_func=constraints_func,
            )
        self._elite_population_selection_strategy = elite_population_selection_strategy
        self._child_generation_strategy = (
            child_generation_strategy
            or NSGAIIChildGenerationStrategy(
                rng=self._rng,
                constraints_func=constraints_func,
                crossover=crossover,
                crossover_prob=crossover_prob,
                swapping_prob=swapping_prob,
                mutation=mutation,
                mutation_prob=mutation_prob,
            )
        )
        self._after_trial_strategy = after_trial_strategy or NSGAIIAfterTrialStrategy(
            constraints_func=constraints_func
        )

    def reseed_rng(self) -> None:
        self._random_sampler.reseed_rng()
        self._rng.rng.seed()

    def infer_relative_search_space(
        self, study: "Study", trial: FrozenTrial
    ) -> dict[str, BaseDistribution]:
        search_space: dict[str, BaseDistribution] = {}
        for name, distribution in self._search_space.calculate(study).items():
            if distribution.single():
                continue
            search_space[name] = distribution
        return search_space

    def select_parent(self, study: "Study", generation: int) -> list[FrozenTrial]:
        return self._elite_population_selection_strategy(
            study,
            self.get_population(study, generation - 1)
            + self.get_parent_population(study, generation - 1),
        )

    def sample_relative(
        self, study: "Study", trial: FrozenTrial, search_space: dict[str, BaseDistribution]
    ) -> dict[str, Any]:
        generation = self.get_trial_generation(study, trial)
        parent_population = self.get_parent_population(study, generation)
        if len(parent_population) == 0:
            return {}
        return self._child_generation_strategy(study, search_space, parent_population)

    def sample_independent(
        self,
        study: "Study",
        trial: FrozenTrial,
        param_name: str,
        param_distribution: BaseDistribution,
    ) -> Any:
        return self._random_sampler.sample_independent(
            study, trial, param_name, param_distribution
        )

    def before_trial(self, study: "Study", trial: FrozenTrial) -> None:
        self._random_sampler.before_trial(study, trial)

    def after_trial(
        self,
        study: "Study",
        trial: FrozenTrial,
        state: TrialState,
        values: Sequence[float] | None,
    ) -> None:
        assert state in [TrialState.COMPLETE, TrialState.FAIL, TrialState.PRUNED]
        self._after_trial_strategy(study, trial, state, values)
        self._random_sampler.after_trial(study, trial, state, values)
