"""BaseGASampler: generation bookkeeping through storage attrs.

Generation of a trial = current max generation, advanced once population_size
trials of that generation complete; parent populations are selected once per
generation and cached in study system attrs so concurrent workers agree.

Parity: reference ``optuna/samplers/_ga/_base.py`` (BaseGASampler :17,
get_trial_generation :87, get_population :133-150, parent cache :152-186).
"""
from __future__ import annotations

import abc
from typing import TYPE_CHECKING, Any

from optuna_amd.samplers._base import BaseSampler
from optuna_amd.trial import FrozenTrial, TrialState


if TYPE_CHECKING:
    from optuna_amd.study import Study


class BaseGASampler(BaseSampler, abc.ABC):
    _GENERATION_KEY = "BaseGASampler:generation"
    _PARENT_CACHE_KEY_PREFIX = "BaseGASampler:parent:"

    def __init_subclass__(cls, **kwargs: Any) -> None:
        super().__init_subclass__(**kwargs)
        cls._GENERATION_KEY = f"{cls.__name__}:generation"
        cls._PARENT_CACHE_KEY_PREFIX = f"{cls.__name__}:parent:"

    @classmethod
    def _get_generation_key(cls) -> str:
        return cls._GENERATION_KEY

    @classmethod
    def _get_parent_cache_key_prefix(cls) -> str:
        return cls._PARENT_CACHE_KEY_PREFIX

    def __init__(self, population_size: int | None) -> None:
        self._population_size = population_size

    @property
    def population_size(self) -> int | None:
        return self._population_size

    @population_size.setter
    def population_size(self, value: int) -> None:
        self._population_size = value

    @abc.abstractmethod
    def select_parent(self, study: "Study", generation: int) -> list[FrozenTrial]:
        """Select the parent population of ``generation`` (called once, then cached)."""
        raise NotImplementedError

    def get_trial_generation(self, study: "Study", trial: FrozenTrial) -> int:
        generation = trial.system_attrs.get(self._get_generation_key(), None)
        if generation is not None:
            return generation

        trials = study._get_trials(deepcopy=False, states=[TrialState.COMPLETE], use_cache=True)
        max_generation, max_generation_count = 0, 0
        for t in reversed(trials):
            gen = t.system_attrs.get(self._get_generation_key(), -1)
            if gen < max_generation:
                continue
            if gen > max_generation:
                max_generation = gen
                max_generation_count = 1
            else:
                max_generation_count += 1

        assert self._population_size is not None, "Population size must be set."
        if max_generation_count < self._population_size:
            generation = max_generation
        else:
            generation = max_generation + 1
        study._storage.set_trial_system_attr(
            trial._trial_id, self._get_generation_key(), generation
        )
        return generation

    def get_population(self, study: "Study", generation: int) -> list[FrozenTrial]:
        return [
            t
            for t in study._get_trials(
                deepcopy=False, states=[TrialState.COMPLETE], use_cache=True
            )
            if t.system_attrs.get(self._get_generation_key(), None) == generation
        ]

    def get_parent_population(self, study: "Study", generation: int) -> list[FrozenTrial]:
        if generation == 0:
            return []
        study_system_attrs = study._storage.get_study_system_attrs(study._study_id)
        cache_key = self._get_parent_cache_key_prefix() + str(generation)
        cached_ids = study_system_attrs.get(cache_key, None)
        if cached_ids is not None:
            trials = study._get_trials(deepcopy=False)
            id_set = set(cached_ids)
            return [t for t in trials if t._trial_id in id_set]
        parent_population = self.select_parent(study, generation)
        study._storage.set_study_system_attr(
            study._study_id, cache_key, [t._trial_id for t in parent_population]
        )
        return parent_population
