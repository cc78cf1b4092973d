"""Grid search sampler.

All combinations of the user-given per-parameter value lists; the target grid id
is assigned in ``before_trial`` as a trial system attr (so distributed workers
coordinate through storage), values are returned from ``sample_independent``.

Parity: reference ``optuna/samplers/_grid.py`` (GridSampler :33, grid assignment
in before_trial :135-176, unvisited scan :241, exhaustion → study.stop :224-231).
"""
from __future__ import annotations

import itertools
import warnings
from numbers import Real
from typing import TYPE_CHECKING, Any, Mapping, Sequence, Union

import numpy as np

from optuna_amd import logging as _logging
from optuna_amd.distributions import BaseDistribution
from optuna_amd.samplers._base import BaseSampler
from optuna_amd.samplers._lazy_random_state import LazyRandomState
from optuna_amd.trial import FrozenTrial, TrialState


if TYPE_CHECKING:
    from optuna_amd.study import Study

_logger = _logging.get_logger(__name__)

GridValueType = Union[str, float, int, bool, None]


class GridSampler(BaseSampler):
    """Exhaustive grid search over explicit candidate lists."""

    def __init__(
        self, search_space: Mapping[str, Sequence[GridValueType]], seed: int | None = None
    ) -> None:
        for param_name, param_values in search_space.items():
            for value in param_values:
                self._check_value(param_name, value)

        self._search_space = {}
        for param_name, param_values in sorted(search_space.items()):
            self._search_space[param_name] = list(param_values)

        self._all_grids = list(itertools.product(*self._search_space.values()))
        self._param_names = sorted(search_space.keys())
        self._n_min_trials = len(self._all_grids)
        self._rng = LazyRandomState(seed or 0)
        # Shuffle so that truncated runs cover the space uniformly.
        self._rng.rng.shuffle(self._all_grids)  # type: ignore[arg-type]

    def reseed_rng(self) -> None:
        self._rng.rng.seed()

    def before_trial(self, study: "Study", trial: FrozenTrial) -> None:
        # Retried or enqueued trials keep their original grid / fixed params.
        if "grid_id" in trial.system_attrs or "fixed_params" in trial.system_attrs:
            return

        if 0 <= trial.number < self._n_min_trials:
            study._storage.set_trial_system_attr(
                trial._trial_id, "search_space", self._search_space
            )
            study._storage.set_trial_system_attr(trial._trial_id, "grid_id", trial.number)
            return

        target_grids = self._get_unvisited_grid_ids(study)
        if len(target_grids) == 0:
            _logger.warning(
                "`GridSampler` is re-evaluating a configuration because the grid has been "
                "exhausted. This may happen due to a timing issue during distributed "
                "optimization or when re-running optimizations on already finished studies."
            )
            target_grids = list(range(len(self._all_grids)))

        # Random choice to reduce duplicate picks among concurrent workers.
        grid_id = int(self._rng.rng.choice(target_grids))
        study._storage.set_trial_system_attr(trial._trial_id, "search_space", self._search_space)
        study._storage.set_trial_system_attr(trial._trial_id, "grid_id", grid_id)

    def infer_relative_search_space(
        self, study: "Study", trial: FrozenTrial
    ) -> dict[str, BaseDistribution]:
        return {}

    def sample_relative(
        self, study: "Study", trial: FrozenTrial, search_space: dict[str, BaseDistribution]
    ) -> dict[str, Any]:
        return {}

    def sample_independent(
        self,
        study: "Study",
        trial: FrozenTrial,
        param_name: str,
        param_distribution: BaseDistribution,
    ) -> Any:
        if "grid_id" not in trial.system_attrs:
            raise ValueError(
                "All parameters must be specified when using GridSampler with enqueue_trial."
            )
        if param_name not in self._search_space:
            raise ValueError(f"The parameter name, {param_name}, is not found in the given grid.")

        grid_id = trial.system_attrs["grid_id"]
        param_value = self._all_grids[grid_id][self._param_names.index(param_name)]
        if not param_distribution._contains(param_distribution.to_internal_repr(param_value)):
            warnings.warn(
                f"The value `{param_value}` is out of range of the parameter `{param_name}`. "
                f"The value will be used but the actual distribution is: `{param_distribution}`."
            )
        return param_value

    def after_trial(
        self,
        study: "Study",
        trial: FrozenTrial,
        state: TrialState,
        values: Sequence[float] | None,
    ) -> None:
        target_grids = self._get_unvisited_grid_ids(study)
        if len(target_grids) == 0:
            study.stop()
        elif len(target_grids) == 1:
            grid_id = study._storage.get_trial_system_attrs(trial._trial_id)["grid_id"]
            if grid_id == target_grids[0]:
                study.stop()

    @staticmethod
    def _check_value(param_name: str, param_value: Any) -> None:
        if param_value is None or isinstance(param_value, (str, int, float, bool)):
            return
        warnings.warn(
            f"{param_name} contains a value with the type of {type(param_value)}, which is "
            "not supported by `GridSampler`. Please make sure a value is `str`, `int`, "
            "`float`, `bool` or `None` for persistent storage."
        )

    def _get_unvisited_grid_ids(self, study: "Study") -> list[int]:
        visited_grids = []
        running_grids = []
        # Query storage directly: bracket-filtered study views must not hide trials.
        trials = study._storage.get_all_trials(study._study_id, deepcopy=False)
        for t in trials:
            if "grid_id" in t.system_attrs and self._same_search_space(
                t.system_attrs["search_space"]
            ):
                if t.state.is_finished():
                    visited_grids.append(t.system_attrs["grid_id"])
                elif t.state == TrialState.RUNNING:
                    running_grids.append(t.system_attrs["grid_id"])

        unvisited = set(range(self._n_min_trials)) - set(visited_grids) - set(running_grids)
        if len(unvisited) == 0:
            # All grids started: re-offer those not yet finished.
            unvisited = set(range(self._n_min_trials)) - set(visited_grids)
        return list(unvisited)

    @staticmethod
    def _grid_value_equal(value1: GridValueType, value2: GridValueType) -> bool:
        value1_is_nan = isinstance(value1, Real) and np.isnan(float(value1))
        value2_is_nan = isinstance(value2, Real) and np.isnan(float(value2))
        return (value1 == value2) or (value1_is_nan and value2_is_nan)

    def _same_search_space(self, search_space: Mapping[str, Sequence[GridValueType]]) -> bool:
        if set(search_space.keys()) != set(self._search_space.keys()):
            return False
        for param_name in search_space.keys():
            if len(search_space[param_name]) != len(self._search_space[param_name]):
                return False
            for i, param_value in enumerate(search_space[param_name]):
                if not self._grid_value_equal(param_value, self._search_space[param_name][i]):
                    return False
        return True

    def is_exhausted(self, study: "Study") -> bool:
        return len(self._get_unvisited_grid_ids(study)) == 0
