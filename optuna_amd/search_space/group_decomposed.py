"""Group-decomposed search space: partition the union of per-trial spaces into
disjoint parameter groups (used by ``TPESampler(group=True)``).

Parity: reference ``optuna/search_space/group_decomposed.py``
(_SearchSpaceGroup.add_distributions :22, _GroupDecomposedSearchSpace :40).
"""
from __future__ import annotations

from typing import TYPE_CHECKING

from optuna_amd.distributions import BaseDistribution
from optuna_amd.trial import TrialState


if TYPE_CHECKING:
    from optuna_amd.study import Study


class _SearchSpaceGroup:
    def __init__(self) -> None:
        self._search_spaces: list[dict[str, BaseDistribution]] = []

    @property
    def search_spaces(self) -> list[dict[str, BaseDistribution]]:
        return self._search_spaces

    def add_distributions(self, distributions: dict[str, BaseDistribution]) -> None:
        dist_keys = set(distributions.keys())
        next_spaces: list[dict[str, BaseDistribution]] = []
        for space in self._search_spaces:
            keys = set(space.keys())
            overlap = keys & dist_keys
            if not overlap:
                next_spaces.append(space)
                continue
            # Split the existing group into (inside-overlap, outside-overlap)
            # parts — intersection first, remainder after, keeping the group's
            # position in the list.
            inside = {k: v for k, v in space.items() if k in overlap}
            outside = {k: v for k, v in space.items() if k not in overlap}
            next_spaces.append(inside)
            if outside:
                next_spaces.append(outside)
            dist_keys -= overlap
        if dist_keys:
            next_spaces.append({k: distributions[k] for k in distributions if k in dist_keys})
        self._search_spaces = next_spaces


class _GroupDecomposedSearchSpace:
    def __init__(self, include_pruned: bool = False) -> None:
        self._search_space = _SearchSpaceGroup()
        self._study_id: int | None = None
        self._include_pruned = include_pruned

    def calculate(self, study: "Study") -> _SearchSpaceGroup:
        if self._study_id is None:
            self._study_id = study._study_id
        else:
            if self._study_id != study._study_id:
                raise ValueError(
                    "`_GroupDecomposedSearchSpace` cannot handle multiple studies."
                )
        states_of_interest = [TrialState.COMPLETE, TrialState.WAITING, TrialState.RUNNING]
        if self._include_pruned:
            states_of_interest.append(TrialState.PRUNED)
        for trial in study._get_trials(deepcopy=False, states=states_of_interest, use_cache=False):
            if not trial.state.is_finished():
                continue
            self._search_space.add_distributions(trial.distributions)
        return self._search_space
