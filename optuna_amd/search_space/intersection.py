"""Intersection search space across finished trials, with incremental caching.

Parity: reference ``optuna/search_space/intersection.py`` (_calculate :14,
IntersectionSearchSpace :58). The intersection keeps only parameters present in
*every* finished trial, with compatible distributions; iteration order follows the
latest trial's parameter order (sorted by name for determinism at the API surface,
matching the reference's sorted output).
"""
from __future__ import annotations

import copy
from typing import TYPE_CHECKING

from optuna_amd.distributions import BaseDistribution
from optuna_amd.trial import FrozenTrial, TrialState


if TYPE_CHECKING:
    from optuna_amd.study import Study


def _calculate(
    trials: list[FrozenTrial],
    include_pruned: bool = False,
    search_space: dict[str, BaseDistribution] | None = None,
) -> dict[str, BaseDistribution] | None:
    states_of_interest = [TrialState.COMPLETE, TrialState.WAITING, TrialState.RUNNING]
    if include_pruned:
        states_of_interest.append(TrialState.PRUNED)

    for trial in trials:
        if trial.state not in states_of_interest:
            continue
        if not trial.state.is_finished():
            continue
        if search_space is None:
            search_space = copy.copy(trial.distributions)
            continue
        search_space = {
            name: distribution
            for name, distribution in search_space.items()
            if trial.distributions.get(name) == distribution
        }
    return search_space


class IntersectionSearchSpace:
    """Caches the intersection computation by the highest trial number seen."""

    def __init__(self, include_pruned: bool = False) -> None:
        self._cursor: int = -1
        self._search_space: dict[str, BaseDistribution] | None = None
        self._study_id: int | None = None
        self._include_pruned = include_pruned
        self._n_finished_seen: int = -1
        self._log_idx: int = 0

    def calculate(self, study: "Study") -> dict[str, BaseDistribution]:
        if self._study_id is None:
            self._study_id = study._study_id
        else:
            if self._study_id != study._study_id:
                raise ValueError("`IntersectionSearchSpace` cannot handle multiple studies.")

        # The intersection depends on finished trials only; an O(1) count check
        # skips the storage read while nothing new finished.
        finished_states = (
            (TrialState.COMPLETE, TrialState.PRUNED)
            if self._include_pruned
            else (TrialState.COMPLETE,)
        )
        n_finished = study._storage.get_n_trials(study._study_id, finished_states)
        if n_finished == self._n_finished_seen and self._search_space is not None:
            return dict(sorted(self._search_space.items(), key=lambda x: x[0]))
        self._n_finished_seen = n_finished

        # Delta path: _calculate only looks at finished trials, and finished
        # trials are immutable — read just the ones logged since the last call.
        delta_read = getattr(study._storage, "get_finished_trials_since", None)
        if delta_read is not None:
            new_finished = delta_read(study._study_id, self._log_idx)
            self._log_idx += len(new_finished)
            self._search_space = _calculate(
                new_finished, self._include_pruned, self._search_space
            )
            search_space = self._search_space or {}
            return dict(sorted(search_space.items(), key=lambda x: x[0]))

        states_of_interest = [TrialState.COMPLETE, TrialState.WAITING, TrialState.RUNNING]
        if self._include_pruned:
            states_of_interest.append(TrialState.PRUNED)
        trials = study._get_trials(deepcopy=False, states=states_of_interest, use_cache=False)
        # Storage returns trials number-ascending: bisect to the cursor boundary.
        import bisect

        lo = bisect.bisect_right(trials, self._cursor, key=lambda t: t.number)
        new_trials = trials[lo:]
        self._search_space = _calculate(new_trials, self._include_pruned, self._search_space)
        # Advance the cursor to the largest prefix of trials that are all finished, so
        # currently-unfinished trials are re-examined once they finish.
        next_cursor = self._cursor
        for t in sorted(new_trials, key=lambda t: t.number):
            if not t.state.is_finished():
                break
            next_cursor = t.number
        self._cursor = next_cursor
        search_space = self._search_space or {}
        return dict(sorted(search_space.items(), key=lambda x: x[0]))


def intersection_search_space(
    trials: list[FrozenTrial], include_pruned: bool = False
) -> dict[str, BaseDistribution]:
    search_space = _calculate(trials, include_pruned)
    search_space = search_space or {}
    return dict(sorted(search_space.items(), key=lambda x: x[0]))
