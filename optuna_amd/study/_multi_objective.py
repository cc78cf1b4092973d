"""Pareto-front and non-domination-rank kernels (host reference implementation).

These are the CPU/numpy versions; on MI355X the same computations run as the K6
HIP kernel family (``optuna_amd/_hip/kernels/pareto.hip``) over the device trial
table for large histories.

Parity: reference ``optuna/study/_multi_objective.py``
(_get_pareto_front_trials_by_trials :19, _fast_non_domination_rank :49,
_is_pareto_front_2d :138, _is_pareto_front_nd :114, _calculate_nondomination_rank
:174, _dominates :209).
"""
from __future__ import annotations

from collections.abc import Sequence
from typing import TYPE_CHECKING

import numpy as np

from optuna_amd.study._study_direction import StudyDirection
from optuna_amd.trial import FrozenTrial, TrialState


# Below this row count the host peel is faster than kernel launches.
_DEVICE_RANK_MIN_ROWS = 4096


def _nondomination_rank_device(
    loss_values: np.ndarray, n_below: int
) -> np.ndarray | None:
    """K6 HIP path; returns per-row ranks (-1 = beyond n_below) or None."""
    from optuna_amd import _hip

    core = _hip.get()
    if core is None or not core.available():
        return None
    return np.asarray(
        core.nondomination_rank(
            np.ascontiguousarray(loss_values, dtype=np.float64), int(n_below)
        )
    )


if TYPE_CHECKING:
    from optuna_amd.study import Study


def _normalize_value(value: float | None, direction: StudyDirection) -> float:
    """Map an objective value onto 'loss' (minimization) orientation; None→inf."""
    if value is None:
        return float("inf")
    return float(value) if direction == StudyDirection.MINIMIZE else -float(value)


def _dominates(
    trial0: FrozenTrial, trial1: FrozenTrial, directions: Sequence[StudyDirection]
) -> bool:
    """True iff trial0 weakly dominates trial1 with at least one strict improvement.

    Any complete trial dominates any incomplete one; two incomplete trials never
    dominate each other (values may be absent for those states).
    """
    if trial0.state != TrialState.COMPLETE:
        return False
    if trial1.state != TrialState.COMPLETE:
        return True
    assert trial0.values is not None and trial1.values is not None
    if len(trial0.values) != len(trial1.values):
        raise ValueError("Trials with different numbers of objectives cannot be compared.")
    if len(trial0.values) != len(directions):
        raise ValueError(
            "The number of the values and the number of the objectives are mismatched."
        )
    values0 = [_normalize_value(v, d) for v, d in zip(trial0.values, directions)]
    values1 = [_normalize_value(v, d) for v, d in zip(trial1.values, directions)]
    if values0 == values1:
        return False
    return all(v0 <= v1 for v0, v1 in zip(values0, values1))


def _is_pareto_front_2d(unique_lexsorted_loss_values: np.ndarray) -> np.ndarray:
    n_trials = unique_lexsorted_loss_values.shape[0]
    on_front = np.zeros(n_trials, dtype=bool)
    nondominated_indices = np.arange(n_trials)
    while len(unique_lexsorted_loss_values):
        # Lexsorted ⇒ first point has minimal obj0 (ties broken by obj1): on front.
        nondominated_and_not_top = np.any(
            unique_lexsorted_loss_values < unique_lexsorted_loss_values[0], axis=1
        )
        on_front[nondominated_indices[0]] = True
        unique_lexsorted_loss_values = unique_lexsorted_loss_values[nondominated_and_not_top]
        nondominated_indices = nondominated_indices[nondominated_and_not_top]
    return on_front


def _is_pareto_front_nd(unique_lexsorted_loss_values: np.ndarray) -> np.ndarray:
    n_trials = unique_lexsorted_loss_values.shape[0]
    on_front = np.zeros(n_trials, dtype=bool)
    nondominated_indices = np.arange(n_trials)
    while len(unique_lexsorted_loss_values):
        nondominated_and_not_top = np.any(
            unique_lexsorted_loss_values < unique_lexsorted_loss_values[0], axis=1
        )
        # The lexicographically smallest point is non-dominated (unique values).
        on_front[nondominated_indices[0]] = True
        unique_lexsorted_loss_values = unique_lexsorted_loss_values[nondominated_and_not_top]
        nondominated_indices = nondominated_indices[nondominated_and_not_top]
    return on_front


def _is_pareto_front(loss_values: np.ndarray, assume_unique_lexsorted: bool) -> np.ndarray:
    apply_unique = not assume_unique_lexsorted
    if apply_unique:
        unique_lexsorted_loss_values, order_inversion = np.unique(
            loss_values, axis=0, return_inverse=True
        )
    else:
        unique_lexsorted_loss_values = loss_values
        order_inversion = None

    n_objectives = unique_lexsorted_loss_values.shape[1]
    if n_objectives == 1:
        on_front = unique_lexsorted_loss_values[:, 0] == unique_lexsorted_loss_values[0, 0]
        if not assume_unique_lexsorted:
            on_front = np.zeros_like(on_front)
            on_front[0] = True
    elif n_objectives == 2:
        on_front = _is_pareto_front_2d(unique_lexsorted_loss_values)
    else:
        on_front = _is_pareto_front_nd(unique_lexsorted_loss_values)

    if order_inversion is not None:
        return on_front[np.asarray(order_inversion).reshape(-1)]
    return on_front


def _fast_non_domination_rank(
    loss_values: np.ndarray, *, penalty: np.ndarray | None = None, n_below: int | None = None
) -> np.ndarray:
    """Non-domination rank per point, feasibility-aware when penalty is given.

    Ranks are guaranteed correct only for the top-``n_below`` points; the rest are
    lumped into the final rank (reference study/_multi_objective.py:49-136).
    Feasible points (penalty<=0) are ranked by objectives first; infeasible points
    get ranks after all feasible ones, ordered by total violation; NaN-penalty
    points come last.
    """
    if penalty is None:
        ranks, _ = _calculate_nondomination_rank(loss_values, n_below=n_below)
        return ranks

    if len(penalty) != len(loss_values):
        raise ValueError(
            "The length of penalty and loss_values must be same, but got "
            f"len(penalty)={len(penalty)} and len(loss_values)={len(loss_values)}."
        )
    n_below = n_below or len(loss_values)
    ranks = np.full(len(loss_values), -1, dtype=int)
    is_nan = np.isnan(penalty)
    is_feasible = np.logical_and(~is_nan, penalty <= 0)
    is_infeasible = np.logical_and(~is_nan, penalty > 0)

    # Feasible: rank by objectives.
    ranks[is_feasible], bottom_rank = _calculate_nondomination_rank(
        loss_values[is_feasible], n_below=n_below
    )
    n_below -= int(np.count_nonzero(is_feasible))

    # Infeasible: rank by constraint violation only.
    top_rank_infeas = bottom_rank + 1
    ranks[is_infeasible], bottom_rank = _calculate_nondomination_rank(
        penalty[is_infeasible][:, np.newaxis], n_below=n_below, base_rank=top_rank_infeas
    )
    n_below -= int(np.count_nonzero(is_infeasible))

    # NaN penalty: worst.
    ranks[is_nan] = bottom_rank + 1
    return ranks


def _calculate_nondomination_rank(
    loss_values: np.ndarray, *, n_below: int | None = None, base_rank: int = 0
) -> tuple[np.ndarray, int]:
    """Peel Pareto fronts; early-stop once n_below points have exact ranks."""
    if n_below is not None and n_below <= 0:
        return np.full(len(loss_values), base_rank, dtype=int), base_rank

    ranks = np.full(len(loss_values), -1, dtype=int)
    n_below = n_below or len(loss_values)

    # ±inf compares naturally under dominance ([1, inf] dominates [inf, inf]);
    # only NaN rows are incomparable and get lumped into the worst front.
    is_valid = ~np.isnan(loss_values).any(axis=1)

    rank = base_rank - 1
    indices = np.arange(len(loss_values))
    remaining_indices = indices[is_valid]
    remaining = loss_values[is_valid]

    # K6 device path: the O(N²M) dominance sweep runs as a HIP bitmatrix kernel
    # with on-device front peeling; the host loop below stays as the fallback
    # and the small-N fast path.
    if len(remaining) >= _DEVICE_RANK_MIN_ROWS:
        device_ranks = _nondomination_rank_device(remaining, n_below)
        if device_ranks is not None:
            ranked = device_ranks >= 0
            ranks[remaining_indices[ranked]] = base_rank + device_ranks[ranked]
            if ranked.any():
                rank = base_rank + int(device_ranks[ranked].max())
            remaining_indices = remaining_indices[~ranked]
            remaining = remaining[~ranked]

    n_assigned = int(np.count_nonzero(ranks >= 0))
    while len(remaining) and n_assigned < n_below:
        rank += 1
        on_front = _is_pareto_front(remaining, assume_unique_lexsorted=False)
        ranks[remaining_indices[on_front]] = rank
        n_assigned += int(np.count_nonzero(on_front))
        remaining_indices = remaining_indices[~on_front]
        remaining = remaining[~on_front]

    if np.any(ranks == -1):
        # Early-stop remainder + invalid rows share one bottom rank.
        bottom = rank + 1
        ranks[ranks == -1] = bottom
        return ranks, bottom
    return ranks, max(rank, base_rank)


def _get_pareto_front_trials_by_trials(
    trials: Sequence[FrozenTrial],
    directions: Sequence[StudyDirection],
    consider_constraint: bool = False,
) -> list[FrozenTrial]:
    from optuna_amd.study._constrained_optimization import _get_feasible_trials

    complete = [t for t in trials if t.state == TrialState.COMPLETE]
    if consider_constraint:
        complete = _get_feasible_trials(complete)
    if len(complete) == 0:
        return []
    loss_values = np.asarray(
        [[_normalize_value(v, d) for v, d in zip(t.values, directions)] for t in complete]
    )
    on_front = _is_pareto_front(loss_values, assume_unique_lexsorted=False)
    return [t for t, f in zip(complete, on_front) if f]


def _get_pareto_front_trials(study: "Study", consider_constraint: bool = False) -> list[FrozenTrial]:
    return _get_pareto_front_trials_by_trials(study.trials, study.directions, consider_constraint)
