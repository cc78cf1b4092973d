"""Box decomposition of the non-dominated region for (q)LogEHVI.

Implements Lacour-Klamroth-Fonseca "A Box Decomposition Algorithm to Compute the
Hypervolume Indicator" (arXiv:1510.01963): Algorithm 2 maintains the upper-bound
set U(N) with its defining points; applying it twice (once on the sign-flipped
upper-bound set) yields a partition of the *non-dominated* region into boxes,
which is what EHVI integrates over.

Parity: reference ``optuna/_hypervolume/box_decomposition.py`` (BoTorch-derived:
_get_upper_bound_set :30, _get_box_bounds :96, get_non_dominated_box_bounds
:138). Minimization convention throughout.
"""
from __future__ import annotations

import warnings

import numpy as np

from optuna_amd.study._multi_objective import _is_pareto_front


def _update_upper_bounds(
    sol: np.ndarray,
    ubs: np.ndarray,
    dps: np.ndarray,
    first_dim_always_updates: np.ndarray,
) -> tuple[np.ndarray, np.ndarray]:
    """One Alg.-2 insertion: replace the bounds strictly dominated by ``sol``.

    ubs: (n_bounds, k) current upper-bound set; dps: (n_bounds, k, k) defining
    points (dps[i, j] is z^j of ubs[i]).
    """
    k = sol.shape[0]
    strictly_below = np.all(sol < ubs, axis=-1)
    if not strictly_below.any():
        return ubs, dps

    kept_ubs = ubs[~strictly_below]
    kept_dps = dps[~strictly_below]
    hit_ubs = ubs[strictly_below]
    hit_dps = dps[strictly_below]

    new_ubs_list = []
    new_dps_list = []
    # max over defining points z^i (i != j) of their j-th coordinate; the first
    # objective always passes because solutions arrive sorted by it (Alg. 2 line 2).
    masked = np.where(first_dim_always_updates, -np.inf, hit_dps)
    thresholds = masked.max(axis=-2)  # (n_hit, k)
    for u_idx in range(len(hit_ubs)):
        for j in range(k):
            if sol[j] >= thresholds[u_idx, j]:
                u_new = hit_ubs[u_idx].copy()
                u_new[j] = sol[j]
                d_new = hit_dps[u_idx].copy()
                d_new[j] = sol
                new_ubs_list.append(u_new)
                new_dps_list.append(d_new)

    if new_ubs_list:
        ubs_out = np.vstack([kept_ubs, np.asarray(new_ubs_list)])
        dps_out = np.vstack([kept_dps, np.asarray(new_dps_list).reshape(-1, k, k)])
    else:
        ubs_out, dps_out = kept_ubs, kept_dps
    return ubs_out, dps_out


def _get_upper_bound_set(
    sorted_pareto_sols: np.ndarray, ref_point: np.ndarray
) -> tuple[np.ndarray, np.ndarray]:
    """U(N) and its defining points for pareto solutions sorted by objective 0."""
    _, k = sorted_pareto_sols.shape
    # mask[i, j] True → skip the inequality for defining-point row i, column j.
    mask = np.eye(k, dtype=bool)
    mask[:, 0] = True

    ubs = np.asarray([ref_point])
    dps = np.full((1, k, k), -np.inf)
    dps[0, np.arange(k), np.arange(k)] = ref_point  # dummy points \hat{z}^k
    for sol in sorted_pareto_sols:
        ubs, dps = _update_upper_bounds(sol, ubs, dps, mask)
    return ubs, dps


def _get_box_bounds(
    upper_bound_set: np.ndarray, def_points: np.ndarray, ref_point: np.ndarray
) -> np.ndarray:
    """Eq. (2) of Lacour17: per-bound boxes [l, u]; empty boxes dropped."""
    k = upper_bound_set.shape[-1]
    assert k > 1, "box decomposition requires n_objectives > 1"
    lower = np.empty_like(upper_bound_set)
    upper = np.empty_like(upper_bound_set)
    lower[:, 0] = def_points[:, 0, 0]
    upper[:, 0] = ref_point[0]
    running_max = np.maximum.accumulate(def_points, axis=-2)
    for j in range(1, k):
        lower[:, j] = running_max[:, j - 1, j]
        upper[:, j] = upper_bound_set[:, j]
    keep = ~np.any(upper <= lower, axis=-1)
    return np.stack([lower[keep], upper[keep]])


def get_non_dominated_box_bounds(
    loss_vals: np.ndarray, ref_point: np.ndarray
) -> tuple[np.ndarray, np.ndarray]:
    """Partition the non-dominated region below ``ref_point`` into boxes.

    Returns (lower_bounds, upper_bounds), each (n_boxes, n_objectives).
    """
    assert np.all(np.isfinite(loss_vals)), "loss_vals must be clipped before box decomposition."
    uniq = np.unique(loss_vals, axis=0)
    sorted_pareto_sols = uniq[_is_pareto_front(uniq, assume_unique_lexsorted=True)]
    k = loss_vals.shape[-1]
    assert k > 1, "box decomposition requires n_objectives > 1"
    if k > 4:
        warnings.warn(
            "Box decomposition (typically used by `GPSampler`) might be significantly "
            "slow for n_objectives > 4. Please consider using another sampler instead."
        )

    # Dual trick: the sign-flipped upper-bound set of U(N) partitions the
    # non-dominated region (see Lacour17 §3 / the reference's derivation).
    neg_ubs = -_get_upper_bound_set(sorted_pareto_sols, ref_point)[0]
    neg_ubs_sorted = np.unique(neg_ubs, axis=0)
    inf_point = np.full_like(ref_point, np.inf)
    neg_lower_set, neg_dps = _get_upper_bound_set(
        neg_ubs_sorted[_is_pareto_front(neg_ubs_sorted, assume_unique_lexsorted=True)],
        inf_point,
    )
    box_upper, box_lower = -_get_box_bounds(neg_lower_set, neg_dps, inf_point)
    return box_lower, box_upper
