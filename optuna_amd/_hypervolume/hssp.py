"""Greedy hypervolume subset selection (HSSP) with the (1−1/e) guarantee.

2-D uses an exact O(k·n) incremental rectangle update; N-D uses greedy selection
with lazily-updated submodular contributions (a candidate's cached contribution
only shrinks as the selected set grows, so the max-heap order can be trusted
until the top element is re-evaluated).

Parity: reference ``optuna/_hypervolume/hssp.py`` (_solve_hssp_2d :10,
_lazy_contribs_update :45, _solve_hssp :143).
"""
from __future__ import annotations

import heapq

import numpy as np

from optuna_amd._hypervolume.wfg import compute_hypervolume


def _solve_hssp_2d(
    rank_i_loss_vals: np.ndarray,
    rank_i_indices: np.ndarray,
    subset_size: int,
    reference_point: np.ndarray,
) -> np.ndarray:
    """Exact greedy for 2-D: contributions are rectangles between sorted neighbors."""
    assert rank_i_loss_vals.shape[1] == 2
    n = len(rank_i_loss_vals)
    order = np.argsort(rank_i_loss_vals[:, 0])
    sorted_vals = rank_i_loss_vals[order]
    sorted_idx = rank_i_indices[order]

    # Doubly-linked neighbor structure over the sorted front.
    left = np.arange(-1, n - 1)
    right = np.arange(1, n + 1)
    selected = np.zeros(n, dtype=bool)

    def contrib(j: int) -> float:
        x_right = reference_point[0] if right[j] >= n else sorted_vals[right[j], 0]
        y_left = reference_point[1] if left[j] < 0 else sorted_vals[left[j], 1]
        return float((x_right - sorted_vals[j, 0]) * (y_left - sorted_vals[j, 1]))

    chosen: list[int] = []
    for _ in range(subset_size):
        best_j, best_c = -1, -np.inf
        for j in range(n):
            if selected[j]:
                continue
            c = contrib(j)
            if c > best_c:
                best_j, best_c = j, c
        assert best_j >= 0
        selected[best_j] = True
        chosen.append(best_j)
        # Splice out of the neighbor list: neighbors' contributions now extend
        # over the removed point's span.
        if left[best_j] >= 0:
            right[left[best_j]] = right[best_j]
        if right[best_j] < n:
            left[right[best_j]] = left[best_j]

    return sorted_idx[np.asarray(chosen)]


def _solve_hssp(
    rank_i_loss_vals: np.ndarray,
    rank_i_indices: np.ndarray,
    subset_size: int,
    reference_point: np.ndarray,
) -> np.ndarray:
    """Greedy HSSP; returns the original indices of the selected subset."""
    assert rank_i_loss_vals.shape[0] == rank_i_indices.shape[0]
    if subset_size >= len(rank_i_indices):
        return rank_i_indices.copy()
    if not np.isfinite(reference_point).all():
        # Degenerate reference: any subset attains infinite HV; pick by objective sum.
        order = np.argsort(rank_i_loss_vals.sum(axis=-1))
        return rank_i_indices[order[:subset_size]]
    if rank_i_loss_vals.shape[1] == 2:
        return _solve_hssp_2d(rank_i_loss_vals, rank_i_indices, subset_size, reference_point)

    n = len(rank_i_loss_vals)
    selected_mask = np.zeros(n, dtype=bool)
    selected_vals: list[np.ndarray] = []
    hv_selected = 0.0

    # Lazy greedy: heap of (-cached_contrib, stamp, j).
    heap: list[tuple[float, int, int]] = []
    for j in range(n):
        c = compute_hypervolume(rank_i_loss_vals[j : j + 1], reference_point)
        heapq.heappush(heap, (-c, 0, j))

    chosen: list[int] = []
    stamp = 0
    while len(chosen) < subset_size:
        neg_c, s, j = heapq.heappop(heap)
        if selected_mask[j]:
            continue
        if s == stamp:
            selected_mask[j] = True
            selected_vals.append(rank_i_loss_vals[j])
            chosen.append(j)
            stamp += 1
            hv_selected = compute_hypervolume(np.asarray(selected_vals), reference_point)
        else:
            # Stale: recompute against the current selected set and push back.
            cand = np.asarray(selected_vals + [rank_i_loss_vals[j]])
            c = compute_hypervolume(cand, reference_point) - hv_selected
            heapq.heappush(heap, (-c, stamp, j))

    return rank_i_indices[np.asarray(chosen)]
