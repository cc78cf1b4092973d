"""Greedy hypervolume subset selection (HSSP) with the (1−1/e) guarantee.

3-objective fronts above a size threshold run the K6b HIP path: every greedy
round evaluates ALL candidates' exclusive contributions in one kernel launch
(one workgroup per candidate, selected set staged in LDS), replacing the
host's lazy-heap sequence of WFG evaluations.

2-D uses an exact O(k·n) shrinking-corner greedy that is valid for arbitrary
inputs (dominated and boundary points included); N-D uses greedy selection
with lazily-updated submodular contributions (a candidate's cached contribution
only shrinks as the selected set grows, so the max-heap order can be trusted
until the top element is re-evaluated).

Loss values are deduplicated up front so duplicated rows can never absorb two
selection slots (their second copy has zero marginal volume).

Parity: reference ``optuna/_hypervolume/hssp.py`` (_solve_hssp_2d :10,
_lazy_contribs_update :45, _solve_hssp :143).
"""
from __future__ import annotations

import heapq
import math

import numpy as np

from optuna_amd._hypervolume.wfg import compute_hypervolume


def _solve_hssp_2d(
    lexsorted_loss_vals: np.ndarray,
    lexsorted_indices: np.ndarray,
    subset_size: int,
    reference_point: np.ndarray,
) -> np.ndarray:
    """Exact O(k·n) greedy for 2-D; rows must be unique-lexsorted.

    Every remaining candidate's marginal contribution is the rectangle spanned
    by the point and its current "free corner". Corners start at the reference
    point; a selected point at sorted position ``m`` caps the x-edge of every
    candidate left of ``m`` and the y-edge of every candidate at/right of
    ``m`` (in lexsorted order those are exactly the candidates whose rectangle
    it cuts). Dominated candidates' corners shrink past them, driving their
    gain to ≤ 0, so no front-filtering is needed.
    """
    assert lexsorted_loss_vals.shape[1] == 2
    vals = lexsorted_loss_vals.copy()
    idx = lexsorted_indices.copy()
    corners = np.tile(np.asarray(reference_point, dtype=float), (len(vals), 1))
    picked = np.empty(subset_size, dtype=idx.dtype)
    for k in range(subset_size):
        gains = (corners[:, 0] - vals[:, 0]) * (corners[:, 1] - vals[:, 1])
        m = int(np.argmax(gains))
        picked[k] = idx[m]
        sx, sy = vals[m, 0], vals[m, 1]
        corners[:m, 0] = np.minimum(corners[:m, 0], sx)
        corners[m:, 1] = np.minimum(corners[m:, 1], sy)
        keep = np.ones(len(vals), dtype=bool)
        keep[m] = False
        vals, idx, corners = vals[keep], idx[keep], corners[keep]
    return picked


def _solve_hssp_on_unique(
    loss_vals: np.ndarray,
    indices: np.ndarray,
    subset_size: int,
    reference_point: np.ndarray,
) -> np.ndarray:
    """Greedy HSSP over unique-lexsorted rows; returns original indices."""
    if not np.isfinite(reference_point).all():
        # Degenerate reference: every nonempty subset attains infinite volume.
        return indices[:subset_size].copy()
    if subset_size >= len(indices):
        return indices.copy()
    if loss_vals.shape[1] == 2:
        return _solve_hssp_2d(loss_vals, indices, subset_size, reference_point)
    if (
        loss_vals.shape[1] == 3
        and len(loss_vals) * subset_size >= _DEVICE_HSSP_MIN_WORK
        and np.isfinite(loss_vals).all()
    ):
        device_choice = _solve_hssp_3d_device(
            loss_vals, indices, subset_size, reference_point
        )
        if device_choice is not None:
            return device_choice

    n = len(loss_vals)
    selected_mask = np.zeros(n, dtype=bool)
    selected_vals: list[np.ndarray] = []
    hv_selected = 0.0

    # Lazy greedy: heap of (-cached_contrib, stamp, j).
    heap: list[tuple[float, int, int]] = []
    for j in range(n):
        c = compute_hypervolume(loss_vals[j : j + 1], reference_point)
        heapq.heappush(heap, (-c, 0, j))

    chosen: list[int] = []
    stamp = 0
    while len(chosen) < subset_size:
        neg_c, s, j = heapq.heappop(heap)
        if selected_mask[j]:
            continue
        if s == stamp or math.isinf(hv_selected):
            # Once the selected volume is infinite every marginal is "inf - inf";
            # all remaining candidates tie at zero effective gain, so cached heap
            # order (singleton volume) is as good a tie-break as any.
            selected_mask[j] = True
            selected_vals.append(loss_vals[j])
            chosen.append(j)
            stamp += 1
            hv_selected = compute_hypervolume(np.asarray(selected_vals), reference_point)
        else:
            # Stale: recompute against the current selected set and push back.
            cand = np.asarray(selected_vals + [loss_vals[j]])
            c = compute_hypervolume(cand, reference_point) - hv_selected
            heapq.heappush(heap, (-c, stamp, j))

    return indices[np.asarray(chosen)]


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

    # Duplicated rows have zero marginal volume once their twin is in; solve on
    # the unique rows and, if that leaves unused slots, pad with duplicates.
    unique_vals, first_pos = np.unique(rank_i_loss_vals, return_index=True, axis=0)
    if len(first_pos) < subset_size:
        take = np.zeros(len(rank_i_indices), dtype=bool)
        take[first_pos] = True
        dup_pos = np.flatnonzero(~take)
        take[dup_pos[: subset_size - len(first_pos)]] = True
        return rank_i_indices[take]

    chosen = _solve_hssp_on_unique(
        unique_vals, first_pos, subset_size, reference_point
    )
    return rank_i_indices[chosen]


# Gate on candidates x selections: the host lazy greedy is fine for tiny
# problems, but its stale re-evaluations blow up with the selection count
# (MO-TPE's 3-objective gamma boundary: ~350 candidates x ~200 selections
# measured ~1 s/suggest on host vs ~50 ms on device).
_DEVICE_HSSP_MIN_WORK = 8192


def _solve_hssp_3d_device(
    vals: np.ndarray,
    indices: np.ndarray,
    subset_size: int,
    reference_point: np.ndarray,
) -> np.ndarray | None:
    """Exact (non-lazy) greedy via the K6b contributions kernel, or None.

    Every round evaluates ALL remaining candidates' exclusive contributions in
    one launch; the selected set travels as pre-sorted views (max(c, .) is
    monotone, so the kernel clamps instead of sorting per candidate). The
    sorted views are maintained incrementally on the host (bisect inserts)."""
    from optuna_amd import _hip

    core = _hip.get()
    if core is None or not core.available() or subset_size > 2000:
        # 2000 = the insert kernel's register-staging bound (32 slots x 64
        # lanes); greedy subsets beyond it fall back to the host path.
        return None
    cand = np.ascontiguousarray(vals, dtype=np.float64)
    rx, ry, rz = (float(v) for v in reference_point)
    # The whole greedy runs as one device launch train: candidates, taken
    # mask, sorted selected-set views and the per-round winner all stay on
    # device; contrib -> masked argmax -> sorted insert repeat subset_size
    # times and a single sync brings back the chosen indices.
    session = core.Hssp3dSession(cand, rx, ry, rz)
    chosen = np.asarray(session.run(int(subset_size)))
    assert (chosen >= 0).all()
    return indices[chosen]
