"""Exact hypervolume (minimization convention) — host implementation.

2-D uses a linear sweep over the lexsorted front; 3-D uses an O(N²) sweep that
maintains the dominated 2-D staircase incrementally; N-D uses the WFG
exclusive-volume recursion with limited-set Pareto filtering.

On device, large-N dominance filtering and the 2-D/3-D paths are the K6 HIP
kernels; the WFG recursion stays on host (irregular control flow) feeding the
device with batched 2-D/3-D subproblems.

Parity: reference ``optuna/_hypervolume/wfg.py`` (_compute_2d :8, _compute_3d :16,
_compute_hv/_compute_exclusive_hv :41-107, compute_hypervolume :110).
"""
from __future__ import annotations

import numpy as np

from optuna_amd.study._multi_objective import _is_pareto_front


def _compute_2d(sorted_pareto_sols: np.ndarray, reference_point: np.ndarray) -> float:
    """Linear sweep: rows lexsorted ⇒ first coord increasing, second decreasing."""
    rect_widths = reference_point[0] - sorted_pareto_sols[:, 0]
    upper_edges = np.concatenate(
        [[reference_point[1]], sorted_pareto_sols[:-1, 1]]
    )
    rect_heights = upper_edges - sorted_pareto_sols[:, 1]
    return float(np.sum(rect_widths * rect_heights))


def _compute_3d(sorted_pareto_sols: np.ndarray, reference_point: np.ndarray) -> float:
    """O(N²): sweep the first axis; maintain the (y,z) staircase via cummax trick.

    For each new point, its exclusive slab volume between x_i and x_{i+1} is the
    area of the region in (y,z) dominated by the first i+1 points; we recompute the
    staircase area in O(i) from points kept sorted by y.
    """
    n = len(sorted_pareto_sols)
    ref_y, ref_z = reference_point[1], reference_point[2]
    hv = 0.0
    # ys sorted ascending; zs the corresponding minimal z "so far" staircase.
    ys: list[float] = []
    zs: list[float] = []

    def staircase_area() -> float:
        area = 0.0
        prev_z = ref_z
        for y, z in zip(ys, zs):
            if z < prev_z:
                area += (ref_y - y) * (prev_z - z)
                prev_z = z
        return area

    for i in range(n):
        x, y, z = sorted_pareto_sols[i]
        # Insert (y, z) keeping ys ascending.
        import bisect

        pos = bisect.bisect_left(ys, y)
        ys.insert(pos, y)
        zs.insert(pos, z)
        next_x = sorted_pareto_sols[i + 1, 0] if i + 1 < n else reference_point[0]
        if next_x > x:
            hv += (next_x - x) * staircase_area()
    return float(hv)


def _compute_exclusive_hv(
    limited_sols: np.ndarray, inclusive_hv: float, reference_point: np.ndarray
) -> float:
    if limited_sols.shape[0] == 0:
        return inclusive_hv
    on_front = _is_pareto_front(limited_sols, assume_unique_lexsorted=False)
    return inclusive_hv - _compute_hv(limited_sols[on_front], reference_point)


def _compute_hv(sorted_pareto_sols: np.ndarray, reference_point: np.ndarray) -> float:
    inclusive_hvs = np.prod(reference_point - sorted_pareto_sols, axis=-1)
    if inclusive_hvs.shape[0] == 1:
        return float(inclusive_hvs[0])
    elif inclusive_hvs.shape[0] == 2:
        # S(A ∪ B) = S(A) + S(B) - S(A ∩ B), the intersection is the pointwise max.
        intersec = float(
            np.prod(reference_point - np.maximum(sorted_pareto_sols[0], sorted_pareto_sols[1]))
        )
        return float(np.sum(inclusive_hvs)) - intersec
    # WFG: sum over points of their exclusive volume w.r.t. later points.
    limited_sols_array = np.maximum(
        sorted_pareto_sols[:, np.newaxis], sorted_pareto_sols
    )
    return float(
        sum(
            _compute_exclusive_hv(
                limited_sols_array[i, i + 1 :], float(inclusive_hv), reference_point
            )
            for i, inclusive_hv in enumerate(inclusive_hvs)
        )
    )


def compute_hypervolume(
    loss_vals: np.ndarray, reference_point: np.ndarray, assume_pareto: bool = False
) -> float:
    """Hypervolume dominated by ``loss_vals`` w.r.t. ``reference_point`` (minimize).

    Every point must dominate or equal the reference point coordinatewise;
    anything else (including NaN anywhere — NaN comparisons are False) is a
    ``ValueError``. Points merely equal to the reference on some axis
    contribute nothing and are dropped.
    """
    if not np.all(loss_vals <= reference_point):
        raise ValueError(
            "All points must be coordinatewise <= the reference point "
            "(NaN in either input also fails this check)."
        )
    if not np.all(np.isfinite(reference_point)):
        # A +inf reference coordinate makes the dominated volume infinite.
        return float("inf")
    assert loss_vals.shape[1] == reference_point.shape[0]

    within = np.all(loss_vals < reference_point, axis=-1)
    loss_vals = loss_vals[within]
    if loss_vals.shape[0] == 0:
        return 0.0

    if not assume_pareto:
        unique_lexsorted = np.unique(loss_vals, axis=0)
        on_front = _is_pareto_front(unique_lexsorted, assume_unique_lexsorted=True)
        sorted_pareto_sols = unique_lexsorted[on_front]
    else:
        sorted_pareto_sols = loss_vals[np.argsort(loss_vals[:, 0], kind="stable")]

    if reference_point.shape[0] == 1:
        return float(reference_point[0] - np.min(sorted_pareto_sols))
    if reference_point.shape[0] == 2:
        return _compute_2d(sorted_pareto_sols, reference_point)
    if reference_point.shape[0] == 3:
        if len(sorted_pareto_sols) >= _DEVICE_HV3D_MIN_ROWS:
            dev = _hv3d_device(sorted_pareto_sols, reference_point)
            if dev is not None:
                return dev
        return _compute_3d(sorted_pareto_sols, reference_point)
    return _compute_hv(sorted_pareto_sols, reference_point)


# The K6a kernel's O(N^2) prefix sweep beats the host Python loop from a few
# hundred points on; launch latency dominates below this.
_DEVICE_HV3D_MIN_ROWS = 512


def _hv3d_device(sorted_pts: np.ndarray, reference_point: np.ndarray) -> float | None:
    """K6a HIP path: per-prefix staircase sweep on the MI355X (or None)."""
    from optuna_amd import _hip

    core = _hip.get()
    if core is None or not core.available():
        return None
    return float(
        core.hv3d(
            np.ascontiguousarray(sorted_pts, dtype=np.float64),
            float(reference_point[0]),
            float(reference_point[1]),
            float(reference_point[2]),
        )
    )
