"""WFG / HSSP correctness vs brute force (Monte Carlo + exhaustive subsets)."""
from __future__ import annotations

import itertools

import numpy as np
import pytest

from optuna_amd._hypervolume import _solve_hssp, compute_hypervolume


def _mc_hypervolume(points: np.ndarray, ref: np.ndarray, n: int = 200_000) -> float:
    rng = np.random.RandomState(0)
    low = points.min(axis=0)
    box = ref - low
    samples = low + rng.rand(n, points.shape[1]) * box
    dominated = np.zeros(n, dtype=bool)
    for p in points:
        dominated |= np.all(samples >= p, axis=1)
    return float(dominated.mean() * np.prod(box))


def test_hv_2d_exact() -> None:
    pts = np.array([[1.0, 3.0], [2.0, 2.0], [3.0, 1.0]])
    ref = np.array([4.0, 4.0])
    # Union of rectangles: 3*1 + 2*1 + 1*1 ... compute by sweep: (4-1)(4-3)+(4-2)(3-2)+(4-3)(2-1)=3+2+1=6
    assert compute_hypervolume(pts, ref) == pytest.approx(6.0)


def test_hv_dominated_points_ignored() -> None:
    pts = np.array([[1.0, 1.0], [2.0, 2.0], [1.5, 1.5]])
    ref = np.array([3.0, 3.0])
    assert compute_hypervolume(pts, ref) == pytest.approx(4.0)


def test_hv_point_outside_ref_raises() -> None:
    pts = np.array([[1.0, 5.0], [2.0, 2.0]])
    ref = np.array([4.0, 4.0])
    with pytest.raises(ValueError):
        compute_hypervolume(pts, ref)


def test_hv_point_on_ref_boundary_contributes_zero() -> None:
    pts = np.array([[1.0, 4.0], [2.0, 2.0]])
    ref = np.array([4.0, 4.0])
    assert compute_hypervolume(pts, ref) == pytest.approx(4.0)


def test_hv_empty() -> None:
    assert compute_hypervolume(np.empty((0, 2)), np.array([1.0, 1.0])) == 0.0


def test_hv_infinite_ref() -> None:
    pts = np.array([[0.0, 0.0]])
    assert compute_hypervolume(pts, np.array([np.inf, 1.0])) == float("inf")


@pytest.mark.parametrize("dim", [2, 3, 4])
def test_hv_matches_monte_carlo(dim: int) -> None:
    rng = np.random.RandomState(dim)
    pts = rng.rand(12, dim)
    ref = np.full(dim, 1.1)
    exact = compute_hypervolume(pts, ref)
    approx = _mc_hypervolume(pts, ref)
    assert exact == pytest.approx(approx, rel=0.05)


@pytest.mark.parametrize("dim", [3, 4])
def test_hv_3d_consistent_with_wfg(dim: int) -> None:
    from optuna_amd._hypervolume.wfg import _compute_hv
    from optuna_amd.study._multi_objective import _is_pareto_front

    rng = np.random.RandomState(7)
    pts = rng.rand(10, dim)
    ref = np.full(dim, 1.2)
    uniq = np.unique(pts, axis=0)
    front = uniq[_is_pareto_front(uniq, assume_unique_lexsorted=True)]
    assert compute_hypervolume(pts, ref) == pytest.approx(_compute_hv(front, ref))


@pytest.mark.parametrize("dim", [2, 3])
def test_hssp_greedy_quality(dim: int) -> None:
    rng = np.random.RandomState(42)
    pts = rng.rand(9, dim)
    ref = np.full(dim, 1.1)
    k = 4
    indices = np.arange(len(pts))
    selected = _solve_hssp(pts, indices, k, ref)
    assert len(selected) == k
    assert len(set(selected.tolist())) == k
    hv_greedy = compute_hypervolume(pts[selected], ref)

    hv_best = max(
        compute_hypervolume(pts[list(sub)], ref)
        for sub in itertools.combinations(range(len(pts)), k)
    )
    # Greedy ≥ (1 - 1/e) · OPT; in practice much closer.
    assert hv_greedy >= (1 - 1 / np.e) * hv_best - 1e-12


def test_hssp_subset_equals_all() -> None:
    pts = np.array([[0.1, 0.9], [0.5, 0.5], [0.9, 0.1]])
    out = _solve_hssp(pts, np.array([10, 20, 30]), 3, np.array([1.0, 1.0]))
    assert set(out.tolist()) == {10, 20, 30}
