"""Mixed continuous/discrete acquisition optimizer.

QMC presample → roulette-wheel warm starts → batched local search alternating
lengthscale-preconditioned L-BFGS-B on the continuous dims with per-dim discrete
moves (exhaustive ≤16 choices, interpolated Brent line search otherwise).

Parity: reference ``optuna/_gp/optim_mixed.py`` (_gradient_ascent_batched :29,
discrete search :97-205, local_search_mixed_batched :232, optimize_acqf_mixed
:280).
"""
from __future__ import annotations

import math
from typing import TYPE_CHECKING

import numpy as np

from optuna_amd import logging as _logging
from optuna_amd._gp import batched_lbfgsb
from optuna_amd._gp.thread_limiting import limit_threads_in_optimization


if TYPE_CHECKING:
    from optuna_amd._gp.acqf import BaseAcquisitionFunc

_logger = _logging.get_logger(__name__)


def _gradient_ascent_batched(
    acqf: "BaseAcquisitionFunc",
    initial_params_batched: np.ndarray,
    initial_fvals: np.ndarray,
    continuous_indices: np.ndarray,
    lengthscales: np.ndarray,
    tol: float,
) -> tuple[np.ndarray, np.ndarray, np.ndarray]:
    """L-BFGS-B on z = x/ℓ (lengthscale preconditioning equalizes curvature)."""
    assert initial_params_batched.ndim == 2
    if len(continuous_indices) == 0:
        return initial_params_batched, initial_fvals, np.zeros(len(initial_fvals), dtype=bool)

    def negative_acqf_with_grad(
        scaled_x: np.ndarray, fixed_params: list[np.ndarray]
    ) -> tuple[np.ndarray, np.ndarray]:
        next_params = np.array(fixed_params)
        next_params[:, continuous_indices] = scaled_x * lengthscales
        fvals, grads = acqf.eval_acqf_batched_with_grad(next_params)
        return (-fvals, -grads[:, continuous_indices] * lengthscales)

    with limit_threads_in_optimization():
        scaled_opt, neg_fvals_opt, n_iterations = batched_lbfgsb.batched_lbfgsb(
            func_and_grad=negative_acqf_with_grad,
            x0_batched=initial_params_batched[:, continuous_indices] / lengthscales,
            batched_args=([p for p in initial_params_batched.copy()],),
            bounds=[(0, 1 / s) for s in lengthscales],
            pgtol=math.sqrt(tol),
            max_iters=200,
        )
    xs_opt = initial_params_batched.copy()
    xs_opt[:, continuous_indices] = scaled_opt * lengthscales
    fvals_opt = -neg_fvals_opt
    updated = (fvals_opt > initial_fvals) & (n_iterations > 0)
    return (
        np.where(updated[:, None], xs_opt, initial_params_batched),
        np.where(updated, fvals_opt, initial_fvals),
        updated,
    )


def _exhaustive_search(
    acqf: "BaseAcquisitionFunc",
    initial_params: np.ndarray,
    initial_fval: float,
    param_idx: int,
    choices: np.ndarray,
) -> tuple[np.ndarray, float, bool]:
    if len(choices) == 1:
        return initial_params, initial_fval, False
    other = choices[choices != initial_params[param_idx]]
    candidates = np.repeat(initial_params[None, :], len(other), axis=0)
    candidates[:, param_idx] = other
    fvals = acqf.eval_acqf_no_grad(candidates)
    best = int(np.argmax(fvals))
    if fvals[best] > initial_fval:
        return candidates[best], float(fvals[best]), True
    return initial_params, initial_fval, False


def _discrete_line_search(
    acqf: "BaseAcquisitionFunc",
    initial_params: np.ndarray,
    initial_fval: float,
    param_idx: int,
    grids: np.ndarray,
    xtol: float,
) -> tuple[np.ndarray, float, bool]:
    """Brent on a piecewise-linear interpolation of the grid acqf values."""
    import scipy.optimize

    if len(grids) == 1:
        return initial_params, initial_fval, False

    def nearest_index(x: float) -> int:
        i = int(np.clip(np.searchsorted(grids, x), 1, len(grids) - 1))
        return i - 1 if abs(x - grids[i - 1]) < abs(x - grids[i]) else i

    current_i = nearest_index(initial_params[param_idx])
    cache = {current_i: -initial_fval}
    probe = initial_params.copy()

    def neg_at(i: int) -> float:
        if i in cache:
            return cache[i]
        probe[param_idx] = grids[i]
        val = -float(acqf.eval_acqf_no_grad(probe))
        cache[i] = val
        return val

    def interpolated(x: float) -> float:
        if x < grids[0] or x > grids[-1]:
            return np.inf
        right = int(np.clip(np.searchsorted(grids, x), 1, len(grids) - 1))
        left = right - 1
        w_left = (grids[right] - x) / (grids[right] - grids[left])
        return w_left * neg_at(left) + (1 - w_left) * neg_at(right)

    EPS = 1e-12
    res = scipy.optimize.minimize_scalar(
        interpolated,
        bracket=(grids[0] - EPS, grids[current_i], grids[-1] + EPS),
        method="brent",
        tol=xtol,
    )
    opt_i = nearest_index(res.x)
    fval_opt = -neg_at(opt_i)
    if opt_i != current_i and fval_opt > initial_fval:
        out = initial_params.copy()
        out[param_idx] = grids[opt_i]
        return out, fval_opt, True
    return initial_params, initial_fval, False


_MAX_EXHAUSTIVE_CHOICES = 16


def _local_search_discrete(
    acqf: "BaseAcquisitionFunc",
    initial_params: np.ndarray,
    initial_fval: float,
    param_idx: int,
    choices: np.ndarray,
    xtol: float,
) -> tuple[np.ndarray, float, bool]:
    if acqf.search_space.is_categorical[param_idx] or len(choices) <= _MAX_EXHAUSTIVE_CHOICES:
        return _exhaustive_search(acqf, initial_params, initial_fval, param_idx, choices)
    return _discrete_line_search(acqf, initial_params, initial_fval, param_idx, choices, xtol)


def local_search_mixed_batched(
    acqf: "BaseAcquisitionFunc", xs0: np.ndarray, *, tol: float = 1e-4, max_iter: int = 100
) -> tuple[np.ndarray, np.ndarray]:
    cont_inds = acqf.search_space.continuous_indices
    lengthscales = acqf.length_scales[cont_inds]
    discrete_indices = acqf.search_space.discrete_indices
    discrete_choices = acqf.search_space.get_choices_of_discrete_params()
    discrete_xtols = [
        np.min(np.diff(choices), initial=np.inf) / 4 for choices in discrete_choices
    ]

    best_xs = xs0.copy()
    best_fvals = acqf.eval_acqf_no_grad(best_xs)
    CONTINUOUS = -1
    last_changed = np.full(len(best_xs), CONTINUOUS, dtype=int)
    remaining = np.arange(len(best_xs))

    for _ in range(max_iter):
        best_xs[remaining], best_fvals[remaining], updated = _gradient_ascent_batched(
            acqf, best_xs[remaining], best_fvals[remaining], cont_inds, lengthscales, tol
        )
        last_changed = np.where(updated, CONTINUOUS, last_changed)

        for i, choices, xtol in zip(discrete_indices, discrete_choices, discrete_xtols):
            converged = last_changed == i
            last_changed = last_changed[~converged]
            remaining = remaining[~converged]
            if remaining.size == 0:
                return best_xs, best_fvals
            updated = np.zeros(len(remaining), dtype=bool)
            for b, row in enumerate(remaining):
                best_xs[row], best_fvals[row], updated[b] = _local_search_discrete(
                    acqf, best_xs[row], best_fvals[row], i, choices, xtol
                )
            last_changed = np.where(updated, i, last_changed)

        converged = last_changed == CONTINUOUS
        remaining = remaining[~converged]
        last_changed = last_changed[~converged]
        if remaining.size == 0:
            return best_xs, best_fvals

    _logger.warning("local_search_mixed: Local search did not converge.")
    return best_xs, best_fvals


def optimize_acqf_mixed(
    acqf: "BaseAcquisitionFunc",
    *,
    warmstart_points: np.ndarray | None = None,
    n_preliminary_samples: int = 2048,
    n_local_search: int = 10,
    tol: float = 1e-4,
    rng: np.random.RandomState | None = None,
) -> tuple[np.ndarray, float]:
    rng = rng or np.random.RandomState()
    if warmstart_points is None:
        warmstart_points = np.empty((0, acqf.search_space.dim))
    assert len(warmstart_points) <= n_local_search - 1

    sampled_xs = acqf.search_space.sample_normalized_params(n_preliminary_samples, rng=rng)
    f_vals = acqf.eval_acqf_no_grad(sampled_xs)
    max_i = int(np.argmax(f_vals))

    # Roulette over exp(f - max): softmax-weighted additional warm starts.
    probs = np.exp(f_vals - f_vals[max_i])
    probs[max_i] = 0.0
    probs /= probs.sum()
    n_improving = int(np.count_nonzero(probs > 0.0))
    n_additional = min(
        n_local_search - len(warmstart_points) - 1, n_improving
    )
    if n_additional == n_improving:
        _logger.warning("Study already converged, so the number of local search is reduced.")
    chosen = np.array([max_i])
    if n_additional > 0:
        chosen = np.append(
            chosen, rng.choice(len(sampled_xs), size=n_additional, replace=False, p=probs)
        )
    x_warmstarts = np.vstack([sampled_xs[chosen, :], warmstart_points])
    best_xs, best_fvals = local_search_mixed_batched(acqf, x_warmstarts, tol=tol)
    best_idx = int(np.argmax(best_fvals))
    return best_xs[best_idx], float(best_fvals[best_idx])
