"""Host (numpy/scipy) truncated standard normal: ppf / rvs / logpdf / log mass.

This is the *reference* implementation the HIP device library
(``optuna_amd/_hip/kernels/truncnorm.hip`` — kernel K3) is tested against: both
implement the same tail-stable branch structure.

Math (not code) follows the classic scipy formulation the reference vendors
(reference ``optuna/samplers/_tpe/_truncnorm.py``): all CDF arithmetic happens in
log space with the left-tail symmetry trick, ``ndtri_exp`` inverts log-CDFs.
We use scipy.special's C implementations directly instead of vendoring.
"""
from __future__ import annotations

import numpy as np
from scipy import special as sc


_LOG_2 = float(np.log(2.0))
_NORM_CONST = -0.5 * float(np.log(2.0 * np.pi))


def _ndtr(x: np.ndarray) -> np.ndarray:
    return sc.ndtr(x)


def _log_ndtr(x: np.ndarray) -> np.ndarray:
    return sc.log_ndtr(x)


def _ndtri_exp(log_p: np.ndarray) -> np.ndarray:
    """Inverse of log_ndtr."""
    return sc.ndtri_exp(log_p)


def _log_gauss_mass(a: np.ndarray, b: np.ndarray) -> np.ndarray:
    """log( Phi(b) - Phi(a) ), elementwise, tail-stable.

    Three regimes (case split on the interval position):
    * b <= 0 (left tail):   log Phi(b) + log1p(-exp(log Phi(a) - log Phi(b)))
    * a > 0  (right tail):  by symmetry = mass(-b, -a) in the left tail
    * straddling zero:      log1p(-Phi(a) - Phi(-b)) in linear space (central mass
                            is large, no cancellation problem)
    """
    a = np.asarray(a, dtype=np.float64)
    b = np.asarray(b, dtype=np.float64)
    a, b = np.broadcast_arrays(a, b)
    out = np.empty(a.shape, dtype=np.float64)

    case_left = b <= 0
    case_right = a > 0
    case_central = ~(case_left | case_right)

    if np.any(case_left):
        a_l, b_l = a[case_left], b[case_left]
        log_b = _log_ndtr(b_l)
        with np.errstate(invalid="ignore"):
            diff = _log_ndtr(a_l) - log_b
            out[case_left] = log_b + np.log1p(-np.exp(diff))
    if np.any(case_right):
        a_r, b_r = a[case_right], b[case_right]
        log_b = _log_ndtr(-a_r)
        with np.errstate(invalid="ignore"):
            diff = _log_ndtr(-b_r) - log_b
            out[case_right] = log_b + np.log1p(-np.exp(diff))
    if np.any(case_central):
        a_c, b_c = a[case_central], b[case_central]
        with np.errstate(divide="ignore"):
            out[case_central] = np.log1p(-_ndtr(a_c) - _ndtr(-b_c))
    # Degenerate interval (a == b, or numerically-empty mass) → -inf.
    out[~(a < b)] = -np.inf
    return out


def ppf(q: np.ndarray, a: np.ndarray, b: np.ndarray) -> np.ndarray:
    """Quantile of the standard normal truncated to [a, b]; q in [0, 1]."""
    q = np.asarray(q, dtype=np.float64)
    a = np.asarray(a, dtype=np.float64)
    b = np.asarray(b, dtype=np.float64)
    q, a, b = np.broadcast_arrays(q, a, b)
    out = np.empty(q.shape, dtype=np.float64)

    case_left = a < 0
    case_right = ~case_left

    def _ppf_left(qq: np.ndarray, aa: np.ndarray, bb: np.ndarray) -> np.ndarray:
        # Phi(x) = Phi(a) + q * mass → log Phi(x) = logaddexp(log Phi(a), log q + log mass)
        with np.errstate(divide="ignore"):
            log_phi_x = np.logaddexp(
                _log_ndtr(aa), np.log(qq) + _log_gauss_mass(aa, bb)
            )
        return _ndtri_exp(log_phi_x)

    def _ppf_right(qq: np.ndarray, aa: np.ndarray, bb: np.ndarray) -> np.ndarray:
        # Symmetric: 1 - Phi(x) side.
        with np.errstate(divide="ignore"):
            log_sf_x = np.logaddexp(
                _log_ndtr(-bb), np.log1p(-qq) + _log_gauss_mass(aa, bb)
            )
        return -_ndtri_exp(log_sf_x)

    if np.any(case_left):
        out[case_left] = _ppf_left(q[case_left], a[case_left], b[case_left])
    if np.any(case_right):
        out[case_right] = _ppf_right(q[case_right], a[case_right], b[case_right])

    out = np.clip(out, a, b)
    # Degenerate (zero-width) interval carries no distribution: NaN, as scipy
    # defines it. np.where keeps 0-d inputs working (clip may return a scalar).
    return np.where(a >= b, np.nan, out)


def rvs(
    a: np.ndarray,
    b: np.ndarray,
    loc: np.ndarray | float = 0.0,
    scale: np.ndarray | float = 1.0,
    random_state: np.random.RandomState | None = None,
) -> np.ndarray:
    rng = random_state or np.random.RandomState()
    size = np.broadcast(a, b, loc, scale).shape
    quantiles = rng.uniform(low=0, high=1, size=size)
    return ppf(quantiles, a, b) * scale + loc


def logpdf(
    x: np.ndarray,
    a: np.ndarray,
    b: np.ndarray,
    loc: np.ndarray | float = 0.0,
    scale: np.ndarray | float = 1.0,
) -> np.ndarray:
    """log density of a normal(loc, scale) truncated to [loc+a*scale, loc+b*scale],
    evaluated at x (x given in the *unstandardized* domain)."""
    x_std = (np.asarray(x, dtype=np.float64) - loc) / scale
    out = (
        _NORM_CONST
        - 0.5 * x_std * x_std
        - _log_gauss_mass(a, b)
        - np.log(np.asarray(scale, dtype=np.float64))
    )
    outside = (x_std < a) | (x_std > b)
    if np.any(outside):
        out = np.where(outside, -np.inf, out)
    # Degenerate (zero-width) truncation has no density: NaN, as scipy defines.
    degenerate = np.asarray(a) >= np.asarray(b)
    if np.any(degenerate):
        out = np.where(degenerate, np.nan, out)
    return out
