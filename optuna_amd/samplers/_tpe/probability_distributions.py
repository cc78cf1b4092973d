"""Batched per-dimension mixture components — reference-compatible internal API.

``_MixtureOfProductDistribution`` is the explicit object form of a TPE KDE:
one weight vector over K components and, per search-space dimension, a batched
(K-sized) marginal. My sampler keeps this state in the SoA layout of
``parzen.py`` (feeding the K1–K3 HIP kernels); this module provides the
reference's object layout (``optuna/samplers/_tpe/probability_distributions.py``)
for code and tests that address the internals directly.
``_ParzenEstimator._mixture_distribution`` converts SoA → this form.

Conventions (observable semantics match the reference):
- log dimensions carry their mixture over the log-transformed coordinate and
  ``sample`` exponentiates back; ``log_pdf`` evaluates at ``log(x)`` without a
  jacobian term (consistent with how the ratio l/g is used).
- discrete dimensions integrate the kernel over ``x ± step/2`` and normalize
  by the kernel's total mass over the adapted range.
"""
from __future__ import annotations

from typing import NamedTuple, Union

import numpy as np

from optuna_amd.samplers._tpe import _truncnorm_np as _tn


class _BatchedCategoricalDistributions(NamedTuple):
    weights: np.ndarray  # (K, n_choices)


class _BatchedTruncNormDistributions(NamedTuple):
    mu: np.ndarray  # (K,)
    sigma: np.ndarray  # (K,)
    low: float
    high: float

    @property
    def adapted_low(self) -> float:
        return self.low

    @property
    def adapted_high(self) -> float:
        return self.high

    @property
    def is_log(self) -> bool:
        return False

    @property
    def step(self) -> float:
        return 0.0


class _BatchedTruncLogNormDistributions(NamedTuple):
    mu: np.ndarray  # (K,) in log space
    sigma: np.ndarray  # (K,)
    low: float  # original domain
    high: float

    @property
    def adapted_low(self) -> float:
        return float(np.log(self.low))

    @property
    def adapted_high(self) -> float:
        return float(np.log(self.high))

    @property
    def is_log(self) -> bool:
        return True

    @property
    def step(self) -> float:
        return 0.0


class _BatchedDiscreteTruncNormDistributions(NamedTuple):
    mu: np.ndarray  # (K,)
    sigma: np.ndarray  # (K,)
    low: float  # grid endpoint (inclusive)
    high: float
    step: float

    @property
    def adapted_low(self) -> float:
        return self.low - self.step / 2

    @property
    def adapted_high(self) -> float:
        return self.high + self.step / 2

    @property
    def is_log(self) -> bool:
        return False


class _BatchedDiscreteTruncLogNormDistributions(NamedTuple):
    mu: np.ndarray  # (K,) in log space
    sigma: np.ndarray  # (K,)
    low: float  # original (un-logged) grid endpoint
    high: float
    step: float

    @property
    def adapted_low(self) -> float:
        return float(np.log(self.low - self.step / 2))

    @property
    def adapted_high(self) -> float:
        return float(np.log(self.high + self.step / 2))

    @property
    def is_log(self) -> bool:
        return True


_BatchedDistributions = Union[
    _BatchedCategoricalDistributions,
    _BatchedTruncNormDistributions,
    _BatchedTruncLogNormDistributions,
    _BatchedDiscreteTruncNormDistributions,
    _BatchedDiscreteTruncLogNormDistributions,
]


class _MixtureOfProductDistribution(NamedTuple):
    weights: np.ndarray  # (K,)
    distributions: list[_BatchedDistributions]

    def sample(self, rng: np.random.RandomState, batch_size: int) -> np.ndarray:
        chosen = rng.choice(len(self.weights), p=self.weights, size=batch_size)
        out = np.empty((batch_size, len(self.distributions)), dtype=np.float64)
        for i, d in enumerate(self.distributions):
            if isinstance(d, _BatchedCategoricalDistributions):
                probs = d.weights[chosen, :]
                cum = np.cumsum(probs, axis=-1)
                cum[:, -1] = 1.0  # guard rounding in the final bin
                q = rng.rand(batch_size)
                out[:, i] = (cum < q[:, None]).sum(axis=-1)
                continue
            mu = d.mu[chosen]
            sigma = d.sigma[chosen]
            draw = _tn.rvs(
                a=(d.adapted_low - mu) / sigma,
                b=(d.adapted_high - mu) / sigma,
                loc=mu,
                scale=sigma,
                random_state=rng,
            )
            if d.is_log:
                draw = np.exp(draw)
            if d.step:
                draw = np.clip(
                    d.low + np.round((draw - d.low) / d.step) * d.step, d.low, d.high
                )
            out[:, i] = draw
        return out

    def log_pdf(self, x: np.ndarray) -> np.ndarray:
        n = len(x)
        K = len(self.weights)
        # (n, K) accumulated log density over the product of dimensions.
        acc = np.zeros((n, K), dtype=np.float64)
        for i, d in enumerate(self.distributions):
            xi = x[:, i]
            if isinstance(d, _BatchedCategoricalDistributions):
                with np.errstate(divide="ignore"):
                    acc += np.log(d.weights.T[xi.astype(np.int64)])
                continue
            a = (d.adapted_low - d.mu) / d.sigma
            b = (d.adapted_high - d.mu) / d.sigma
            if d.step:
                # Mass of the kernel over the grid cell around xi.
                half = d.step / 2
                lo = np.log(xi - half) if d.is_log else xi - half
                hi = np.log(xi + half) if d.is_log else xi + half
                cell = _tn._log_gauss_mass(
                    (lo[:, None] - d.mu) / d.sigma, (hi[:, None] - d.mu) / d.sigma
                )
                acc += cell - _tn._log_gauss_mass(a, b)[None, :]
            else:
                xv = np.log(xi) if d.is_log else xi
                acc += _tn.logpdf(xv[:, None], a=a, b=b, loc=d.mu, scale=d.sigma)
        with np.errstate(divide="ignore"):
            acc += np.log(self.weights)[None, :]
        peak = acc.max(axis=1)
        peak[np.isneginf(peak)] = 0.0  # all-zero rows: avoid (-inf) - (-inf)
        with np.errstate(divide="ignore"):
            return np.log(np.exp(acc - peak[:, None]).sum(axis=1)) + peak
