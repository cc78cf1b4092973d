"""Device dispatch for the TPE hot path (K1 fit + K2 mixture log-pdf on gfx950).

Used by ``TPESampler._sample`` for the "above" KDE when the history is large:
the (K≈N)×D Parzen fit and the S×K×D log-pdf run in ``optuna_amd._hip._hipcore``
kernels; the host never materializes mus/sigmas for the big estimator at all.

Eligibility: all dimensions continuous (optionally log-scaled). Discrete and
categorical dimensions currently keep the (already vectorized) host path.
"""
from __future__ import annotations

import math
from typing import TYPE_CHECKING

import numpy as np

from optuna_amd import _hip
from optuna_amd.distributions import (
    BaseDistribution,
    FloatDistribution,
    IntDistribution,
)


# Below this kernel count the host fit is cheaper than a kernel launch round trip.
DEVICE_MIN_KERNELS = 512


def space_is_device_eligible(space: dict[str, BaseDistribution]) -> bool:
    if not space:
        return False
    for dist in space.values():
        if not isinstance(dist, (FloatDistribution, IntDistribution)):
            return False
        if isinstance(dist, IntDistribution):
            return False  # int domains are discrete cells
        if dist.step is not None:
            return False
    return True


def device_ready(n_kernels: int) -> bool:
    return n_kernels >= DEVICE_MIN_KERNELS and _hip.is_available()


def kde_logpdf(
    space: dict[str, BaseDistribution],
    observations: dict[str, np.ndarray],
    orders: dict[str, np.ndarray] | None,
    weights: np.ndarray,
    samples: dict[str, np.ndarray],
    consider_endpoints: bool,
    consider_magic_clip: bool,
) -> np.ndarray:
    """log mixture pdf of `samples` under the KDE fit to `observations` (device)."""
    core = _hip.get()
    assert core is not None
    names = list(space.keys())
    D = len(names)
    N = len(observations[names[0]])

    obs = np.empty((N, D), dtype=np.float64)
    x = np.column_stack([np.asarray(samples[n], dtype=np.float64) for n in names])
    alow = np.empty(D)
    ahigh = np.empty(D)
    sorted_pos = np.empty((N, D), dtype=np.int64)
    for c, name in enumerate(names):
        dist = space[name]
        assert isinstance(dist, FloatDistribution) and dist.step is None
        col = np.asarray(observations[name], dtype=np.float64)
        if dist.log:
            obs[:, c] = np.log(col)
            x[:, c] = np.log(x[:, c])
            alow[c] = math.log(dist.low)
            ahigh[c] = math.log(dist.high)
        else:
            obs[:, c] = col
            alow[c] = dist.low
            ahigh[c] = dist.high
        if orders is not None:
            sorted_pos[:, c] = orders[name]
        else:
            sorted_pos[:, c] = np.argsort(col, kind="stable")

    with np.errstate(divide="ignore"):
        logw = np.log(weights)
    return core.kde_logpdf(
        obs,
        sorted_pos,
        logw,
        alow,
        ahigh,
        x,
        consider_endpoints,
        consider_magic_clip,
    )
