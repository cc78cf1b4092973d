"""Scrambled-Sobol standard-normal samples, device-resident (K9).

The reference draws torch's CPU SobolEngine and uploads
(reference ``optuna/_gp/qmc.py`` :19-27). Here the Sobol32 digital sequence is
built from scipy's Joe–Kuo direction numbers with plain torch integer ops —
30 XOR steps over the Gray-coded index — so the same code path produces
bit-identical draws on the host and on the MI355X, and a device-resident GP's
fantasy/EHVI sampling never stages through host memory. Scrambling is a
seeded per-dimension random digital shift (a standard scrambled-QMC
construction; the reference's Owen scrambling differs in the exact point set
but not in the QMC properties the samplers rely on).
"""
from __future__ import annotations

import math
from typing import TYPE_CHECKING

import numpy as np


if TYPE_CHECKING:
    import torch
else:
    from optuna_amd._imports import _LazyImport

    torch = _LazyImport("torch")

_SQRT_2 = math.sqrt(2)
_BITS = 30

# Direction-number cache per dimension count (host int64 (dim, _BITS)).
_v_cache: dict[int, np.ndarray] = {}


def _direction_numbers(dim: int) -> np.ndarray:
    v = _v_cache.get(dim)
    if v is None:
        import scipy.stats._sobol as _sobol

        v = np.zeros((dim, _BITS), dtype=np.uint64)
        _sobol._initialize_v(v, dim=dim, bits=_BITS)
        v = v.astype(np.int64)
        _v_cache[dim] = v
    return v


def sobol_uniform(
    dim: int, n_samples: int, seed: int, device: "torch.device | None" = None
) -> "torch.Tensor":
    """(n_samples, dim) scrambled-Sobol uniforms in (0, 1), fp64, on ``device``."""
    v = torch.from_numpy(_direction_numbers(dim))
    if device is not None:
        v = v.to(device)
    idx = torch.arange(n_samples, dtype=torch.int64, device=v.device)
    gray = idx ^ (idx >> 1)
    x = torch.zeros((n_samples, dim), dtype=torch.int64, device=v.device)
    for j in range(_BITS):
        bit = ((gray >> j) & 1).to(torch.bool)
        x = torch.where(bit.unsqueeze(-1), x ^ v[:, j], x)
    # Seeded per-dim digital shift (scramble) + half-ulp centering so 0 is
    # never emitted (ndtri(-inf) guards).
    shift = np.random.RandomState(seed).randint(0, 1 << _BITS, size=dim, dtype=np.int64)
    x = x ^ torch.from_numpy(shift).to(v.device)
    return (x.to(torch.float64) + 0.5) / float(1 << _BITS)


def sample_from_normal_sobol(
    dim: int, n_samples: int, seed: int, device: "torch.device | None" = None
) -> "torch.Tensor":
    """(n_samples, dim) standard-normal scrambled-QMC draws on ``device``."""
    u = sobol_uniform(dim, n_samples, seed, device)
    # (0,1) → (-1,1) → standard normal via erfinv.
    return torch.erfinv(2.0 * (u - 0.5)) * _SQRT_2
