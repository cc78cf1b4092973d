"""Scrambled-Sobol standard-normal samples (parity: reference ``optuna/_gp/qmc.py``)."""
from __future__ import annotations

import math
from typing import TYPE_CHECKING


if TYPE_CHECKING:
    import torch
else:
    from optuna_amd._imports import _LazyImport

    torch = _LazyImport("torch")

_SQRT_2 = math.sqrt(2)


def sample_from_normal_sobol(dim: int, n_samples: int, seed: int) -> "torch.Tensor":
    sobol = torch.quasirandom.SobolEngine(dimension=dim, scramble=True, seed=seed).draw(
        n_samples, dtype=torch.float64
    )
    # [0,1) → (-1,1) → standard normal via erfinv.
    return torch.erfinv(2.0 * (sobol - 0.5)) * _SQRT_2
