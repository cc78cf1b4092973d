"""Sampling-only acquisition optimizer (parity: reference ``optuna/_gp/optim_sample.py``)."""
from __future__ import annotations

from typing import TYPE_CHECKING

import numpy as np


if TYPE_CHECKING:
    from optuna_amd._gp import acqf as acqf_module


def optimize_acqf_sample(
    acqf: "acqf_module.BaseAcquisitionFunc",
    *,
    n_samples: int = 2048,
    rng: np.random.RandomState | None = None,
) -> tuple[np.ndarray, float]:
    xs = acqf.search_space.sample_normalized_params(n_samples, rng=rng)
    vals = acqf.eval_acqf_no_grad(xs)
    best = int(np.argmax(vals))
    return xs[best, :], float(vals[best])
