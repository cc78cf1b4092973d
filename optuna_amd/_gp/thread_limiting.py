"""Prevent BLAS/torch thread oversubscription during scipy-driven optimization.

Parity: reference ``optuna/_gp/thread_limiting.py`` (limit_threads_in_optimization
:23 — torch intra-op threads → 1 and OPENBLAS_NUM_THREADS → 1 while optimizing).
On a GPU box the HIP stream scheduler replaces this concern for device work; the
limiter still applies to the host-side scipy L-BFGS-B driver.
"""
from __future__ import annotations

import contextlib
import os
from typing import Generator


@contextlib.contextmanager
def limit_threads_in_optimization() -> Generator[None, None, None]:
    import torch

    n_torch_threads = torch.get_num_threads()
    old_openblas = os.environ.get("OPENBLAS_NUM_THREADS")
    try:
        torch.set_num_threads(1)
        os.environ["OPENBLAS_NUM_THREADS"] = "1"
        yield
    finally:
        torch.set_num_threads(n_torch_threads)
        if old_openblas is None:
            os.environ.pop("OPENBLAS_NUM_THREADS", None)
        else:
            os.environ["OPENBLAS_NUM_THREADS"] = old_openblas
