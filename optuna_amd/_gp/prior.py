"""Log-prior over GP kernel hyperparameters.

Parity: reference ``optuna/_gp/prior.py`` :16-33 (gamma(2,1) on kernel scale,
gamma(1.1,30) on noise, hand-crafted prior on inverse squared lengthscales).
"""
from __future__ import annotations

from typing import TYPE_CHECKING


if TYPE_CHECKING:
    import torch

    from optuna_amd._gp import gp
else:
    from optuna_amd._imports import _LazyImport

    torch = _LazyImport("torch")


DEFAULT_MINIMUM_NOISE_VAR = 1e-6


def default_log_prior(gpr: "gp.GPRegressor") -> "torch.Tensor":
    def gamma_log_pdf_unnormalized(
        x: "torch.Tensor", concentration: float, rate: float
    ) -> "torch.Tensor":
        return (concentration - 1) * torch.log(x) - rate * x

    # Penalize both extremely large and extremely small inverse squared
    # lengthscales (i.e., keep lengthscales in a sane band).
    return (
        -(
            0.1 / gpr.inverse_squared_lengthscales
            + 0.1 * gpr.inverse_squared_lengthscales
        ).sum()
        + gamma_log_pdf_unnormalized(gpr.kernel_scale, 2, 1)
        + gamma_log_pdf_unnormalized(gpr.noise_var, 1.1, 30)
    )
