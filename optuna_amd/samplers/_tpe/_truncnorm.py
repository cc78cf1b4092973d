"""Truncated standard normal — reference-compatible module path.

The implementation lives in ``_truncnorm_np`` (host numpy/scipy path, the
golden model for the K3 HIP kernel); this module mirrors the reference's
``optuna/samplers/_tpe/_truncnorm.py`` import location.
"""
from optuna_amd.samplers._tpe._truncnorm_np import (  # noqa: F401
    _log_gauss_mass,
    _log_ndtr,
    _ndtr,
    _ndtri_exp,
    logpdf,
    ppf,
    rvs,
)
