"""Parzen estimator — reference-compatible module path.

The implementation lives in ``parzen.py`` (SoA layout feeding the K1-K3 HIP
kernels); this module mirrors the reference's
``optuna/samplers/_tpe/parzen_estimator.py`` import location.
"""
from optuna_amd.samplers._tpe.parzen import (  # noqa: F401
    _ParzenEstimator,
    _ParzenEstimatorParameters,
)
