"""Lazy integration shim.

The reference delegates framework integrations (LightGBM/PyTorch/MLflow/...)
to the external ``optuna-integration`` distribution and keeps only a lazy
re-export shim here (reference ``optuna/integration/__init__.py`` :12-32).
This build keeps the same surface: attribute access attempts the external
package and raises a clear error when it is not installed.
"""
from __future__ import annotations

import importlib
from typing import Any


_INTEGRATION_MODULES = {
    "BoTorchSampler": "botorch",
    "CatBoostPruningCallback": "catboost",
    "DaskStorage": "dask",
    "FastAIPruningCallback": "fastai",
    "LightGBMPruningCallback": "lightgbm",
    "LightGBMTuner": "lightgbm",
    "LightGBMTunerCV": "lightgbm",
    "MLflowCallback": "mlflow",
    "OptunaSearchCV": "sklearn",
    "PyTorchIgnitePruningHandler": "pytorch_ignite",
    "PyTorchLightningPruningCallback": "pytorch_lightning",
    "TensorBoardCallback": "tensorboard",
    "TorchDistributedTrial": "pytorch_distributed",
    "WeightsAndBiasesCallback": "wandb",
    "XGBoostPruningCallback": "xgboost",
}

__all__ = list(_INTEGRATION_MODULES)


def __getattr__(name: str) -> Any:
    if name in _INTEGRATION_MODULES:
        submodule = _INTEGRATION_MODULES[name]
        try:
            mod = importlib.import_module(f"optuna_integration.{submodule}")
            return getattr(mod, name)
        except ImportError as e:
            raise ImportError(
                f"`optuna_amd.integration.{name}` requires the `optuna-integration` "
                f"package (submodule `{submodule}`), which is not installed in this "
                "environment."
            ) from e
    raise AttributeError(f"module {__name__!r} has no attribute {name!r}")
