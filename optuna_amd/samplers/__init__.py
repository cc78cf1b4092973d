"""Sampler registry.

Parity: reference ``optuna/samplers/__init__.py``.
"""
from optuna_amd.samplers._base import BaseSampler
from optuna_amd.samplers._lazy_random_state import LazyRandomState  # noqa: F401
from optuna_amd.samplers._random import RandomSampler
from optuna_amd.samplers._tpe.sampler import TPESampler


__all__ = [
    "nsgaii",
    "BaseGASampler",
    "BaseSampler",
    "BruteForceSampler",
    "CmaEsSampler",
    "GPSampler",
    "GridSampler",
    "NSGAIISampler",
    "NSGAIIISampler",
    "PartialFixedSampler",
    "QMCSampler",
    "RandomSampler",
    "TPESampler",
]


def __getattr__(name: str):  # lazy heavy/optional samplers
    if name == "GridSampler":
        from optuna_amd.samplers._grid import GridSampler

        return GridSampler
    if name == "QMCSampler":
        from optuna_amd.samplers._qmc import QMCSampler

        return QMCSampler
    if name == "BruteForceSampler":
        from optuna_amd.samplers._brute_force import BruteForceSampler

        return BruteForceSampler
    if name == "PartialFixedSampler":
        from optuna_amd.samplers._partial_fixed import PartialFixedSampler

        return PartialFixedSampler
    if name == "CmaEsSampler":
        from optuna_amd.samplers._cmaes import CmaEsSampler

        return CmaEsSampler
    if name == "GPSampler":
        from optuna_amd.samplers._gp.sampler import GPSampler

        return GPSampler
    if name == "NSGAIISampler":
        from optuna_amd.samplers.nsgaii import NSGAIISampler

        return NSGAIISampler
    if name == "NSGAIIISampler":
        from optuna_amd.samplers._nsgaiii import NSGAIIISampler

        return NSGAIIISampler
    if name == "BaseGASampler":
        from optuna_amd.samplers._ga._base import BaseGASampler

        return BaseGASampler
    raise AttributeError(f"module {__name__!r} has no attribute {name!r}")
