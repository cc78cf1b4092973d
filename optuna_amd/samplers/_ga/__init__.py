from optuna_amd.samplers._ga._base import BaseGASampler


__all__ = ["BaseGASampler"]
