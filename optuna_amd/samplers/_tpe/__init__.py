from optuna_amd.samplers._tpe.sampler import TPESampler


__all__ = ["TPESampler"]
