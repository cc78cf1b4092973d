from optuna_amd.samplers._cmaes._sampler import CmaEsSampler


__all__ = ["CmaEsSampler"]
