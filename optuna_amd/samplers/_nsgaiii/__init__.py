from optuna_amd.samplers._nsgaiii._sampler import NSGAIIISampler


__all__ = ["NSGAIIISampler"]
