from optuna_amd.samplers._gp.sampler import GPSampler


__all__ = ["GPSampler"]
