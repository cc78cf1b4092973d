from optuna_amd.parallel.collective import CollectiveOpPlane

__all__ = ["CollectiveOpPlane"]
