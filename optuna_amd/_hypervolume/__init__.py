from optuna_amd._hypervolume.hssp import _solve_hssp
from optuna_amd._hypervolume.wfg import compute_hypervolume


__all__ = ["compute_hypervolume", "_solve_hssp"]
