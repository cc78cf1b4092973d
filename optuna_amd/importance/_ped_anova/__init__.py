"""PED-ANOVA package facade; implementation in ``_ped_anova_impl``."""
from optuna_amd.importance._ped_anova_impl import (  # noqa: F401
    PedAnovaImportanceEvaluator,
    _QuantileFilter,
)
