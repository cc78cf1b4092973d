"""Pruner registry and the Hyperband study-filter hook.

Parity: reference ``optuna/pruners/__init__.py`` (_filter_study :32-37 — samplers
must observe the bracket-filtered study when HyperbandPruner is active).
"""
from __future__ import annotations

from typing import TYPE_CHECKING

from optuna_amd.pruners._base import BasePruner
from optuna_amd.pruners._hyperband import HyperbandPruner
from optuna_amd.pruners._median import MedianPruner
from optuna_amd.pruners._nop import NopPruner
from optuna_amd.pruners._patient import PatientPruner
from optuna_amd.pruners._percentile import PercentilePruner
from optuna_amd.pruners._successive_halving import SuccessiveHalvingPruner
from optuna_amd.pruners._threshold import ThresholdPruner
from optuna_amd.pruners._wilcoxon import WilcoxonPruner


if TYPE_CHECKING:
    from optuna_amd.study import Study
    from optuna_amd.trial import FrozenTrial

__all__ = [
    "BasePruner",
    "HyperbandPruner",
    "MedianPruner",
    "NopPruner",
    "PatientPruner",
    "PercentilePruner",
    "SuccessiveHalvingPruner",
    "ThresholdPruner",
    "WilcoxonPruner",
]


def _filter_study(study: "Study", trial: "FrozenTrial") -> "Study":
    if isinstance(study.pruner, HyperbandPruner):
        # Create a bracket-local study view so samplers only see sibling trials.
        pruner: HyperbandPruner = study.pruner
        return pruner._create_bracket_study(study, pruner._get_bracket_id(study, trial))
    return study
