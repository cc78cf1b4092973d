"""Deterministic pruner fake (parity: reference ``optuna/testing/pruners.py`` :6-11)."""
from __future__ import annotations

from typing import TYPE_CHECKING

from optuna_amd.pruners import BasePruner
from optuna_amd.trial import FrozenTrial


if TYPE_CHECKING:
    from optuna_amd.study import Study


class DeterministicPruner(BasePruner):
    def __init__(self, is_pruning: bool) -> None:
        self.is_pruning = is_pruning

    def prune(self, study: "Study", trial: FrozenTrial) -> bool:
        return self.is_pruning
