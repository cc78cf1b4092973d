"""Pruner that never prunes.

Parity: reference ``optuna/pruners/_nop.py`` :13.
"""
from __future__ import annotations

from typing import TYPE_CHECKING

from optuna_amd.pruners._base import BasePruner
from optuna_amd.trial import FrozenTrial


if TYPE_CHECKING:
    from optuna_amd.study import Study


class NopPruner(BasePruner):
    def prune(self, study: "Study", trial: FrozenTrial) -> bool:
        return False
