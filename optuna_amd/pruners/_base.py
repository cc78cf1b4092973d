"""BasePruner.

Parity: reference ``optuna/pruners/_base.py`` :11-15.
"""
from __future__ import annotations

import abc
from typing import TYPE_CHECKING

from optuna_amd.trial import FrozenTrial


if TYPE_CHECKING:
    from optuna_amd.study import Study


class BasePruner(abc.ABC):
    @abc.abstractmethod
    def prune(self, study: "Study", trial: FrozenTrial) -> bool:
        """Judge whether the trial should be pruned at its last reported step."""
        raise NotImplementedError
