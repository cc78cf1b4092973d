"""Canned objectives (parity: reference ``optuna/testing/objectives.py``)."""
from __future__ import annotations

from typing import TYPE_CHECKING

from optuna_amd.exceptions import TrialPruned


if TYPE_CHECKING:
    from optuna_amd.trial import Trial


def fail_objective(_: "Trial") -> float:
    raise ValueError("intentional failure")


def pruned_objective(trial: "Trial") -> float:
    raise TrialPruned()
