"""Abstract trial interface shared by live, frozen and fixed trials.

Parity: reference ``optuna/trial/_base.py`` (BaseTrial :12).
"""
from __future__ import annotations

import abc
from datetime import datetime
from typing import Any, Sequence

from optuna_amd.distributions import BaseDistribution, CategoricalChoiceType


class BaseTrial(abc.ABC):
    """Base class for trials: the ``suggest_*`` surface user objectives program against."""

    def suggest_float(
        self,
        name: str,
        low: float,
        high: float,
        *,
        step: float | None = None,
        log: bool = False,
    ) -> float:
        raise NotImplementedError

    def suggest_uniform(self, name: str, low: float, high: float) -> float:
        raise NotImplementedError

    def suggest_loguniform(self, name: str, low: float, high: float) -> float:
        raise NotImplementedError

    def suggest_discrete_uniform(self, name: str, low: float, high: float, q: float) -> float:
        raise NotImplementedError

    def suggest_int(self, name: str, low: int, high: int, step: int = 1, log: bool = False) -> int:
        raise NotImplementedError

    def suggest_categorical(
        self, name: str, choices: Sequence[CategoricalChoiceType]
    ) -> CategoricalChoiceType:
        raise NotImplementedError

    def report(self, value: float, step: int) -> None:
        raise NotImplementedError

    def should_prune(self) -> bool:
        raise NotImplementedError

    def set_user_attr(self, key: str, value: Any) -> None:
        raise NotImplementedError

    def set_system_attr(self, key: str, value: Any) -> None:
        raise NotImplementedError

    @property
    def params(self) -> dict[str, Any]:
        raise NotImplementedError

    @property
    def distributions(self) -> dict[str, BaseDistribution]:
        raise NotImplementedError

    @property
    def user_attrs(self) -> dict[str, Any]:
        raise NotImplementedError

    @property
    def system_attrs(self) -> dict[str, Any]:
        raise NotImplementedError

    @property
    def datetime_start(self) -> datetime | None:
        raise NotImplementedError

    @property
    def number(self) -> int:
        raise NotImplementedError
