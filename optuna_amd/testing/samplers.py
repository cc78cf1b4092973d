"""Deterministic sampler fakes for tests.

Parity (pattern): reference ``optuna/testing/samplers.py`` (DeterministicSampler
:13-39) and ``optuna/testing/pytest_samplers.py`` (FixedSampler :50-80).
"""
from __future__ import annotations

from typing import TYPE_CHECKING, Any

from optuna_amd.distributions import BaseDistribution
from optuna_amd.samplers import BaseSampler
from optuna_amd.trial import FrozenTrial


if TYPE_CHECKING:
    from optuna_amd.study import Study


class DeterministicSampler(BaseSampler):
    """Always samples from the given relative params dict."""

    def __init__(self, params: dict[str, Any]) -> None:
        self.params = params

    def infer_relative_search_space(
        self, study: "Study", trial: FrozenTrial
    ) -> dict[str, BaseDistribution]:
        return {}

    def sample_relative(
        self, study: "Study", trial: FrozenTrial, search_space: dict[str, BaseDistribution]
    ) -> dict[str, Any]:
        return {}

    def sample_independent(
        self,
        study: "Study",
        trial: FrozenTrial,
        param_name: str,
        param_distribution: BaseDistribution,
    ) -> Any:
        return self.params[param_name]


class FixedSampler(BaseSampler):
    """Relative sampler returning fixed relative params over a fixed search space."""

    def __init__(
        self,
        relative_search_space: dict[str, BaseDistribution],
        relative_params: dict[str, Any],
        unknown_param_value: float,
    ) -> None:
        self.relative_search_space = relative_search_space
        self.relative_params = relative_params
        self.unknown_param_value = unknown_param_value

    def infer_relative_search_space(
        self, study: "Study", trial: FrozenTrial
    ) -> dict[str, BaseDistribution]:
        return self.relative_search_space

    def sample_relative(
        self, study: "Study", trial: FrozenTrial, search_space: dict[str, BaseDistribution]
    ) -> dict[str, Any]:
        return self.relative_params

    def sample_independent(
        self,
        study: "Study",
        trial: FrozenTrial,
        param_name: str,
        param_distribution: BaseDistribution,
    ) -> Any:
        return self.unknown_param_value
