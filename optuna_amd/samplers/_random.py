"""Uniform random sampler via the search-space transform round trip.

Parity: reference ``optuna/samplers/_random.py`` (RandomSampler :19,
sample_independent :61-71).
"""
from __future__ import annotations

from typing import TYPE_CHECKING, Any

from optuna_amd import distributions as _distributions
from optuna_amd._transform import _SearchSpaceTransform
from optuna_amd.samplers._base import BaseSampler
from optuna_amd.samplers._lazy_random_state import LazyRandomState
from optuna_amd.trial import FrozenTrial


if TYPE_CHECKING:
    from optuna_amd.study import Study


class RandomSampler(BaseSampler):
    """Independent uniform sampling over each parameter's domain."""

    def __init__(self, seed: int | None = None) -> None:
        self._rng = LazyRandomState(seed)

    def reseed_rng(self) -> None:
        self._rng.rng.seed()

    def infer_relative_search_space(
        self, study: "Study", trial: FrozenTrial
    ) -> dict[str, _distributions.BaseDistribution]:
        return {}

    def sample_relative(
        self,
        study: "Study",
        trial: FrozenTrial,
        search_space: dict[str, _distributions.BaseDistribution],
    ) -> dict[str, Any]:
        return {}

    def sample_independent(
        self,
        study: "Study",
        trial: FrozenTrial,
        param_name: str,
        param_distribution: _distributions.BaseDistribution,
    ) -> Any:
        search_space = {param_name: param_distribution}
        trans = _SearchSpaceTransform(search_space)
        trans_params = self._rng.rng.uniform(trans.bounds[:, 0], trans.bounds[:, 1])
        return trans.untransform(trans_params)[param_name]
