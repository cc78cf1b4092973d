"""Sampler conformance suites: subclass with a ``sampler`` fixture.

Any sampler must produce in-domain, correctly-typed values over the
distribution matrix, behave deterministically under a fixed seed, and survive
conditional search spaces.

Parity (pattern): reference ``optuna/testing/pytest_samplers.py``
(BasicSamplerTestCase / RelativeSamplerTestCase / MultiObjectiveSamplerTestCase
:88-156, conditional-space regressions :180+).
"""
from __future__ import annotations

import warnings
from typing import Any, Callable

import pytest

import optuna_amd
from optuna_amd.distributions import (
    BaseDistribution,
    CategoricalDistribution,
    FloatDistribution,
    IntDistribution,
)
from optuna_amd.samplers import BaseSampler


DISTRIBUTION_MATRIX: list[BaseDistribution] = [
    FloatDistribution(-5.0, 5.0),
    FloatDistribution(1e-5, 1e5, log=True),
    FloatDistribution(-2.0, 2.0, step=0.5),
    IntDistribution(-10, 10),
    IntDistribution(1, 1024, log=True),
    IntDistribution(0, 100, step=5),
    CategoricalDistribution(("a", "b", "c")),
    CategoricalDistribution((1, 2.5, None)),
]


def _suggest(trial: optuna_amd.Trial, name: str, dist: BaseDistribution) -> Any:
    if isinstance(dist, FloatDistribution):
        return trial.suggest_float(name, dist.low, dist.high, log=dist.log, step=dist.step)
    if isinstance(dist, IntDistribution):
        return trial.suggest_int(name, dist.low, dist.high, log=dist.log, step=dist.step)
    assert isinstance(dist, CategoricalDistribution)
    return trial.suggest_categorical(name, dist.choices)


class BasicSamplerTestCase:
    """Domain/type/determinism contract for single-objective samplers."""

    n_trials: int = 10

    @pytest.fixture
    def sampler_factory(self) -> Callable[[int], BaseSampler]:
        raise NotImplementedError

    @pytest.mark.parametrize(
        "dist", DISTRIBUTION_MATRIX, ids=[repr(d)[:40] for d in DISTRIBUTION_MATRIX]
    )
    def test_values_in_domain(
        self, sampler_factory: Callable[[int], BaseSampler], dist: BaseDistribution
    ) -> None:
        with warnings.catch_warnings():
            warnings.simplefilter("ignore")
            study = optuna_amd.create_study(sampler=sampler_factory(3))

            def objective(trial: optuna_amd.Trial) -> float:
                v = _suggest(trial, "p", dist)
                assert dist._contains(dist.to_internal_repr(v))
                if isinstance(dist, IntDistribution):
                    assert isinstance(v, int)
                elif isinstance(dist, FloatDistribution):
                    assert isinstance(v, float)
                return 0.0 if isinstance(v, str) or v is None else float(v) * 0 + 1.0

            study.optimize(objective, n_trials=self.n_trials)
        assert len(study.trials) == self.n_trials

    def test_seed_determinism(self, sampler_factory: Callable[[int], BaseSampler]) -> None:
        def run(seed: int) -> list[dict[str, Any]]:
            with warnings.catch_warnings():
                warnings.simplefilter("ignore")
                study = optuna_amd.create_study(sampler=sampler_factory(seed))
                study.optimize(
                    lambda t: t.suggest_float("x", -1, 1) ** 2
                    + t.suggest_int("i", 0, 10),
                    n_trials=self.n_trials,
                )
            return [t.params for t in study.trials]

        assert run(42) == run(42)

    def test_conditional_search_space(
        self, sampler_factory: Callable[[int], BaseSampler]
    ) -> None:
        with warnings.catch_warnings():
            warnings.simplefilter("ignore")
            study = optuna_amd.create_study(sampler=sampler_factory(0))

            def objective(trial: optuna_amd.Trial) -> float:
                branch = trial.suggest_categorical("branch", ("l", "r"))
                if branch == "l":
                    return trial.suggest_float("left", 0, 1)
                return float(trial.suggest_int("right", 0, 10))

            study.optimize(objective, n_trials=self.n_trials)
        assert len(study.trials) == self.n_trials


class MultiObjectiveSamplerTestCase:
    """Contract for samplers that support multi-objective studies."""

    n_trials: int = 12

    @pytest.fixture
    def sampler_factory(self) -> Callable[[int], BaseSampler]:
        raise NotImplementedError

    def test_multi_objective_runs(
        self, sampler_factory: Callable[[int], BaseSampler]
    ) -> None:
        with warnings.catch_warnings():
            warnings.simplefilter("ignore")
            study = optuna_amd.create_study(
                directions=["minimize", "maximize"], sampler=sampler_factory(1)
            )

            def objective(trial: optuna_amd.Trial) -> tuple[float, float]:
                x = trial.suggest_float("x", 0, 1)
                y = trial.suggest_float("y", 0, 1)
                return x + y, x - y

            study.optimize(objective, n_trials=self.n_trials)
        assert len(study.trials) == self.n_trials
        assert len(study.best_trials) >= 1
