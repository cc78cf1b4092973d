"""Run the sampler conformance suites over every shipped sampler."""
from __future__ import annotations

from typing import Callable

import pytest

import optuna_amd
from optuna_amd.samplers import BaseSampler
from optuna_amd.testing.pytest_samplers import (
    BasicSamplerTestCase as _RefBasic,
    ExtendedSamplerTestCase,
    MultiObjectiveSamplerTestCase as _RefMO,
    RelativeSamplerTestCase as _RefRelative,
)


class _SeededBridge:
    """Derive the reference suites' nullary ``sampler`` fixture from this
    file's seeded ``sampler_factory`` fixture."""

    @pytest.fixture
    def sampler(
        self, sampler_factory: Callable[[int], BaseSampler]
    ) -> Callable[[], BaseSampler]:
        return lambda: sampler_factory(0)


class BasicSamplerTestCase(_SeededBridge, _RefBasic):
    pass


class MultiObjectiveSamplerTestCase(_SeededBridge, _RefMO):
    pass


class RelativeSamplerTestCase(_SeededBridge, _RefRelative):
    pass


class TestRandomSampler(BasicSamplerTestCase, ExtendedSamplerTestCase, MultiObjectiveSamplerTestCase):
    @pytest.fixture
    def sampler_factory(self) -> Callable[[int], BaseSampler]:
        return lambda seed: optuna_amd.samplers.RandomSampler(seed=seed)


class TestTPESampler(BasicSamplerTestCase, ExtendedSamplerTestCase, MultiObjectiveSamplerTestCase):
    @pytest.fixture
    def sampler_factory(self) -> Callable[[int], BaseSampler]:
        return lambda seed: optuna_amd.samplers.TPESampler(seed=seed, n_startup_trials=3)


class TestTPEMultivariate(BasicSamplerTestCase, ExtendedSamplerTestCase, RelativeSamplerTestCase):
    @pytest.fixture
    def sampler_factory(self) -> Callable[[int], BaseSampler]:
        return lambda seed: optuna_amd.samplers.TPESampler(
            seed=seed, n_startup_trials=3, multivariate=True, group=True
        )

    @pytest.fixture
    def sampler(self) -> Callable[[], BaseSampler]:
        # The relative suite samples right after ONE finished trial.
        return lambda: optuna_amd.samplers.TPESampler(
            seed=0, n_startup_trials=0, multivariate=True
        )


class TestCmaEsSampler(BasicSamplerTestCase, ExtendedSamplerTestCase, RelativeSamplerTestCase):
    @pytest.fixture
    def sampler_factory(self) -> Callable[[int], BaseSampler]:
        return lambda seed: optuna_amd.samplers.CmaEsSampler(
            seed=seed, n_startup_trials=2, warn_independent_sampling=False
        )

    @pytest.fixture
    def sampler(self) -> Callable[[], BaseSampler]:
        return lambda: optuna_amd.samplers.CmaEsSampler(
            seed=0, n_startup_trials=0, warn_independent_sampling=False
        )


class TestCmaEsSamplerWithMargin(BasicSamplerTestCase):
    @pytest.fixture
    def sampler_factory(self) -> Callable[[int], BaseSampler]:
        import warnings

        def make(seed: int) -> BaseSampler:
            with warnings.catch_warnings():
                warnings.simplefilter("ignore")
                return optuna_amd.samplers.CmaEsSampler(
                    seed=seed,
                    n_startup_trials=2,
                    with_margin=True,
                    warn_independent_sampling=False,
                )

        return make


class TestCmaEsSamplerSeparable(BasicSamplerTestCase):
    @pytest.fixture
    def sampler_factory(self) -> Callable[[int], BaseSampler]:
        import warnings

        def make(seed: int) -> BaseSampler:
            with warnings.catch_warnings():
                warnings.simplefilter("ignore")
                return optuna_amd.samplers.CmaEsSampler(
                    seed=seed,
                    n_startup_trials=2,
                    use_separable_cma=True,
                    warn_independent_sampling=False,
                )

        return make


class TestGPSampler(BasicSamplerTestCase, RelativeSamplerTestCase):
    n_trials = 6

    @pytest.fixture
    def sampler_factory(self) -> Callable[[int], BaseSampler]:
        return lambda seed: optuna_amd.samplers.GPSampler(
            seed=seed, n_startup_trials=3, warn_independent_sampling=False
        )

    @pytest.fixture
    def sampler(self) -> Callable[[], BaseSampler]:
        # The relative suite samples right after ONE finished trial.
        return lambda: optuna_amd.samplers.GPSampler(
            seed=0, n_startup_trials=0, warn_independent_sampling=False
        )


class TestNSGAIISampler(BasicSamplerTestCase, ExtendedSamplerTestCase, MultiObjectiveSamplerTestCase):
    @pytest.fixture
    def sampler_factory(self) -> Callable[[int], BaseSampler]:
        return lambda seed: optuna_amd.samplers.NSGAIISampler(seed=seed, population_size=4)


class TestNSGAIIISampler(MultiObjectiveSamplerTestCase):
    @pytest.fixture
    def sampler_factory(self) -> Callable[[int], BaseSampler]:
        return lambda seed: optuna_amd.samplers.NSGAIIISampler(seed=seed, population_size=4)


class TestQMCSampler(BasicSamplerTestCase, ExtendedSamplerTestCase):
    @pytest.fixture
    def sampler_factory(self) -> Callable[[int], BaseSampler]:
        return lambda seed: optuna_amd.samplers.QMCSampler(
            seed=seed, scramble=True, warn_independent_sampling=False
        )
