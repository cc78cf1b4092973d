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


class ExtendedSamplerTestCase:
    """Behavioral cases beyond the basics: NaN objectives, single/dynamic
    spaces, mixed-distribution objectives, partial fixing, reproducibility in
    another process. Mix into a sampler's test class alongside
    BasicSamplerTestCase."""

    n_trials: int = 10

    @pytest.fixture
    def sampler_factory(self) -> Callable[[int], BaseSampler]:
        raise NotImplementedError

    def test_nan_objective_then_recovery(
        self, sampler_factory: Callable[[int], BaseSampler]
    ) -> None:
        # A NaN objective marks the trial FAIL; the sampler must keep working.
        with warnings.catch_warnings():
            warnings.simplefilter("ignore")
            study = optuna_amd.create_study(sampler=sampler_factory(7))

            def objective(trial: optuna_amd.Trial) -> float:
                x = trial.suggest_float("x", 0, 1)
                if trial.number % 3 == 1:
                    return float("nan")
                return x

            study.optimize(objective, n_trials=self.n_trials, catch=())
        states = [t.state for t in study.trials]
        assert optuna_amd.trial.TrialState.FAIL in states
        assert optuna_amd.trial.TrialState.COMPLETE in states
        assert study.best_trial.value is not None

    def test_single_value_distributions(
        self, sampler_factory: Callable[[int], BaseSampler]
    ) -> None:
        # low == high collapses to the single value for every dist kind.
        with warnings.catch_warnings():
            warnings.simplefilter("ignore")
            study = optuna_amd.create_study(sampler=sampler_factory(5))

            def objective(trial: optuna_amd.Trial) -> float:
                f = trial.suggest_float("f", 3.25, 3.25)
                i = trial.suggest_int("i", 7, 7)
                c = trial.suggest_categorical("c", ("only",))
                assert f == 3.25 and i == 7 and c == "only"
                return trial.suggest_float("x", 0, 1)

            study.optimize(objective, n_trials=self.n_trials)
        assert all(t.params["i"] == 7 for t in study.trials)

    def test_single_parameter_objective(
        self, sampler_factory: Callable[[int], BaseSampler]
    ) -> None:
        with warnings.catch_warnings():
            warnings.simplefilter("ignore")
            study = optuna_amd.create_study(sampler=sampler_factory(2))
            study.optimize(lambda t: t.suggest_float("x", -1, 1) ** 2, n_trials=self.n_trials)
        assert len(study.trials) == self.n_trials
        assert 0 <= study.best_value <= 1

    def test_mixed_distribution_objective(
        self, sampler_factory: Callable[[int], BaseSampler]
    ) -> None:
        with warnings.catch_warnings():
            warnings.simplefilter("ignore")
            study = optuna_amd.create_study(sampler=sampler_factory(9))

            def objective(trial: optuna_amd.Trial) -> float:
                total = trial.suggest_float("f", -1, 1)
                total += trial.suggest_float("flog", 1e-3, 10, log=True) * 0
                total += trial.suggest_float("fstep", 0, 1, step=0.25)
                total += trial.suggest_int("i", 0, 16)
                total += trial.suggest_int("ilog", 1, 64, log=True) * 0
                total += {"a": 0.0, "b": 1.0}[trial.suggest_categorical("c", ("a", "b"))]
                return total

            study.optimize(objective, n_trials=self.n_trials)
        for t in study.trials:
            assert t.params["fstep"] in [0.0, 0.25, 0.5, 0.75, 1.0]
            assert isinstance(t.params["i"], int)

    def test_dynamic_value_range(
        self, sampler_factory: Callable[[int], BaseSampler]
    ) -> None:
        # The range of "x" shifts between trials — samplers must tolerate it
        # (the storage records a widened/compatible distribution).
        with warnings.catch_warnings():
            warnings.simplefilter("ignore")
            study = optuna_amd.create_study(sampler=sampler_factory(11))

            def objective(trial: optuna_amd.Trial) -> float:
                width = 1.0 + (trial.number % 3)
                return trial.suggest_float("x", -width, width) ** 2

            study.optimize(objective, n_trials=self.n_trials)
        assert len(study.trials) == self.n_trials

    def test_partial_fixed_wrapper(
        self, sampler_factory: Callable[[int], BaseSampler]
    ) -> None:
        from optuna_amd.samplers import PartialFixedSampler

        with warnings.catch_warnings():
            warnings.simplefilter("ignore")
            base = sampler_factory(3)
            study = optuna_amd.create_study(
                sampler=PartialFixedSampler({"y": 0.5}, base)
            )

            def objective(trial: optuna_amd.Trial) -> float:
                x = trial.suggest_float("x", -1, 1)
                y = trial.suggest_float("y", -1, 1)
                return x * x + y

            study.optimize(objective, n_trials=self.n_trials)
        assert all(t.params["y"] == 0.5 for t in study.trials)

    def test_reproducible_across_processes(
        self, sampler_factory: Callable[[int], BaseSampler]
    ) -> None:
        # Same seed in a spawned subprocess must yield the identical sequence
        # (no hidden process-local entropy).
        import multiprocessing

        factory = sampler_factory
        ctx = multiprocessing.get_context("spawn")
        queue = ctx.Queue()
        proc = ctx.Process(
            target=_child_param_sequence, args=(type(self), queue)
        )
        proc.start()
        child = queue.get(timeout=120)
        proc.join(timeout=60)
        here = _run_param_sequence(factory, self.n_trials)
        assert child == here


def _run_param_sequence(
    factory: Callable[[int], BaseSampler], n_trials: int
) -> list[dict[str, Any]]:
    with warnings.catch_warnings():
        warnings.simplefilter("ignore")
        study = optuna_amd.create_study(sampler=factory(1234))
        study.optimize(
            lambda t: t.suggest_float("x", -1, 1) + t.suggest_int("i", 0, 9),
            n_trials=n_trials,
        )
    return [t.params for t in study.trials]


def _child_param_sequence(case_cls: type, queue: Any) -> None:
    case = case_cls()
    # Re-derive the factory from the fixture definition in the subclass.
    factory = case.__class__.sampler_factory.__wrapped__(case)  # type: ignore[attr-defined]
    queue.put(_run_param_sequence(factory, case_cls.n_trials))


class RelativeSamplerTestCase:
    """For samplers with a real relative stage (TPE-multivariate, GP, CMA-ES):
    the relative search space must be inferred and sampled within bounds over
    numerical, categorical and mixed spaces."""

    n_trials: int = 14

    @pytest.fixture
    def sampler_factory(self) -> Callable[[int], BaseSampler]:
        raise NotImplementedError

    def _run(self, factory: Callable[[int], BaseSampler], objective) -> "optuna_amd.Study":
        with warnings.catch_warnings():
            warnings.simplefilter("ignore")
            study = optuna_amd.create_study(sampler=factory(17))
            study.optimize(objective, n_trials=self.n_trials)
        return study

    def test_relative_numerical(self, sampler_factory: Callable[[int], BaseSampler]) -> None:
        def objective(trial: optuna_amd.Trial) -> float:
            a = trial.suggest_float("a", -2.0, 2.0)
            b = trial.suggest_float("b", 1e-2, 1e2, log=True)
            c = trial.suggest_int("c", 0, 20)
            assert -2.0 <= a <= 2.0 and 1e-2 <= b <= 1e2 and 0 <= c <= 20
            return a * a + abs(np_log10(b)) + c

        study = self._run(sampler_factory, objective)
        assert len(study.trials) == self.n_trials
        # After startup, the relative stage must actually provide params (the
        # suggest flow records identical distributions for every trial).
        last = study.trials[-1]
        assert set(last.params) == {"a", "b", "c"}

    def test_relative_categorical_mixed(
        self, sampler_factory: Callable[[int], BaseSampler]
    ) -> None:
        def objective(trial: optuna_amd.Trial) -> float:
            x = trial.suggest_float("x", 0.0, 1.0)
            k = trial.suggest_categorical("k", ("lo", "mid", "hi"))
            bump = {"lo": 0.0, "mid": 0.3, "hi": 0.9}[k]
            return (x - bump) ** 2

        study = self._run(sampler_factory, objective)
        assert all(t.params["k"] in ("lo", "mid", "hi") for t in study.trials)

    def test_relative_params_with_n_jobs(
        self, sampler_factory: Callable[[int], BaseSampler]
    ) -> None:
        with warnings.catch_warnings():
            warnings.simplefilter("ignore")
            study = optuna_amd.create_study(sampler=sampler_factory(23))
            study.optimize(
                lambda t: t.suggest_float("x", -1, 1) ** 2
                + (t.suggest_float("y", -1, 1) - 0.3) ** 2,
                n_trials=16,
                n_jobs=4,
            )
        assert len(study.trials) == 16
        assert all(set(t.params) == {"x", "y"} for t in study.trials)


def np_log10(v: float) -> float:
    import math

    return math.log10(v)
