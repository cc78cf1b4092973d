"""Sampler conformance suites: subclass and provide a ``sampler`` fixture.

The ``sampler`` fixture returns a NULLARY factory (``Callable[[], BaseSampler]``);
each test constructs a fresh sampler from it. The extended suite additionally
uses a SEEDED factory fixture ``sampler_factory`` (``Callable[[int], BaseSampler]``)
for determinism / cross-process checks.

Any sampler must produce in-domain, correctly-typed values over the
distribution matrix, survive conditional and dynamic search spaces, NaN
objectives and single-valued distributions.

Parity: reference ``optuna/testing/pytest_samplers.py`` (BasicSamplerTestCase,
RelativeSamplerTestCase, MultiObjectiveSamplerTestCase,
SingleOnlySamplerTestCase, parametrize_suggest_method, _create_new_trial).
"""
from __future__ import annotations

import warnings
from typing import TYPE_CHECKING, Any, Callable, Sequence

import numpy as np
import pytest

import optuna_amd as optuna
from optuna_amd.distributions import (
    BaseDistribution,
    CategoricalChoiceType,
    CategoricalDistribution,
    FloatDistribution,
    IntDistribution,
)
from optuna_amd.samplers import BaseSampler
from optuna_amd.testing.samplers import FixedSampler  # noqa: F401 — re-export
from optuna_amd.trial import FrozenTrial, Trial, TrialState


if TYPE_CHECKING:
    from optuna_amd.study import Study


def parametrize_suggest_method(name: str) -> Any:
    """One suggest call per distribution family, addressed by fixture name."""
    return pytest.mark.parametrize(
        f"suggest_method_{name}",
        [
            lambda t: t.suggest_float(name, 0, 10),
            lambda t: t.suggest_int(name, 0, 10),
            lambda t: t.suggest_categorical(name, [0, 1, 2]),
            lambda t: t.suggest_float(name, 0, 10, step=0.5),
            lambda t: t.suggest_float(name, 1e-7, 10, log=True),
            lambda t: t.suggest_int(name, 1, 10, log=True),
        ],
    )


def _create_new_trial(study: "Study") -> FrozenTrial:
    trial_id = study._storage.create_new_trial(study._study_id)
    return study._storage.get_trial(trial_id)


class _BaseSamplerTestCase:
    @pytest.fixture
    def sampler(self) -> Callable[[], BaseSampler]:
        raise NotImplementedError


class BasicSamplerTestCase(_BaseSamplerTestCase):
    """Domain/type contract for the independent-sampling surface."""

    @pytest.mark.parametrize(
        "distribution",
        [
            FloatDistribution(-1.0, 1.0),
            FloatDistribution(0.0, 1.0),
            FloatDistribution(-1.0, 0.0),
            FloatDistribution(1e-7, 1.0, log=True),
            FloatDistribution(-10, 10, step=0.1),
            FloatDistribution(-10.2, 10.2, step=0.1),
        ],
    )
    def test_float(
        self, sampler: Callable[[], BaseSampler], distribution: FloatDistribution
    ) -> None:
        study = optuna.create_study(sampler=sampler())
        points = np.array(
            [
                study.sampler.sample_independent(
                    study, _create_new_trial(study), "x", distribution
                )
                for _ in range(100)
            ]
        )
        assert np.all(points >= distribution.low)
        assert np.all(points <= distribution.high)
        # Plain Python floats, not numpy scalars (they leak into user code).
        one = study.sampler.sample_independent(
            study, _create_new_trial(study), "x", distribution
        )
        assert not isinstance(one, np.floating)
        if distribution.step is not None:
            on_grid = (points - distribution.low) / distribution.step
            np.testing.assert_almost_equal(np.round(on_grid), on_grid)

    @pytest.mark.parametrize(
        "distribution",
        [
            IntDistribution(-10, 10),
            IntDistribution(0, 10),
            IntDistribution(-10, 0),
            IntDistribution(-10, 10, step=2),
            IntDistribution(0, 10, step=2),
            IntDistribution(-10, 0, step=2),
            IntDistribution(1, 100, log=True),
        ],
    )
    def test_int(
        self, sampler: Callable[[], BaseSampler], distribution: IntDistribution
    ) -> None:
        study = optuna.create_study(sampler=sampler())
        points = np.array(
            [
                study.sampler.sample_independent(
                    study, _create_new_trial(study), "x", distribution
                )
                for _ in range(100)
            ]
        )
        assert np.all(points >= distribution.low)
        assert np.all(points <= distribution.high)
        one = study.sampler.sample_independent(
            study, _create_new_trial(study), "x", distribution
        )
        assert not isinstance(one, np.integer)

    @pytest.mark.parametrize("choices", [(1, 2, 3), ("a", "b", "c"), (1, "a")])
    def test_categorical(
        self, sampler: Callable[[], BaseSampler], choices: Sequence[CategoricalChoiceType]
    ) -> None:
        distribution = CategoricalDistribution(choices)
        study = optuna.create_study(sampler=sampler())

        def one_index() -> float:
            value = study.sampler.sample_independent(
                study, _create_new_trial(study), "x", distribution
            )
            return float(distribution.to_internal_repr(value))

        points = np.asarray([one_index() for _ in range(100)])
        assert np.all(points >= 0)
        assert np.all(points <= len(distribution.choices) - 1)
        np.testing.assert_almost_equal(np.round(points), points)

    def test_conditional_sample_independent(
        self, sampler: Callable[[], BaseSampler]
    ) -> None:
        # Two finished trials with DIFFERENT conditional branches: sampling the
        # shared and the branch-specific params must both work (GH #2734).
        study = optuna.create_study(sampler=sampler())
        cat = CategoricalDistribution(choices=["x", "y"])
        dep = CategoricalDistribution(choices=["a", "b"])
        study.add_trial(
            optuna.create_trial(
                params={"category": "x", "x": "a"},
                distributions={"category": cat, "x": dep},
                value=0.1,
            )
        )
        study.add_trial(
            optuna.create_trial(
                params={"category": "y", "y": "b"},
                distributions={"category": cat, "y": dep},
                value=0.1,
            )
        )
        trial = _create_new_trial(study)
        category = study.sampler.sample_independent(study, trial, "category", cat)
        assert category in ["x", "y"]
        value = study.sampler.sample_independent(study, trial, category, dep)
        assert value in ["a", "b"]

    def test_nan_objective_value(self, sampler: Callable[[], BaseSampler]) -> None:
        study = optuna.create_study(sampler=sampler())

        def objective(trial: Trial, base_value: float) -> float:
            return trial.suggest_float("x", 0.1, 0.2) + base_value

        for i in range(10, 1, -1):
            study.optimize(lambda t: objective(t, i), n_trials=1, catch=())
        assert int(study.best_value) == 2
        # A NaN objective fails that trial but must not disturb the incumbent.
        study.optimize(lambda t: objective(t, float("nan")), n_trials=1, catch=())
        assert int(study.best_value) == 2
        study.optimize(lambda t: objective(t, 1), n_trials=1, catch=())
        assert int(study.best_value) == 1

    def test_partial_fixed_sampling(self, sampler: Callable[[], BaseSampler]) -> None:
        study = optuna.create_study(sampler=sampler())

        def objective(trial: Trial) -> float:
            x = trial.suggest_float("x", -1, 1)
            y = trial.suggest_int("y", -1, 1)
            z = trial.suggest_float("z", -1, 1)
            return x + y + z

        study.optimize(objective, n_trials=1)
        fixed_params = {"y": 0}
        with warnings.catch_warnings():
            warnings.simplefilter("ignore", optuna.exceptions.ExperimentalWarning)
            study.sampler = optuna.samplers.PartialFixedSampler(
                fixed_params, study.sampler
            )
        study.optimize(objective, n_trials=1)
        assert study.trials[-1].params["y"] == fixed_params["y"]

    def test_sample_single_distribution(
        self, sampler: Callable[[], BaseSampler]
    ) -> None:
        relative_search_space: dict[str, BaseDistribution] = {
            "a": CategoricalDistribution([1]),
            "b": IntDistribution(low=1, high=1),
            "c": IntDistribution(low=1, high=1, log=True),
            "d": FloatDistribution(low=1.0, high=1.0),
            "e": FloatDistribution(low=1.0, high=1.0, log=True),
            "f": FloatDistribution(low=1.0, high=1.0, step=1.0),
        }
        with warnings.catch_warnings():
            warnings.simplefilter("ignore", optuna.exceptions.ExperimentalWarning)
            sampler_ = sampler()
        study = optuna.create_study(sampler=sampler_)
        # Two rounds so model-based samplers also construct their model.
        for _ in range(2):
            trial = study.ask(fixed_distributions=relative_search_space)
            study.tell(trial, 1.0)
            for param_name in relative_search_space:
                assert trial.params[param_name] == 1

    @parametrize_suggest_method("x")
    def test_single_parameter_objective(
        self,
        sampler: Callable[[], BaseSampler],
        suggest_method_x: Callable[[Trial], float],
    ) -> None:
        with warnings.catch_warnings():
            warnings.simplefilter("ignore", optuna.exceptions.ExperimentalWarning)
            sampler_ = sampler()
        study = optuna.create_study(sampler=sampler_)
        study.optimize(lambda t: float(suggest_method_x(t)), n_trials=10)
        assert len(study.trials) == 10
        assert all(t.state == TrialState.COMPLETE for t in study.trials)

    def test_conditional_parameter_objective(
        self, sampler: Callable[[], BaseSampler]
    ) -> None:
        def objective(trial: Trial) -> float:
            x = trial.suggest_categorical("x", [True, False])
            if x:
                return trial.suggest_float("y", 0, 1)
            return trial.suggest_float("z", 0, 1)

        with warnings.catch_warnings():
            warnings.simplefilter("ignore", optuna.exceptions.ExperimentalWarning)
            sampler_ = sampler()
        study = optuna.create_study(sampler=sampler_)
        study.optimize(objective, n_trials=10)
        assert len(study.trials) == 10
        assert all(t.state == TrialState.COMPLETE for t in study.trials)

    @parametrize_suggest_method("x")
    @parametrize_suggest_method("y")
    def test_combination_of_different_distributions_objective(
        self,
        sampler: Callable[[], BaseSampler],
        suggest_method_x: Callable[[Trial], float],
        suggest_method_y: Callable[[Trial], float],
    ) -> None:
        with warnings.catch_warnings():
            warnings.simplefilter("ignore", optuna.exceptions.ExperimentalWarning)
            sampler_ = sampler()
        study = optuna.create_study(sampler=sampler_)
        study.optimize(
            lambda t: float(suggest_method_x(t)) + float(suggest_method_y(t)), n_trials=3
        )
        assert len(study.trials) == 3
        assert all(t.state == TrialState.COMPLETE for t in study.trials)

    @pytest.mark.parametrize(
        "second_low,second_high",
        [(0, 5), (0, 20), (20, 30)],  # narrow / expand / disjoint
    )
    def test_dynamic_range_objective(
        self, sampler: Callable[[], BaseSampler], second_low: int, second_high: int
    ) -> None:
        def objective(trial: Trial, low: int, high: int) -> float:
            v = trial.suggest_float("x", low, high)
            v += trial.suggest_int("y", low, high)
            return v

        with warnings.catch_warnings():
            warnings.simplefilter("ignore", optuna.exceptions.ExperimentalWarning)
            sampler_ = sampler()
        study = optuna.create_study(sampler=sampler_)
        study.optimize(lambda t: objective(t, 0, 10), n_trials=10)
        study.optimize(lambda t: objective(t, second_low, second_high), n_trials=10)
        assert len(study.trials) == 20
        assert all(t.state == TrialState.COMPLETE for t in study.trials)


_NUMERICAL_MATRIX: list[BaseDistribution] = [
    FloatDistribution(-1.0, 1.0),
    FloatDistribution(1e-7, 1.0, log=True),
    FloatDistribution(-10, 10, step=0.5),
    IntDistribution(3, 10),
    IntDistribution(1, 100, log=True),
    IntDistribution(3, 9, step=2),
]


class RelativeSamplerTestCase(_BaseSamplerTestCase):
    """For samplers with a real relative (joint) stage."""

    @pytest.mark.parametrize("x_distribution", _NUMERICAL_MATRIX)
    @pytest.mark.parametrize("y_distribution", _NUMERICAL_MATRIX)
    def test_sample_relative_numerical(
        self,
        sampler: Callable[[], BaseSampler],
        x_distribution: BaseDistribution,
        y_distribution: BaseDistribution,
    ) -> None:
        search_space: dict[str, BaseDistribution] = dict(
            x=x_distribution, y=y_distribution
        )
        study = optuna.create_study(sampler=sampler())
        trial = study.ask(search_space)
        study.tell(trial, sum(trial.params.values()))

        def sample() -> list[Any]:
            params = study.sampler.sample_relative(
                study, _create_new_trial(study), search_space
            )
            return [params[name] for name in search_space]

        points = np.array([sample() for _ in range(10)])
        for i, distribution in enumerate(search_space.values()):
            assert isinstance(distribution, (FloatDistribution, IntDistribution))
            assert np.all(points[:, i] >= distribution.low)
            assert np.all(points[:, i] <= distribution.high)
        for param_value, distribution in zip(sample(), search_space.values()):
            assert not isinstance(param_value, np.floating)
            assert not isinstance(param_value, np.integer)
            if isinstance(distribution, IntDistribution):
                assert isinstance(param_value, int)
            else:
                assert isinstance(param_value, float)

    def test_sample_relative_categorical(
        self, sampler: Callable[[], BaseSampler]
    ) -> None:
        search_space: dict[str, BaseDistribution] = dict(
            x=CategoricalDistribution([1, 10, 100]),
            y=CategoricalDistribution([-1, -10, -100]),
        )
        study = optuna.create_study(sampler=sampler())
        trial = study.ask(search_space)
        study.tell(trial, sum(trial.params.values()))

        def sample() -> list[Any]:
            params = study.sampler.sample_relative(
                study, _create_new_trial(study), search_space
            )
            return [params[name] for name in search_space]

        points = np.array([sample() for _ in range(10)])
        for i, distribution in enumerate(search_space.values()):
            assert isinstance(distribution, CategoricalDistribution)
            assert np.all([v in distribution.choices for v in points[:, i]])
        for param_value in sample():
            assert not isinstance(param_value, np.floating)
            assert not isinstance(param_value, np.integer)
            assert isinstance(param_value, int)

    @pytest.mark.parametrize(
        "x_distribution",
        [
            FloatDistribution(-1.0, 1.0),
            FloatDistribution(1e-7, 1.0, log=True),
            FloatDistribution(-10, 10, step=0.5),
            IntDistribution(1, 10),
            IntDistribution(1, 100, log=True),
        ],
    )
    def test_sample_relative_mixed(
        self, sampler: Callable[[], BaseSampler], x_distribution: BaseDistribution
    ) -> None:
        search_space: dict[str, BaseDistribution] = dict(
            x=x_distribution, y=CategoricalDistribution([-1, -10, -100])
        )
        study = optuna.create_study(sampler=sampler())
        trial = study.ask(search_space)
        study.tell(trial, sum(trial.params.values()))

        def sample() -> list[Any]:
            params = study.sampler.sample_relative(
                study, _create_new_trial(study), search_space
            )
            return [params[name] for name in search_space]

        points = np.array([sample() for _ in range(10)])
        assert isinstance(search_space["x"], (FloatDistribution, IntDistribution))
        assert np.all(points[:, 0] >= search_space["x"].low)
        assert np.all(points[:, 0] <= search_space["x"].high)
        assert isinstance(search_space["y"], CategoricalDistribution)
        assert np.all([v in search_space["y"].choices for v in points[:, 1]])
        for param_value, distribution in zip(sample(), search_space.values()):
            assert not isinstance(param_value, np.floating)
            assert not isinstance(param_value, np.integer)
            if isinstance(distribution, (IntDistribution, CategoricalDistribution)):
                assert isinstance(param_value, int)
            else:
                assert isinstance(param_value, float)

    @pytest.mark.parametrize("n_jobs", [1, 2])
    def test_trial_relative_params(
        self, n_jobs: int, sampler: Callable[[], BaseSampler]
    ) -> None:
        study = optuna.create_study(sampler=sampler())

        def objective(trial: Trial) -> float:
            assert trial._relative_params is None
            trial.suggest_float("x", -10, 10)
            trial.suggest_float("y", -10, 10)
            assert trial._relative_params is not None
            return -1

        study.optimize(objective, n_trials=10, n_jobs=n_jobs)


class MultiObjectiveSamplerTestCase(_BaseSamplerTestCase):
    @pytest.mark.parametrize(
        "distribution",
        [
            FloatDistribution(-1.0, 1.0),
            FloatDistribution(0.0, 1.0),
            FloatDistribution(-1.0, 0.0),
            FloatDistribution(1e-7, 1.0, log=True),
            FloatDistribution(-10, 10, step=0.1),
            FloatDistribution(-10.2, 10.2, step=0.1),
            IntDistribution(-10, 10),
            IntDistribution(0, 10),
            IntDistribution(-10, 0),
            IntDistribution(-10, 10, step=2),
            IntDistribution(0, 10, step=2),
            IntDistribution(-10, 0, step=2),
            IntDistribution(1, 100, log=True),
            CategoricalDistribution((1, 2, 3)),
            CategoricalDistribution(("a", "b", "c")),
            CategoricalDistribution((1, "a")),
        ],
    )
    def test_multi_objective_sample_independent(
        self, sampler: Callable[[], BaseSampler], distribution: BaseDistribution
    ) -> None:
        study = optuna.create_study(
            directions=["minimize", "maximize"], sampler=sampler()
        )
        for _ in range(100):
            value = study.sampler.sample_independent(
                study, _create_new_trial(study), "x", distribution
            )
            assert distribution._contains(distribution.to_internal_repr(value))
            if not isinstance(distribution, CategoricalDistribution):
                assert not isinstance(value, np.floating)
            if isinstance(distribution, FloatDistribution) and distribution.step:
                on_grid = (value - distribution.low) / distribution.step
                np.testing.assert_almost_equal(np.round(on_grid), on_grid)


class SingleOnlySamplerTestCase(_BaseSamplerTestCase):
    def test_raise_error_for_samplers_during_multi_objectives(
        self, sampler: Callable[[], BaseSampler]
    ) -> None:
        study = optuna.create_study(
            directions=["maximize", "maximize"], sampler=sampler()
        )
        distribution = FloatDistribution(0.0, 1.0)
        with pytest.raises(ValueError):
            study.sampler.sample_independent(
                study, _create_new_trial(study), "x", distribution
            )
        with pytest.raises(ValueError):
            trial = _create_new_trial(study)
            study.sampler.sample_relative(
                study, trial, study.sampler.infer_relative_search_space(study, trial)
            )


class ExtendedSamplerTestCase:
    """Extra coverage beyond the reference matrix: determinism under a fixed
    seed, NaN-recovery, single-value distributions, dynamic ranges, partial
    fixing, cross-process reproducibility. Uses a SEEDED factory fixture."""

    n_trials: int = 10

    @pytest.fixture
    def sampler_factory(self) -> Callable[[int], BaseSampler]:
        raise NotImplementedError

    def test_seed_determinism(self, sampler_factory: Callable[[int], BaseSampler]) -> None:
        def run(seed: int) -> list[dict[str, Any]]:
            with warnings.catch_warnings():
                warnings.simplefilter("ignore")
                study = optuna.create_study(sampler=sampler_factory(seed))
                study.optimize(
                    lambda t: t.suggest_float("x", -1, 1) ** 2
                    + t.suggest_int("i", 0, 10),
                    n_trials=self.n_trials,
                )
            return [t.params for t in study.trials]

        assert run(42) == run(42)

    def test_nan_objective_then_recovery(
        self, sampler_factory: Callable[[int], BaseSampler]
    ) -> None:
        # A NaN objective marks the trial FAIL; the sampler must keep working.
        with warnings.catch_warnings():
            warnings.simplefilter("ignore")
            study = optuna.create_study(sampler=sampler_factory(7))

            def objective(trial: Trial) -> float:
                x = trial.suggest_float("x", 0, 1)
                if trial.number % 3 == 1:
                    return float("nan")
                return x

            study.optimize(objective, n_trials=self.n_trials, catch=())
        states = [t.state for t in study.trials]
        assert TrialState.FAIL in states
        assert TrialState.COMPLETE in states
        assert study.best_trial.value is not None

    def test_single_value_distributions(
        self, sampler_factory: Callable[[int], BaseSampler]
    ) -> None:
        with warnings.catch_warnings():
            warnings.simplefilter("ignore")
            study = optuna.create_study(sampler=sampler_factory(5))

            def objective(trial: Trial) -> float:
                f = trial.suggest_float("f", 3.25, 3.25)
                i = trial.suggest_int("i", 7, 7)
                c = trial.suggest_categorical("c", ("only",))
                assert f == 3.25 and i == 7 and c == "only"
                return trial.suggest_float("x", 0, 1)

            study.optimize(objective, n_trials=self.n_trials)
        assert all(t.params["i"] == 7 for t in study.trials)

    def test_mixed_distribution_objective(
        self, sampler_factory: Callable[[int], BaseSampler]
    ) -> None:
        with warnings.catch_warnings():
            warnings.simplefilter("ignore")
            study = optuna.create_study(sampler=sampler_factory(9))

            def objective(trial: Trial) -> float:
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
        with warnings.catch_warnings():
            warnings.simplefilter("ignore")
            study = optuna.create_study(sampler=sampler_factory(11))

            def objective(trial: Trial) -> float:
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
            study = optuna.create_study(sampler=PartialFixedSampler({"y": 0.5}, base))

            def objective(trial: Trial) -> float:
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
        proc = ctx.Process(target=_child_param_sequence, args=(type(self), queue))
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
        study = optuna.create_study(sampler=factory(1234))
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
