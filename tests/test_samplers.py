"""Sampler behavior: ranges, types, seed-reproducibility, TPE semantics."""
from __future__ import annotations

import multiprocessing
import warnings
from typing import Any, Callable

import numpy as np
import pytest

import optuna_amd
from optuna_amd.distributions import (
    BaseDistribution,
    CategoricalDistribution,
    FloatDistribution,
    IntDistribution,
)
from optuna_amd.trial import TrialState


optuna_amd.logging.set_verbosity(optuna_amd.logging.WARNING)

DIST_MATRIX: list[BaseDistribution] = [
    FloatDistribution(-5.0, 5.0),
    FloatDistribution(1e-5, 1e5, log=True),
    FloatDistribution(-2.0, 2.0, step=0.5),
    IntDistribution(-10, 10),
    IntDistribution(1, 1024, log=True),
    IntDistribution(0, 100, step=5),
    CategoricalDistribution(("a", "b", "c")),
    CategoricalDistribution((1, 2.5, None)),
]


def _sampler_factories() -> list[Callable[[int], optuna_amd.samplers.BaseSampler]]:
    return [
        lambda seed: optuna_amd.samplers.RandomSampler(seed=seed),
        lambda seed: optuna_amd.samplers.TPESampler(seed=seed, n_startup_trials=3),
        lambda seed: optuna_amd.samplers.TPESampler(
            seed=seed, n_startup_trials=3, multivariate=True
        ),
    ]


def _value_in_domain(value: Any, dist: BaseDistribution) -> bool:
    return dist._contains(dist.to_internal_repr(value))


@pytest.mark.parametrize("make_sampler", _sampler_factories())
@pytest.mark.parametrize("dist", DIST_MATRIX, ids=[repr(d)[:40] for d in DIST_MATRIX])
def test_sample_within_domain(make_sampler, dist) -> None:
    with warnings.catch_warnings():
        warnings.simplefilter("ignore")
        study = optuna_amd.create_study(sampler=make_sampler(7))

        def objective(trial: optuna_amd.Trial) -> float:
            if isinstance(dist, FloatDistribution):
                v: Any = trial.suggest_float(
                    "p", dist.low, dist.high, log=dist.log, step=dist.step
                )
            elif isinstance(dist, IntDistribution):
                v = trial.suggest_int("p", dist.low, dist.high, log=dist.log, step=dist.step)
            else:
                v = trial.suggest_categorical("p", dist.choices)
            assert _value_in_domain(v, dist)
            if isinstance(v, str):
                return float(len(v))
            return float(v if v is not None else 0.0)

        study.optimize(objective, n_trials=12)
    assert len(study.trials) == 12


@pytest.mark.parametrize("make_sampler", _sampler_factories())
def test_seed_reproducibility(make_sampler) -> None:
    def run(seed: int) -> list[dict[str, Any]]:
        with warnings.catch_warnings():
            warnings.simplefilter("ignore")
            study = optuna_amd.create_study(sampler=make_sampler(seed))

            def objective(trial: optuna_amd.Trial) -> float:
                x = trial.suggest_float("x", -10, 10)
                y = trial.suggest_int("y", 1, 100, log=True)
                c = trial.suggest_categorical("c", ("a", "b"))
                return x**2 + y + (0 if c == "a" else 1)

            study.optimize(objective, n_trials=12)
        return [t.params for t in study.trials]

    assert run(42) == run(42)
    assert run(42) != run(43)


def test_tpe_multivariate_uses_relative() -> None:
    sampler = optuna_amd.samplers.TPESampler(seed=0, n_startup_trials=5, multivariate=True)
    study = optuna_amd.create_study(sampler=sampler)

    def objective(trial: optuna_amd.Trial) -> float:
        x = trial.suggest_float("x", -5, 5)
        y = trial.suggest_float("y", -5, 5)
        return x**2 + y**2

    study.optimize(objective, n_trials=25)
    assert study.best_value < 25.0


def test_tpe_group_decomposition() -> None:
    with warnings.catch_warnings():
        warnings.simplefilter("ignore")
        sampler = optuna_amd.samplers.TPESampler(seed=0, n_startup_trials=3, group=True)
        study = optuna_amd.create_study(sampler=sampler)

        def objective(trial: optuna_amd.Trial) -> float:
            a = trial.suggest_float("a", 0, 1)
            if trial.number % 2 == 0:
                b = trial.suggest_float("b", 0, 1)
                return a + b
            c = trial.suggest_float("c", 0, 1)
            return a + c

        study.optimize(objective, n_trials=15)
    assert len(study.trials) == 15


def test_tpe_constant_liar_shares_params() -> None:
    sampler = optuna_amd.samplers.TPESampler(
        seed=0, n_startup_trials=2, constant_liar=True, multivariate=True
    )
    study = optuna_amd.create_study(sampler=sampler)
    study.optimize(lambda t: t.suggest_float("x", 0, 1) ** 2, n_trials=5)
    # Ask (not tell) so the trial stays RUNNING with shared relative params.
    t = study.ask()
    t.suggest_float("x", 0, 1)
    attrs = study._storage.get_trial(t._trial_id).system_attrs
    assert any(k.startswith("tpe:relative_params") for k in attrs)


def test_tpe_multi_objective_split() -> None:
    sampler = optuna_amd.samplers.TPESampler(seed=1, n_startup_trials=5)
    study = optuna_amd.create_study(directions=["minimize", "minimize"], sampler=sampler)

    def mo(trial: optuna_amd.Trial) -> tuple[float, float]:
        x = trial.suggest_float("x", 0, 1)
        y = trial.suggest_float("y", 0, 1)
        return x, y

    study.optimize(mo, n_trials=20)
    assert len(study.trials) == 20


def test_tpe_with_conditional_space_falls_back() -> None:
    sampler = optuna_amd.samplers.TPESampler(
        seed=0, n_startup_trials=2, warn_independent_sampling=False
    )
    study = optuna_amd.create_study(sampler=sampler)

    def objective(trial: optuna_amd.Trial) -> float:
        kind = trial.suggest_categorical("kind", ("lin", "quad"))
        x = trial.suggest_float(f"x_{kind}", 0, 1)
        return x if kind == "lin" else x**2

    study.optimize(objective, n_trials=12)
    assert len(study.trials) == 12


def _worker_sample_params(seed_and_queue) -> None:
    seed, queue = seed_and_queue
    study = optuna_amd.create_study(
        sampler=optuna_amd.samplers.TPESampler(seed=seed, n_startup_trials=2)
    )
    study.optimize(lambda t: t.suggest_float("x", 0, 1) ** 2, n_trials=6)
    queue.put([t.params["x"] for t in study.trials])


def test_reproducible_in_other_process() -> None:
    ctx = multiprocessing.get_context("spawn")
    queue = ctx.Manager().Queue()
    p = ctx.Process(target=_worker_sample_params, args=((11, queue),))
    p.start()
    p.join()
    assert p.exitcode == 0
    remote = queue.get()

    study = optuna_amd.create_study(
        sampler=optuna_amd.samplers.TPESampler(seed=11, n_startup_trials=2)
    )
    study.optimize(lambda t: t.suggest_float("x", 0, 1) ** 2, n_trials=6)
    local = [t.params["x"] for t in study.trials]
    assert local == pytest.approx(remote)


def test_after_trial_constraints_recorded() -> None:
    def constraints(trial: optuna_amd.trial.FrozenTrial) -> list[float]:
        return [trial.params["x"] - 0.5]

    with warnings.catch_warnings():
        warnings.simplefilter("ignore")
        sampler = optuna_amd.samplers.TPESampler(
            seed=0, n_startup_trials=2, constraints_func=constraints
        )
        study = optuna_amd.create_study(sampler=sampler)
        study.optimize(lambda t: t.suggest_float("x", 0, 1), n_trials=4)
    for t in study.trials:
        assert "constraints" in t.system_attrs
