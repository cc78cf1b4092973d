"""GPSampler end-to-end behavior (small budgets; CPU torch)."""
from __future__ import annotations

import warnings

import numpy as np
import pytest

import optuna_amd


optuna_amd.logging.set_verbosity(optuna_amd.logging.WARNING)


def test_gp_sampler_optimizes_quadratic() -> None:
    sampler = optuna_amd.samplers.GPSampler(seed=0, n_startup_trials=5)
    study = optuna_amd.create_study(sampler=sampler)

    def objective(trial: optuna_amd.Trial) -> float:
        x = trial.suggest_float("x", -5, 5)
        y = trial.suggest_float("y", -5, 5)
        return (x - 1) ** 2 + (y + 2) ** 2

    study.optimize(objective, n_trials=25)
    assert study.best_value < 1.5
    assert abs(study.best_params["x"] - 1) < 1.5


def test_gp_sampler_mixed_space() -> None:
    sampler = optuna_amd.samplers.GPSampler(seed=1, n_startup_trials=4)
    study = optuna_amd.create_study(sampler=sampler)

    def objective(trial: optuna_amd.Trial) -> float:
        x = trial.suggest_float("x", 0.0, 1.0)
        i = trial.suggest_int("i", 0, 10)
        c = trial.suggest_categorical("c", ("a", "b"))
        lg = trial.suggest_float("lg", 1e-3, 1e1, log=True)
        return x + i * 0.1 + (0 if c == "a" else 0.5) + abs(np.log10(lg))

    study.optimize(objective, n_trials=14)
    assert len(study.trials) == 14


def test_gp_sampler_maximize() -> None:
    sampler = optuna_amd.samplers.GPSampler(seed=2, n_startup_trials=4)
    study = optuna_amd.create_study(direction="maximize", sampler=sampler)
    study.optimize(lambda t: -((t.suggest_float("x", -3, 3) - 1) ** 2), n_trials=18)
    assert study.best_value > -1.0


def test_gp_sampler_multi_objective_ehvi() -> None:
    sampler = optuna_amd.samplers.GPSampler(seed=3, n_startup_trials=4)
    study = optuna_amd.create_study(directions=["minimize", "minimize"], sampler=sampler)

    def mo(trial: optuna_amd.Trial) -> tuple[float, float]:
        x = trial.suggest_float("x", 0, 1)
        y = trial.suggest_float("y", 0, 1)
        return x + 0.1 * y, (1 - x) + 0.1 * y

    study.optimize(mo, n_trials=12)
    assert len(study.best_trials) >= 2


def test_gp_sampler_with_running_trials_uses_q_variant() -> None:
    sampler = optuna_amd.samplers.GPSampler(seed=4, n_startup_trials=3)
    study = optuna_amd.create_study(sampler=sampler)
    study.optimize(lambda t: t.suggest_float("x", -1, 1) ** 2, n_trials=5)
    # Leave an asked trial RUNNING with shared relative params, then ask again.
    t1 = study.ask()
    t1.suggest_float("x", -1, 1)
    t2 = study.ask()
    v = t2.suggest_float("x", -1, 1)
    assert -1 <= v <= 1
    attrs = study._storage.get_trial(t1._trial_id).system_attrs
    assert any(k.startswith("gp:relative_params") for k in attrs)


def test_gp_sampler_constraints() -> None:
    def constraints(trial: optuna_amd.trial.FrozenTrial) -> list[float]:
        return [trial.params["x"] - 0.5]

    with warnings.catch_warnings():
        warnings.simplefilter("ignore")
        sampler = optuna_amd.samplers.GPSampler(
            seed=5, n_startup_trials=4, constraints_func=constraints
        )
        study = optuna_amd.create_study(sampler=sampler)
        study.optimize(lambda t: -t.suggest_float("x", 0, 1), n_trials=12)
    assert all("constraints" in t.system_attrs for t in study.trials)


def test_gp_sampler_deterministic_objective_flag() -> None:
    sampler = optuna_amd.samplers.GPSampler(
        seed=6, n_startup_trials=3, deterministic_objective=True
    )
    study = optuna_amd.create_study(sampler=sampler)
    study.optimize(lambda t: t.suggest_float("x", -1, 1) ** 2, n_trials=8)
    assert len(study.trials) == 8


def test_gp_sampler_seed_reproducible() -> None:
    def run(seed: int) -> list[float]:
        sampler = optuna_amd.samplers.GPSampler(seed=seed, n_startup_trials=3)
        study = optuna_amd.create_study(sampler=sampler)
        study.optimize(lambda t: t.suggest_float("x", -1, 1) ** 2, n_trials=8)
        return [t.params["x"] for t in study.trials]

    assert run(11) == pytest.approx(run(11))
