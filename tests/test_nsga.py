"""NSGA-II / NSGA-III behavior."""
from __future__ import annotations

import warnings

import numpy as np
import pytest

import optuna_amd
from optuna_amd.samplers.nsgaii import (
    BLXAlphaCrossover,
    NSGAIISampler,
    PolynomialMutation,
    SBXCrossover,
    SPXCrossover,
    UNDXCrossover,
    UniformCrossover,
    VSBXCrossover,
)
from optuna_amd.samplers._nsgaiii import NSGAIIISampler
from optuna_amd.samplers._nsgaiii._elite_selection import (
    _generate_default_reference_point,
)
from optuna_amd.trial import TrialState


optuna_amd.logging.set_verbosity(optuna_amd.logging.WARNING)


def _binh_korn(trial: optuna_amd.Trial) -> tuple[float, float]:
    x = trial.suggest_float("x", 0, 5)
    y = trial.suggest_float("y", 0, 3)
    return 4 * x**2 + 4 * y**2, (x - 5) ** 2 + (y - 5) ** 2


def test_nsgaii_finds_pareto_spread() -> None:
    sampler = NSGAIISampler(population_size=20, seed=1)
    study = optuna_amd.create_study(directions=["minimize", "minimize"], sampler=sampler)
    study.optimize(_binh_korn, n_trials=120)
    front = study.best_trials
    assert len(front) >= 5
    v0 = sorted(t.values[0] for t in front)
    assert v0[0] < 15 and v0[-1] > 50  # spread along the front


def test_nsgaii_generation_bookkeeping() -> None:
    sampler = NSGAIISampler(population_size=10, seed=0)
    study = optuna_amd.create_study(directions=["minimize", "minimize"], sampler=sampler)
    study.optimize(_binh_korn, n_trials=35)
    gens = [t.system_attrs.get("NSGAIISampler:generation") for t in study.trials]
    assert all(g is not None for g in gens)
    assert max(g for g in gens if g is not None) >= 2
    # Parent caches recorded in study system attrs.
    attrs = study._storage.get_study_system_attrs(study._study_id)
    assert any(k.startswith("NSGAIISampler:parent:") for k in attrs)


@pytest.mark.parametrize(
    "crossover",
    [
        UniformCrossover(),
        BLXAlphaCrossover(),
        SBXCrossover(),
        VSBXCrossover(),
        SPXCrossover(),
        UNDXCrossover(),
    ],
    ids=lambda c: type(c).__name__,
)
def test_nsgaii_crossovers_run(crossover) -> None:
    with warnings.catch_warnings():
        warnings.simplefilter("ignore")
        sampler = NSGAIISampler(population_size=8, seed=0, crossover=crossover)
        study = optuna_amd.create_study(directions=["minimize", "minimize"], sampler=sampler)
        study.optimize(_binh_korn, n_trials=30)
    assert len(study.trials) == 30
    for t in study.trials:
        assert 0 <= t.params["x"] <= 5
        assert 0 <= t.params["y"] <= 3


def test_nsgaii_with_mutation_and_categorical() -> None:
    sampler = NSGAIISampler(
        population_size=8, seed=0, mutation=PolynomialMutation(eta=20)
    )
    study = optuna_amd.create_study(directions=["minimize", "minimize"], sampler=sampler)

    def objective(trial: optuna_amd.Trial) -> tuple[float, float]:
        x = trial.suggest_float("x", 0, 5)
        c = trial.suggest_categorical("c", ("a", "b"))
        return x + (0 if c == "a" else 1), 5 - x

    study.optimize(objective, n_trials=30)
    assert len(study.trials) == 30


def test_nsgaii_constraints() -> None:
    def constraints(trial: optuna_amd.trial.FrozenTrial) -> list[float]:
        return [trial.params["x"] - 3.0]  # x <= 3 feasible

    with warnings.catch_warnings():
        warnings.simplefilter("ignore")
        sampler = NSGAIISampler(population_size=10, seed=2, constraints_func=constraints)
        study = optuna_amd.create_study(directions=["minimize", "minimize"], sampler=sampler)
        study.optimize(_binh_korn, n_trials=50)
    feasible_front = [t for t in study.best_trials if t.params["x"] <= 3.0 + 1e-9]
    assert len(study.trials) == 50
    assert all("constraints" in t.system_attrs for t in study.trials)


def test_nsgaii_invalid_args() -> None:
    with pytest.raises(ValueError):
        NSGAIISampler(population_size=1)
    with pytest.raises(ValueError):
        NSGAIISampler(population_size=2, crossover=SPXCrossover())  # needs 3 parents


def test_das_dennis_reference_points() -> None:
    pts = _generate_default_reference_point(3, 3)
    assert pts.shape[1] == 3
    np.testing.assert_allclose(pts.sum(axis=1), 3.0)
    assert len(np.unique(pts, axis=0)) == len(pts)


def test_nsgaiii_three_objectives() -> None:
    sampler = NSGAIIISampler(population_size=12, seed=3)
    study = optuna_amd.create_study(
        directions=["minimize", "minimize", "minimize"], sampler=sampler
    )

    def dtlz2ish(trial: optuna_amd.Trial) -> tuple[float, float, float]:
        x = trial.suggest_float("x", 0, 1)
        y = trial.suggest_float("y", 0, 1)
        z = trial.suggest_float("z", 0, 1)
        return x, y, (1 - x) * (1 - y) + z * 0.1

    study.optimize(dtlz2ish, n_trials=60)
    assert len(study.trials) == 60
    assert len(study.best_trials) >= 3
    gens = [t.system_attrs.get("NSGAIIISampler:generation") for t in study.trials]
    assert max(g for g in gens if g is not None) >= 2


def test_nsgaii_seed_reproducible() -> None:
    def run(seed: int) -> list[tuple]:
        sampler = NSGAIISampler(population_size=8, seed=seed)
        study = optuna_amd.create_study(directions=["minimize", "minimize"], sampler=sampler)
        study.optimize(_binh_korn, n_trials=25)
        return [(t.params["x"], t.params["y"]) for t in study.trials]

    assert run(5) == run(5)
