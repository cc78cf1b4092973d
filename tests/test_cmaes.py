"""Native CMA-ES: core convergence + sampler driver protocol."""
from __future__ import annotations

import warnings

import numpy as np
import pytest

import optuna_amd
from optuna_amd.samplers._cmaes._core import CMA, SepCMA, get_warm_start_mgd


optuna_amd.logging.set_verbosity(optuna_amd.logging.WARNING)


def _sphere(x: np.ndarray) -> float:
    return float(np.sum(x**2))


def _rosenbrock(x: np.ndarray) -> float:
    return float(np.sum(100 * (x[1:] - x[:-1] ** 2) ** 2 + (1 - x[:-1]) ** 2))


@pytest.mark.parametrize("cls", [CMA, SepCMA])
def test_core_converges_on_sphere(cls) -> None:
    dim = 6
    opt = cls(mean=np.full(dim, 3.0), sigma=2.0, seed=1)
    best = np.inf
    for _ in range(200):
        sols = []
        for _ in range(opt.population_size):
            x = opt.ask()
            v = _sphere(x)
            best = min(best, v)
            sols.append((x, v))
        opt.tell(sols)
        if best < 1e-8:
            break
    assert best < 1e-6


def test_core_converges_on_rosenbrock() -> None:
    dim = 4
    opt = CMA(mean=np.zeros(dim), sigma=0.5, seed=3)
    best = np.inf
    for _ in range(500):
        sols = []
        for _ in range(opt.population_size):
            x = opt.ask()
            v = _rosenbrock(x)
            best = min(best, v)
            sols.append((x, v))
        opt.tell(sols)
        if best < 1e-6:
            break
    assert best < 1e-3


def test_core_respects_bounds() -> None:
    bounds = np.array([[0.0, 1.0]] * 3)
    opt = CMA(mean=np.full(3, 0.5), sigma=5.0, bounds=bounds, seed=0)
    for _ in range(20):
        x = opt.ask()
        assert np.all(x >= 0.0) and np.all(x <= 1.0)


def test_core_pickle_roundtrip() -> None:
    import pickle

    opt = CMA(mean=np.zeros(3), sigma=1.0, seed=0)
    sols = [(opt.ask(), float(i)) for i in range(opt.population_size)]
    opt.tell(sols)
    clone = pickle.loads(pickle.dumps(opt))
    assert clone.generation == opt.generation
    np.testing.assert_array_equal(clone.mean, opt.mean)


def test_warm_start_mgd() -> None:
    rng = np.random.RandomState(0)
    target = np.array([0.7, 0.3])
    sols = []
    for _ in range(200):
        x = rng.rand(2)
        sols.append((x, _sphere(x - target)))
    mean, sigma, cov = get_warm_start_mgd(sols)
    assert np.linalg.norm(mean - target) < 0.2
    assert sigma > 0
    assert cov.shape == (2, 2)


def test_sampler_optimizes() -> None:
    sampler = optuna_amd.samplers.CmaEsSampler(seed=1, n_startup_trials=2)
    study = optuna_amd.create_study(sampler=sampler)

    def objective(trial: optuna_amd.Trial) -> float:
        x = trial.suggest_float("x", -5, 5)
        y = trial.suggest_float("y", -5, 5)
        return (x - 2) ** 2 + (y + 1) ** 2

    study.optimize(objective, n_trials=120)
    assert study.best_value < 0.5


def test_sampler_stores_state_in_system_attrs() -> None:
    sampler = optuna_amd.samplers.CmaEsSampler(seed=1, n_startup_trials=1, popsize=4)
    study = optuna_amd.create_study(sampler=sampler)
    study.optimize(
        lambda t: t.suggest_float("x", -1, 1) ** 2 + t.suggest_float("y", -1, 1) ** 2,
        n_trials=12,
    )
    gen_tagged = [t for t in study.trials if "cma:generation" in t.system_attrs]
    assert len(gen_tagged) >= 8
    state_holders = [
        t
        for t in study.trials
        if any(k.startswith("cma:optimizer") for k in t.system_attrs)
    ]
    assert len(state_holders) >= 1
    gens = {t.system_attrs.get("cma:generation") for t in gen_tagged}
    assert len(gens) >= 2  # the strategy actually advanced generations


def test_sampler_resumes_from_stored_state() -> None:
    storage = optuna_amd.storages.InMemoryStorage()
    study = optuna_amd.create_study(
        study_name="resume", storage=storage,
        sampler=optuna_amd.samplers.CmaEsSampler(seed=1, n_startup_trials=1, popsize=4),
    )
    obj = lambda t: t.suggest_float("x", -1, 1) ** 2 + t.suggest_float("y", -1, 1) ** 2
    study.optimize(obj, n_trials=10)
    # A fresh sampler instance must pick up the stored optimizer state.
    study2 = optuna_amd.load_study(
        study_name="resume", storage=storage,
        sampler=optuna_amd.samplers.CmaEsSampler(seed=2, n_startup_trials=1, popsize=4),
    )
    study2.optimize(obj, n_trials=6)
    gens = {
        t.system_attrs.get("cma:generation")
        for t in study2.trials
        if "cma:generation" in t.system_attrs
    }
    assert max(g for g in gens if g is not None) >= 2


def test_sampler_categorical_falls_back() -> None:
    with warnings.catch_warnings():
        warnings.simplefilter("ignore")
        sampler = optuna_amd.samplers.CmaEsSampler(
            seed=0, n_startup_trials=1, warn_independent_sampling=False
        )
        study = optuna_amd.create_study(sampler=sampler)

        def objective(trial: optuna_amd.Trial) -> float:
            c = trial.suggest_categorical("c", ("a", "b"))
            x = trial.suggest_float("x", -1, 1)
            return x**2 + (0 if c == "a" else 1)

        study.optimize(objective, n_trials=10)
    assert len(study.trials) == 10


def test_sampler_multiobjective_rejected() -> None:
    sampler = optuna_amd.samplers.CmaEsSampler(seed=0)
    study = optuna_amd.create_study(directions=["minimize", "minimize"], sampler=sampler)
    with pytest.raises(ValueError):
        study.optimize(lambda t: (t.suggest_float("x", 0, 1), 1.0), n_trials=2)


def test_sampler_seed_reproducible() -> None:
    def run(seed: int) -> list[float]:
        sampler = optuna_amd.samplers.CmaEsSampler(seed=seed, n_startup_trials=1, popsize=4)
        study = optuna_amd.create_study(sampler=sampler)
        study.optimize(lambda t: t.suggest_float("x", -1, 1) ** 2, n_trials=10)
        return [t.params["x"] for t in study.trials]

    assert run(7) == run(7)


def test_warm_start_sampler() -> None:
    src = optuna_amd.create_study(sampler=optuna_amd.samplers.RandomSampler(seed=0))
    src.optimize(lambda t: (t.suggest_float("x", -1, 1) - 0.5) ** 2, n_trials=40)
    sampler = optuna_amd.samplers.CmaEsSampler(
        seed=1, n_startup_trials=1, source_trials=src.trials
    )
    study = optuna_amd.create_study(sampler=sampler)
    study.optimize(lambda t: (t.suggest_float("x", -1, 1) - 0.5) ** 2, n_trials=30)
    assert study.best_value < 0.05


def test_cmawm_core_mixed_integer() -> None:
    """CMAwM optimizes a mixed int/float sphere; discrete dims stay on-grid and
    keep at least `margin` probability of leaving the mean's cell."""
    from optuna_amd.samplers._cmaes._core import CMAwM

    rng_bounds = np.array([[0.0, 1.0]] * 4)
    steps = np.array([0.25, 0.0, 0.125, 0.0])  # dims 0,2 discrete
    opt = CMAwM(
        mean=np.full(4, 0.5), sigma=0.3, bounds=rng_bounds, steps=steps, seed=3,
        population_size=8,
    )
    target = np.array([0.75, 0.2, 0.5, 0.9])
    best = np.inf
    for _ in range(60):
        sols = []
        for _ in range(opt.population_size):
            x_eval, x_tell = opt.ask()
            # discrete dims land exactly on the grid
            assert abs(x_eval[0] / 0.25 - round(x_eval[0] / 0.25)) < 1e-12
            assert abs(x_eval[2] / 0.125 - round(x_eval[2] / 0.125)) < 1e-12
            assert 0.0 <= x_eval[1] <= 1.0 and 0.0 <= x_eval[3] <= 1.0
            val = float(np.sum((x_eval - target) ** 2))
            best = min(best, val)
            sols.append((x_tell, val))
        opt.tell(sols)
    assert best < 1e-3


def test_cmawm_margin_keeps_discrete_mutation_alive() -> None:
    """With a binary dim whose optimum is at one end, margin correction must keep
    the other value reachable (plain CMA would collapse sigma and freeze it)."""
    from optuna_amd.samplers._cmaes._core import CMAwM

    opt = CMAwM(
        mean=np.array([0.5, 0.5]), sigma=0.2,
        bounds=np.array([[0.0, 1.0], [0.0, 1.0]]),
        steps=np.array([1.0, 0.0]), seed=0, population_size=6,
    )
    flips = 0
    for g in range(50):
        sols = []
        seen = set()
        for _ in range(opt.population_size):
            x_eval, x_tell = opt.ask()
            seen.add(x_eval[0])
            sols.append((x_tell, float(x_eval[0] + (x_eval[1] - 0.3) ** 2)))
        if len(seen) > 1:
            flips += 1
        opt.tell(sols)
    # The non-optimal discrete value keeps being proposed occasionally.
    assert flips >= 3


def test_sampler_with_margin_end_to_end() -> None:
    sampler = optuna_amd.samplers.CmaEsSampler(
        seed=5, n_startup_trials=2, with_margin=True, popsize=6
    )
    study = optuna_amd.create_study(sampler=sampler)

    def objective(trial):
        i = trial.suggest_int("i", 0, 10)
        f = trial.suggest_float("f", -2.0, 2.0)
        return (i - 7) ** 2 + f * f

    study.optimize(objective, n_trials=80)
    assert study.best_value < 2.0
    # the raw sample is recorded for generation replay
    assert any("x_for_tell" in t.system_attrs for t in study.trials)


def test_sampler_with_margin_state_roundtrip() -> None:
    from optuna_amd.samplers._cmaes._core import CMAwM
    import pickle

    opt = CMAwM(
        mean=np.array([0.5, 0.5]), sigma=0.3,
        bounds=np.array([[0.0, 1.0], [0.0, 1.0]]),
        steps=np.array([0.5, 0.0]), seed=1, population_size=4,
    )
    for _ in range(3):
        sols = [(opt.ask()[1], float(i)) for i in range(4)]
        opt.tell(sols)
    clone = pickle.loads(pickle.dumps(opt))
    np.testing.assert_array_equal(clone._A, opt._A)
    np.testing.assert_array_equal(clone.mean, opt.mean)


@pytest.mark.parametrize("kwargs", [
    {"with_margin": True},
    {"lr_adapt": True},
    {"use_separable_cma": True},
    {"consider_pruned_trials": True},
])
def test_experimental_argument_warnings(kwargs) -> None:
    from optuna_amd.exceptions import ExperimentalWarning

    with pytest.warns(ExperimentalWarning):
        optuna_amd.samplers.CmaEsSampler(**kwargs)


def test_lr_adapt_actually_adapts() -> None:
    """LRA-CMA: learning rates must move off 1.0 on a noisy objective and the
    sampler round-trips the adapted state through system attrs."""
    import numpy as np

    import optuna_amd
    from optuna_amd.samplers._cmaes._core import CMA

    rng = np.random.RandomState(3)
    opt = CMA(mean=np.full(6, 2.0), sigma=1.5, seed=5, lr_adapt=True)
    for _ in range(80):
        sols = []
        for _ in range(opt.population_size):
            x = opt.ask()
            f = float(np.sum(x * x) + 2.0 * rng.randn())
            sols.append((x, f))
        opt.tell(sols)
    assert opt._eta_m != 1.0 or opt._eta_c != 1.0

    # End-to-end through the sampler (state pickled into system attrs).
    import warnings

    with warnings.catch_warnings():
        warnings.simplefilter("ignore")
        sampler = optuna_amd.samplers.CmaEsSampler(seed=0, lr_adapt=True, n_startup_trials=1)
        study = optuna_amd.create_study(sampler=sampler)
        study.optimize(
            lambda t: sum(t.suggest_float(f"x{i}", -3, 3) ** 2 for i in range(4)),
            n_trials=25,
        )
    assert len(study.trials) == 25
