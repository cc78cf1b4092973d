from __future__ import annotations

import warnings

import pytest

import optuna_amd
from optuna_amd.distributions import FloatDistribution, IntDistribution
from optuna_amd.testing.pruners import DeterministicPruner
from optuna_amd.testing.samplers import DeterministicSampler, FixedSampler
from optuna_amd.trial import FixedTrial, TrialState, create_trial


optuna_amd.logging.set_verbosity(optuna_amd.logging.WARNING)


def test_suggest_float_variants() -> None:
    study = optuna_amd.create_study(sampler=optuna_amd.samplers.RandomSampler(seed=0))
    t = study.ask()
    x = t.suggest_float("x", 0.0, 1.0)
    assert 0.0 <= x <= 1.0
    xl = t.suggest_float("xl", 1e-5, 1e-1, log=True)
    assert 1e-5 <= xl <= 1e-1
    xs = t.suggest_float("xs", 0.0, 1.0, step=0.1)
    assert round(xs / 0.1) * 0.1 == pytest.approx(xs)
    i = t.suggest_int("i", 1, 10)
    assert 1 <= i <= 10 and isinstance(i, int)
    il = t.suggest_int("il", 1, 1000, log=True)
    assert 1 <= il <= 1000
    c = t.suggest_categorical("c", ["a", "b"])
    assert c in ("a", "b")


def test_suggest_is_cached() -> None:
    study = optuna_amd.create_study(sampler=optuna_amd.samplers.RandomSampler(seed=0))
    t = study.ask()
    x1 = t.suggest_float("x", 0, 1)
    x2 = t.suggest_float("x", 0, 1)
    assert x1 == x2


def test_suggest_incompatible_redefinition() -> None:
    study = optuna_amd.create_study()
    t = study.ask()
    t.suggest_float("x", 0, 1)
    with pytest.raises(ValueError):
        t.suggest_int("x", 0, 1)


def test_single_distribution_shortcut() -> None:
    study = optuna_amd.create_study(
        sampler=DeterministicSampler({})  # never called for single() dists
    )
    t = study.ask()
    assert t.suggest_float("x", 3.0, 3.0) == 3.0
    assert t.suggest_int("i", 5, 5) == 5
    assert t.suggest_categorical("c", ["only"]) == "only"


def test_relative_sampling_used() -> None:
    space = {"x": FloatDistribution(0, 10)}
    sampler = FixedSampler(space, {"x": 7.5}, unknown_param_value=1.0)
    study = optuna_amd.create_study(sampler=sampler)
    t = study.ask()
    assert t.suggest_float("x", 0, 10) == 7.5
    # Param outside relative space falls through to sample_independent.
    assert t.suggest_float("y", 0, 10) == 1.0


def test_report_and_should_prune() -> None:
    study = optuna_amd.create_study(pruner=DeterministicPruner(True))
    t = study.ask()
    t.report(1.0, step=0)
    assert t.should_prune()
    study2 = optuna_amd.create_study(pruner=DeterministicPruner(False))
    t2 = study2.ask()
    t2.report(1.0, step=0)
    assert not t2.should_prune()


def test_report_validation() -> None:
    study = optuna_amd.create_study()
    t = study.ask()
    with pytest.raises(TypeError):
        t.report("bad", 0)  # type: ignore[arg-type]
    with pytest.raises(ValueError):
        t.report(1.0, -1)
    t.report(1.0, 0)
    with warnings.catch_warnings(record=True) as w:
        warnings.simplefilter("always")
        t.report(2.0, 0)  # duplicate step ignored with warning
        assert any("already reported" in str(x.message) for x in w)
    assert study._storage.get_trial(t._trial_id).intermediate_values[0] == 1.0


def test_user_attrs_and_system_attrs() -> None:
    study = optuna_amd.create_study()
    t = study.ask()
    t.set_user_attr("k", [1])
    assert t.user_attrs == {"k": [1]}
    with pytest.warns(FutureWarning):
        t.set_system_attr("s", 2)
    assert t.system_attrs["s"] == 2


def test_set_constraint() -> None:
    study = optuna_amd.create_study()
    t = study.ask()
    t.set_constraint([-1.0, 0.5])
    frozen = study._storage.get_trial(t._trial_id)
    assert frozen.system_attrs["constraints"] == [-1.0, 0.5]
    assert frozen.constraints == {"0": -1.0, "1": 0.5}


def test_fixed_trial() -> None:
    t = FixedTrial({"x": 0.5, "c": "b", "i": 3})
    assert t.suggest_float("x", 0, 1) == 0.5
    assert t.suggest_categorical("c", ["a", "b"]) == "b"
    assert t.suggest_int("i", 0, 10) == 3
    with pytest.raises(ValueError):
        t.suggest_float("missing", 0, 1)
    with pytest.warns(UserWarning):
        # Out of range: warned, but the fixed value is still returned.
        assert t.suggest_float("x", 0.6, 1.0) == 0.5
    assert not t.should_prune()
    assert t.params == {"x": 0.5, "c": "b", "i": 3}


def test_frozen_trial_replay_and_validate() -> None:
    frozen = create_trial(
        value=1.0,
        params={"x": 0.25},
        distributions={"x": FloatDistribution(0, 1)},
    )
    assert frozen.suggest_float("x", 0, 1) == 0.25
    with pytest.raises(ValueError):
        frozen.suggest_float("y", 0, 1)
    assert frozen.value == 1.0
    assert frozen.last_step is None
    assert frozen.duration is not None


def test_create_trial_validation() -> None:
    with pytest.raises(ValueError):
        create_trial(state=TrialState.COMPLETE)  # no value
    with pytest.raises(ValueError):
        create_trial(value=1.0, params={"x": 1.0}, distributions={})
    with pytest.raises(ValueError):
        create_trial(
            value=1.0,
            params={"x": 5.0},
            distributions={"x": FloatDistribution(0, 1)},
        )


def test_multi_objective_report_rejected() -> None:
    study = optuna_amd.create_study(directions=["minimize", "minimize"])
    t = study.ask()
    with pytest.raises(NotImplementedError):
        t.report(1.0, 0)
    with pytest.raises(NotImplementedError):
        t.should_prune()


def test_deprecated_suggest_aliases() -> None:
    study = optuna_amd.create_study(sampler=optuna_amd.samplers.RandomSampler(seed=0))
    t = study.ask()
    with pytest.warns(FutureWarning):
        t.suggest_uniform("a", 0, 1)
    with pytest.warns(FutureWarning):
        t.suggest_loguniform("b", 1e-3, 1)
    with pytest.warns(FutureWarning):
        t.suggest_discrete_uniform("c", 0, 1, 0.5)


def test_suggest_precedence_enqueued_over_sampler() -> None:
    """_suggest precedence: cached > fixed (enqueued) > single > relative >
    independent (reference trial/_trial.py:620-651)."""
    import warnings

    import optuna_amd

    with warnings.catch_warnings():
        warnings.simplefilter("ignore")
        study = optuna_amd.create_study(
            sampler=optuna_amd.samplers.RandomSampler(seed=0)
        )
        study.enqueue_trial({"x": 0.123})
        got = {}

        def objective(trial):
            got[trial.number] = trial.suggest_float("x", 0, 1)
            return got[trial.number]

        study.optimize(objective, n_trials=2)
    assert got[0] == 0.123  # enqueued fixed param wins over the sampler
    assert got[1] != 0.123  # next trial samples freely


def test_suggest_precedence_fixed_out_of_range_warns_but_returns() -> None:
    import warnings

    import optuna_amd

    study = optuna_amd.create_study(sampler=optuna_amd.samplers.RandomSampler(seed=0))
    study.enqueue_trial({"x": 7.0})  # outside [0, 1]

    def objective(trial):
        with warnings.catch_warnings(record=True) as caught:
            warnings.simplefilter("always")
            v = trial.suggest_float("x", 0, 1)
        assert any("out of range" in str(w.message) for w in caught)
        return v

    with warnings.catch_warnings():
        warnings.simplefilter("ignore")
        study.optimize(objective, n_trials=1)
    # Reference semantics: an out-of-range fixed param warns but is still used
    # (the enqueued value always wins over the sampler).
    assert study.trials[0].params["x"] == 7.0


def test_suggest_precedence_cached_over_everything() -> None:
    import optuna_amd

    study = optuna_amd.create_study(sampler=optuna_amd.samplers.RandomSampler(seed=0))
    t = study.ask()
    first = t.suggest_float("x", 0, 1)
    again = t.suggest_float("x", 0, 1)
    assert first == again  # same trial re-ask returns the cached value


def test_suggest_single_value_beats_sampler() -> None:
    import optuna_amd

    class ExplodingSampler(optuna_amd.samplers.RandomSampler):
        def sample_independent(self, *a, **k):  # pragma: no cover
            raise AssertionError("single() distributions never reach the sampler")

    study = optuna_amd.create_study(sampler=ExplodingSampler(seed=0))
    t = study.ask()
    assert t.suggest_float("x", 2.5, 2.5) == 2.5


def test_waiting_claim_race_under_threads() -> None:
    """N threads pop the WAITING queue concurrently: every enqueued trial is
    claimed exactly once (the storage CAS settles races)."""
    import threading

    import optuna_amd

    study = optuna_amd.create_study(sampler=optuna_amd.samplers.RandomSampler(seed=0))
    n_queued = 24
    for i in range(n_queued):
        study.enqueue_trial({"x": float(i)})

    claimed: list[float] = []
    lock = threading.Lock()

    def worker() -> None:
        while True:
            t = study.ask()
            x = t.suggest_float("x", -1e9, 1e9)
            with lock:
                fresh = len(claimed) < n_queued
                if fresh:
                    claimed.append(x)
            study.tell(t, 0.0)
            if not fresh:
                return

    threads = [threading.Thread(target=worker) for _ in range(6)]
    for th in threads:
        th.start()
    for th in threads:
        th.join()
    # every enqueued value claimed exactly once
    assert sorted(claimed[:n_queued]) == [float(i) for i in range(n_queued)]


def test_frozen_trial_equality_and_repr() -> None:
    import optuna_amd
    from optuna_amd.distributions import FloatDistribution

    t1 = optuna_amd.create_trial(
        value=1.0, params={"x": 0.5}, distributions={"x": FloatDistribution(0, 1)}
    )
    t2 = optuna_amd.create_trial(
        value=1.0, params={"x": 0.5}, distributions={"x": FloatDistribution(0, 1)}
    )
    t2._trial_id = t1._trial_id
    t2.number = t1.number
    t2.datetime_start = t1.datetime_start
    t2.datetime_complete = t1.datetime_complete
    assert t1 == t2
    t2.params = {"x": 0.7}
    assert t1 != t2
    assert "x" in repr(t1)


def test_frozen_trial_value_values_exclusive() -> None:
    import optuna_amd

    with pytest.raises(ValueError):
        optuna_amd.create_trial(value=1.0, values=[1.0, 2.0])
    t = optuna_amd.create_trial(values=[1.0, 2.0])
    assert t.values == [1.0, 2.0]
    with pytest.raises(RuntimeError):
        _ = t.value  # multi-objective trials expose .values only


def test_frozen_trial_duration_and_last_step() -> None:
    import datetime as dt

    import optuna_amd

    t = optuna_amd.create_trial(value=0.0, intermediate_values={0: 1.0, 7: 2.0})
    assert t.last_step == 7
    t.datetime_start = dt.datetime(2026, 1, 1, 0, 0, 0)
    t.datetime_complete = dt.datetime(2026, 1, 1, 0, 1, 30)
    assert t.duration == dt.timedelta(seconds=90)
