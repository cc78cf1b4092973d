from __future__ import annotations

import math
import warnings

import pytest

import optuna_amd
from optuna_amd.exceptions import DuplicatedStudyError
from optuna_amd.study import StudyDirection
from optuna_amd.testing.objectives import fail_objective, pruned_objective
from optuna_amd.trial import TrialState


optuna_amd.logging.set_verbosity(optuna_amd.logging.WARNING)


def _square(trial: optuna_amd.Trial) -> float:
    x = trial.suggest_float("x", -10, 10)
    return x**2


def test_create_study_directions() -> None:
    assert optuna_amd.create_study().direction == StudyDirection.MINIMIZE
    assert optuna_amd.create_study(direction="maximize").direction == StudyDirection.MAXIMIZE
    study = optuna_amd.create_study(directions=["minimize", "maximize"])
    assert study.directions == [StudyDirection.MINIMIZE, StudyDirection.MAXIMIZE]
    with pytest.raises(ValueError):
        optuna_amd.create_study(direction="invalid")
    with pytest.raises(ValueError):
        optuna_amd.create_study(direction="minimize", directions=["minimize"])


def test_optimize_and_best() -> None:
    study = optuna_amd.create_study(sampler=optuna_amd.samplers.RandomSampler(seed=0))
    study.optimize(_square, n_trials=20)
    assert len(study.trials) == 20
    assert study.best_value == min(t.value for t in study.trials)
    assert study.best_trial.params == study.best_params


def test_optimize_with_catch() -> None:
    study = optuna_amd.create_study()
    with pytest.raises(ValueError):
        study.optimize(fail_objective, n_trials=3)
    assert len(study.trials) == 1
    study.optimize(fail_objective, n_trials=3, catch=(ValueError,))
    assert len(study.trials) == 4
    assert all(t.state == TrialState.FAIL for t in study.trials)


def test_optimize_pruned() -> None:
    study = optuna_amd.create_study()
    study.optimize(pruned_objective, n_trials=2)
    assert all(t.state == TrialState.PRUNED for t in study.trials)


def test_ask_tell() -> None:
    study = optuna_amd.create_study(direction="maximize")
    trial = study.ask()
    x = trial.suggest_float("x", 0, 1)
    frozen = study.tell(trial, x)
    assert frozen.state == TrialState.COMPLETE
    assert frozen.value == x
    # tell by number
    t2 = study.ask()
    t2.suggest_float("x", 0, 1)
    study.tell(t2.number, 0.25)
    assert study.trials[1].value == 0.25


def test_tell_validation() -> None:
    study = optuna_amd.create_study()
    t = study.ask()
    with pytest.raises(ValueError):
        study.tell(t, state=TrialState.COMPLETE)  # no values
    with pytest.raises(ValueError):
        study.tell(t, values=1.0, state=TrialState.PRUNED)
    with pytest.raises(ValueError):
        # Wrong length with an explicit COMPLETE state raises ...
        study.tell(t, values=[1.0, 2.0], state=TrialState.COMPLETE)
    with pytest.warns(UserWarning):
        # ... but without a state it fails the trial with a warning.
        assert study.tell(t, values=[1.0, 2.0]).state == TrialState.FAIL
    t = study.ask()
    study.tell(t, 1.0)
    with pytest.raises(ValueError):
        study.tell(1000, 1.0)  # unknown trial number


def test_tell_infeasible_values_become_fail() -> None:
    study = optuna_amd.create_study()
    t = study.ask()
    with warnings.catch_warnings():
        warnings.simplefilter("ignore")
        frozen = study.tell(t, "not-a-float")  # type: ignore[arg-type]
    assert frozen.state == TrialState.FAIL


def test_tell_pruned_promotes_last_intermediate() -> None:
    study = optuna_amd.create_study()
    t = study.ask()
    t.suggest_float("x", 0, 1)
    t.report(0.7, step=0)
    t.report(0.3, step=5)
    frozen = study.tell(t, state=TrialState.PRUNED)
    assert frozen.state == TrialState.PRUNED
    assert frozen.value == 0.3


def test_tell_skip_if_finished() -> None:
    study = optuna_amd.create_study()
    t = study.ask()
    study.tell(t, 1.0)
    frozen = study.tell(t, 2.0, skip_if_finished=True)
    assert frozen.value == 1.0


def test_enqueue_trial() -> None:
    study = optuna_amd.create_study()
    study.enqueue_trial({"x": 5.0})
    study.enqueue_trial({"x": -5.0})
    study.optimize(_square, n_trials=2)
    assert study.trials[0].params["x"] == 5.0
    assert study.trials[1].params["x"] == -5.0


def test_enqueue_skip_if_exists() -> None:
    study = optuna_amd.create_study()
    study.enqueue_trial({"x": 1.0})
    study.enqueue_trial({"x": 1.0}, skip_if_exists=True)
    assert len(study.get_trials(states=(TrialState.WAITING,))) == 1


def test_add_trial_and_load() -> None:
    study = optuna_amd.create_study()
    study.add_trial(
        optuna_amd.create_trial(
            value=0.5,
            params={"x": 1.0},
            distributions={"x": optuna_amd.distributions.FloatDistribution(-10, 10)},
        )
    )
    assert study.best_value == 0.5


def test_stop_in_objective() -> None:
    def objective(trial: optuna_amd.Trial) -> float:
        if trial.number >= 4:
            trial.study.stop()
        return float(trial.number)

    study = optuna_amd.create_study()
    study.optimize(objective, n_trials=100)
    assert len(study.trials) == 5


def test_stop_outside_optimize_raises() -> None:
    study = optuna_amd.create_study()
    with pytest.raises(RuntimeError):
        study.stop()


def test_callbacks() -> None:
    seen: list[int] = []
    study = optuna_amd.create_study()
    study.optimize(
        _square, n_trials=3, callbacks=[lambda s, t: seen.append(t.number)]
    )
    assert seen == [0, 1, 2]


def test_max_trials_callback() -> None:
    study = optuna_amd.create_study()
    study.optimize(
        _square,
        n_trials=100,
        callbacks=[optuna_amd.MaxTrialsCallback(5, states=(TrialState.COMPLETE,))],
    )
    assert len(study.trials) == 5


def test_user_attrs() -> None:
    study = optuna_amd.create_study()
    study.set_user_attr("k", "v")
    assert study.user_attrs == {"k": "v"}


def test_metric_names() -> None:
    study = optuna_amd.create_study()
    study.set_metric_names(["loss"])
    assert study.metric_names == ["loss"]
    with pytest.raises(ValueError):
        study.set_metric_names(["a", "b"])


def test_load_study_and_duplicates() -> None:
    storage = optuna_amd.storages.InMemoryStorage()
    optuna_amd.create_study(study_name="s1", storage=storage)
    with pytest.raises(DuplicatedStudyError):
        optuna_amd.create_study(study_name="s1", storage=storage)
    study = optuna_amd.create_study(study_name="s1", storage=storage, load_if_exists=True)
    assert study.study_name == "s1"
    loaded = optuna_amd.load_study(study_name="s1", storage=storage)
    assert loaded._study_id == study._study_id
    loaded2 = optuna_amd.load_study(study_name=None, storage=storage)
    assert loaded2.study_name == "s1"


def test_delete_and_copy_study() -> None:
    storage = optuna_amd.storages.InMemoryStorage()
    study = optuna_amd.create_study(study_name="src", storage=storage)
    study.set_user_attr("k", 1)
    study.optimize(_square, n_trials=3)
    dst_storage = optuna_amd.storages.InMemoryStorage()
    optuna_amd.copy_study(
        from_study_name="src", from_storage=storage, to_storage=dst_storage
    )
    copied = optuna_amd.load_study(study_name="src", storage=dst_storage)
    assert len(copied.trials) == 3
    assert copied.user_attrs == {"k": 1}
    optuna_amd.delete_study(study_name="src", storage=storage)
    assert "src" not in optuna_amd.get_all_study_names(storage)


def test_study_summaries() -> None:
    storage = optuna_amd.storages.InMemoryStorage()
    study = optuna_amd.create_study(study_name="sum", storage=storage)
    study.optimize(_square, n_trials=2)
    summaries = optuna_amd.get_all_study_summaries(storage)
    assert len(summaries) == 1
    assert summaries[0].study_name == "sum"
    assert summaries[0].n_trials == 2
    assert summaries[0].best_trial is not None


def test_multi_objective_best_trials() -> None:
    study = optuna_amd.create_study(directions=["minimize", "minimize"])

    def mo(trial: optuna_amd.Trial) -> tuple[float, float]:
        x = trial.suggest_float("x", 0, 1)
        return x, 1 - x

    study.optimize(mo, n_trials=20)
    best = study.best_trials
    assert len(best) >= 1
    with pytest.raises(RuntimeError):
        study.best_trial
    with pytest.raises(RuntimeError):
        study.direction


def test_nan_objective_fails_trial() -> None:
    study = optuna_amd.create_study()
    study.optimize(lambda t: float("nan"), n_trials=2, catch=())
    assert all(t.state == TrialState.FAIL for t in study.trials)
    study.optimize(_square, n_trials=1)
    assert study.best_trial.value is not None and not math.isnan(study.best_trial.value)


def test_trials_dataframe() -> None:
    study = optuna_amd.create_study()
    study.optimize(_square, n_trials=3)
    df = study.trials_dataframe()
    assert len(df) == 3
    assert "params_x" in df.columns
    assert "value" in df.columns


@pytest.mark.parametrize("mode", ["sqlite", "journal"])
def test_copy_study_across_backends(mode, tmp_path) -> None:
    """copy_study moves a full study (params, values, attrs) between storages."""
    import warnings

    from optuna_amd.testing.storages import StorageSupplier

    with warnings.catch_warnings():
        warnings.simplefilter("ignore")
        src = optuna_amd.create_study(study_name="copy-src")
        src.set_user_attr("tag", "v")
        src.optimize(lambda t: t.suggest_float("x", 0, 1) ** 2, n_trials=5)
        with StorageSupplier(mode) as dst_storage:
            optuna_amd.copy_study(
                from_study_name="copy-src",
                from_storage=src._storage,
                to_storage=dst_storage,
            )
            copied = optuna_amd.load_study(study_name="copy-src", storage=dst_storage)
            assert [t.value for t in copied.trials] == [t.value for t in src.trials]
            assert copied.user_attrs["tag"] == "v"


def test_get_all_study_summaries_over_rdb(tmp_path) -> None:
    import warnings

    from optuna_amd.testing.storages import StorageSupplier

    with warnings.catch_warnings():
        warnings.simplefilter("ignore")
        with StorageSupplier("sqlite") as storage:
            for name in ("sum-a", "sum-b"):
                s = optuna_amd.create_study(study_name=name, storage=storage)
                s.optimize(lambda t: t.suggest_float("x", 0, 1), n_trials=2)
            summaries = optuna_amd.get_all_study_summaries(storage)
            names = sorted(sm.study_name for sm in summaries)
            assert names == ["sum-a", "sum-b"]
            assert all(sm.n_trials == 2 for sm in summaries)
