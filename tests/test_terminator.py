"""Terminator: evaluators + termination decision."""
from __future__ import annotations

import warnings

import numpy as np
import pytest

import optuna_amd
from optuna_amd.terminator import (
    BestValueStagnationEvaluator,
    CrossValidationErrorEvaluator,
    EMMREvaluator,
    MedianErrorEvaluator,
    RegretBoundEvaluator,
    StaticErrorEvaluator,
    Terminator,
    TerminatorCallback,
    report_cross_validation_scores,
)
from optuna_amd.study import StudyDirection


optuna_amd.logging.set_verbosity(optuna_amd.logging.WARNING)


def _converged_study(n: int = 25) -> optuna_amd.Study:
    study = optuna_amd.create_study(sampler=optuna_amd.samplers.RandomSampler(seed=0))
    study.optimize(lambda t: t.suggest_float("x", -0.001, 0.001) ** 2, n_trials=n)
    return study


def test_static_error_evaluator() -> None:
    assert StaticErrorEvaluator(1.5).evaluate([], StudyDirection.MINIMIZE) == 1.5


def test_cross_validation_error_evaluator() -> None:
    study = optuna_amd.create_study()
    t = study.ask()
    t.suggest_float("x", 0, 1)
    report_cross_validation_scores(t, [0.8, 0.9, 0.85, 0.95])
    study.tell(t, 0.875)
    err = CrossValidationErrorEvaluator().evaluate(study.trials, StudyDirection.MINIMIZE)
    scores = np.array([0.8, 0.9, 0.85, 0.95])
    expected = (1 / 4 + 1 / 3) * scores.var(ddof=1)
    assert err == pytest.approx(expected)


def test_cross_validation_error_requires_report() -> None:
    study = optuna_amd.create_study()
    study.optimize(lambda t: t.suggest_float("x", 0, 1), n_trials=1)
    with pytest.raises(ValueError):
        CrossValidationErrorEvaluator().evaluate(study.trials, StudyDirection.MINIMIZE)


def test_best_value_stagnation() -> None:
    evaluator = BestValueStagnationEvaluator(max_stagnation_trials=5)
    study = optuna_amd.create_study()
    # Improvement at every step → budget stays full.
    for v in [5.0, 4.0, 3.0]:
        t = study.ask()
        t.suggest_float("x", 0, 1)
        study.tell(t, v)
    assert evaluator.evaluate(study.trials, StudyDirection.MINIMIZE) == 5
    # Stagnation eats the budget.
    for v in [3.5, 3.6, 3.7]:
        t = study.ask()
        t.suggest_float("x", 0, 1)
        study.tell(t, v)
    assert evaluator.evaluate(study.trials, StudyDirection.MINIMIZE) == 2


def test_regret_bound_evaluator_shrinks_with_convergence() -> None:
    converged = _converged_study()
    wide = optuna_amd.create_study(sampler=optuna_amd.samplers.RandomSampler(seed=0))
    wide.optimize(lambda t: t.suggest_float("x", -100, 100) ** 2, n_trials=25)
    evaluator = RegretBoundEvaluator(min_n_trials=5, seed=0)
    with warnings.catch_warnings():
        warnings.simplefilter("ignore")
        r_converged = evaluator.evaluate(converged.trials, StudyDirection.MINIMIZE)
        r_wide = RegretBoundEvaluator(min_n_trials=5, seed=0).evaluate(
            wide.trials, StudyDirection.MINIMIZE
        )
    assert r_converged < r_wide


def test_terminator_with_stagnation() -> None:
    terminator = Terminator(
        improvement_evaluator=BestValueStagnationEvaluator(max_stagnation_trials=3),
        min_n_trials=5,
    )
    study = optuna_amd.create_study()
    # 10 non-improving trials after the first.
    values = [1.0] + [2.0] * 10
    for v in values:
        t = study.ask()
        t.suggest_float("x", 0, 1)
        study.tell(t, v)
    assert terminator.should_terminate(study)


def test_terminator_min_trials_guard() -> None:
    terminator = Terminator(
        improvement_evaluator=BestValueStagnationEvaluator(max_stagnation_trials=0),
        min_n_trials=100,
    )
    study = _converged_study(5)
    assert not terminator.should_terminate(study)


def test_terminator_callback_stops_study() -> None:
    terminator = Terminator(
        improvement_evaluator=BestValueStagnationEvaluator(max_stagnation_trials=2),
        min_n_trials=3,
    )
    callback = TerminatorCallback(terminator)
    study = optuna_amd.create_study()
    # Constant objective: stagnates immediately after min_n_trials.
    study.optimize(
        lambda t: 1.0 + 0 * t.suggest_float("x", 0, 1),
        n_trials=100,
        callbacks=[callback],
    )
    assert len(study.trials) < 100


def test_emmr_and_median_error_run() -> None:
    study = _converged_study(8)
    emmr = EMMREvaluator(min_n_trials=3, seed=0)
    with warnings.catch_warnings():
        warnings.simplefilter("ignore")
        value = emmr.evaluate(study.trials, StudyDirection.MINIMIZE)
    assert np.isfinite(value)
    median_err = MedianErrorEvaluator(
        EMMREvaluator(min_n_trials=3, seed=0), warm_up_trials=1, n_initial_trials=3
    )
    with warnings.catch_warnings():
        warnings.simplefilter("ignore")
        threshold = median_err.evaluate(study.trials, StudyDirection.MINIMIZE)
    assert np.isfinite(threshold)


def test_evaluator_validation() -> None:
    with pytest.raises(ValueError):
        BestValueStagnationEvaluator(max_stagnation_trials=-1)
    with pytest.raises(ValueError):
        EMMREvaluator(min_n_trials=1)
    with pytest.raises(ValueError):
        Terminator(min_n_trials=0)
    with pytest.raises(ValueError):
        MedianErrorEvaluator(EMMREvaluator(), warm_up_trials=-1)
