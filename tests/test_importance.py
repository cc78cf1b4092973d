"""Importance evaluators: sanity (dominant param ranks first) + API contract."""
from __future__ import annotations

import warnings

import numpy as np
import pytest

import optuna_amd
from optuna_amd.importance import (
    FanovaImportanceEvaluator,
    MeanDecreaseImpurityImportanceEvaluator,
    PedAnovaImportanceEvaluator,
    get_param_importances,
)


optuna_amd.logging.set_verbosity(optuna_amd.logging.WARNING)


def _make_study(n_trials: int = 60) -> optuna_amd.Study:
    study = optuna_amd.create_study(sampler=optuna_amd.samplers.RandomSampler(seed=0))

    def objective(trial):
        big = trial.suggest_float("big", -10, 10)
        small = trial.suggest_float("small", -10, 10)
        cat = trial.suggest_categorical("cat", ("a", "b"))
        return big**2 + 0.01 * small + (0.1 if cat == "b" else 0.0)

    study.optimize(objective, n_trials=n_trials)
    return study


EVALUATORS = [
    PedAnovaImportanceEvaluator,
    FanovaImportanceEvaluator,
    MeanDecreaseImpurityImportanceEvaluator,
]


@pytest.mark.parametrize("evaluator_cls", EVALUATORS)
def test_dominant_param_ranks_first(evaluator_cls) -> None:
    study = _make_study()
    with warnings.catch_warnings():
        warnings.simplefilter("ignore")
        imp = get_param_importances(study, evaluator=evaluator_cls())
    assert set(imp) == {"big", "small", "cat"}
    assert max(imp, key=imp.get) == "big"  # type: ignore[arg-type]
    assert sum(imp.values()) == pytest.approx(1.0)
    assert all(v >= 0 for v in imp.values())
    # Sorted descending.
    vals = list(imp.values())
    assert vals == sorted(vals, reverse=True)


def test_default_evaluator_is_pedanova() -> None:
    study = _make_study(40)
    imp = get_param_importances(study)
    assert max(imp, key=imp.get) == "big"  # type: ignore[arg-type]


def test_unnormalized() -> None:
    study = _make_study(40)
    with warnings.catch_warnings():
        warnings.simplefilter("ignore")
        imp = get_param_importances(study, normalize=False)
    assert sum(imp.values()) != pytest.approx(1.0) or max(imp.values()) < 1.0


def test_params_subset() -> None:
    study = _make_study(40)
    imp = get_param_importances(study, params=["big", "small"])
    assert set(imp) == {"big", "small"}


def test_target_callable() -> None:
    study = _make_study(40)
    with warnings.catch_warnings():
        warnings.simplefilter("ignore")
        imp = get_param_importances(
            study,
            evaluator=MeanDecreaseImpurityImportanceEvaluator(),
            target=lambda t: t.params["small"],
        )
    assert max(imp, key=imp.get) == "small"  # type: ignore[arg-type]


def test_no_completed_trials_returns_empty() -> None:
    study = optuna_amd.create_study()
    study.add_trial(
        optuna_amd.create_trial(
            state=optuna_amd.trial.TrialState.RUNNING,
        )
    )
    # Reference semantics: nothing to attribute yet -> empty mapping.
    assert optuna_amd.importance.get_param_importances(study) == {}


def test_pedanova_conditional_params() -> None:
    study = optuna_amd.create_study(sampler=optuna_amd.samplers.RandomSampler(seed=1))

    def objective(trial):
        kind = trial.suggest_categorical("kind", ("lin", "quad"))
        if kind == "lin":
            return trial.suggest_float("a", -5, 5)
        return trial.suggest_float("b", -5, 5) ** 2

    study.optimize(objective, n_trials=50)
    with warnings.catch_warnings():
        warnings.simplefilter("ignore")
        imp = get_param_importances(study)
    # PedAnova covers conditional params too.
    assert {"kind", "a", "b"} <= set(imp)


def test_evaluator_type_check() -> None:
    study = _make_study(20)
    with pytest.raises(TypeError):
        get_param_importances(study, evaluator="bad")  # type: ignore[arg-type]
