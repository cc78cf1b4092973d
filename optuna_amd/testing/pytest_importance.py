"""Importance-evaluator conformance suites: subclass with an ``evaluator``
fixture returning a factory.

Parity (pattern): reference ``optuna/testing/pytest_importance.py``
(BasicImportanceEvaluatorTestCase and the conditional/non-conditional splits).
"""
from __future__ import annotations

import math
import warnings
from typing import Any, Callable

import pytest

import optuna_amd
from optuna_amd.importance._base import BaseImportanceEvaluator
from optuna_amd.trial import Trial


# ---------------------------------------------------------------------------
# Reference-contract suites (``evaluator`` fixture returns a nullary factory).
# Parity: reference optuna/testing/pytest_importance.py:19-374.
# ---------------------------------------------------------------------------
import numpy as np  # noqa: E402

import optuna_amd as _optuna  # noqa: E402
from optuna_amd.distributions import (  # noqa: E402
    BaseDistribution as _BaseDist,
    FloatDistribution as _FloatDist,
)
from optuna_amd.importance import get_param_importances as _get_importances  # noqa: E402
from optuna_amd.samplers import RandomSampler as _RandomSampler  # noqa: E402
from optuna_amd.testing.objectives import pruned_objective as _pruned_objective  # noqa: E402


def _objective(trial: "Trial") -> float:
    x1 = trial.suggest_float("x1", 0.1, 3)
    x2 = trial.suggest_float("x2", 0.1, 3, log=True)
    x3 = trial.suggest_float("x3", 2, 4, log=True)
    return x1 + x2 * x3


def _multi_objective_function(trial: "Trial") -> tuple[float, float]:
    x1 = trial.suggest_float("x1", 0.1, 3)
    x2 = trial.suggest_float("x2", 0.1, 3, log=True)
    x3 = trial.suggest_float("x3", 2, 4, log=True)
    return x1, x2 * x3


def _get_study(seed: int, n_trials: int, is_multi_obj: bool):
    directions = ["minimize", "minimize"] if is_multi_obj else ["minimize"]
    study = _optuna.create_study(
        sampler=_RandomSampler(seed=seed), directions=directions
    )
    study.optimize(_multi_objective_function if is_multi_obj else _objective, n_trials=n_trials)
    return study


class _BaseImportanceEvaluatorTestCase:
    @pytest.fixture
    def evaluator(self):
        raise NotImplementedError


class BasicImportanceEvaluatorTestCase(_BaseImportanceEvaluatorTestCase):
    def test_get_param_importances_invalid_empty_study(self, evaluator) -> None:
        study = _optuna.create_study()
        assert _get_importances(study, evaluator=evaluator()) == {}
        study.optimize(_pruned_objective, n_trials=3)
        assert _get_importances(study, evaluator=evaluator()) == {}

    def test_get_param_importances_invalid_single_trial(self, evaluator) -> None:
        study = _optuna.create_study()
        study.optimize(lambda t: t.suggest_float("x1", 0.1, 3) ** 2, n_trials=1)
        assert _get_importances(study, evaluator=evaluator()) == {"x1": 1.0}

    def test_get_param_importances_invalid_no_completed_trials_params(
        self, evaluator
    ) -> None:
        def objective(trial) -> float:
            x1 = trial.suggest_float("x1", 0.1, 3)
            if trial.number % 2 == 0:
                trial.suggest_float("x2", 0.1, 3, log=True)
                raise _optuna.TrialPruned
            return x1 ** 2

        study = _optuna.create_study()
        study.optimize(objective, n_trials=3)
        for params in (["x2"], ["x1", "x2"], ["x3"]):
            with pytest.raises(ValueError):
                _get_importances(study, evaluator=evaluator(), params=params)

    def test_get_param_importances_empty_search_space(self, evaluator) -> None:
        def objective(trial) -> float:
            x = trial.suggest_float("x", 0, 5)
            y = trial.suggest_float("y", 1, 1)
            return 4 * x ** 2 + 4 * y ** 2

        study = _optuna.create_study()
        study.optimize(objective, n_trials=3)
        importance = _get_importances(study, evaluator=evaluator())
        assert set(importance) == {"x", "y"}
        assert importance["x"] > 0.0
        assert importance["y"] == 0.0

    @pytest.mark.filterwarnings("ignore::UserWarning")
    @pytest.mark.parametrize("inf_value", [float("inf"), -float("inf")])
    @pytest.mark.parametrize("target_idx", [0, 1, None])
    def test_evaluator_with_infinite(self, evaluator, inf_value, target_idx) -> None:
        # Rows with non-finite objectives must not change the result at all.
        evaluator_instance = evaluator()
        is_multi_obj = target_idx is not None
        study = _get_study(seed=13, n_trials=10, is_multi_obj=is_multi_obj)
        target = (lambda t: t.values[target_idx]) if is_multi_obj else None
        without_inf = evaluator_instance.evaluate(study, target=target)
        study.add_trial(
            _optuna.create_trial(
                values=[inf_value, inf_value] if is_multi_obj else [inf_value],
                params={"x1": 1.0, "x2": 1.0, "x3": 3.0},
                distributions={
                    "x1": _FloatDist(low=0.1, high=3),
                    "x2": _FloatDist(low=0.1, high=3, log=True),
                    "x3": _FloatDist(low=2, high=4, log=True),
                },
            )
        )
        with_inf = evaluator_instance.evaluate(study, target=target)
        assert with_inf == without_inf

    def test_evaluator_with_only_single_dists(self, evaluator) -> None:
        study = _optuna.create_study(sampler=_RandomSampler(seed=0))
        study.optimize(lambda t: t.suggest_float("a", 0.0, 0.0), n_trials=3)
        assert evaluator().evaluate(study) == {"a": 0.0}

    def test_importance_evaluator_with_target(self, evaluator) -> None:
        study = _optuna.create_study(sampler=_RandomSampler(seed=0))
        study.optimize(_objective, n_trials=3)
        evaluator_instance = evaluator()
        plain = evaluator_instance.evaluate(study)
        targeted = evaluator_instance.evaluate(study, target=lambda t: t.params["x3"])
        assert plain != targeted

    @pytest.mark.parametrize("params", [[], ["x1"], ["x1", "x3"], ["x1", "x4"]])
    @pytest.mark.parametrize("normalize", [True, False])
    def test_get_param_importances_with_params(self, evaluator, params, normalize) -> None:
        def objective(trial) -> float:
            x1 = trial.suggest_float("x1", 0.1, 3)
            x2 = trial.suggest_float("x2", 0.1, 3, log=True)
            x3 = trial.suggest_float("x3", 0, 3, step=1)
            value = x1 ** 4 + x2 + x3
            if trial.number % 2 == 0:
                value += trial.suggest_float("x4", 0.1, 3)
            return value

        study = _optuna.create_study()
        study.optimize(objective, n_trials=10)
        importance = _get_importances(
            study, evaluator=evaluator(), params=params, normalize=normalize
        )
        assert set(importance) == set(params)
        assert all(isinstance(v, float) and 0 <= v < float("inf") for v in importance.values())
        if normalize and importance:
            assert np.isclose(sum(importance.values()), 1.0)


class ConditionalImportanceEvaluatorTestCase(_BaseImportanceEvaluatorTestCase):
    @pytest.mark.parametrize(
        "params",
        [None, [], ["c"], ["x"], ["c", "x"], ["x", "y"], ["c", "x", "y"], ["d"], ["c", "d"]],
    )
    def test_conditional_parameters(self, evaluator, params) -> None:
        study = _optuna.create_study()
        dists_cx: dict[str, _BaseDist] = {
            "c": _FloatDist(0.0, 1.0),
            "x": _FloatDist(-2.0, 0.0),
        }
        dists_cy: dict[str, _BaseDist] = {
            "c": _FloatDist(0.0, 1.0),
            "y": _FloatDist(0.0, 2.0),
        }
        rows = [
            ({"c": 1.0, "x": -1.0}, dists_cx, -1.0),
            ({"c": 0.0, "y": 1.0}, dists_cy, 1.0),
            ({"c": 0.8, "x": -0.8}, dists_cx, -0.8),
            ({"c": 0.2, "y": 0.2}, dists_cy, 0.2),
            ({"c": 0.8, "x": -0.6}, dists_cx, -0.6),
            ({"c": 0.2, "y": 0.3}, dists_cy, 0.3),
        ]
        study.add_trials(
            [_optuna.create_trial(params=p, distributions=d, value=v) for p, d, v in rows]
        )
        if params and "d" in params:
            with pytest.raises(ValueError):
                evaluator().evaluate(study, params=params)
            return
        importance = evaluator().evaluate(study, params=params)
        if params == []:
            assert importance == {}
            return
        assert set(importance) == set(params or ["c", "x", "y"])
        assert not all(v == 0.0 for v in importance.values()), f"{importance=}"


class NonConditionalImportanceEvaluatorTestCase(_BaseImportanceEvaluatorTestCase):
    @pytest.mark.parametrize("normalize", [True, False])
    def test_get_param_importances_non_conditional(self, evaluator, normalize) -> None:
        def objective(trial) -> float:
            x1 = trial.suggest_float("x1", 0.1, 3)
            x2 = trial.suggest_float("x2", 0.1, 3, log=True)
            x3 = trial.suggest_float("x3", 0, 3, step=1)
            x4 = trial.suggest_int("x4", -3, 3)
            x5 = trial.suggest_int("x5", 1, 5, log=True)
            x6 = trial.suggest_categorical("x6", [1.0, 1.1, 1.2])
            value = x1 ** 4 + x2 + x3 - x4 ** 2 - x5 + x6
            if trial.number % 2 == 0:
                value += trial.suggest_float("x7", 0.1, 3)
            return value

        study = _optuna.create_study(sampler=_RandomSampler())
        study.optimize(objective, n_trials=3)
        importance = _get_importances(study, evaluator=evaluator(), normalize=normalize)
        assert set(importance) == {"x1", "x2", "x3", "x4", "x5", "x6"}
        values = list(importance.values())
        assert values == sorted(values, reverse=True)  # descending order
        assert all(isinstance(v, float) and 0 <= v < float("inf") for v in values)
        if normalize:
            assert np.isclose(sum(values), 1.0)

    def test_get_param_importances_invalid_dynamic_search_space_params(
        self, evaluator
    ) -> None:
        study = _optuna.create_study()
        study.optimize(
            lambda t: t.suggest_float("x1", 0.1, t.number + 0.1) ** 2, n_trials=3
        )
        with pytest.raises(ValueError):
            _get_importances(study, evaluator=evaluator(), params=["x1"])

    @pytest.mark.parametrize("normalize", [True, False])
    def test_get_param_importances_with_target(self, evaluator, normalize) -> None:
        def objective(trial) -> float:
            x1 = trial.suggest_float("x1", 0.1, 3)
            x2 = trial.suggest_float("x2", 0.1, 3, log=True)
            x3 = trial.suggest_float("x3", 0, 3, step=1)
            value = x1 ** 4 + x2 + x3
            if trial.number % 2 == 0:
                value += trial.suggest_float("x4", 0.1, 3)
            return value

        study = _optuna.create_study()
        study.optimize(objective, n_trials=3)
        importance = _get_importances(
            study,
            evaluator=evaluator(),
            target=lambda t: t.params["x1"] + t.params["x2"],
            normalize=normalize,
        )
        assert set(importance) == {"x1", "x2", "x3"}
        values = list(importance.values())
        assert values == sorted(values, reverse=True)
        assert all(isinstance(v, float) and 0 <= v < float("inf") for v in values)
        if normalize:
            assert np.isclose(sum(values), 1.0)


class MultiObjectiveImportanceEvaluatorTestCase(_BaseImportanceEvaluatorTestCase):
    def test_get_param_importance_target_is_none_and_study_is_multi_obj(
        self, evaluator
    ) -> None:
        def objective(trial) -> tuple[float, float]:
            x1 = trial.suggest_float("x1", 0.1, 3)
            x2 = trial.suggest_float("x2", 0.1, 3, log=True)
            x3 = trial.suggest_float("x3", 0, 3, step=1)
            x4 = trial.suggest_int("x4", -3, 3)
            x5 = trial.suggest_int("x5", 1, 5, log=True)
            x6 = trial.suggest_categorical("x6", [1.0, 1.1, 1.2])
            value = x1 ** 4 + x2 + x3 - x4 ** 2 - x5 + x6
            if trial.number % 2 == 0:
                value += trial.suggest_float("x7", 0.1, 3)
            return value, 0.0

        study = _optuna.create_study(directions=["minimize", "minimize"])
        study.optimize(objective, n_trials=3)
        assert isinstance(_get_importances(study, evaluator=evaluator()), dict)
