"""Run the importance conformance suites over every shipped evaluator."""
from __future__ import annotations

from typing import Callable

import pytest

from optuna_amd.importance import (
    FanovaImportanceEvaluator,
    MeanDecreaseImpurityImportanceEvaluator,
    PedAnovaImportanceEvaluator,
)
from optuna_amd.importance._base import BaseImportanceEvaluator
from optuna_amd.testing.pytest_importance import (
    BasicImportanceEvaluatorTestCase,
    ConditionalImportanceEvaluatorTestCase,
    NonConditionalImportanceEvaluatorTestCase,
)


class TestPedAnova(BasicImportanceEvaluatorTestCase, ConditionalImportanceEvaluatorTestCase):
    @pytest.fixture
    def evaluator(self) -> Callable[..., BaseImportanceEvaluator]:
        return lambda: PedAnovaImportanceEvaluator()


class TestFanova(BasicImportanceEvaluatorTestCase, NonConditionalImportanceEvaluatorTestCase):
    @pytest.fixture
    def evaluator(self) -> Callable[..., BaseImportanceEvaluator]:
        return lambda: FanovaImportanceEvaluator(n_trees=16, seed=0)


class TestMeanDecreaseImpurity(
    BasicImportanceEvaluatorTestCase, NonConditionalImportanceEvaluatorTestCase
):
    @pytest.fixture
    def evaluator(self) -> Callable[..., BaseImportanceEvaluator]:
        return lambda: MeanDecreaseImpurityImportanceEvaluator(n_trees=16, seed=0)
