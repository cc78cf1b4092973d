"""Mean-decrease-impurity importance via sklearn random forests.

Parity: reference ``optuna/importance/_mean_decrease_impurity.py``.
"""
from __future__ import annotations

from typing import TYPE_CHECKING, Callable

import numpy as np

from optuna_amd._imports import try_import
from optuna_amd._transform import _SearchSpaceTransform
from optuna_amd.importance._base import (
    BaseImportanceEvaluator,
    _get_distributions,
    _get_filtered_trials,
    _get_target_values,
    _sort_dict_by_importance,
)
from optuna_amd.trial import FrozenTrial


if TYPE_CHECKING:
    from optuna_amd.study import Study

with try_import() as _imports:
    from sklearn.ensemble import RandomForestRegressor


class MeanDecreaseImpurityImportanceEvaluator(BaseImportanceEvaluator):
    def __init__(
        self, *, n_trees: int = 64, max_depth: int = 64, seed: int | None = None
    ) -> None:
        _imports.check()
        self._forest = RandomForestRegressor(
            n_estimators=n_trees, max_depth=max_depth, random_state=seed
        )

    def evaluate(
        self,
        study: "Study",
        params: list[str] | None = None,
        *,
        target: Callable[[FrozenTrial], float] | None = None,
    ) -> dict[str, float]:
        distributions = _get_distributions(study, params=params)
        if params is None:
            params = list(distributions.keys())
        if len(params) == 0:
            return {}
        distributions = {k: v for k, v in distributions.items() if not v.single()}
        if len(distributions) == 0:
            return {k: 0.0 for k in params}

        trials = [
            t for t in _get_filtered_trials(study, target)
            if all(name in t.params for name in distributions)
        ]
        trans = _SearchSpaceTransform(distributions)
        X = np.stack([trans.transform(t.params) for t in trials])
        y = np.asarray(_get_target_values(trials, target), dtype=np.float64)

        self._forest.fit(X, y)
        feature_importances = self._forest.feature_importances_

        # Sum one-hot columns back onto their categorical parameter.
        importances = {}
        for i, name in enumerate(distributions.keys()):
            cols = trans.column_to_encoded_columns[i]
            importances[name] = float(feature_importances[cols].sum())
        return _sort_dict_by_importance(
            {**{k: 0.0 for k in params}, **importances}
        )
