"""fANOVA importance: per-feature marginal variance fractions on random forests.

Functional ANOVA on tree ensembles (Hutter, Hoos & Leyton-Brown, "An Efficient
Approach for Assessing Hyperparameter Importance", ICML 2014): each tree defines a
piecewise-constant function on the (transformed) search-space box; the importance
of parameter i is the fraction of the tree's total variance explained by the
marginal over dimension i, averaged over trees.

Parity: reference ``optuna/importance/_fanova/`` (_evaluator.py:25-132,
_tree.py:14-319 — n_trees=64, depth=64, per-node subspace statistics). The tree
walk here is a direct recursive formulation instead of the reference's
precomputed node tables.
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


class _TreeMarginals:
    """Marginal statistics of one fitted sklearn decision tree over a box domain."""

    def __init__(self, tree: "object", bounds: np.ndarray) -> None:
        self._left = tree.children_left  # type: ignore[attr-defined]
        self._right = tree.children_right  # type: ignore[attr-defined]
        self._feature = tree.feature  # type: ignore[attr-defined]
        self._threshold = tree.threshold  # type: ignore[attr-defined]
        self._value = tree.value.reshape(-1)  # type: ignore[attr-defined]
        self._bounds = bounds
        self._sizes = bounds[:, 1] - bounds[:, 0]
        self._sizes[self._sizes == 0] = 1.0

    def _leaf_stats(
        self, node: int, box: np.ndarray, out: list[tuple[float, float]]
    ) -> None:
        """Collect (volume fraction, leaf value) of every leaf's subdomain."""
        if self._left[node] == -1:  # leaf
            frac = float(np.prod((box[:, 1] - box[:, 0]) / self._sizes))
            out.append((frac, float(self._value[node])))
            return
        f = self._feature[node]
        t = self._threshold[node]
        lo, hi = box[f]
        if t > lo:
            left_box = box.copy()
            left_box[f, 1] = min(hi, t)
            self._leaf_stats(self._left[node], left_box, out)
        if t < hi:
            right_box = box.copy()
            right_box[f, 0] = max(lo, t)
            self._leaf_stats(self._right[node], right_box, out)

    def total_variance(self) -> float:
        out: list[tuple[float, float]] = []
        self._leaf_stats(0, self._bounds.copy(), out)
        w = np.array([o[0] for o in out])
        v = np.array([o[1] for o in out])
        w = w / w.sum()
        mean = float(w @ v)
        return float(w @ (v - mean) ** 2)

    def marginal_variance(self, feature: int) -> float:
        """Var over x_feature of E[f | x_feature] for one dimension."""
        # Split points of this tree along `feature` define the piecewise intervals.
        lo, hi = self._bounds[feature]
        cuts = sorted(
            {lo, hi}
            | {
                float(t)
                for f, t in zip(self._feature, self._threshold)
                if f == feature and lo < t < hi
            }
        )
        cuts_arr = np.asarray(cuts)
        mids = (cuts_arr[:-1] + cuts_arr[1:]) / 2
        widths = np.diff(cuts_arr)
        if widths.sum() == 0:
            return 0.0

        def marginal(node: int, x: float, box: np.ndarray) -> tuple[float, float]:
            """(weight, weighted value) of the subtree with x_feature pinned to x."""
            if self._left[node] == -1:
                others = np.arange(len(self._sizes)) != feature
                frac = float(
                    np.prod((box[others, 1] - box[others, 0]) / self._sizes[others])
                )
                return frac, frac * float(self._value[node])
            f = self._feature[node]
            t = self._threshold[node]
            if f == feature:
                child = self._left[node] if x <= t else self._right[node]
                return marginal(child, x, box)
            w_total, v_total = 0.0, 0.0
            b_lo, b_hi = box[f]
            if t > b_lo:
                left_box = box.copy()
                left_box[f, 1] = min(b_hi, t)
                w, v = marginal(self._left[node], x, left_box)
                w_total += w
                v_total += v
            if t < b_hi:
                right_box = box.copy()
                right_box[f, 0] = max(b_lo, t)
                w, v = marginal(self._right[node], x, right_box)
                w_total += w
                v_total += v
            return w_total, v_total

        values = np.empty(len(mids))
        for i, x in enumerate(mids):
            w, v = marginal(0, float(x), self._bounds.copy())
            values[i] = v / w if w > 0 else 0.0
        probs = widths / widths.sum()
        mean = float(probs @ values)
        return float(probs @ (values - mean) ** 2)


class FanovaImportanceEvaluator(BaseImportanceEvaluator):
    """fANOVA on a random forest (n_trees=64, max_depth=64 by default)."""

    def __init__(
        self, *, n_trees: int = 64, max_depth: int = 64, seed: int | None = None
    ) -> None:
        _imports.check()
        self._n_trees = n_trees
        self._max_depth = max_depth
        self._seed = seed

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
        non_single = {k: v for k, v in distributions.items() if not v.single()}
        if len(non_single) == 0:
            return {k: 0.0 for k in params}

        trials = [
            t for t in _get_filtered_trials(study, target)
            if all(name in t.params for name in non_single)
        ]
        trans = _SearchSpaceTransform(non_single, transform_log=True, transform_step=True)
        X = np.stack([trans.transform(t.params) for t in trials])
        y = np.asarray(_get_target_values(trials, target), dtype=np.float64)

        forest = RandomForestRegressor(
            n_estimators=self._n_trees,
            max_depth=self._max_depth,
            min_samples_leaf=1,
            random_state=self._seed,
        )
        forest.fit(X, y)

        # Importance of param p = mean over trees of V_marginal(cols of p) / V_total.
        importances = {name: 0.0 for name in non_single}
        n_effective = 0
        for estimator in forest.estimators_:
            marg = _TreeMarginals(estimator.tree_, trans.bounds.copy())
            v_total = marg.total_variance()
            if v_total <= 0:
                continue
            n_effective += 1
            for i, name in enumerate(non_single):
                cols = trans.column_to_encoded_columns[i]
                v = sum(marg.marginal_variance(int(c)) for c in cols)
                importances[name] += v / v_total
        if n_effective > 0:
            importances = {k: v / n_effective for k, v in importances.items()}
        return _sort_dict_by_importance({**{k: 0.0 for k in params}, **importances})
