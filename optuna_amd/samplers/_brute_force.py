"""BruteForceSampler: exhaustive search over a (possibly conditional) finite space.

The observed trials define a lazily-discovered tree: each internal node is a
parameter, edges are its candidate values (internal repr), leaves are complete
parameter assignments. Sampling picks children proportionally to the number of
unexpanded descendants (blended 50/50 with a flat distribution so unexplored
branches never starve), avoiding running trials' nodes when possible; the study
stops when the tree has no expandable node left.

Parity: reference ``optuna/samplers/_brute_force.py`` (_TreeNode :54 state
machine, sample_child weighting :186-215, after_trial early-return stop logic
:365-392, avoid_premature_stop).
"""
from __future__ import annotations

import decimal
import math
from dataclasses import dataclass
from functools import lru_cache
from numbers import Real
from typing import TYPE_CHECKING, Any, Sequence

import numpy as np

from optuna_amd.distributions import (
    BaseDistribution,
    CategoricalDistribution,
    FloatDistribution,
    IntDistribution,
)
from optuna_amd.samplers._base import BaseSampler
from optuna_amd.samplers._lazy_random_state import LazyRandomState
from optuna_amd.trial import FrozenTrial, TrialState, create_trial


if TYPE_CHECKING:
    from optuna_amd.study import Study

# (low, high, step) — candidate-enumeration arguments for one parameter.
ChoicesArgs = tuple


@lru_cache
def _enumerate_candidates(low, high, step) -> tuple[float, ...]:
    if step is None:
        raise ValueError(
            "FloatDistribution.step must be given for BruteForceSampler "
            "(otherwise, the search space will be infinite)."
        )
    if isinstance(low, int) and isinstance(high, int) and isinstance(step, int):
        return tuple(range(low, high + 1, step))
    lo = decimal.Decimal(str(low))
    hi = decimal.Decimal(str(high))
    st = decimal.Decimal(str(step))
    out = []
    while lo <= hi:
        out.append(float(lo))
        lo += st
    return tuple(out)


def _choices_args_of(dist: BaseDistribution) -> ChoicesArgs:
    if isinstance(dist, CategoricalDistribution):
        return (0, len(dist.choices) - 1, 1)
    assert isinstance(dist, (IntDistribution, FloatDistribution))
    return (dist.low, dist.high, dist.step)


@dataclass
class _Node:
    """States: Unexpanded (children None, not running) / Running (children None) /
    Leaf (children == {}) / Internal (children non-empty)."""

    param_name: str | None = None
    children: dict[float, "_Node | None"] | None = None  # None value = unexpanded child
    is_running: bool = False
    choices_args: ChoicesArgs | None = None

    def expand(self, param_name: str | None, choices_args: ChoicesArgs) -> None:
        if self.children is None:
            self.param_name = param_name
            self.children = {v: None for v in _enumerate_candidates(*choices_args)}
            self.choices_args = choices_args
        else:
            self._check_consistent(param_name, choices_args)

    def _check_consistent(self, param_name: str | None, choices_args: ChoicesArgs | None) -> None:
        if self.param_name != param_name:
            raise ValueError(f"param_name mismatch: {self.param_name} != {param_name}")
        if choices_args != self.choices_args:
            assert self.children is not None and choices_args is not None
            raise ValueError(
                f"search_space mismatch in {param_name}: "
                f"{list(self.children)} != {list(_enumerate_candidates(*choices_args))}"
            )

    def set_leaf(self) -> None:
        if self.children is not None:
            self._check_consistent(None, None)
        self.children = {}

    def add_path(self, path: list[tuple[str, ChoicesArgs, float]]) -> "_Node | None":
        node = self
        for param_name, choices_args, value in path:
            node.expand(param_name, choices_args)
            children = node.children
            if not children:
                return None  # a finished leaf crossed this prefix: off-grid
            if value not in children:
                return None  # off-grid value (e.g. out-of-range enqueue)
            child = children[value]
            if child is None:
                child = _Node()
                children[value] = child
            node = child
        return node

    def count_unexpanded(self, exclude_running: bool) -> int:
        if self.children is None:
            return 0 if exclude_running and self.is_running else 1
        return sum(
            1 if c is None else c.count_unexpanded(exclude_running)
            for c in self.children.values()
        )

    def is_any_expandable(self, exclude_running: bool) -> bool:
        if self.children is None:
            return not exclude_running or not self.is_running
        return any(
            True if c is None else c.is_any_expandable(exclude_running)
            for c in self.children.values()
        )

    def sample_child(self, rng: np.random.RandomState, exclude_running: bool) -> float:
        assert self.children is not None
        keys = list(self.children.keys())
        counts = np.array(
            [
                1 if c is None else c.count_unexpanded(exclude_running)
                for c in self.children.values()
            ],
            dtype=float,
        )
        # Blend proportional with flat weights so shallow branches aren't starved.
        alpha = 0.5
        weights = (1 - alpha) * counts / counts.sum()
        flat = np.where(counts > 0, 1.0, 0.0)
        weights += alpha * flat / flat.sum()
        # Prefer children without running trials when any such child has weight.
        child_running = [c is not None and c.is_running for c in self.children.values()]
        if any(w > 0 and not r for w, r in zip(weights, child_running)):
            weights = np.where(child_running, 0.0, weights)
        weights /= weights.sum()
        return rng.choice(keys, p=weights).item()


def _is_nan(v: Any) -> bool:
    return isinstance(v, Real) and math.isnan(float(v))


# Reference-compatible names: the reference models an unexpanded child as a
# dedicated sentinel object; this implementation uses None for the same state.
_TreeNode = _Node
_UNEXPANDED_NODE = None


class _UnexpandedTreeNode:
    """Placeholder type for the reference's unexpanded-child sentinel."""

    is_running: bool = False


class BruteForceSampler(BaseSampler):
    def __init__(self, seed: int | None = None, avoid_premature_stop: bool = False) -> None:
        self._rng = LazyRandomState(seed)
        self._avoid_premature_stop = avoid_premature_stop

    def reseed_rng(self) -> None:
        self._rng.rng.seed()

    def infer_relative_search_space(
        self, study: "Study", trial: FrozenTrial
    ) -> dict[str, BaseDistribution]:
        return {}

    def sample_relative(
        self, study: "Study", trial: FrozenTrial, search_space: dict[str, BaseDistribution]
    ) -> dict[str, Any]:
        return {}

    @staticmethod
    def _trials_and_current_index(
        study: "Study", current_trial_number: int
    ) -> tuple[list[FrozenTrial], int]:
        states = (TrialState.COMPLETE, TrialState.PRUNED, TrialState.RUNNING, TrialState.FAIL)
        # Bypass bracket-filtered views: the tree must see all trials.
        trials = study._storage.get_all_trials(study._study_id, deepcopy=False, states=states)
        for i in range(1, len(trials) + 1):
            if trials[-i].number == current_trial_number:
                return trials, len(trials) - i
        raise AssertionError("current trial not found")

    @staticmethod
    def _populate_tree(tree: _Node, trials: list[FrozenTrial], params: dict[str, Any]) -> None:
        """Insert every trial consistent with `params` (the already-fixed prefix)."""
        nonnan_items = {k: v for k, v in params.items() if not _is_nan(v)}.items()
        nan_names = [k for k, v in params.items() if _is_nan(v)]

        for trial in trials:
            if params:
                tp = trial.params
                if not (nonnan_items <= tp.items()):
                    continue
                if not all(_is_nan(tp.get(p)) for p in nan_names):
                    continue
            path = []
            for name, dist in trial.distributions.items():
                if name in params:
                    continue
                value = dist.to_internal_repr(trial.params[name])
                path.append((name, _choices_args_of(dist), value))
            leaf = tree.add_path(path)
            if leaf is not None:
                if trial.state.is_finished():
                    leaf.set_leaf()
                else:
                    leaf.is_running = True

    def sample_independent(
        self,
        study: "Study",
        trial: FrozenTrial,
        param_name: str,
        param_distribution: BaseDistribution,
    ) -> Any:
        exclude_running = not self._avoid_premature_stop
        trials, current_idx = self._trials_and_current_index(study, trial.number)
        trials.pop(current_idx)
        tree = _Node()
        c_args = _choices_args_of(param_distribution)
        tree.expand(param_name, c_args)
        self._populate_tree(tree, trials, trial.params)
        if tree.is_any_expandable(exclude_running):
            value = tree.sample_child(self._rng.rng, exclude_running)
        else:
            value = self._rng.rng.choice(_enumerate_candidates(*c_args)).item()
        return param_distribution.to_external_repr(value)

    def after_trial(
        self,
        study: "Study",
        trial: FrozenTrial,
        state: TrialState,
        values: Sequence[float] | None,
    ) -> None:
        exclude_running = not self._avoid_premature_stop
        trials, current_idx = self._trials_and_current_index(study, trial.number)
        trials[current_idx] = create_trial(
            state=state, values=values, params=trial.params, distributions=trial.distributions
        )
        # Walk prefixes from deepest to shallowest: if any prefix still has an
        # expandable continuation, more evaluations are needed; otherwise stop.
        params = trial.params.copy()
        for param_name in reversed(list(trial.params.keys())):
            params.pop(param_name)
            tree = _Node()
            self._populate_tree(tree, trials, params)
            if tree.is_any_expandable(exclude_running):
                return
        study.stop()
