"""Coverage for utility layers: logging, decorators, callbacks, search-space
helpers, and multi-objective primitives (reference tests/test_logging.py,
test_experimental.py, test_deprecated.py, test_convert_positional_args.py,
test_callbacks.py, search_space_tests/, test_multi_objective.py)."""
from __future__ import annotations

import logging as py_logging
import warnings

import numpy as np
import pytest

import optuna_amd
from optuna_amd.distributions import FloatDistribution, IntDistribution
from optuna_amd.trial import TrialState


def test_logging_verbosity_roundtrip() -> None:
    import optuna_amd.logging as olog

    old = olog.get_verbosity()
    try:
        olog.set_verbosity(olog.DEBUG)
        assert olog.get_verbosity() == py_logging.DEBUG
        olog.set_verbosity(olog.WARNING)
        assert olog.get_verbosity() == py_logging.WARNING
    finally:
        olog.set_verbosity(old)


def test_logging_handler_toggles() -> None:
    import optuna_amd.logging as olog

    olog.disable_default_handler()
    root = olog._get_library_root_logger()
    assert olog._default_handler not in root.handlers
    olog.enable_default_handler()
    assert olog._default_handler in root.handlers
    olog.enable_propagation()
    assert root.propagate
    olog.disable_propagation()
    assert not root.propagate


def test_experimental_func_warns() -> None:
    from optuna_amd._experimental import experimental_func

    @experimental_func("1.2.3")
    def f(x: int) -> int:
        """Docstring."""
        return x + 1

    with pytest.warns(Warning):
        assert f(1) == 2
    assert "1.2.3" in (f.__doc__ or "")


def test_deprecated_func_warns() -> None:
    from optuna_amd._deprecated import deprecated_func

    @deprecated_func("1.0.0", "9.0.0")
    def g() -> int:
        return 7

    with pytest.warns(FutureWarning):
        assert g() == 7


def test_convert_positional_args() -> None:
    from optuna_amd._convert_positional_args import convert_positional_args

    @convert_positional_args(previous_positional_arg_names=["a", "b"])
    def h(a: int, *, b: int = 2) -> int:
        return a * 10 + b

    with pytest.warns(FutureWarning):
        assert h(3, 4) == 34  # now-keyword-only arg passed positionally: warns
    assert h(3, b=4) == 34
    with pytest.raises(TypeError):
        h(1, 2, 3)  # type: ignore[call-arg]


def test_max_trials_callback_counts_states() -> None:
    study = optuna_amd.create_study()
    from optuna_amd.study import MaxTrialsCallback

    cb = MaxTrialsCallback(3, states=(TrialState.COMPLETE,))
    study.optimize(lambda t: t.suggest_float("x", 0, 1), n_trials=10, callbacks=[cb])
    assert len(study.get_trials(states=(TrialState.COMPLETE,))) == 3


def test_retry_failed_trial_callback() -> None:
    from optuna_amd.storages import RetryFailedTrialCallback

    cb = RetryFailedTrialCallback(max_retry=2, inherit_intermediate_values=True)
    study = optuna_amd.create_study()
    t = study.ask()
    t.suggest_float("x", 0, 1)
    t.report(0.5, step=1)
    frozen = study._storage.get_trial(t._trial_id)
    study.tell(t, state=TrialState.FAIL)
    frozen = study._storage.get_trial(t._trial_id)
    cb(study, frozen)
    waiting = study.get_trials(states=(TrialState.WAITING,), deepcopy=False)
    assert len(waiting) == 1
    assert waiting[0].system_attrs.get("failed_trial") == frozen.number
    assert waiting[0].system_attrs.get("retry_history") == [frozen.number]
    assert RetryFailedTrialCallback.retried_trial_number(waiting[0]) == frozen.number
    assert waiting[0].params == frozen.params
    assert waiting[0].intermediate_values == {1: 0.5}


def test_intersection_search_space_incremental() -> None:
    from optuna_amd.search_space import IntersectionSearchSpace, intersection_search_space

    study = optuna_amd.create_study()
    study.optimize(lambda t: t.suggest_float("a", 0, 1), n_trials=2)
    study.optimize(
        lambda t: t.suggest_float("a", 0, 1) + t.suggest_float("b", 0, 1), n_trials=2
    )
    space = IntersectionSearchSpace()
    got = space.calculate(study)
    assert list(got) == ["a"]
    # Repeated call with no new finished trials hits the cached result.
    assert space.calculate(study) == got
    assert intersection_search_space(study.get_trials(deepcopy=False)) == got
    # Another study (same storage, different id) on the same object is rejected.
    other = optuna_amd.create_study(storage=study._storage, study_name="other")
    with pytest.raises(ValueError):
        space.calculate(other)


def test_group_decomposed_search_space() -> None:
    from optuna_amd.search_space.group_decomposed import _GroupDecomposedSearchSpace

    study = optuna_amd.create_study()

    def obj(t):
        if t.number % 2 == 0:
            return t.suggest_float("a", 0, 1) + t.suggest_float("b", 0, 1)
        return t.suggest_float("c", 0, 1)

    study.optimize(obj, n_trials=4)
    groups = _GroupDecomposedSearchSpace().calculate(study).search_spaces
    keysets = sorted(tuple(sorted(g)) for g in groups)
    assert keysets == [("a", "b"), ("c",)]


def test_fast_non_domination_rank_with_penalty_and_nan() -> None:
    from optuna_amd.study._multi_objective import _fast_non_domination_rank

    vals = np.array([[0.0, 0.0], [1.0, 1.0], [0.5, 0.5], [2.0, 2.0]])
    penalty = np.array([0.0, 0.0, 1.0, np.nan])
    ranks = _fast_non_domination_rank(vals, penalty=penalty)
    # Feasible ranked first by objectives, infeasible after, NaN last.
    assert ranks[0] == 0 and ranks[1] == 1
    assert ranks[2] == 2 and ranks[3] == 3

    plain = _fast_non_domination_rank(vals)
    assert plain[0] == 0 and plain[2] == 1 and plain[1] == 2 and plain[3] == 3


def test_dominates_semantics() -> None:
    from optuna_amd.study._multi_objective import _dominates
    from optuna_amd.study import StudyDirection
    from optuna_amd.trial import create_trial

    d2 = [StudyDirection.MINIMIZE, StudyDirection.MAXIMIZE]
    t = lambda v: create_trial(values=list(v), state=TrialState.COMPLETE)
    assert _dominates(t([0.0, 2.0]), t([1.0, 1.0]), d2)
    assert not _dominates(t([0.0, 1.0]), t([1.0, 2.0]), d2)
    assert not _dominates(t([0.0, 1.0]), t([0.0, 1.0]), d2)


def test_optimize_catch_and_failed_state() -> None:
    study = optuna_amd.create_study()

    def obj(t):
        t.suggest_float("x", 0, 1)
        if t.number % 2 == 0:
            raise ValueError("boom")
        return 1.0

    study.optimize(obj, n_trials=6, catch=(ValueError,))
    states = [t.state for t in study.trials]
    assert states.count(TrialState.FAIL) == 3
    assert states.count(TrialState.COMPLETE) == 3
    # without catch, the exception propagates
    with pytest.raises(ValueError):
        study.optimize(obj, n_trials=1)


def test_optimize_callbacks_see_final_state() -> None:
    seen = []
    study = optuna_amd.create_study()

    def cb(st, trial):
        seen.append((trial.number, trial.state))

    def obj(t):
        t.suggest_float("x", 0, 1)
        if t.number == 1:
            raise optuna_amd.TrialPruned()
        return 0.0

    study.optimize(obj, n_trials=3, callbacks=[cb])
    assert [s for _, s in seen] == [
        TrialState.COMPLETE,
        TrialState.PRUNED,
        TrialState.COMPLETE,
    ]


def test_failed_trial_state_recorded() -> None:
    study = optuna_amd.create_study()

    def obj(t):
        t.suggest_float("x", 0, 1)
        raise RuntimeError("xyz failure")

    study.optimize(obj, n_trials=1, catch=(RuntimeError,))
    t = study.trials[0]
    assert t.state == TrialState.FAIL
    # No extra system attrs are written for plain failures (the reference keeps
    # the trial record clean; failure details go to the log).
    assert "fail_reason" not in t.system_attrs
