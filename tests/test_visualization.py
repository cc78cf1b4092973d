"""Visualization: info-layer correctness + both renderers run on every plot."""
from __future__ import annotations

import math
import warnings

import numpy as np
import pytest

import optuna_amd
from optuna_amd import visualization as vis
from optuna_amd.trial import TrialState


optuna_amd.logging.set_verbosity(optuna_amd.logging.WARNING)


@pytest.fixture(scope="module")
def study() -> optuna_amd.Study:
    study = optuna_amd.create_study(sampler=optuna_amd.samplers.RandomSampler(seed=0))

    def objective(trial):
        x = trial.suggest_float("x", -5, 5)
        lg = trial.suggest_float("lg", 1e-3, 1e2, log=True)
        c = trial.suggest_categorical("c", ("a", "b"))
        for step in range(3):
            trial.report(x**2 + step, step)
        return x**2 + 0.1 * np.log10(lg) + (0.5 if c == "b" else 0.0)

    study.optimize(objective, n_trials=25)
    return study


@pytest.fixture(scope="module")
def mo_study() -> optuna_amd.Study:
    study = optuna_amd.create_study(
        directions=["minimize", "minimize"],
        sampler=optuna_amd.samplers.RandomSampler(seed=1),
    )
    study.optimize(
        lambda t: (t.suggest_float("x", 0, 1), 1 - t.suggest_float("x", 0, 1)),
        n_trials=20,
    )
    return study


def test_is_available() -> None:
    assert vis.is_available()
    assert vis.matplotlib.is_available()


def test_optimization_history_info(study) -> None:
    from optuna_amd.visualization._optimization_history import (
        _get_optimization_history_info_list,
    )

    (info,) = _get_optimization_history_info_list(study, None, "Objective Value", False)
    assert info.trial_numbers == list(range(25))
    assert info.best_values_info is not None
    bests = info.best_values_info.values
    assert all(b2 <= b1 for b1, b2 in zip(bests, bests[1:]))  # monotone minimize
    assert bests[-1] == pytest.approx(study.best_value)


def test_edf_info(study) -> None:
    from optuna_amd.visualization._edf import _get_edf_info

    info = _get_edf_info(study)
    assert len(info.lines) == 1
    y = info.lines[0].y_values
    assert y[-1] == pytest.approx(1.0)
    assert np.all(np.diff(y) >= 0)


def test_slice_info(study) -> None:
    from optuna_amd.visualization._slice import _get_slice_plot_info

    info = _get_slice_plot_info(study, None, None, "Objective Value")
    names = [s.param_name for s in info.subplots]
    assert names == ["c", "lg", "x"]
    lg = info.subplots[1]
    assert lg.is_log
    assert len(lg.x) == 25


def test_pareto_front_info(mo_study) -> None:
    from optuna_amd.visualization._pareto_front import _get_pareto_front_info

    info = _get_pareto_front_info(mo_study)
    assert info.n_targets == 2
    assert len(info.best_trials_with_values) >= 1
    n_total = len(info.best_trials_with_values) + len(info.non_best_trials_with_values)
    assert n_total == 20


def test_hypervolume_history_info(mo_study) -> None:
    from optuna_amd.visualization._hypervolume_history import (
        _get_hypervolume_history_info,
    )

    info = _get_hypervolume_history_info(mo_study, np.array([2.0, 2.0]))
    assert len(info.values) == 20
    assert all(b >= a - 1e-12 for a, b in zip(info.values, info.values[1:]))


def test_timeline_info(study) -> None:
    from optuna_amd.visualization._timeline import _get_timeline_info

    info = _get_timeline_info(study)
    assert len(info.bars) == 25
    assert all(b.complete >= b.start for b in info.bars)


def test_intermediate_info(study) -> None:
    from optuna_amd.visualization._intermediate_values import (
        _get_intermediate_plot_info,
    )

    info = _get_intermediate_plot_info(study)
    assert len(info.trial_infos) == 25
    assert info.trial_infos[0].sorted_intermediate_values[0][0] == 0


PLOTLY_FUNCS = [
    lambda s: vis.plot_optimization_history(s),
    lambda s: vis.plot_slice(s),
    lambda s: vis.plot_contour(s, params=["x", "lg"]),
    lambda s: vis.plot_parallel_coordinate(s),
    lambda s: vis.plot_param_importances(s),
    lambda s: vis.plot_rank(s),
    lambda s: vis.plot_edf(s),
    lambda s: vis.plot_intermediate_values(s),
    lambda s: vis.plot_timeline(s),
]


@pytest.mark.parametrize("func", PLOTLY_FUNCS)
def test_plotly_renderers_run(study, func) -> None:
    with warnings.catch_warnings():
        warnings.simplefilter("ignore")
        fig = func(study)
    assert fig is not None


MPL_FUNCS = [
    lambda s: vis.matplotlib.plot_optimization_history(s),
    lambda s: vis.matplotlib.plot_slice(s),
    lambda s: vis.matplotlib.plot_contour(s, params=["x", "lg"]),
    lambda s: vis.matplotlib.plot_parallel_coordinate(s),
    lambda s: vis.matplotlib.plot_param_importances(s),
    lambda s: vis.matplotlib.plot_rank(s),
    lambda s: vis.matplotlib.plot_edf(s),
    lambda s: vis.matplotlib.plot_intermediate_values(s),
    lambda s: vis.matplotlib.plot_timeline(s),
]


@pytest.mark.parametrize("func", MPL_FUNCS)
def test_matplotlib_renderers_run(study, func) -> None:
    with warnings.catch_warnings():
        warnings.simplefilter("ignore")
        ax = func(study)
    assert ax is not None


def test_mo_plots(mo_study) -> None:
    assert vis.plot_pareto_front(mo_study) is not None
    assert vis.plot_hypervolume_history(mo_study, [2.0, 2.0]) is not None
    assert vis.matplotlib.plot_pareto_front(mo_study) is not None
    assert vis.matplotlib.plot_hypervolume_history(mo_study, [2.0, 2.0]) is not None


def test_single_objective_hypervolume_rejected(study) -> None:
    with pytest.raises(ValueError):
        vis.plot_hypervolume_history(study, [1.0])


def test_mo_requires_target(mo_study) -> None:
    with pytest.raises(ValueError):
        vis.plot_optimization_history(mo_study)
    fig = vis.plot_optimization_history(
        mo_study, target=lambda t: t.values[0], target_name="obj0"
    )
    assert fig is not None


def test_contour_info(study) -> None:
    from optuna_amd.visualization._contour import _get_contour_info

    info = _get_contour_info(study)
    assert info.sorted_params == ["c", "lg", "x"]
    n = len(info.sorted_params)
    assert len(info.sub_plot_infos) == n and all(len(r) == n for r in info.sub_plot_infos)
    # off-diagonal cells pair distinct params
    for i, row in enumerate(info.sub_plot_infos):
        for j, cell in enumerate(row):
            if i != j:
                assert cell.xaxis.name != cell.yaxis.name


def test_parallel_coordinate_info(study) -> None:
    from optuna_amd.visualization._parallel_coordinate import (
        _get_parallel_coordinate_info,
    )

    info = _get_parallel_coordinate_info(study)
    assert len(info.dims_params) >= 2
    assert info.dim_objective is not None


def test_rank_info(study) -> None:
    from optuna_amd.visualization._rank import _get_rank_info

    info = _get_rank_info(study, params=None, target=None, target_name="Objective Value")
    assert len(info.params) >= 2


def test_param_importances_info(study) -> None:
    from optuna_amd.visualization._param_importances import _get_importances_info
    from optuna_amd.importance import PedAnovaImportanceEvaluator

    info = _get_importances_info(
        study, PedAnovaImportanceEvaluator(), params=None, target=None,
        target_name="Objective Value",
    )
    assert set(info.param_names) <= {"x", "lg", "c"}
    assert len(info.importance_values) == len(info.param_names)


def test_contour_with_categorical_and_log(study) -> None:
    import optuna_amd
    from optuna_amd.visualization import plot_contour

    fig = plot_contour(study, params=["x", "lg"])
    assert fig is not None


def test_edf_multiple_studies() -> None:
    import optuna_amd
    from optuna_amd.visualization._edf import _get_edf_info

    studies = []
    for seed in (1, 2):
        s = optuna_amd.create_study(
            sampler=optuna_amd.samplers.RandomSampler(seed=seed)
        )
        s.optimize(lambda t: t.suggest_float("x", 0, 1) ** 2, n_trials=6)
        studies.append(s)
    info = _get_edf_info(studies)
    assert len(info.lines) == 2
    for line in info.lines:
        assert (np.diff(line.y_values) >= 0).all()  # EDFs are nondecreasing


def test_timeline_states_and_order(study) -> None:
    from optuna_amd.visualization._timeline import _get_timeline_info

    info = _get_timeline_info(study)
    assert len(info.bars) == len(study.trials)
    for bar in info.bars:
        assert bar.complete >= bar.start


def test_terminator_improvement_info() -> None:
    import optuna_amd
    from optuna_amd.visualization._terminator_improvement import (
        _get_improvement_info,
    )

    s = optuna_amd.create_study(sampler=optuna_amd.samplers.RandomSampler(seed=3))
    s.optimize(
        lambda t: t.suggest_float("a", 0, 1) + t.suggest_float("b", 0, 1),
        n_trials=6,
    )
    info = _get_improvement_info(s)
    assert len(info.trial_numbers) == 6
