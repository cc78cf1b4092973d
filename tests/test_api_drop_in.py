"""Drop-in compatibility: code written for the reference's README/tutorials
runs unchanged with `import optuna_amd as optuna`."""
from __future__ import annotations

import warnings

import optuna_amd as optuna


def test_reference_readme_example() -> None:
    # The reference repo's front-page example, verbatim apart from the import.
    def objective(trial):
        x = trial.suggest_float("x", -10, 10)
        return (x - 2) ** 2

    with warnings.catch_warnings():
        warnings.simplefilter("ignore")
        study = optuna.create_study()
        study.optimize(objective, n_trials=40)
    assert isinstance(study.best_params["x"], float)
    assert study.best_value < 25


def test_reference_tutorial_surface() -> None:
    # The canonical distributed/tutorial idioms: named RDB study,
    # load_if_exists, pruning report loop, user attrs, dataframe export.
    import tempfile

    with warnings.catch_warnings():
        warnings.simplefilter("ignore")
        with tempfile.TemporaryDirectory() as d:
            url = f"sqlite:///{d}/example.db"
            study = optuna.create_study(
                study_name="tut",
                storage=url,
                load_if_exists=True,
                pruner=optuna.pruners.MedianPruner(n_startup_trials=2),
            )

            def objective(trial):
                lr = trial.suggest_float("lr", 1e-5, 1e-1, log=True)
                layers = trial.suggest_int("layers", 1, 3)
                err = 1.0
                for step in range(5):
                    err = err * 0.8 + lr * layers * 0.01
                    trial.report(err, step)
                    if trial.should_prune():
                        raise optuna.TrialPruned()
                return err

            study.optimize(objective, n_trials=12)
            again = optuna.load_study(study_name="tut", storage=url)
            assert len(again.trials) == 12
            again.set_user_attr("dataset", "synthetic")
            df = again.trials_dataframe()
            assert {"number", "value", "state"} <= set(df.columns)


def test_reference_multiobjective_idiom() -> None:
    with warnings.catch_warnings():
        warnings.simplefilter("ignore")
        study = optuna.create_study(directions=["minimize", "minimize"])
        study.optimize(
            lambda t: (t.suggest_float("x", 0, 2), t.suggest_float("y", 0, 2)),
            n_trials=15,
        )
    assert len(study.best_trials) >= 1
    for t in study.best_trials:
        assert len(t.values) == 2
