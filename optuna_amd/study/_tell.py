"""Trial completion: value validation, state inference, sampler.after_trial, commit.

Parity: reference ``optuna/study/_tell.py`` (_tell_with_warning :80,
_check_values_are_feasible :60; pruned→last-intermediate-value promotion).
"""
from __future__ import annotations

import copy
import math
from typing import TYPE_CHECKING, Any, Sequence

from optuna_amd import logging as _logging
from optuna_amd.trial import FrozenTrial, Trial, TrialState


if TYPE_CHECKING:
    from optuna_amd.study import Study

_logger = _logging.get_logger(__name__)


def _get_frozen_trial(study: "Study", trial: Trial | int) -> FrozenTrial:
    if isinstance(trial, Trial):
        trial_id = trial._trial_id
    elif isinstance(trial, int):
        trial_number = trial
        try:
            trial_id = study._storage.get_trial_id_from_study_id_trial_number(
                study._study_id, trial_number
            )
        except KeyError as e:
            raise ValueError(
                f"Cannot tell for trial with number {trial_number} since it has not been "
                "created."
            ) from e
    else:
        raise TypeError("Trial must be a trial object or trial number.")
    return study._storage.get_trial(trial_id)


def _check_values_are_feasible(study: "Study", values: Sequence[float]) -> str | None:
    """NaN and non-float-castable values are infeasible → the trial FAILs
    (parity: reference study/_tell.py:60-77)."""
    errors = []
    for v in values:
        try:
            float(v)
        except (ValueError, TypeError):
            errors.append(f"The value {v!r} could not be cast to float")
            continue
        if math.isnan(float(v)):
            errors.append(f"The value {v} is not acceptable")
    if errors:
        return "; ".join(errors)
    if len(values) != len(study.directions):
        return (
            f"The number of the values {len(values)} did not match the number of the "
            f"objectives {len(study.directions)}"
        )
    return None


def _tell_with_warning(
    study: "Study",
    trial: Trial | int,
    value_or_values: float | Sequence[float] | None = None,
    state: TrialState | None = None,
    skip_if_finished: bool = False,
    suppress_warning: bool = False,
) -> tuple[TrialState, list[float] | None, str | None]:
    """Validate and commit the trial; returns (state, values, failure message).

    Returning the commit outcome (rather than re-reading the stored trial) lets
    the optimize loop log without a storage round trip per trial.
    """
    frozen_trial = _get_frozen_trial(study, trial)
    warning_message = None

    if frozen_trial.state.is_finished() and skip_if_finished:
        _logger.info(
            f"Skipped telling trial {frozen_trial.number} with values "
            f"{value_or_values} and state {state} since trial was already finished. "
            f"Finished trial has values {frozen_trial.values} and state {frozen_trial.state}."
        )
        return frozen_trial.state, frozen_trial.values, None
    if frozen_trial.state != TrialState.RUNNING:
        raise ValueError(f"Cannot tell a {frozen_trial.state.name} trial.")

    if state == TrialState.COMPLETE and value_or_values is None:
        raise ValueError(
            "No values were told. Values are required when state is TrialState.COMPLETE."
        )
    if state in (TrialState.PRUNED, TrialState.FAIL) and value_or_values is not None:
        raise ValueError(
            "Values were told. Values cannot be specified when state is "
            "TrialState.PRUNED or TrialState.FAIL."
        )
    if state is not None and state not in (
        TrialState.COMPLETE,
        TrialState.PRUNED,
        TrialState.FAIL,
    ):
        raise ValueError(f"Cannot tell with state {state}.")

    values: list[float] | None
    if value_or_values is None:
        values = None
    elif isinstance(value_or_values, Sequence) and not isinstance(value_or_values, str):
        values = list(value_or_values)
    else:
        values = [value_or_values]  # type: ignore[list-item]

    if state == TrialState.PRUNED:
        # Promote the last reported intermediate value to the final value —
        # but only if it is usable as one (NaN stays out).
        assert values is None
        last_step = frozen_trial.last_step
        if last_step is not None:
            last_intermediate = frozen_trial.intermediate_values[last_step]
            if _check_values_are_feasible(study, [last_intermediate]) is None:
                values = [last_intermediate]

    if state is None:
        # The optimize path: infeasible values fail the trial with a warning
        # rather than raising (an objective returning NaN must not kill the
        # whole optimize loop).
        if values is None:
            warning_message = "The value None could not be cast to float."
        else:
            warning_message = _check_values_are_feasible(study, values)
        if warning_message is None:
            state = TrialState.COMPLETE
        else:
            state = TrialState.FAIL
            values = None
    elif state == TrialState.COMPLETE:
        assert values is not None
        feasibility_message = _check_values_are_feasible(study, values)
        if feasibility_message is not None:
            raise ValueError(feasibility_message)

    assert state is not None
    if values is not None:
        values = [float(v) for v in values]

    if warning_message is not None:
        # Record while the trial is still mutable (RUNNING).
        study._storage.set_trial_system_attr(
            frozen_trial._trial_id, "study:tell_warning", warning_message
        )
        if not suppress_warning:
            import warnings

            warnings.warn(warning_message)
            warning_message = None

    try:
        # Hyperband needs samplers to observe the bracket-filtered study.
        from optuna_amd.pruners import _filter_study

        filtered_study = _filter_study(study, frozen_trial)
        study.sampler.after_trial(filtered_study, frozen_trial, state, values)
    finally:
        study._storage.set_trial_state_values(frozen_trial._trial_id, state, values)

    return state, values, warning_message
