"""Optimize-loop callbacks: MaxTrialsCallback and storage retry callbacks.

Parity: reference ``optuna/_callbacks.py`` (MaxTrialsCallback :15) and
``optuna/storages/_callbacks.py`` (RetryFailedTrialCallback).
"""
from __future__ import annotations

from typing import TYPE_CHECKING, Container

from optuna_amd.trial import FrozenTrial, TrialState


if TYPE_CHECKING:
    from optuna_amd.study import Study


class MaxTrialsCallback:
    """Stop the study once ``n_trials`` trials in the given states exist (cross-process)."""

    def __init__(
        self, n_trials: int, states: Container[TrialState] | None = (TrialState.COMPLETE,)
    ) -> None:
        self._n_trials = n_trials
        self._states = states

    def __call__(self, study: "Study", trial: FrozenTrial) -> None:
        trials = study.get_trials(deepcopy=False, states=self._states)
        n_complete = len(trials)
        if n_complete >= self._n_trials:
            study.stop()


class RetryFailedTrialCallback:
    """Re-enqueue a failed trial's parameters as a new WAITING trial.

    Used as ``failed_trial_callback`` of a heartbeat-enabled storage
    (parity: reference storages/_callbacks.py:17-90).
    """

    def __init__(self, max_retry: int | None = None, inherit_intermediate_values: bool = False) -> None:
        self._max_retry = max_retry
        self._inherit_intermediate_values = inherit_intermediate_values

    def __call__(self, study: "Study", trial: FrozenTrial) -> None:
        from optuna_amd.trial import create_trial

        # Attr names match the reference exactly (storages/_callbacks.py:72-77):
        # "failed_trial" keeps the ORIGINAL failed number (existing attrs win in
        # the merge), "retry_history" accumulates every retried number.
        system_attrs: dict = {
            "failed_trial": trial.number,
            "retry_history": [],
            **trial.system_attrs,
        }
        system_attrs["retry_history"] = list(system_attrs["retry_history"])
        system_attrs["retry_history"].append(trial.number)
        if self._max_retry is not None:
            if self._max_retry < len(system_attrs["retry_history"]):
                return

        study.add_trial(
            create_trial(
                state=TrialState.WAITING,
                params=trial.params,
                distributions=trial.distributions,
                user_attrs=trial.user_attrs,
                system_attrs=system_attrs,
                intermediate_values=(
                    trial.intermediate_values if self._inherit_intermediate_values else None
                ),
            )
        )

    @staticmethod
    def retried_trial_number(trial: FrozenTrial) -> int | None:
        """Number of the first failed trial in the retry series (or None)."""
        return trial.system_attrs.get("failed_trial", None)

    @staticmethod
    def retry_history(trial: FrozenTrial) -> list[int]:
        """Retried trial numbers in series order, oldest first."""
        return trial.system_attrs.get("retry_history", [])
