"""The trial-execution loop: n_jobs threading, exception→state mapping, heartbeat.

Parity: reference ``optuna/study/_optimize.py`` (_optimize :39, thread pool :87-121,
_optimize_sequential :127, _run_trial :186).
"""
from __future__ import annotations

import copy
import datetime
import gc
import itertools
import os
import sys
from concurrent.futures import FIRST_COMPLETED, Future, ThreadPoolExecutor, wait
from typing import TYPE_CHECKING, Any, Callable, Sequence

from optuna_amd import exceptions, logging as _logging
from optuna_amd.study._tell import _tell_with_warning
from optuna_amd.progress_bar import _ProgressBar
from optuna_amd.storages._heartbeat import (
    fail_stale_trials,
    get_heartbeat_thread,
    is_heartbeat_enabled,
)
from optuna_amd.trial import FrozenTrial, Trial, TrialState


if TYPE_CHECKING:
    from optuna_amd.study import Study
    from optuna_amd.study.study import ObjectiveFuncType

_logger = _logging.get_logger(__name__)


def _optimize(
    study: "Study",
    func: "ObjectiveFuncType",
    n_trials: int | None = None,
    timeout: float | None = None,
    n_jobs: int = 1,
    catch: tuple[type[Exception], ...] = (),
    callbacks: list[Callable[["Study", FrozenTrial], None]] | None = None,
    gc_after_trial: bool = False,
    show_progress_bar: bool = False,
) -> None:
    if not isinstance(catch, tuple):
        raise TypeError("The catch argument is of type '{}' but must be a tuple.".format(
            type(catch).__name__
        ))
    if study._thread_local.in_optimize_loop:
        raise RuntimeError("Nested invocation of `Study.optimize` method isn't allowed.")
    if show_progress_bar and n_trials is None and timeout is not None and n_jobs != 1:
        import warnings as _warnings

        _warnings.warn("The timeout-based progress bar is not supported with n_jobs != 1.")
        show_progress_bar = False

    progress_bar = _ProgressBar(show_progress_bar, n_trials, timeout)
    study._stop_flag = False

    try:
        if n_jobs == 1:
            _optimize_sequential(
                study,
                func,
                n_trials,
                timeout,
                catch,
                callbacks,
                gc_after_trial,
                reseed_sampler_rng=False,
                time_start=None,
                progress_bar=progress_bar,
            )
        else:
            if n_jobs == -1:
                n_jobs = os.cpu_count() or 1
            time_start = datetime.datetime.now()
            futures: set[Future] = set()

            with ThreadPoolExecutor(max_workers=n_jobs) as executor:
                for n_submitted_trials in itertools.count():
                    if study._stop_flag:
                        break
                    if (
                        timeout is not None
                        and (datetime.datetime.now() - time_start).total_seconds() > timeout
                    ):
                        break
                    if n_trials is not None and n_submitted_trials >= n_trials:
                        break
                    if len(futures) >= n_jobs:
                        completed, futures = wait(futures, return_when=FIRST_COMPLETED)
                        for f in completed:
                            f.result()  # re-raise
                    futures.add(
                        executor.submit(
                            _optimize_sequential,
                            study,
                            func,
                            1,  # one trial per job
                            timeout,
                            catch,
                            callbacks,
                            gc_after_trial,
                            True,  # reseed per worker thread
                            time_start,
                            progress_bar,
                        )
                    )
                for f in futures:
                    f.result()
    finally:
        study._thread_local.in_optimize_loop = False
        progress_bar.close()


def _optimize_sequential(
    study: "Study",
    func: "ObjectiveFuncType",
    n_trials: int | None,
    timeout: float | None,
    catch: tuple[type[Exception], ...],
    callbacks: list[Callable[["Study", FrozenTrial], None]] | None,
    gc_after_trial: bool,
    reseed_sampler_rng: bool,
    time_start: datetime.datetime | None,
    progress_bar: _ProgressBar | None,
) -> None:
    study._thread_local.in_optimize_loop = True
    if reseed_sampler_rng:
        study.sampler.reseed_rng()

    i_trial = 0
    if time_start is None:
        time_start = datetime.datetime.now()

    while True:
        if study._stop_flag:
            break
        if n_trials is not None and i_trial >= n_trials:
            break
        if timeout is not None:
            elapsed = (datetime.datetime.now() - time_start).total_seconds()
            if elapsed > timeout:
                break
        try:
            frozen_trial_id = _run_trial(study, func, catch)
        finally:
            if gc_after_trial:
                gc.collect()
        if callbacks is not None:
            frozen_trial = study._storage.get_trial(frozen_trial_id)
            for callback in callbacks:
                callback(study, copy.deepcopy(frozen_trial))
        if progress_bar is not None:
            elapsed = (datetime.datetime.now() - time_start).total_seconds()
            progress_bar.update(elapsed, study)
        i_trial += 1

    study._storage.remove_session()


def _run_trial(
    study: "Study",
    func: "ObjectiveFuncType",
    catch: tuple[type[Exception], ...],
) -> int:
    """Run one trial; returns the finished trial's id (reference :186-190)."""
    if is_heartbeat_enabled(study._storage):
        fail_stale_trials(study)

    trial = study.ask()

    state: TrialState | None = None
    value_or_values: Any = None
    func_err: Exception | KeyboardInterrupt | None = None
    func_err_fail_exc_info: Any = None

    with get_heartbeat_thread(trial._trial_id, study._storage):
        try:
            value_or_values = func(trial)
        except exceptions.TrialPruned as e:
            state = TrialState.PRUNED
            func_err = e
        except (Exception, KeyboardInterrupt) as e:
            state = TrialState.FAIL
            func_err = e
            func_err_fail_exc_info = sys.exc_info()

    # Commit the trial (validations and pruned-value promotion happen in _tell).
    updated_state, values, warning_message = _tell_with_warning(
        study=study,
        trial=trial,
        value_or_values=value_or_values,
        state=state,
        suppress_warning=True,
    )

    if updated_state == TrialState.COMPLETE:
        assert values is not None
        study._log_completed_trial(values, trial.number, trial.params)
    elif updated_state == TrialState.PRUNED:
        _logger.info(f"Trial {trial.number} pruned. {str(func_err)}")
    elif updated_state == TrialState.FAIL:
        if func_err is not None:
            _log_failed_trial(
                trial.number,
                trial.params,
                repr(func_err),
                exc_info=func_err_fail_exc_info,
                value_or_values=value_or_values,
            )
        elif warning_message is not None:
            _log_failed_trial(
                trial.number, trial.params, warning_message, value_or_values=value_or_values
            )

    if func_err is not None and not isinstance(func_err, (*catch, exceptions.TrialPruned)):
        raise func_err
    return trial._trial_id


def _log_failed_trial(
    trial_number: int,
    trial_params: Any,
    message: Any,
    exc_info: Any = None,
    value_or_values: Any = None,
) -> None:
    _logger.warning(
        f"Trial {trial_number} failed with parameters: {trial_params} "
        f"because of the following error: {message}.",
        exc_info=exc_info,
    )
    _logger.warning(f"Trial {trial_number} failed with value {value_or_values!r}.")
