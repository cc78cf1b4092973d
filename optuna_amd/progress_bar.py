"""tqdm progress bar for ``Study.optimize`` (n_trials or timeout mode).

Parity: reference ``optuna/progress_bar.py`` (_ProgressBar :20, log-redirect :32).
"""
from __future__ import annotations

import logging
from typing import TYPE_CHECKING, Any

from optuna_amd import logging as _logging


if TYPE_CHECKING:
    from optuna_amd.study import Study

try:
    from tqdm.auto import tqdm

    _tqdm_available = True
except ImportError:  # pragma: no cover
    _tqdm_available = False


class _TqdmLoggingHandler(logging.StreamHandler):
    def emit(self, record: Any) -> None:
        try:
            msg = self.format(record)
            tqdm.write(msg)
            self.flush()
        except Exception:
            self.handleError(record)


class _ProgressBar:
    """Progress bar, enabled only when requested and tqdm is importable."""

    def __init__(
        self,
        is_valid: bool,
        n_trials: int | None = None,
        timeout: float | None = None,
    ) -> None:
        if is_valid and not _tqdm_available:
            _logging.get_logger(__name__).warning(
                "Progress bar requested but tqdm is not installed."
            )
            is_valid = False
        if is_valid and n_trials is None and timeout is None:
            import warnings

            warnings.warn(
                "Progress bar won't be displayed because n_trials and timeout are None."
            )
        self._is_valid = is_valid and (n_trials or timeout) is not None
        self._n_trials = n_trials
        self._timeout = timeout
        self._last_elapsed_seconds = 0.0
        if self._is_valid:
            if self._n_trials is not None:
                self._progress_bar = tqdm(total=self._n_trials)
            else:
                total = tqdm.format_interval(self._timeout)
                fmt = "{desc} {percentage:3.0f}%|{bar}| {elapsed}/" + total
                self._progress_bar = tqdm(total=self._timeout, bar_format=fmt)
            # Redirect library log lines above the bar.
            self._handler = _TqdmLoggingHandler()
            self._handler.setLevel(logging.INFO)
            self._handler.setFormatter(_logging.create_default_formatter())
            _logging.disable_default_handler()
            _logging._get_library_root_logger().addHandler(self._handler)

    def update(self, elapsed_seconds: float, study: "Study") -> None:
        if not self._is_valid:
            return
        if not study._is_multi_objective():
            try:
                msg = f"Best trial: {study.best_trial.number}. Best value: {study.best_value:.6g}"
            except ValueError:
                msg = "Best trial: None. Best value: None"
            self._progress_bar.set_description(msg)
        if self._n_trials is not None:
            self._progress_bar.update(1)
        elif self._timeout is not None:
            increment = elapsed_seconds - self._last_elapsed_seconds
            if increment > 0:
                self._progress_bar.update(increment)
                self._last_elapsed_seconds = elapsed_seconds

    def close(self) -> None:
        if self._is_valid:
            self._progress_bar.close()
            _logging._get_library_root_logger().removeHandler(self._handler)
            _logging.enable_default_handler()
