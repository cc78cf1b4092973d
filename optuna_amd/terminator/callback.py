"""TerminatorCallback: run a terminator after every trial of ``Study.optimize``.

Parity: reference ``optuna/terminator/callback.py``.
"""
from __future__ import annotations

from typing import TYPE_CHECKING

from optuna_amd import logging as _logging
from optuna_amd.terminator.terminator import BaseTerminator, Terminator
from optuna_amd.trial import FrozenTrial


if TYPE_CHECKING:
    from optuna_amd.study import Study

_logger = _logging.get_logger(__name__)


class TerminatorCallback:
    def __init__(self, terminator: BaseTerminator | None = None) -> None:
        self._terminator = terminator or Terminator()

    def __call__(self, study: "Study", trial: FrozenTrial) -> None:
        should_terminate = self._terminator.should_terminate(study=study)
        if should_terminate:
            _logger.info("The study has been stopped by the terminator.")
            study.stop()
