"""Warning helper that attributes warnings to user code, not framework frames.

Parity: reference ``optuna/_warnings.py`` (optuna_warn :13-40). On Python 3.12+
``skip_file_prefixes`` points the warning at the caller outside the package; on
older interpreters it is a plain ``warnings.warn``.
"""
from __future__ import annotations

import os
import sys
import warnings
from pathlib import Path


_MODULE_ROOT: str = str(Path(__file__).resolve().parent) + os.sep


def optuna_warn(
    message: str,
    category: type[Warning] = UserWarning,
    stacklevel: int = 1,
) -> None:
    if sys.version_info >= (3, 12):
        warnings.warn(
            message,
            category,
            skip_file_prefixes=(_MODULE_ROOT,),  # type: ignore[call-arg]
        )
    else:
        warnings.warn(message, category, stacklevel=stacklevel + 1)
