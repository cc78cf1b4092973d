"""Library-wide logging with an opt-out default handler.

Parity: reference ``optuna/logging.py`` (get_logger :96, set_verbosity :146,
disable/enable_default_handler :169/:181, disable/enable_propagation :193/:207).
colorlog is replaced by a tiny ANSI formatter (no external dependency).
"""
from __future__ import annotations

import logging
import os
import sys
import threading
from logging import CRITICAL, DEBUG, ERROR, FATAL, INFO, WARN, WARNING  # noqa: F401


_lock = threading.Lock()
_default_handler: logging.Handler | None = None

_LEVEL_COLORS = {
    logging.DEBUG: "\x1b[36m",
    logging.INFO: "\x1b[32m",
    logging.WARNING: "\x1b[33m",
    logging.ERROR: "\x1b[31m",
    logging.CRITICAL: "\x1b[35m",
}
_RESET = "\x1b[0m"


class _ColorFormatter(logging.Formatter):
    def __init__(self, use_color: bool) -> None:
        super().__init__("[%(levelname)1.1s %(asctime)s,%(msecs)03d] %(message)s", "%Y-%m-%d %H:%M:%S")
        self._use_color = use_color

    def format(self, record: logging.LogRecord) -> str:
        text = super().format(record)
        if self._use_color:
            color = _LEVEL_COLORS.get(record.levelno, "")
            if color:
                return f"{color}{text}{_RESET}"
        return text


def create_default_formatter() -> logging.Formatter:
    use_color = sys.stderr.isatty() and os.environ.get("NO_COLOR") is None
    return _ColorFormatter(use_color)


def _get_library_name() -> str:
    return __name__.split(".")[0]


def _get_library_root_logger() -> logging.Logger:
    return logging.getLogger(_get_library_name())


def _configure_library_root_logger() -> None:
    global _default_handler
    with _lock:
        if _default_handler is not None:
            return
        _default_handler = logging.StreamHandler()
        _default_handler.setFormatter(create_default_formatter())
        root = _get_library_root_logger()
        root.addHandler(_default_handler)
        root.setLevel(logging.INFO)
        root.propagate = False


def _reset_library_root_logger() -> None:
    """Undo ``_configure_library_root_logger`` (test/teardown hook)."""
    global _default_handler
    with _lock:
        if _default_handler is None:
            return
        root = _get_library_root_logger()
        root.removeHandler(_default_handler)
        root.setLevel(logging.NOTSET)
        _default_handler = None


def get_logger(name: str) -> logging.Logger:
    """Return a logger under the library namespace with the default handler set up.

    Names under the reference package's namespace (``optuna`` / ``optuna.*``)
    are remapped onto this library's namespace so drop-in callers keep
    participating in :func:`set_verbosity` / propagation controls.
    """
    _configure_library_root_logger()
    lib = _get_library_name()
    if name == "optuna" or name.startswith("optuna."):
        name = lib + name[len("optuna"):]
    return logging.getLogger(name)


def get_verbosity() -> int:
    _configure_library_root_logger()
    return _get_library_root_logger().getEffectiveLevel()


def set_verbosity(verbosity: int) -> None:
    _configure_library_root_logger()
    _get_library_root_logger().setLevel(verbosity)


def disable_default_handler() -> None:
    _configure_library_root_logger()
    assert _default_handler is not None
    _get_library_root_logger().removeHandler(_default_handler)


def enable_default_handler() -> None:
    _configure_library_root_logger()
    assert _default_handler is not None
    _get_library_root_logger().addHandler(_default_handler)


def disable_propagation() -> None:
    _configure_library_root_logger()
    _get_library_root_logger().propagate = False


def enable_propagation() -> None:
    _configure_library_root_logger()
    _get_library_root_logger().propagate = True


def is_default_handler_enabled() -> bool:
    _configure_library_root_logger()
    return _default_handler in _get_library_root_logger().handlers
