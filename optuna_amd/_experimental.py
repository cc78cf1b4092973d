"""``@experimental_func`` / ``@experimental_class`` decorators.

Parity: reference ``optuna/_experimental.py`` (experimental_func :41, experimental_class :74).
Stamps the docstring and emits ``ExperimentalWarning`` on first use of a class.
"""
from __future__ import annotations

import functools
import textwrap
import warnings
from typing import Any, Callable, TypeVar

from optuna_amd.exceptions import ExperimentalWarning


FT = TypeVar("FT", bound=Callable[..., Any])
CT = TypeVar("CT")

_NOTE_TMPL = """

.. note::
    Added in v{ver} as an experimental feature. The interface may change in newer versions
    without prior notice.
"""


def _validate_version(version: str) -> None:
    if not isinstance(version, str) or len(version.split(".")) != 3:
        raise ValueError(f"Invalid semantic version: {version}")


def _stamp_doc(doc: str | None, version: str) -> str:
    base = textwrap.dedent(doc or "")
    return base + _NOTE_TMPL.format(ver=version)


def experimental_func(version: str, name: str | None = None) -> Callable[[FT], FT]:
    _validate_version(version)

    def decorator(func: FT) -> FT:
        @functools.wraps(func)
        def wrapper(*args: Any, **kwargs: Any) -> Any:
            warnings.warn(
                f"{name or func.__name__} is experimental (supported from v{version}). "
                "The interface can change in the future.",
                ExperimentalWarning,
                stacklevel=2,
            )
            return func(*args, **kwargs)

        wrapper.__doc__ = _stamp_doc(func.__doc__, version)
        return wrapper  # type: ignore[return-value]

    return decorator


def experimental_class(version: str, name: str | None = None) -> Callable[[CT], CT]:
    _validate_version(version)

    def decorator(cls: CT) -> CT:
        init = cls.__init__  # type: ignore[misc]

        @functools.wraps(init)
        def wrapped_init(self: Any, *args: Any, **kwargs: Any) -> None:
            warnings.warn(
                f"{name or cls.__name__} is experimental (supported from v{version}). "  # type: ignore[attr-defined]
                "The interface can change in the future.",
                ExperimentalWarning,
                stacklevel=2,
            )
            init(self, *args, **kwargs)

        cls.__init__ = wrapped_init  # type: ignore[misc]
        cls.__doc__ = _stamp_doc(cls.__doc__, version)  # type: ignore[attr-defined]
        return cls

    return decorator


def warn_experimental_argument(option_name: str) -> None:
    """Emit an ExperimentalWarning for an experimental keyword argument
    (parity: reference optuna/_experimental.py warn_experimental_argument)."""
    import warnings

    from optuna_amd.exceptions import ExperimentalWarning

    warnings.warn(
        f"Argument ``{option_name}`` is an experimental feature."
        " The interface can change in the future.",
        ExperimentalWarning,
        stacklevel=2,
    )
