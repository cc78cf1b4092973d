"""``@deprecated_func`` / ``@deprecated_class`` decorators.

Parity: reference ``optuna/_deprecated.py`` (deprecated_func :49, deprecated_class :117).
"""
from __future__ import annotations

import functools
import textwrap
import warnings
from typing import Any, Callable, TypeVar


FT = TypeVar("FT", bound=Callable[..., Any])
CT = TypeVar("CT")

_NOTE_TMPL = """

.. warning::
    Deprecated in v{dep}. This feature will be removed in v{rem}.
"""


def _default_removed(deprecated_version: str) -> str:
    major = int(deprecated_version.split(".")[0])
    return f"{major + 2}.0.0"


def _make_message(name: str, deprecated_version: str, removed_version: str, text: str | None) -> str:
    message = (
        f"{name} has been deprecated in v{deprecated_version}. "
        f"This feature will be removed in v{removed_version}."
    )
    if text:
        message += " " + text
    return message


def deprecated_func(
    deprecated_version: str,
    removed_version: str | None = None,
    name: str | None = None,
    text: str | None = None,
) -> Callable[[FT], FT]:
    removed = removed_version or _default_removed(deprecated_version)

    def decorator(func: FT) -> FT:
        @functools.wraps(func)
        def wrapper(*args: Any, **kwargs: Any) -> Any:
            warnings.warn(
                _make_message(name or func.__name__, deprecated_version, removed, text),
                FutureWarning,
                stacklevel=2,
            )
            return func(*args, **kwargs)

        wrapper.__doc__ = textwrap.dedent(func.__doc__ or "") + _NOTE_TMPL.format(
            dep=deprecated_version, rem=removed
        )
        return wrapper  # type: ignore[return-value]

    return decorator


def deprecated_class(
    deprecated_version: str,
    removed_version: str | None = None,
    name: str | None = None,
    text: str | None = None,
) -> Callable[[CT], CT]:
    removed = removed_version or _default_removed(deprecated_version)

    def decorator(cls: CT) -> CT:
        init = cls.__init__  # type: ignore[misc]

        @functools.wraps(init)
        def wrapped_init(self: Any, *args: Any, **kwargs: Any) -> None:
            warnings.warn(
                _make_message(name or cls.__name__, deprecated_version, removed, text),  # type: ignore[attr-defined]
                FutureWarning,
                stacklevel=2,
            )
            init(self, *args, **kwargs)

        cls.__init__ = wrapped_init  # type: ignore[misc]
        cls.__doc__ = textwrap.dedent(cls.__doc__ or "") + _NOTE_TMPL.format(  # type: ignore[attr-defined]
            dep=deprecated_version, rem=removed
        )
        return cls

    return decorator
