"""Decorator that converts legacy positional arguments to keyword arguments.

Parity: reference ``optuna/_convert_positional_args.py`` (convert_positional_args :32).
"""
from __future__ import annotations

import functools
import warnings
from inspect import Parameter, signature
from typing import Any, Callable, Sequence, TypeVar


FT = TypeVar("FT", bound=Callable[..., Any])


def _validate_version(version: str) -> None:
    parts = version.split(".")
    if len(parts) < 2 or not all(p.split("a")[0].split("b")[0].isdigit() for p in parts if p):
        raise ValueError(f"Invalid version string: {version!r}.")


def convert_positional_args(
    *,
    previous_positional_arg_names: Sequence[str],
    deprecated_version: str | None = None,
    removed_version: str | None = None,
    warning_stacklevel: int = 2,
) -> Callable[[FT], FT]:
    if deprecated_version is not None or removed_version is not None:
        if deprecated_version is None:
            raise ValueError(
                "deprecated_version must not be None when removed_version is specified."
            )
        if removed_version is None:
            raise ValueError(
                "removed_version must not be None when deprecated_version is specified."
            )
        _validate_version(deprecated_version)
        _validate_version(removed_version)
    def decorator(func: FT) -> FT:
        sig = signature(func)
        kwonly = {
            name
            for name, p in sig.parameters.items()
            if p.kind == Parameter.KEYWORD_ONLY
        }
        missing = set(previous_positional_arg_names) - set(sig.parameters)
        assert not missing, f"{missing} not in the signature of {func.__name__}"

        @functools.wraps(func)
        def wrapper(*args: Any, **kwargs: Any) -> Any:
            if len(args) > len(previous_positional_arg_names):
                raise TypeError(
                    f"{func.__name__}() takes {len(previous_positional_arg_names)} positional"
                    f" arguments but {len(args)} were given."
                )
            for name, value in zip(previous_positional_arg_names, args):
                if name in kwargs:
                    raise TypeError(f"{func.__name__}() got multiple values for argument '{name}'.")
                if name in kwonly:
                    warnings.warn(
                        f"{name} is specified positionally to {func.__name__}(). "
                        "Positional use is deprecated; pass it as a keyword argument.",
                        FutureWarning,
                        stacklevel=warning_stacklevel,
                    )
                kwargs[name] = value
            return func(**kwargs)

        return wrapper  # type: ignore[return-value]

    return decorator
