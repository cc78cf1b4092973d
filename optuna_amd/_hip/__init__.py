"""Loader/dispatch for the MI355X HIP extension (``_hipcore``).

The extension is built in-tree by ``optuna_amd/_hip/build.py`` (invoked from the
repo-level ``__graft_entry__.build()``); the ``.so`` lives next to this file so it
travels with the source tree to GPU boxes.

Policy: on a machine WITH a visible GPU the HIP path is mandatory — a missing or
unloadable extension raises rather than silently falling back to numpy (set
``OPTUNA_AMD_ALLOW_CPU_FALLBACK=1`` to override). On CPU-only machines everything
transparently uses the numpy reference implementations.
"""
from __future__ import annotations

import os
from typing import Any


_core: Any = None
_import_error: Exception | None = None
_checked = False


def _load() -> Any:
    global _core, _import_error, _checked
    if _checked:
        return _core
    _checked = True
    if os.environ.get("OPTUNA_AMD_DISABLE_HIP"):
        return None
    try:
        from optuna_amd._hip import _hipcore as core  # type: ignore[attr-defined]

        _core = core
    except ImportError as e:
        _import_error = e
        _core = None
    return _core


def _gpu_visible() -> bool:
    # Cheap probe that avoids importing torch: the HIP runtime in the extension
    # is authoritative; before it's loaded, look for kfd render nodes.
    try:
        import torch

        return bool(torch.cuda.is_available())
    except Exception:
        return False


def is_available() -> bool:
    """True iff the extension is importable AND a GPU is visible."""
    core = _load()
    if core is None:
        return False
    try:
        return bool(core.available())
    except Exception:
        return False


def require() -> Any:
    """Return the extension module; on a GPU machine a missing build is an error."""
    core = _load()
    if core is not None:
        return core
    if _gpu_visible() and not os.environ.get("OPTUNA_AMD_ALLOW_CPU_FALLBACK"):
        raise RuntimeError(
            "optuna_amd: an AMD GPU is visible but the HIP extension "
            "optuna_amd._hip._hipcore is not built/importable "
            f"(import error: {_import_error}). Build it with "
            "`python -m optuna_amd._hip.build`, or set "
            "OPTUNA_AMD_ALLOW_CPU_FALLBACK=1 to run the numpy path."
        )
    return None


def get() -> Any:
    """Extension module or None; raises on GPU machines without the build."""
    if _gpu_visible():
        return require()
    return _load()
