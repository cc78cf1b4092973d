"""Improvement-evaluator package facade; implementation in ``_improvement_impl``."""
from optuna_amd.terminator._improvement_impl import *  # noqa: F401,F403
from optuna_amd.terminator import _improvement_impl as _impl

__all__ = [n for n in dir(_impl) if not n.startswith("__")]
