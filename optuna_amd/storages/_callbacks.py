"""Heartbeat-stale trial retry callbacks (module location parity).

The implementation lives in ``optuna_amd._callbacks``; this module mirrors the
reference's ``optuna.storages._callbacks`` import path and its current naming
(``RetryHeartbeatStaleTrialCallback`` is the name since reference v4.9,
``RetryFailedTrialCallback`` the historical alias).
"""
from __future__ import annotations

from optuna_amd._callbacks import RetryFailedTrialCallback


# Current reference name; the historical class is the implementation.
RetryHeartbeatStaleTrialCallback = RetryFailedTrialCallback

__all__ = ["RetryFailedTrialCallback", "RetryHeartbeatStaleTrialCallback"]
