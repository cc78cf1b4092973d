"""Trial lifecycle states.

Parity: reference ``optuna/trial/_state.py`` :4-35. The integer values are part of
the RDB checkpoint format (``trials.state`` column stores the enum name, but the
journal format stores the int), so they must not change.
"""
import enum


class TrialState(enum.IntEnum):
    """State of a trial.

    RUNNING is the only mutable state; WAITING trials are queued (``enqueue_trial``)
    and claimed by a compare-and-swap to RUNNING.
    """

    RUNNING = 0
    COMPLETE = 1
    PRUNED = 2
    FAIL = 3
    WAITING = 4

    def __repr__(self) -> str:
        return str(self)

    def is_finished(self) -> bool:
        return self != TrialState.RUNNING and self != TrialState.WAITING
