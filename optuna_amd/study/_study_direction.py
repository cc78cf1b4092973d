"""Optimization direction enum.

Parity: reference ``optuna/study/_study_direction.py``. The int values are part of
the RDB checkpoint format (``study_directions.direction``).
"""
import enum


class StudyDirection(enum.IntEnum):
    """Direction of a study: minimize or maximize the objective value."""

    NOT_SET = 0
    MINIMIZE = 1
    MAXIMIZE = 2

    def __repr__(self) -> str:
        return str(self)
