"""Thread wrapper that re-raises worker exceptions on join.

Parity: reference ``optuna/testing/threading.py`` (_TestableThread :12-26).
"""
from __future__ import annotations

import threading
from typing import Any


class _TestableThread(threading.Thread):
    def __init__(self, target: Any, args: tuple[Any, ...] = ()) -> None:
        super().__init__(target=target, args=args)
        self.exc: BaseException | None = None

    def run(self) -> None:
        try:
            super().run()
        except BaseException as e:
            self.exc = e

    def join(self, timeout: float | None = None) -> None:
        super().join(timeout)
        if self.exc is not None:
            raise self.exc
