"""RNG wrapper created on first use, so deep-copied samplers don't share streams.

Parity: reference ``optuna/samplers/_lazy_random_state.py`` (LazyRandomState :6).
"""
from __future__ import annotations

import numpy as np


class LazyRandomState:
    def __init__(self, seed: int | None = None) -> None:
        self._rng: np.random.RandomState | None = None
        self._seed = seed

    def seed(self, seed: int | None) -> None:
        self._seed = seed
        self._rng = None

    @property
    def rng(self) -> np.random.RandomState:
        if self._rng is None:
            self._rng = np.random.RandomState(self._seed)
        return self._rng
