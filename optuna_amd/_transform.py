"""Search-space ⇄ ℝ^d vectorization.

``_SearchSpaceTransform`` maps a dict of distributions onto a real vector space:
numerical parameters become one axis each (optionally log-scaled, with step domains
widened by ±step/2 so untransform rounds back onto the grid), categorical
parameters become one-hot blocks. This is the host-side reference implementation;
the HIP kernel K7 (``optuna_amd/_hip/kernels/transform.hip``) applies the same
per-dimension descriptors to whole trial tables on device.

Parity: reference ``optuna/_transform.py`` (_SearchSpaceTransform :18, transform
:99, untransform :137, _transform_search_space :176).
"""
from __future__ import annotations

import math
from typing import Any

import numpy as np

from optuna_amd.distributions import (
    BaseDistribution,
    CategoricalDistribution,
    FloatDistribution,
    IntDistribution,
)


class _SearchSpaceTransform:
    """Bidirectional mapping between parameter dicts and real vectors.

    Args:
        search_space: Ordered dict of ``name -> BaseDistribution``.
        transform_log: Map log-scaled domains through ``log``.
        transform_step: Widen stepped domains by half a step at both ends so that a
            uniform draw rounds onto the grid with uniform mass per point.
        transform_0_1: Rescale every axis to the unit interval (used by CMA-ES).
    """

    def __init__(
        self,
        search_space: dict[str, BaseDistribution],
        transform_log: bool = True,
        transform_step: bool = True,
        transform_0_1: bool = False,
    ) -> None:
        self._search_space = search_space
        self._transform_log = transform_log
        self._transform_step = transform_step
        self._transform_0_1 = transform_0_1

        n_cols = 0
        # column_to_encoded_columns[i] = array of encoded column indices of param i.
        column_to_encoded_columns: list[np.ndarray] = []
        encoded_column_to_column: list[int] = []
        bounds_list: list[tuple[float, float]] = []

        for i, distribution in enumerate(search_space.values()):
            if isinstance(distribution, CategoricalDistribution):
                n_choices = len(distribution.choices)
                cols = np.arange(n_cols, n_cols + n_choices)
                bounds_list.extend([(0.0, 1.0)] * n_choices)
                encoded_column_to_column.extend([i] * n_choices)
                n_cols += n_choices
            else:
                cols = np.array([n_cols])
                bounds_list.append(self._numerical_bounds(distribution))
                encoded_column_to_column.append(i)
                n_cols += 1
            column_to_encoded_columns.append(cols)

        self.column_to_encoded_columns = column_to_encoded_columns
        self.encoded_column_to_column = np.array(encoded_column_to_column, dtype=np.int64)
        self._raw_bounds = np.array(bounds_list, dtype=np.float64).reshape(-1, 2)
        if transform_0_1:
            self._bounds = np.zeros_like(self._raw_bounds)
            self._bounds[:, 1] = 1.0
        else:
            self._bounds = self._raw_bounds

    @property
    def bounds(self) -> np.ndarray:
        return self._bounds

    def _numerical_bounds(self, distribution: BaseDistribution) -> tuple[float, float]:
        assert isinstance(distribution, (FloatDistribution, IntDistribution))
        low: float = distribution.low
        high: float = distribution.high
        step: float | None
        if isinstance(distribution, FloatDistribution):
            step = distribution.step
        else:
            step = float(distribution.step)
        if distribution.log and self._transform_log:
            if isinstance(distribution, IntDistribution) and self._transform_step:
                half = 0.5
                return math.log(low - half), math.log(high + half)
            return math.log(low), math.log(high)
        if step is not None and self._transform_step:
            half = step / 2
            return low - half, high + half
        return low, high

    def transform(self, params: dict[str, Any]) -> np.ndarray:
        """Parameter dict (external repr) → real vector."""
        trans_params = np.zeros(len(self._raw_bounds), dtype=np.float64)
        for i, (name, distribution) in enumerate(self._search_space.items()):
            cols = self.column_to_encoded_columns[i]
            if isinstance(distribution, CategoricalDistribution):
                choice_index = int(distribution.to_internal_repr(params[name]))
                trans_params[cols[choice_index]] = 1.0
            else:
                trans_params[cols[0]] = self._transform_numerical_param(
                    params[name], distribution
                )
        if self._transform_0_1:
            spread = self._raw_bounds[:, 1] - self._raw_bounds[:, 0]
            spread[spread == 0.0] = 1.0
            trans_params = (trans_params - self._raw_bounds[:, 0]) / spread
        return trans_params

    def untransform(self, trans_params: np.ndarray) -> dict[str, Any]:
        """Real vector → parameter dict (external repr)."""
        assert trans_params.shape == (len(self._raw_bounds),)
        if self._transform_0_1:
            trans_params = (
                trans_params * (self._raw_bounds[:, 1] - self._raw_bounds[:, 0])
                + self._raw_bounds[:, 0]
            )
        params: dict[str, Any] = {}
        for i, (name, distribution) in enumerate(self._search_space.items()):
            cols = self.column_to_encoded_columns[i]
            if isinstance(distribution, CategoricalDistribution):
                index = int(np.argmax(trans_params[cols]))
                params[name] = distribution.to_external_repr(index)
            else:
                params[name] = self._untransform_numerical_param(
                    float(trans_params[cols[0]]), distribution
                )
        return params

    def _transform_numerical_param(
        self, value: float, distribution: BaseDistribution
    ) -> float:
        return _transform_numerical_param(value, distribution, self._transform_log)

    def _untransform_numerical_param(
        self, trans_value: float, distribution: BaseDistribution
    ) -> float | int:
        return _untransform_numerical_param(trans_value, distribution, self._transform_log)


def _transform_numerical_param(
    param: int | float, distribution: BaseDistribution, transform_log: bool
) -> float:
    assert isinstance(distribution, (FloatDistribution, IntDistribution))
    if distribution.log and transform_log:
        return math.log(param)
    return float(param)


def _untransform_numerical_param(
    trans_param: float, distribution: BaseDistribution, transform_log: bool
) -> int | float:
    """Map one transformed axis value back into the distribution's domain.

    Semantics match the reference (_transform.py:269-306): plain/log floats are
    capped strictly below ``high`` (``nextafter``) so downstream half-open-range
    code never sees the boundary, stepped domains snap onto the grid with
    half-up rounding (NOT banker's rounding — a draw exactly between two grid
    points goes up, giving every grid point equal width), and log ints round in
    linear space after exponentiation.
    """
    assert isinstance(distribution, (FloatDistribution, IntDistribution))
    if isinstance(distribution, FloatDistribution):
        if distribution.log:
            value = math.exp(trans_param) if transform_log else trans_param
            if not distribution.single():
                value = float(min(value, np.nextafter(distribution.high, -math.inf)))
            return float(value)
        if distribution.step is not None:
            k = math.floor((trans_param - distribution.low) / distribution.step + 0.5)
            value = distribution.low + k * distribution.step
            return float(min(max(value, distribution.low), distribution.high))
        if distribution.single():
            return float(trans_param)
        return float(min(trans_param, np.nextafter(distribution.high, -math.inf)))
    else:
        if distribution.log:
            if not transform_log:
                return int(trans_param)
            int_value = int(round(math.exp(trans_param)))
            return min(max(int_value, distribution.low), distribution.high)
        k = math.floor((trans_param - distribution.low) / distribution.step + 0.5)
        int_value = int(distribution.low + k * distribution.step)
        return min(max(int_value, distribution.low), distribution.high)
