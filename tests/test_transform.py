from __future__ import annotations

import math

import numpy as np
import pytest

from optuna_amd._transform import _SearchSpaceTransform
from optuna_amd.distributions import (
    CategoricalDistribution,
    FloatDistribution,
    IntDistribution,
)


def test_bounds_shapes() -> None:
    space = {
        "f": FloatDistribution(0.0, 2.0),
        "c": CategoricalDistribution(("a", "b", "c")),
        "i": IntDistribution(1, 10),
    }
    trans = _SearchSpaceTransform(space)
    assert trans.bounds.shape == (5, 2)  # 1 + 3 one-hot + 1


def test_roundtrip_numerical() -> None:
    space = {
        "f": FloatDistribution(-1.0, 3.0),
        "flog": FloatDistribution(1e-4, 10.0, log=True),
        "fstep": FloatDistribution(0.0, 1.0, step=0.25),
        "i": IntDistribution(2, 12, step=2),
        "ilog": IntDistribution(1, 100, log=True),
    }
    trans = _SearchSpaceTransform(space)
    params = {"f": 0.5, "flog": 0.1, "fstep": 0.75, "i": 6, "ilog": 10}
    out = trans.untransform(trans.transform(params))
    assert out["f"] == pytest.approx(0.5)
    assert out["flog"] == pytest.approx(0.1)
    assert out["fstep"] == pytest.approx(0.75)
    assert out["i"] == 6
    assert out["ilog"] == 10


def test_roundtrip_categorical() -> None:
    space = {"c": CategoricalDistribution((True, "x", None))}
    trans = _SearchSpaceTransform(space)
    for choice in (True, "x", None):
        assert trans.untransform(trans.transform({"c": choice}))["c"] == choice


def test_log_bounds() -> None:
    space = {"flog": FloatDistribution(0.1, 10.0, log=True)}
    trans = _SearchSpaceTransform(space)
    np.testing.assert_allclose(trans.bounds[0], [math.log(0.1), math.log(10.0)])


def test_step_widened_bounds() -> None:
    space = {"i": IntDistribution(0, 10)}
    trans = _SearchSpaceTransform(space)
    np.testing.assert_allclose(trans.bounds[0], [-0.5, 10.5])


def test_untransform_clips_and_rounds() -> None:
    space = {"i": IntDistribution(0, 10)}
    trans = _SearchSpaceTransform(space)
    assert trans.untransform(np.array([10.49]))["i"] == 10
    assert trans.untransform(np.array([-0.49]))["i"] == 0
    assert trans.untransform(np.array([3.2]))["i"] == 3


def test_transform_0_1() -> None:
    space = {"f": FloatDistribution(10.0, 20.0)}
    trans = _SearchSpaceTransform(space, transform_0_1=True)
    np.testing.assert_allclose(trans.bounds[0], [0.0, 1.0])
    v = trans.transform({"f": 15.0})
    assert v[0] == pytest.approx(0.5)
    assert trans.untransform(np.array([0.5]))["f"] == pytest.approx(15.0)
