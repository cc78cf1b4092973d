from __future__ import annotations

import json
import math
import warnings

import pytest

from optuna_amd.distributions import (
    CategoricalDistribution,
    FloatDistribution,
    IntDistribution,
    check_distribution_compatibility,
    distribution_to_json,
    json_to_distribution,
)


def test_float_basic() -> None:
    d = FloatDistribution(0.0, 2.0)
    assert not d.single()
    assert d._contains(0.0) and d._contains(2.0) and not d._contains(2.1)
    assert d.to_external_repr(1.5) == 1.5


def test_float_log_requires_positive_low() -> None:
    with pytest.raises(ValueError):
        FloatDistribution(0.0, 1.0, log=True)
    with pytest.raises(ValueError):
        FloatDistribution(1.0, 2.0, log=True, step=0.1)


def test_float_low_gt_high() -> None:
    with pytest.raises(ValueError):
        FloatDistribution(2.0, 1.0)


def test_float_step_adjust_high() -> None:
    with pytest.warns(UserWarning):
        d = FloatDistribution(0.0, 1.0, step=0.3)
    assert d.high == pytest.approx(0.9)
    assert d._contains(0.6)
    assert not d._contains(0.65)


def test_float_single() -> None:
    assert FloatDistribution(1.0, 1.0).single()
    assert FloatDistribution(1.0, 1.2, step=0.5).single()
    assert not FloatDistribution(1.0, 1.5, step=0.5).single()


def test_int_basic() -> None:
    d = IntDistribution(1, 9, step=2)
    assert d._contains(3) and not d._contains(4)
    assert d.to_external_repr(3.0) == 3
    assert d.to_internal_repr(3) == 3.0


def test_int_log_constraints() -> None:
    with pytest.raises(ValueError):
        IntDistribution(0, 10, log=True)
    with pytest.raises(ValueError):
        IntDistribution(1, 10, log=True, step=2)


def test_categorical() -> None:
    d = CategoricalDistribution(("a", None, 3, 2.5))
    assert d.to_internal_repr("a") == 0
    assert d.to_internal_repr(None) == 1
    assert d.to_external_repr(2) == 3
    assert d._contains(0) and d._contains(3) and not d._contains(4)
    with pytest.raises(ValueError):
        d.to_internal_repr("missing")


def test_categorical_nan_choice() -> None:
    d = CategoricalDistribution((1.0, float("nan")))
    assert d.to_internal_repr(float("nan")) == 1


def test_json_roundtrip() -> None:
    dists = [
        FloatDistribution(0.0, 1.5),
        FloatDistribution(1e-3, 1e2, log=True),
        FloatDistribution(0.0, 1.0, step=0.25),
        IntDistribution(-3, 7),
        IntDistribution(1, 1024, log=True),
        IntDistribution(0, 10, step=2),
        CategoricalDistribution(("x", None, 1, 2.5, True)),
    ]
    for d in dists:
        assert json_to_distribution(distribution_to_json(d)) == d


def test_json_format_compatibility() -> None:
    # The on-disk format must match the reference codec byte structure.
    payload = json.loads(distribution_to_json(FloatDistribution(0.0, 1.0, log=False)))
    assert payload["name"] == "FloatDistribution"
    assert payload["attributes"] == {"low": 0.0, "high": 1.0, "log": False, "step": None}


def test_json_legacy_names_decodable() -> None:
    legacy = json.dumps({"name": "UniformDistribution", "attributes": {"low": 0.0, "high": 1.0}})
    assert json_to_distribution(legacy) == FloatDistribution(0.0, 1.0)
    legacy = json.dumps(
        {"name": "IntLogUniformDistribution", "attributes": {"low": 1, "high": 10}}
    )
    assert json_to_distribution(legacy) == IntDistribution(1, 10, log=True)


def test_compatibility_check() -> None:
    check_distribution_compatibility(FloatDistribution(0, 1), FloatDistribution(0, 2))
    with pytest.raises(ValueError):
        check_distribution_compatibility(FloatDistribution(0, 1), IntDistribution(0, 1))
    with pytest.raises(ValueError):
        check_distribution_compatibility(
            CategoricalDistribution(("a",)), CategoricalDistribution(("b",))
        )


def test_eq_hash() -> None:
    assert FloatDistribution(0, 1) == FloatDistribution(0, 1)
    assert hash(FloatDistribution(0, 1)) == hash(FloatDistribution(0, 1))
    assert FloatDistribution(0, 1) != FloatDistribution(0, 2)


def test_float_log_and_step_mutually_exclusive() -> None:
    with pytest.raises(ValueError):
        FloatDistribution(1.0, 10.0, log=True, step=0.5)


def test_int_log_and_step_mutually_exclusive() -> None:
    with pytest.raises(ValueError):
        IntDistribution(1, 10, log=True, step=2)


def test_contains_edges() -> None:
    d = FloatDistribution(-1.0, 1.0)
    assert d._contains(-1.0) and d._contains(1.0)
    assert not d._contains(-1.0000001) and not d._contains(1.0000001)
    di = IntDistribution(2, 8, step=3)  # {2, 5, 8}
    assert di._contains(2) and di._contains(8)
    assert not di._contains(9)
    c = CategoricalDistribution(("a", "b"))
    assert c._contains(0) and c._contains(1) and not c._contains(2)


def test_repr_round_trip() -> None:
    for d in (
        FloatDistribution(-2.0, 3.5),
        FloatDistribution(1e-4, 1e2, log=True),
        FloatDistribution(0.0, 1.0, step=0.25),
        IntDistribution(-3, 12),
        IntDistribution(1, 512, log=True),
        CategoricalDistribution(("x", 1, None)),
    ):
        assert eval(repr(d)) == d  # noqa: S307 — repr is the constructor form


def test_int_to_external_repr_is_int() -> None:
    d = IntDistribution(0, 10)
    v = d.to_external_repr(3.0)
    assert isinstance(v, int) and v == 3


def test_single_variants() -> None:
    assert FloatDistribution(2.0, 2.0).single()
    assert IntDistribution(5, 5).single()
    assert CategoricalDistribution(("only",)).single()
    assert not FloatDistribution(0.0, 1e-9).single()
    assert FloatDistribution(1.0, 1.4, step=0.5).single()  # only one grid point
