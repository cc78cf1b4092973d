from optuna_amd.search_space.group_decomposed import _GroupDecomposedSearchSpace
from optuna_amd.search_space.intersection import (
    IntersectionSearchSpace,
    intersection_search_space,
)


__all__ = [
    "_SearchSpaceGroup",
    "IntersectionSearchSpace",
    "intersection_search_space",
    "_GroupDecomposedSearchSpace",
]


def __getattr__(name: str):
    if name == "_SearchSpaceGroup":
        from optuna_amd.search_space.group_decomposed import _SearchSpaceGroup

        return _SearchSpaceGroup
    raise AttributeError(f"module {__name__!r} has no attribute {name!r}")
