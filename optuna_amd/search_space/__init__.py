from optuna_amd.search_space.group_decomposed import _GroupDecomposedSearchSpace
from optuna_amd.search_space.intersection import (
    IntersectionSearchSpace,
    intersection_search_space,
)


__all__ = [
    "IntersectionSearchSpace",
    "intersection_search_space",
    "_GroupDecomposedSearchSpace",
]
