"""Reference-compatible module path; implementation in ``_mutations.py``."""
from optuna_amd.samplers.nsgaii._mutations import (  # noqa: F401
    BaseMutation,
    perform_mutation,
)
