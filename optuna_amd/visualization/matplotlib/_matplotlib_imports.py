"""Guarded matplotlib imports (parity: reference matplotlib/_matplotlib_imports.py)."""
from optuna_amd._imports import try_import


with try_import() as _imports:
    import matplotlib

    matplotlib.use("Agg", force=False)
    from matplotlib import pyplot as plt
    from matplotlib.axes import Axes


__all__ = ["_imports", "plt", "Axes"]


def is_available() -> bool:
    return _imports.is_successful()
