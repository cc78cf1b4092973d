"""Guarded plotly imports (parity: reference ``optuna/visualization/_plotly_imports.py``)."""
from optuna_amd._imports import try_import


with try_import() as _imports:
    import plotly
    import plotly.graph_objects as go
    from plotly import __version__ as plotly_version
    from plotly.graph_objects import Figure, Scatter
    from plotly.subplots import make_subplots


__all__ = ["_imports", "go", "plotly", "Figure", "Scatter", "make_subplots"]


def is_available() -> bool:
    """Whether plotly-based visualization is importable."""
    return _imports.is_successful()
