"""Device dispatch for the TPE hot path (K1 fit + K2 mixture log-pdf on gfx950).

Two device modes:

* **Resident history** (the fast path): each (study, search-space) keeps a
  ``TpeDeviceHistory`` whose parameter table lives in HBM and grows append-only
  in lockstep with the host mirror (``_history.py``). A suggest uploads only the
  per-dim sorted order (i32), a subset position map, the mixture weights and the
  24 candidates — the fp64 observation matrix never crosses PCIe again. Subset
  compaction, the K1 fit and the K2 scoring all run on device.
* **Stateless scoring** (`kde_logpdf`): used by golden tests and the
  constant-liar path (whose observation set includes rows not in the table).

Eligibility: any numerical dimensions — continuous, log, int, or
step-discretized (the scoring kernel integrates the step cell). Categorical
dimensions keep the (already vectorized) host path.
"""
from __future__ import annotations

import math
from typing import TYPE_CHECKING

import numpy as np

from optuna_amd import _hip
from optuna_amd.distributions import (
    BaseDistribution,
    CategoricalDistribution,
    FloatDistribution,
    IntDistribution,
)


if TYPE_CHECKING:
    from optuna_amd.samplers._tpe._history import _SpaceCache

# Below this kernel count the host fit is cheaper than a kernel launch round trip.
DEVICE_MIN_KERNELS = 512


def space_is_device_eligible(space: dict[str, BaseDistribution]) -> bool:
    # Float / int / discrete dims use the (possibly cell-integrated) truncnorm
    # kernels; categorical dims use the closed-form smoothed one-hot weights.
    return bool(space)


def device_ready(n_kernels: int) -> bool:
    return n_kernels >= DEVICE_MIN_KERNELS and _hip.is_available()


def _space_domains(
    space: dict[str, BaseDistribution],
) -> tuple[np.ndarray, np.ndarray, np.ndarray, np.ndarray]:
    """(is_log, alow, ahigh, steps, n_choices) per dim, KDE domain.

    Step-discretized dims widen the domain by half a step on each side before
    the log transform — identical to the host estimator's adapted bounds
    (parzen.py `_numerical_kernels_batched`).
    """
    dists = list(space.values())
    is_cat = np.array([isinstance(d, CategoricalDistribution) for d in dists])
    is_log = np.array(
        [False if c else bool(d.log) for c, d in zip(is_cat, dists)]
    )
    steps = np.array(
        [
            0.0 if c else (float(d.step) if d.step is not None else 0.0)
            for c, d in zip(is_cat, dists)
        ],
        dtype=np.float64,
    )
    n_choices = np.array(
        [float(len(d.choices)) if c else 0.0 for c, d in zip(is_cat, dists)],
        dtype=np.float64,
    )
    alow = np.zeros(len(dists))
    ahigh = np.ones(len(dists))
    for c, d in enumerate(dists):
        if is_cat[c]:
            continue  # unused by the categorical branch
        lo = float(d.low) - steps[c] / 2 if steps[c] else float(d.low)
        hi = float(d.high) + steps[c] / 2 if steps[c] else float(d.high)
        if d.log:
            lo, hi = math.log(lo), math.log(hi)
        alow[c] = lo
        ahigh[c] = hi
    return is_log, alow, ahigh, steps, n_choices


def _cell_edges(
    x_raw: np.ndarray, is_log: np.ndarray, steps: np.ndarray
) -> np.ndarray:
    """(S, 2D) per-candidate step-cell edges [lo | hi] in KDE domain.

    Continuous dims get zeros (the kernel never reads them); discrete dims get
    x ∓ step/2, logged for log-discrete dims — exactly the host integration
    cell (parzen.py `_log_pdf_array`).
    """
    S, D = x_raw.shape
    edges = np.zeros((S, 2 * D), dtype=np.float64)
    disc = steps > 0
    if disc.any():
        half = steps[disc] / 2
        lo = x_raw[:, disc] - half
        hi = x_raw[:, disc] + half
        log_disc = is_log[disc]
        if log_disc.any():
            with np.errstate(divide="ignore", invalid="ignore"):
                lo[:, log_disc] = np.log(lo[:, log_disc])
                hi[:, log_disc] = np.log(hi[:, log_disc])
        idx = np.nonzero(disc)[0]
        edges[:, idx] = lo
        edges[:, D + idx] = hi
    return edges


class _SpaceDeviceMirror:
    """HBM-resident copy of one space's parameter table (KDE domain)."""

    def __init__(self, space: dict[str, BaseDistribution]) -> None:
        core = _hip.get()
        assert core is not None
        (
            self._is_log,
            self._alow,
            self._ahigh,
            self._steps,
            self._n_choices,
        ) = _space_domains(space)
        self._hist = core.TpeDeviceHistory(len(space))
        self._n_appended = 0
        # Device-resident sorted index bookkeeping: entries of the host cache's
        # insert log already replayed, and the resident index length.
        self._sorted_log_cursor = 0
        self._n_sorted_synced = -1

    def sync(self, cache: "_SpaceCache") -> None:
        n_total = len(cache.valid)
        if n_total > self._n_appended:
            block = np.array(cache.params[self._n_appended :], dtype=np.float64)
            if self._is_log.any():
                with np.errstate(invalid="ignore", divide="ignore"):
                    block[:, self._is_log] = np.log(block[:, self._is_log])
            self._hist.append(np.ascontiguousarray(block))
            self._n_appended = n_total
        self._sync_sorted(cache)

    def _sync_sorted(self, cache: "_SpaceCache") -> None:
        log = cache._insert_log
        if self._n_sorted_synced == cache._n_sorted:
            return
        pending = log[self._sorted_log_cursor :]
        # Bulk (re-)upload when starting fresh or when a large batch arrived
        # (the sequential device insert is O(batch · N)); per-tell single-row
        # inserts replay incrementally.
        if (
            self._n_sorted_synced < 0
            or len(pending) > 16
            or any(len(rows) > 64 for _, rows in pending)
        ):
            self._hist.upload_sorted(list(cache.sorted_rows))
        else:
            for pos, rows in pending:
                self._hist.insert_sorted(pos, rows)
        self._sorted_log_cursor = len(log)
        self._n_sorted_synced = cache._n_sorted

    def score(
        self,
        cache: "_SpaceCache",
        sel: np.ndarray,
        weights: np.ndarray,
        samples: dict[str, np.ndarray],
        consider_endpoints: bool,
        consider_magic_clip: bool,
        prior_weight: float = 1.0,
        extras: np.ndarray | None = None,
        per_dim: bool = False,
    ) -> np.ndarray:
        """Score candidates against the resident table.

        ``extras`` ((L, D), internal repr) are constant-liar rows from other
        workers' RUNNING trials: they are appended to the mixture after the
        ``sel`` kernels (matching the host estimator's observation order) and
        merged into the per-dim sorted subsets on device.
        """
        self.sync(cache)
        n_total = len(cache.valid)
        pos = np.full(n_total, -1, dtype=np.int32)
        pos[sel] = np.arange(len(sel), dtype=np.int32)
        # The per-dim sorted index is device-resident (kept current by
        # _sync_sorted); pass no columns unless the resident copy somehow
        # lagged, in which case fall back to a one-off upload.
        if self._n_sorted_synced == cache._n_sorted:
            sorted_cols: list = []
        else:
            sorted_cols = list(cache.sorted_rows)
        x_raw = np.column_stack(
            [np.asarray(samples[n], dtype=np.float64) for n in cache.names]
        )
        xedges = _cell_edges(x_raw, self._is_log, self._steps)
        x = x_raw.copy()
        if self._is_log.any():
            x[:, self._is_log] = np.log(x[:, self._is_log])
        with np.errstate(divide="ignore"):
            logw = np.log(weights)
        kwargs = {}
        if extras is not None and len(extras):
            E = np.array(extras, dtype=np.float64)
            if self._is_log.any():
                E[:, self._is_log] = np.log(E[:, self._is_log])
            order = np.argsort(E, axis=0, kind="stable")  # (L, D) per column
            kwargs = dict(
                extras_raw=np.ascontiguousarray(E),
                extras_sorted=np.ascontiguousarray(
                    np.take_along_axis(E, order, axis=0).T  # (D, L)
                ),
                extras_sorted_idx=np.ascontiguousarray(
                    order.T.astype(np.int32)  # (D, L)
                ),
            )
        return self._hist.score(
            sorted_cols,
            pos,
            int(len(sel)),
            logw,
            self._alow,
            self._ahigh,
            self._steps,
            self._n_choices,
            float(prior_weight),
            np.ascontiguousarray(x),
            np.ascontiguousarray(xedges),
            consider_endpoints,
            consider_magic_clip,
            per_dim=per_dim,
            **kwargs,
        )


def score_above_resident(
    cache: "_SpaceCache",
    sel: np.ndarray,
    weights: np.ndarray,
    samples: dict[str, np.ndarray],
    consider_endpoints: bool,
    consider_magic_clip: bool,
    prior_weight: float = 1.0,
    extras: np.ndarray | None = None,
    per_dim: bool = False,
) -> np.ndarray:
    """log g(x) for candidates via the device-resident table (creates the mirror
    on first use; attached to the host space cache so lifetimes match).
    ``per_dim=True`` returns the (S, D) per-dimension 1-D mixture log-pdfs
    (independent-mode TPE) instead of the joint (S,) product score."""
    mirror = getattr(cache, "_device_mirror", None)
    if mirror is None:
        mirror = _SpaceDeviceMirror(cache.space)
        cache._device_mirror = mirror  # type: ignore[attr-defined]
    return mirror.score(
        cache, sel, weights, samples, consider_endpoints, consider_magic_clip,
        prior_weight=prior_weight, extras=extras, per_dim=per_dim,
    )


def kde_logpdf(
    space: dict[str, BaseDistribution],
    observations: dict[str, np.ndarray],
    orders: dict[str, np.ndarray] | None,
    weights: np.ndarray,
    samples: dict[str, np.ndarray],
    consider_endpoints: bool,
    consider_magic_clip: bool,
    prior_weight: float = 1.0,
) -> np.ndarray:
    """Stateless device scoring (observation matrix uploaded per call)."""
    core = _hip.get()
    assert core is not None
    names = list(space.keys())
    D = len(names)
    N = len(observations[names[0]])

    is_log, alow, ahigh, steps, n_choices = _space_domains(space)
    obs = np.empty((N, D), dtype=np.float64)
    x_raw = np.column_stack(
        [np.asarray(samples[n], dtype=np.float64) for n in names]
    )
    sorted_pos = np.empty((N, D), dtype=np.int64)
    for c, name in enumerate(names):
        col = np.asarray(observations[name], dtype=np.float64)
        obs[:, c] = np.log(col) if is_log[c] else col
        if orders is not None:
            sorted_pos[:, c] = orders[name]
        else:
            sorted_pos[:, c] = np.argsort(col, kind="stable")

    xedges = _cell_edges(x_raw, is_log, steps)
    x = x_raw.copy()
    if is_log.any():
        x[:, is_log] = np.log(x[:, is_log])

    with np.errstate(divide="ignore"):
        logw = np.log(weights)
    return core.kde_logpdf(
        obs,
        sorted_pos,
        logw,
        alow,
        ahigh,
        steps,
        n_choices,
        float(prior_weight),
        x,
        np.ascontiguousarray(xedges),
        consider_endpoints,
        consider_magic_clip,
    )
