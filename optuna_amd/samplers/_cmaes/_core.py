"""Native CMA-ES core: mean/σ/C updates, CSA step-size control, eigendecomposition.

The reference delegates these equations to the external ``cmaes`` package
(reference ``optuna/samplers/_cmaes.py`` :34-41); this build implements them
directly following Hansen's tutorial formulation ("The CMA Evolution Strategy: A
Tutorial", 2016): rank-1 + rank-μ covariance updates with active (negative-weight)
recombination, cumulative step-size adaptation, and lazy eigendecomposition of C.

``SepCMA`` keeps a diagonal covariance (O(n) per update, for very high dim);
``get_warm_start_mgd`` estimates a promising multivariate Gaussian from source
solutions (WS-CMA-ES).

At bench dimensions (n ≤ a few hundred) the n×n symmetric eigendecomposition is
microseconds on host; the K8 HIP one-workgroup Jacobi eigensolver takes over for
batched/large-n studies (see _hip/csrc — planned batching path).
"""
from __future__ import annotations

import math
from typing import Any

import numpy as np


_EPS = 1e-8
_SIGMA_MAX = 1e32


class CMA:
    """(μ/μ_w, λ)-CMA-ES with active covariance adaptation."""

    def __init__(
        self,
        mean: np.ndarray,
        sigma: float,
        bounds: np.ndarray | None = None,
        n_max_resampling: int = 100,
        seed: int | None = None,
        population_size: int | None = None,
        cov: np.ndarray | None = None,
        lr_adapt: bool = False,
    ) -> None:
        n_dim = len(mean)
        if n_dim < 1:
            raise ValueError("The dimension of mean must be larger than 0")
        if sigma <= 0:
            raise ValueError("sigma must be non-zero positive value")

        popsize = population_size or (4 + math.floor(3 * math.log(n_dim)))
        if popsize < 2:
            raise ValueError("popsize must be non-zero positive value.")
        mu = popsize // 2

        # Raw log-rank weights; positive for the top half, negative below.
        w_raw = np.log((popsize + 1) / 2) - np.log(np.arange(1, popsize + 1))
        w_pos = w_raw[:mu]
        w_neg = w_raw[mu:]
        mu_eff = float(w_pos.sum() ** 2 / (w_pos**2).sum())
        mu_eff_minus = float(w_neg.sum() ** 2 / (w_neg**2).sum()) if len(w_neg) else 0.0

        alpha_cov = 2.0
        c1 = alpha_cov / ((n_dim + 1.3) ** 2 + mu_eff)
        cmu = min(
            1 - c1 - 1e-8,
            alpha_cov
            * (0.25 + mu_eff + 1 / mu_eff - 2)
            / ((n_dim + 2) ** 2 + alpha_cov * mu_eff / 2),
        )
        c_sigma = (mu_eff + 2) / (n_dim + mu_eff + 5)
        d_sigma = 1 + 2 * max(0.0, math.sqrt((mu_eff - 1) / (n_dim + 1)) - 1) + c_sigma
        cc = (4 + mu_eff / n_dim) / (n_dim + 4 + 2 * mu_eff / n_dim)

        # Scale negative weights (active CMA) so C stays positive definite.
        if len(w_neg):
            alpha_mu_minus = 1 + c1 / cmu
            alpha_mueff_minus = 1 + 2 * mu_eff_minus / (mu_eff + 2)
            alpha_posdef_minus = (1 - c1 - cmu) / (n_dim * cmu)
            neg_scale = min(alpha_mu_minus, alpha_mueff_minus, alpha_posdef_minus) / abs(
                w_neg.sum()
            )
            weights = np.concatenate([w_pos / w_pos.sum(), w_neg * neg_scale])
        else:
            weights = w_pos / w_pos.sum()

        self._n_dim = n_dim
        self._popsize = popsize
        self._mu = mu
        self._mu_eff = mu_eff
        self._weights = weights
        self._c1 = c1
        self._cmu = cmu
        self._c_sigma = c_sigma
        self._d_sigma = d_sigma
        self._cc = cc
        self._cm = 1.0
        # E||N(0, I)||
        self._chi_n = math.sqrt(n_dim) * (1.0 - 1.0 / (4 * n_dim) + 1.0 / (21 * n_dim**2))

        self._mean = np.array(mean, dtype=np.float64).copy()
        self._sigma = float(sigma)
        self._C = np.array(cov, dtype=np.float64) if cov is not None else np.eye(n_dim)
        self._p_sigma = np.zeros(n_dim)
        self._pc = np.zeros(n_dim)
        self._g = 0
        self._rng = np.random.RandomState(seed)
        self._bounds = np.array(bounds, dtype=np.float64) if bounds is not None else None
        self._n_max_resampling = n_max_resampling
        # LRA-CMA (SNR-driven learning-rate adaptation, Nomura et al. GECCO
        # 2023 — reference drives it via the external cmaes package,
        # optuna/samplers/_cmaes.py:591-600). Reconstructed from the published
        # mechanism: per-generation mean/covariance updates are damped by
        # multiplicative rates eta_m/eta_c, each adapted from an estimated
        # signal-to-noise ratio of its own normalized update stream. Exact
        # numeric parity with the external package is not claimed.
        self._lr_adapt = lr_adapt
        if lr_adapt:
            self._eta_m = 1.0
            self._eta_c = 1.0
            self._lr_Em = np.zeros(n_dim)
            self._lr_Vm = 0.0
            self._lr_EC = np.zeros((n_dim, n_dim))
            self._lr_VC = 0.0

        self._B: np.ndarray | None = None
        self._D: np.ndarray | None = None

    # ---- properties -----------------------------------------------------------------

    @property
    def dim(self) -> int:
        return self._n_dim

    @property
    def population_size(self) -> int:
        return self._popsize

    @property
    def generation(self) -> int:
        return self._g

    @property
    def mean(self) -> np.ndarray:
        return self._mean.copy()

    # ---- sampling -------------------------------------------------------------------

    # K8: below this dimension the host eigh wins (an n=100 symmetric
    # eigendecomposition is ~0.5 ms on CPU, amortized over a whole generation,
    # while a device round trip costs more in launch latency than it saves);
    # above it the O(n³) work dominates and rocSOLVER syevd takes over.
    _DEVICE_EIGH_MIN_DIM = 256

    def _eigen(self) -> tuple[np.ndarray, np.ndarray]:
        if self._B is None or self._D is None:
            self._C = (self._C + self._C.T) / 2  # enforce symmetry
            d2, B = self._eigh(self._C)
            D = np.sqrt(np.maximum(d2, _EPS**2))
            self._C = B @ np.diag(D**2) @ B.T
            self._B, self._D = B, D
        return self._B, self._D

    def _eigh(self, C: np.ndarray) -> tuple[np.ndarray, np.ndarray]:
        if self._n_dim >= self._DEVICE_EIGH_MIN_DIM:
            try:
                import torch

                if torch.cuda.is_available():
                    d2, B = torch.linalg.eigh(
                        torch.from_numpy(C).to("cuda")
                    )
                    return d2.cpu().numpy(), B.cpu().numpy()
            except ImportError:
                pass
        return np.linalg.eigh(C)

    def _sample_one(self) -> np.ndarray:
        B, D = self._eigen()
        z = self._rng.randn(self._n_dim)
        return self._mean + self._sigma * (B @ (D * z))

    def _in_bounds(self, x: np.ndarray) -> bool:
        if self._bounds is None:
            return True
        return bool(np.all(x >= self._bounds[:, 0]) and np.all(x <= self._bounds[:, 1]))

    def _repair(self, x: np.ndarray) -> np.ndarray:
        if self._bounds is None:
            return x
        return np.clip(x, self._bounds[:, 0], self._bounds[:, 1])

    def ask(self) -> np.ndarray:
        for _ in range(self._n_max_resampling):
            x = self._sample_one()
            if self._in_bounds(x):
                return x
        return self._repair(self._sample_one())

    # ---- update ---------------------------------------------------------------------

    def tell(self, solutions: list[tuple[np.ndarray, float]]) -> None:
        if len(solutions) != self._popsize:
            raise ValueError("Must tell popsize-length solutions.")
        self._g += 1
        solutions.sort(key=lambda s: s[1])

        B, D = self._eigen()
        self._B, self._D = None, None  # recomputed after the C update

        xs = np.array([s[0] for s in solutions])  # (λ, n)
        ys = (xs - self._mean) / self._sigma  # (λ, n)

        # Mean update from the top-μ ranks.
        w = self._weights
        y_w = w[: self._mu] @ ys[: self._mu]
        mean_shift = self._cm * self._sigma * y_w
        if self._lr_adapt:
            self._mean = self._mean + self._eta_m * mean_shift
        else:
            self._mean = self._mean + mean_shift

        # CSA path (whitened by C^-1/2 = B D^-1 B^T).
        C_inv_half = B @ np.diag(1.0 / D) @ B.T
        self._p_sigma = (1 - self._c_sigma) * self._p_sigma + math.sqrt(
            self._c_sigma * (2 - self._c_sigma) * self._mu_eff
        ) * (C_inv_half @ y_w)

        norm_p_sigma = float(np.linalg.norm(self._p_sigma))
        self._sigma *= math.exp(
            (self._c_sigma / self._d_sigma) * (norm_p_sigma / self._chi_n - 1)
        )
        self._sigma = min(self._sigma, _SIGMA_MAX)

        # Covariance path with stall indicator h_σ.
        h_sigma_cond = norm_p_sigma / math.sqrt(
            1 - (1 - self._c_sigma) ** (2 * (self._g + 1))
        )
        h_sigma = 1.0 if h_sigma_cond < (1.4 + 2 / (self._n_dim + 1)) * self._chi_n else 0.0
        self._pc = (1 - self._cc) * self._pc + h_sigma * math.sqrt(
            self._cc * (2 - self._cc) * self._mu_eff
        ) * y_w

        # Active rank-μ: negative weights rescaled by n/||C^-1/2 y||².
        w_circ = np.where(
            w >= 0,
            w,
            w * self._n_dim / (np.linalg.norm(ys @ C_inv_half.T, axis=1) ** 2 + _EPS),
        )
        delta_h = (1 - h_sigma) * self._cc * (2 - self._cc)
        rank_one = np.outer(self._pc, self._pc)
        rank_mu = (ys.T * w_circ) @ ys
        C_old = self._C
        C_new = (
            (1 + self._c1 * delta_h - self._c1 - self._cmu * w.sum()) * C_old
            + self._c1 * rank_one
            + self._cmu * rank_mu
        )
        if self._lr_adapt:
            self._C = C_old + self._eta_c * (C_new - C_old)
            self._lr_adaptation(C_inv_half, mean_shift, C_new - C_old, C_old)
        else:
            self._C = C_new

    # LRA constants (paper defaults): target SNR alpha, EMA rates beta, and
    # the relative update cap gamma.
    _LRA_ALPHA = 1.4
    _LRA_BETA_MEAN = 0.1
    _LRA_BETA_COV = 0.03
    _LRA_GAMMA = 0.1
    _LRA_MIN_ETA = 1e-10

    def _lr_adaptation(
        self,
        C_inv_half: np.ndarray,
        mean_shift: np.ndarray,
        C_delta: np.ndarray,
        C_old: np.ndarray,
    ) -> None:
        """One SNR-tracking step for eta_m and eta_c.

        The raw (eta-free) updates are expressed in the local coordinates of
        the pre-update distribution — Delta_m = C^{-1/2} dm / sigma and
        Delta_C = C^{-1/2} dC C^{-1/2} / sqrt(2) — where successive
        generations are approximately iid. An exponential moving average E of
        the update and V of its squared norm give the SNR estimate
        (|E|^2 - beta/(2-beta) V) / (V - |E|^2); each eta moves
        multiplicatively toward SNR == alpha and is clipped into
        (MIN_ETA, 1]."""
        dm = C_inv_half @ mean_shift / max(self._sigma, _EPS)
        dC = (C_inv_half @ C_delta @ C_inv_half) / math.sqrt(2.0)

        b = self._LRA_BETA_MEAN
        self._lr_Em = (1 - b) * self._lr_Em + b * dm
        self._lr_Vm = (1 - b) * self._lr_Vm + b * float(dm @ dm)
        self._eta_m = self._lr_eta_step(
            self._eta_m, float(self._lr_Em @ self._lr_Em), self._lr_Vm, b
        )

        b = self._LRA_BETA_COV
        self._lr_EC = (1 - b) * self._lr_EC + b * dC
        self._lr_VC = (1 - b) * self._lr_VC + b * float(np.sum(dC * dC))
        self._eta_c = self._lr_eta_step(
            self._eta_c, float(np.sum(self._lr_EC * self._lr_EC)), self._lr_VC, b
        )

    def _lr_eta_step(self, eta: float, e_sq: float, v: float, beta: float) -> float:
        noise = v - e_sq
        if noise <= _EPS:
            snr = self._LRA_ALPHA  # no measurable noise: hold eta
        else:
            snr = max(0.0, e_sq - beta / (2 - beta) * v) / noise
        drive = min(1.0, max(-1.0, snr / self._LRA_ALPHA - 1.0))
        step = min(self._LRA_GAMMA * eta, beta)
        return float(min(1.0, max(self._LRA_MIN_ETA, eta * math.exp(step * drive))))

    def should_stop(self) -> bool:
        B, D = self._eigen()
        if self._sigma * float(D.max()) > 1e32:
            return True
        if np.all(self._sigma * np.sqrt(np.diag(self._C)) < 1e-12):
            return True
        return False


class SepCMA(CMA):
    """Separable CMA-ES: diagonal covariance, O(n) update per generation."""

    def __init__(
        self,
        mean: np.ndarray,
        sigma: float,
        bounds: np.ndarray | None = None,
        n_max_resampling: int = 100,
        seed: int | None = None,
        population_size: int | None = None,
    ) -> None:
        super().__init__(
            mean,
            sigma,
            bounds=bounds,
            n_max_resampling=n_max_resampling,
            seed=seed,
            population_size=population_size,
        )
        # Separable speedup: larger rank-μ learning rate (Ros & Hansen 2008).
        n = self._n_dim
        self._cmu = min(1 - self._c1, (self._cmu) * (n + 2) / 3)
        self._diag_C = np.ones(n)
        del self._C  # diagonal representation only

    def _eigen(self) -> tuple[np.ndarray, np.ndarray]:  # type: ignore[override]
        D = np.sqrt(np.maximum(self._diag_C, _EPS**2))
        return np.eye(self._n_dim), D  # B = I

    def _sample_one(self) -> np.ndarray:
        D = np.sqrt(np.maximum(self._diag_C, _EPS**2))
        z = self._rng.randn(self._n_dim)
        return self._mean + self._sigma * D * z

    def tell(self, solutions: list[tuple[np.ndarray, float]]) -> None:
        if len(solutions) != self._popsize:
            raise ValueError("Must tell popsize-length solutions.")
        self._g += 1
        solutions.sort(key=lambda s: s[1])

        D = np.sqrt(np.maximum(self._diag_C, _EPS**2))
        xs = np.array([s[0] for s in solutions])
        ys = (xs - self._mean) / self._sigma

        w = self._weights
        y_w = w[: self._mu] @ ys[: self._mu]
        mean_shift = self._cm * self._sigma * y_w
        if self._lr_adapt:
            self._mean = self._mean + self._eta_m * mean_shift
        else:
            self._mean = self._mean + mean_shift

        self._p_sigma = (1 - self._c_sigma) * self._p_sigma + math.sqrt(
            self._c_sigma * (2 - self._c_sigma) * self._mu_eff
        ) * (y_w / D)
        norm_p_sigma = float(np.linalg.norm(self._p_sigma))
        self._sigma *= math.exp(
            (self._c_sigma / self._d_sigma) * (norm_p_sigma / self._chi_n - 1)
        )
        self._sigma = min(self._sigma, _SIGMA_MAX)

        h_sigma_cond = norm_p_sigma / math.sqrt(
            1 - (1 - self._c_sigma) ** (2 * (self._g + 1))
        )
        h_sigma = 1.0 if h_sigma_cond < (1.4 + 2 / (self._n_dim + 1)) * self._chi_n else 0.0
        self._pc = (1 - self._cc) * self._pc + h_sigma * math.sqrt(
            self._cc * (2 - self._cc) * self._mu_eff
        ) * y_w

        w_circ = np.where(
            w >= 0, w, w * self._n_dim / (np.linalg.norm(ys / D, axis=1) ** 2 + _EPS)
        )
        delta_h = (1 - h_sigma) * self._cc * (2 - self._cc)
        rank_one = self._pc**2
        rank_mu = (w_circ[:, None] * ys**2).sum(axis=0)
        self._diag_C = (
            (1 + self._c1 * delta_h - self._c1 - self._cmu * w.sum()) * self._diag_C
            + self._c1 * rank_one
            + self._cmu * rank_mu
        )
        self._diag_C = np.maximum(self._diag_C, _EPS**2)

    def should_stop(self) -> bool:
        if np.all(self._sigma * np.sqrt(self._diag_C) < 1e-12):
            return True
        return False


class CMAwM(CMA):
    """CMA-ES with Margin for mixed-integer spaces (Hamano et al., GECCO 2022).

    Discrete coordinates (``steps[i] > 0``) are optimized in the continuous
    relaxation; ``ask`` returns both the discretized point (for evaluation) and
    the raw sample (for ``tell``). After every generation update, the margin
    correction lower-bounds the probability that a sample leaves the mean's
    current discrete cell by ``margin`` (default ``1/(n_dim·λ)``), adjusting the
    mean and a per-coordinate sampling expansion factor ``A``.

    Interface parity: reference ``optuna/samplers/_cmaes.py`` :432-456 (ask
    returns ``(x_for_eval, x_for_tell)``; tell consumes the raw samples), with
    ``steps`` given in the 0-1-transformed space (:568-579).
    """

    def __init__(
        self,
        mean: np.ndarray,
        sigma: float,
        bounds: np.ndarray,
        steps: np.ndarray,
        n_max_resampling: int = 100,
        seed: int | None = None,
        population_size: int | None = None,
        cov: np.ndarray | None = None,
        margin: float | None = None,
    ) -> None:
        super().__init__(
            mean,
            sigma,
            bounds=np.asarray(bounds, dtype=np.float64),
            n_max_resampling=n_max_resampling,
            seed=seed,
            population_size=population_size,
            cov=cov,
        )
        steps = np.asarray(steps, dtype=np.float64)
        if len(steps) != self._n_dim:
            raise ValueError("steps must have one entry per dimension")
        self._steps = steps
        self._disc_idx = np.nonzero(steps > 0)[0]
        self._cont_idx = np.nonzero(steps == 0)[0]
        # Candidate grids and midpoint thresholds per discrete dim.
        self._z_space: list[np.ndarray] = []
        self._z_lims: list[np.ndarray] = []
        for i in self._disc_idx:
            lo, hi = self._bounds[i]
            n_cand = max(int(round((hi - lo) / steps[i])) + 1, 2)
            cand = lo + steps[i] * np.arange(n_cand)
            self._z_space.append(cand)
            self._z_lims.append((cand[:-1] + cand[1:]) / 2.0)
        self._A = np.ones(self._n_dim)
        self._margin = (
            margin if margin is not None else 1.0 / (self._n_dim * self._popsize)
        )

    # ---- sampling -------------------------------------------------------------------

    def _sample_one(self) -> np.ndarray:
        B, D = self._eigen()
        z = self._rng.randn(self._n_dim)
        return self._mean + self._sigma * self._A * (B @ (D * z))

    def _in_bounds(self, x: np.ndarray) -> bool:
        # Discrete dims are snapped onto the grid, so only continuous dims
        # constrain resampling.
        c = self._cont_idx
        if len(c) == 0:
            return True
        return bool(
            np.all(x[c] >= self._bounds[c, 0]) and np.all(x[c] <= self._bounds[c, 1])
        )

    def _encode(self, x: np.ndarray) -> np.ndarray:
        out = x.copy()
        c = self._cont_idx
        out[c] = np.clip(out[c], self._bounds[c, 0], self._bounds[c, 1])
        for k, i in enumerate(self._disc_idx):
            cand = self._z_space[k]
            out[i] = cand[np.argmin(np.abs(cand - x[i]))]
        return out

    def ask(self) -> tuple[np.ndarray, np.ndarray]:  # type: ignore[override]
        for _ in range(self._n_max_resampling):
            x = self._sample_one()
            if self._in_bounds(x):
                return self._encode(x), x
        x = self._sample_one()
        x[self._cont_idx] = np.clip(
            x[self._cont_idx],
            self._bounds[self._cont_idx, 0],
            self._bounds[self._cont_idx, 1],
        )
        return self._encode(x), x

    # ---- update ---------------------------------------------------------------------

    def tell(self, solutions: list[tuple[np.ndarray, float]]) -> None:
        if len(solutions) != self._popsize:
            raise ValueError("Must tell popsize-length solutions.")
        self._g += 1
        solutions.sort(key=lambda s: s[1])

        B, D = self._eigen()
        self._B, self._D = None, None

        xs = np.array([s[0] for s in solutions])  # raw samples (λ, n)
        # Underlying Gaussian variates: x = m + σ A∘(BDz) ⇒ y = (x−m)/(σA).
        ys = (xs - self._mean) / (self._sigma * self._A)

        w = self._weights
        y_w = w[: self._mu] @ ys[: self._mu]
        # Mean moves in the sampled (A-scaled) space.
        self._mean = self._mean + self._cm * self._sigma * self._A * y_w

        C_inv_half = B @ np.diag(1.0 / D) @ B.T
        self._p_sigma = (1 - self._c_sigma) * self._p_sigma + math.sqrt(
            self._c_sigma * (2 - self._c_sigma) * self._mu_eff
        ) * (C_inv_half @ y_w)
        norm_p_sigma = float(np.linalg.norm(self._p_sigma))
        self._sigma *= math.exp(
            (self._c_sigma / self._d_sigma) * (norm_p_sigma / self._chi_n - 1)
        )
        self._sigma = min(self._sigma, _SIGMA_MAX)

        h_sigma_cond = norm_p_sigma / math.sqrt(
            1 - (1 - self._c_sigma) ** (2 * (self._g + 1))
        )
        h_sigma = 1.0 if h_sigma_cond < (1.4 + 2 / (self._n_dim + 1)) * self._chi_n else 0.0
        self._pc = (1 - self._cc) * self._pc + h_sigma * math.sqrt(
            self._cc * (2 - self._cc) * self._mu_eff
        ) * y_w

        w_circ = np.where(
            w >= 0,
            w,
            w * self._n_dim / (np.linalg.norm(ys @ C_inv_half.T, axis=1) ** 2 + _EPS),
        )
        delta_h = (1 - h_sigma) * self._cc * (2 - self._cc)
        rank_one = np.outer(self._pc, self._pc)
        rank_mu = (ys.T * w_circ) @ ys
        C_old = self._C
        C_new = (
            (1 + self._c1 * delta_h - self._c1 - self._cmu * w.sum()) * C_old
            + self._c1 * rank_one
            + self._cmu * rank_mu
        )
        if self._lr_adapt:
            self._C = C_old + self._eta_c * (C_new - C_old)
            self._lr_adaptation(C_inv_half, mean_shift, C_new - C_old, C_old)
        else:
            self._C = C_new

    # LRA constants (paper defaults): target SNR alpha, EMA rates beta, and
    # the relative update cap gamma.
    _LRA_ALPHA = 1.4
    _LRA_BETA_MEAN = 0.1
    _LRA_BETA_COV = 0.03
    _LRA_GAMMA = 0.1
    _LRA_MIN_ETA = 1e-10

    def _lr_adaptation(
        self,
        C_inv_half: np.ndarray,
        mean_shift: np.ndarray,
        C_delta: np.ndarray,
        C_old: np.ndarray,
    ) -> None:
        """One SNR-tracking step for eta_m and eta_c.

        The raw (eta-free) updates are expressed in the local coordinates of
        the pre-update distribution — Delta_m = C^{-1/2} dm / sigma and
        Delta_C = C^{-1/2} dC C^{-1/2} / sqrt(2) — where successive
        generations are approximately iid. An exponential moving average E of
        the update and V of its squared norm give the SNR estimate
        (|E|^2 - beta/(2-beta) V) / (V - |E|^2); each eta moves
        multiplicatively toward SNR == alpha and is clipped into
        (MIN_ETA, 1]."""
        dm = C_inv_half @ mean_shift / max(self._sigma, _EPS)
        dC = (C_inv_half @ C_delta @ C_inv_half) / math.sqrt(2.0)

        b = self._LRA_BETA_MEAN
        self._lr_Em = (1 - b) * self._lr_Em + b * dm
        self._lr_Vm = (1 - b) * self._lr_Vm + b * float(dm @ dm)
        self._eta_m = self._lr_eta_step(
            self._eta_m, float(self._lr_Em @ self._lr_Em), self._lr_Vm, b
        )

        b = self._LRA_BETA_COV
        self._lr_EC = (1 - b) * self._lr_EC + b * dC
        self._lr_VC = (1 - b) * self._lr_VC + b * float(np.sum(dC * dC))
        self._eta_c = self._lr_eta_step(
            self._eta_c, float(np.sum(self._lr_EC * self._lr_EC)), self._lr_VC, b
        )

    def _lr_eta_step(self, eta: float, e_sq: float, v: float, beta: float) -> float:
        noise = v - e_sq
        if noise <= _EPS:
            snr = self._LRA_ALPHA  # no measurable noise: hold eta
        else:
            snr = max(0.0, e_sq - beta / (2 - beta) * v) / noise
        drive = min(1.0, max(-1.0, snr / self._LRA_ALPHA - 1.0))
        step = min(self._LRA_GAMMA * eta, beta)
        return float(min(1.0, max(self._LRA_MIN_ETA, eta * math.exp(step * drive))))

        self._margin_correction()

    def _margin_correction(self) -> None:
        from scipy.special import ndtr, ndtri

        alpha = self._margin
        diag_C = np.diag(self._C)
        for k, i in enumerate(self._disc_idx):
            lims = self._z_lims[k]
            m = float(self._mean[i])
            s_dev = self._sigma * self._A[i] * math.sqrt(max(diag_C[i], _EPS**2))
            pos = int(np.searchsorted(lims, m))
            n_cand = len(lims) + 1
            if pos == 0:
                # Mean sits in the lowest cell: keep P(x > lims[0]) ≥ α.
                lo_lim = lims[0]
                if 1.0 - ndtr((lo_lim - m) / s_dev) < alpha:
                    self._mean[i] = lo_lim - s_dev * ndtri(1.0 - alpha)
            elif pos == n_cand - 1:
                # Highest cell: keep P(x < lims[-1]) ≥ α.
                up_lim = lims[-1]
                if ndtr((up_lim - m) / s_dev) < alpha:
                    self._mean[i] = up_lim + s_dev * ndtri(1.0 - alpha)
            else:
                lo_lim = lims[pos - 1]
                up_lim = lims[pos]
                p_low = float(ndtr((lo_lim - m) / s_dev))
                p_up = 1.0 - float(ndtr((up_lim - m) / s_dev))
                if p_low >= alpha / 2 and p_up >= alpha / 2:
                    continue
                p_low_t = max(p_low, alpha / 2)
                p_up_t = max(p_up, alpha / 2)
                q_low = float(ndtri(p_low_t))
                q_up = float(ndtri(1.0 - p_up_t))
                if not q_up > q_low:
                    continue  # degenerate (extremely wide cell); leave untouched
                s_new = (up_lim - lo_lim) / (q_up - q_low)
                self._mean[i] = lo_lim - s_new * q_low
                self._A[i] = s_new / (
                    self._sigma * math.sqrt(max(diag_C[i], _EPS**2))
                )

def get_warm_start_mgd(
    source_solutions: list[tuple[np.ndarray, float]],
    gamma: float = 0.1,
    alpha: float = 0.1,
) -> tuple[np.ndarray, float, np.ndarray]:
    """Estimate a promising (mean, sigma, cov) from source-task solutions.

    WS-CMA-ES (Nomura et al., AAAI 2021): fit a multivariate Gaussian to the
    top-γ fraction of the source solutions; σ carries the scale (det-normalized)
    and cov the shape.
    """
    if len(source_solutions) == 0:
        raise ValueError("solutions should contain one or more items.")
    n_dim = len(source_solutions[0][0])
    top_k = max(int(math.ceil(len(source_solutions) * gamma)), n_dim + 1)
    top_k = min(top_k, len(source_solutions))
    best = sorted(source_solutions, key=lambda s: s[1])[:top_k]
    X = np.array([s[0] for s in best])  # (k, n)

    mean = X.mean(axis=0)
    centered = X - mean
    cov_full = (centered.T @ centered) / max(len(X) - 1, 1) + alpha**2 * np.eye(n_dim)
    # Split scale (σ) and shape (unit-determinant covariance).
    sign, logdet = np.linalg.slogdet(cov_full)
    assert sign > 0
    sigma = float(np.exp(logdet / (2 * n_dim)))
    cov = cov_full / (sigma**2)
    return mean, sigma, cov
