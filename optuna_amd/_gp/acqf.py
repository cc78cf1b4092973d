"""Acquisition functions over fitted GPRegressors (K5 target).

Parity: reference ``optuna/_gp/acqf.py`` (standard_logei :96, logei :121, LogEI
:156, qLogEI :180, LogPI :215 with Kriging-Believer append, qLogPI :257,
UCB/LCB :288/:304, LogCEI :320, qLogCEI :353, LogEHVI :404 with non-dominated
box decomposition, qLogEHVI :473, LogCEHVI :534). All math in fp64.
"""
from __future__ import annotations

import math
from abc import ABC, abstractmethod
from typing import TYPE_CHECKING

import numpy as np

from optuna_amd._gp.gp import ConditionalGPRegressor, GPRegressor
from optuna_amd._gp.qmc import sample_from_normal_sobol
from optuna_amd._hypervolume.box_decomposition import get_non_dominated_box_bounds
from optuna_amd.study._multi_objective import _is_pareto_front


if TYPE_CHECKING:
    import torch

    from optuna_amd._gp.search_space import SearchSpace
else:
    from optuna_amd._imports import _LazyImport

    torch = _LazyImport("torch")

_EPS = 1e-12  # zero eps makes gradients NaN
_HALF_LOG_2PI = 0.5 * math.log(2.0 * math.pi)
# Below this z the two terms of z*Phi(z)+phi(z) cancel; switch to the erfcx form.
_TAIL_Z = -1.0


def standard_logei(z: "torch.Tensor") -> "torch.Tensor":
    """log E_{x~N(0,1)}[max(0, x+z)] = log(z*Phi(z) + phi(z)), tail-stable.

    Two algebraically-equivalent evaluations selected by ``torch.where``:

    * right half (z >= _TAIL_Z): the textbook sum, whose terms are the same
      order of magnitude there, logged directly;
    * left half: factor phi(z) out first.  Since Phi(z) = phi(z) *
      sqrt(pi/2) * erfcx(-z/sqrt(2)), the whole expectation is
      phi(z) * (1 + sqrt(pi/2) * z * erfcx(-z/sqrt(2))) and its log is the
      log-density plus a log1p of a quantity in (-1, 0] — no cancellation
      however deep the tail.

    Both branches are evaluated on range-clamped copies of ``z`` so the
    inactive branch can neither overflow nor poison gradients through where().
    """
    inv_sqrt2 = math.sqrt(0.5)
    zr = z.clamp(min=_TAIL_Z - 0.5)
    phi_r = torch.exp(-0.5 * zr * zr - _HALF_LOG_2PI)
    right = torch.log(zr * torch.special.ndtr(zr) + phi_r)

    zl = z.clamp(max=_TAIL_Z + 0.5)
    scaled_mills = math.sqrt(0.5 * math.pi) * zl * torch.special.erfcx(-zl * inv_sqrt2)
    left = (-0.5 * zl * zl - _HALF_LOG_2PI) + torch.log1p(scaled_mills)

    return torch.where(z >= _TAIL_Z, right, left)


def logei(mean: "torch.Tensor", var: "torch.Tensor", f0: float) -> "torch.Tensor":
    """log E_{y~N(mean, var)}[max(0, y - f0)] elementwise."""
    half_log_var = 0.5 * torch.log(var)
    z = (mean - f0) * torch.exp(-half_log_var)
    return half_log_var + standard_logei(z)


def logehvi(
    Y_post: "torch.Tensor",  # (..., n_qmc_samples, n_objectives)
    non_dominated_box_lower_bounds: "torch.Tensor",
    non_dominated_box_intervals: "torch.Tensor",
) -> "torch.Tensor":
    log_n = float(np.log(Y_post.shape[-2]))
    diff = (Y_post.unsqueeze(-2) - non_dominated_box_lower_bounds).clamp_min_(_EPS)
    diff = torch.minimum(diff, non_dominated_box_intervals)
    return torch.special.logsumexp(diff.log().sum(dim=-1), dim=(-2, -1)) - log_n


def _per_sample_log_hvi(
    Y_post: "torch.Tensor",
    non_dominated_box_lower_bounds: "torch.Tensor",
    non_dominated_box_intervals: "torch.Tensor",
) -> "torch.Tensor":
    diff = (Y_post.unsqueeze(-2) - non_dominated_box_lower_bounds).clamp_min_(_EPS)
    diff = torch.minimum(diff, non_dominated_box_intervals)
    return torch.special.logsumexp(diff.log().sum(dim=-1), dim=-1)


def _get_reference_point(Y: "torch.Tensor") -> np.ndarray:
    # Work in loss (minimization) space: worst observed loss per objective,
    # pushed 10% further from the origin, then one ulp more so every kept
    # point is strictly inside the reference box.
    worst = np.asarray((-Y).amax(dim=0))
    pushed = np.where(worst >= 0.0, 1.1 * worst, 0.9 * worst)
    return np.nextafter(pushed, np.inf)


def _get_boxes(Y: "torch.Tensor", ref_point: np.ndarray) -> tuple["torch.Tensor", "torch.Tensor"]:
    """Non-dominated boxes of -Y under ``ref_point``, returned in max space.

    Negating swaps the roles of the bounds, so the decomposition's (lower,
    upper) pairs come back as (-upper, -lower).
    """
    losses = np.asarray(-Y)
    kept = losses[(losses < ref_point).all(axis=-1)]
    front = kept[_is_pareto_front(kept, assume_unique_lexsorted=False)]
    box_lo, box_hi = get_non_dominated_box_bounds(front, ref_point)
    return torch.from_numpy(-box_hi), torch.from_numpy(-box_lo)


def _fantasize(
    gpr: GPRegressor,
    X_running: np.ndarray,
    n_qmc_samples: int,
    qmc_seed: int,
    stabilizing_noise: float,
) -> ConditionalGPRegressor:
    """Condition ``gpr`` on QMC fantasy outcomes at the running trials' points."""
    return ConditionalGPRegressor(
        gpr=gpr,
        X_running=torch.from_numpy(X_running),
        n_qmc_samples=n_qmc_samples,
        qmc_seed=qmc_seed,
        stabilizing_noise=stabilizing_noise,
    )


def _mean_max_log_utility(log_utils: "torch.Tensor") -> "torch.Tensor":
    # MC estimate in log space: per posterior draw keep the best of the
    # q-batch, then average the draws.
    per_draw_best = log_utils.amax(dim=-1)
    n_draws = per_draw_best.shape[-1]
    return per_draw_best.logsumexp(dim=-1) - math.log(n_draws)


class BaseAcquisitionFunc(ABC):
    # Device the underlying GP tensors live on (None = CPU). The numpy entry
    # points below are the only host/device boundary: candidates go up, scalars
    # and gradients come back.
    _device: "torch.device | None" = None

    def __init__(
        self,
        length_scales: np.ndarray,
        search_space: "SearchSpace",
        device: "torch.device | None" = None,
    ) -> None:
        self.length_scales = length_scales
        self.search_space = search_space
        if device is not None and device.type != "cpu":
            self._device = device

    def set_device(self, device: "torch.device | None") -> None:
        self._device = device

    def _up(self, x: np.ndarray) -> "torch.Tensor":
        t = torch.from_numpy(x)
        return t.to(self._device) if self._device is not None else t

    @abstractmethod
    def eval_acqf(self, x: "torch.Tensor") -> "torch.Tensor":
        raise NotImplementedError

    def eval_acqf_no_grad(self, x: np.ndarray) -> np.ndarray:
        with torch.no_grad():
            return self.eval_acqf(self._up(x)).detach().cpu().numpy()

    def eval_acqf_with_grad(self, x: np.ndarray) -> tuple[float, np.ndarray]:
        assert x.ndim == 1
        x_tensor = self._up(x).requires_grad_(True)
        val = self.eval_acqf(x_tensor)
        val.backward()
        return val.item(), x_tensor.grad.detach().cpu().numpy()  # type: ignore[union-attr]

    def eval_acqf_batched_with_grad(self, x: np.ndarray) -> tuple[np.ndarray, np.ndarray]:
        """(fvals, grads) for a (B, D) batch — one graph, one transfer each way."""
        x_tensor = self._up(x).requires_grad_(True)
        fvals = self.eval_acqf(x_tensor)
        fvals.sum().backward()
        return (
            np.atleast_1d(fvals.detach().cpu().numpy()),
            x_tensor.grad.detach().cpu().numpy(),  # type: ignore[union-attr]
        )


class LogEI(BaseAcquisitionFunc):
    def __init__(
        self,
        gpr: GPRegressor,
        search_space: "SearchSpace",
        threshold: float,
        stabilizing_noise: float = 1e-12,
    ) -> None:
        self._gpr = gpr
        self._stabilizing_noise = stabilizing_noise
        self._threshold = threshold
        self._fused = None
        super().__init__(gpr.length_scales, search_space, gpr.device)

    def eval_acqf(self, x: "torch.Tensor") -> "torch.Tensor":
        if np.isneginf(self._threshold):
            return torch.zeros(x.shape[:-1], dtype=torch.float64, device=x.device)
        mean, var = self._gpr.posterior(x)
        return logei(mean=mean, var=var + self._stabilizing_noise, f0=self._threshold)

    # ---- fused K5 device path -------------------------------------------------------
    # On the MI355X the torch evaluation costs ~50 kernel launches per call
    # (posterior GEMMs + erfc branches + autograd); the K5 kernel computes the
    # cross-covariance, one rocBLAS dgemm against the resident explicit
    # inverse, and the tail-stable log-EI with its CLOSED-FORM gradient — four
    # launches total, operating directly on the torch-owned device tensors.

    def _fused_session(self):
        if self._fused is not None:
            return self._fused or None
        gpr = self._gpr
        if (
            self._device is None
            or np.isneginf(self._threshold)
            or gpr._cov_Y_Y_inv is None
            or gpr._X_all is not gpr._X_train
            or bool(gpr._is_categorical.any())
            or gpr._X_train.shape[1] > 64
        ):
            self._fused = False
            return None
        from optuna_amd import _hip

        core = _hip.get()
        if core is None or not core.available():
            self._fused = False
            return None
        # Our stream must not race torch's producers of these tensors.
        torch.cuda.synchronize()
        self._fused_refs = (
            gpr._X_train.contiguous(),
            gpr._cov_Y_Y_inv_Y.contiguous(),
            gpr._cov_Y_Y_inv.contiguous(),
            gpr.inverse_squared_lengthscales.contiguous(),
        )
        X, alpha, cinv, eta = self._fused_refs
        self._fused = core.GpLogEiSession(
            X.data_ptr(),
            alpha.data_ptr(),
            cinv.data_ptr(),
            eta.data_ptr(),
            int(X.shape[0]),
            int(X.shape[1]),
            float(self._gpr.kernel_scale.item()),
            float(self._stabilizing_noise),
            float(self._threshold),
        )
        return self._fused

    def eval_acqf_no_grad(self, x: np.ndarray) -> np.ndarray:
        session = self._fused_session()
        if session is not None:
            x2 = np.ascontiguousarray(np.atleast_2d(x), dtype=np.float64)
            f, _ = session.eval(x2, with_grad=False)
            return np.asarray(f) if x.ndim > 1 else np.asarray(f)[0]
        return super().eval_acqf_no_grad(x)

    def eval_acqf_batched_with_grad(self, x: np.ndarray) -> tuple[np.ndarray, np.ndarray]:
        session = self._fused_session()
        if session is not None:
            f, g = session.eval(np.ascontiguousarray(x, dtype=np.float64), with_grad=True)
            return np.asarray(f), np.asarray(g)
        return super().eval_acqf_batched_with_grad(x)

    def eval_acqf_with_grad(self, x: np.ndarray) -> tuple[float, np.ndarray]:
        session = self._fused_session()
        if session is not None:
            f, g = session.eval(
                np.ascontiguousarray(x[None, :], dtype=np.float64), with_grad=True
            )
            return float(np.asarray(f)[0]), np.asarray(g)[0]
        return super().eval_acqf_with_grad(x)


class qLogEI(BaseAcquisitionFunc):
    def __init__(
        self,
        gpr: GPRegressor,
        search_space: "SearchSpace",
        threshold: float,
        normalized_params_of_running_trials: np.ndarray,
        n_qmc_samples: int,
        qmc_seed: int,
        stabilizing_noise: float = 1e-12,
    ) -> None:
        self._threshold = threshold
        self._cond_gpr = _fantasize(
            gpr, normalized_params_of_running_trials, n_qmc_samples, qmc_seed, stabilizing_noise
        )
        n_running = len(normalized_params_of_running_trials)
        self._per_sample_shape = (n_qmc_samples, n_running + 1)
        super().__init__(gpr.length_scales, search_space, gpr.device)

    def compute_per_sample_log_utility(self, x: "torch.Tensor") -> "torch.Tensor":
        if np.isneginf(self._threshold):
            return torch.zeros(
                x.shape[:-1] + self._per_sample_shape, dtype=torch.float64, device=x.device
            )
        y_post = self._cond_gpr.sample_joint_posterior(x)
        return (y_post - self._threshold).clamp_min_(_EPS).log()

    def eval_acqf(self, x: "torch.Tensor") -> "torch.Tensor":
        return _mean_max_log_utility(self.compute_per_sample_log_utility(x))


class LogPI(BaseAcquisitionFunc):
    def __init__(
        self,
        gpr: GPRegressor,
        search_space: "SearchSpace",
        threshold: float,
        normalized_params_of_running_trials: np.ndarray | None = None,
        stabilizing_noise: float = 1e-12,
    ) -> None:
        self._gpr = gpr
        self._stabilizing_noise = stabilizing_noise
        self._threshold = threshold
        if normalized_params_of_running_trials is not None:
            # Kriging Believer: append running points at their posterior mean
            # (copy-on-write — the sampler's cached regressor stays pristine).
            X_running = torch.from_numpy(normalized_params_of_running_trials).to(gpr.device)
            self._gpr = gpr.cloned_with_running(X_running, gpr.posterior(X_running)[0])
        super().__init__(gpr.length_scales, search_space, gpr.device)

    def eval_acqf(self, x: "torch.Tensor") -> "torch.Tensor":
        mean, var = self._gpr.posterior(x)
        sigma = torch.sqrt(var + self._stabilizing_noise)
        return torch.special.log_ndtr((mean - self._threshold) / sigma)


class qLogPI(BaseAcquisitionFunc):
    def __init__(
        self,
        gpr: GPRegressor,
        search_space: "SearchSpace",
        threshold: float,
        normalized_params_of_running_trials: np.ndarray,
        n_qmc_samples: int,
        qmc_seed: int,
        stabilizing_noise: float = 1e-12,
        tau: float = 1e-2,
    ) -> None:
        self._threshold = threshold
        self._tau = tau
        self._cond_gpr = _fantasize(
            gpr, normalized_params_of_running_trials, n_qmc_samples, qmc_seed, stabilizing_noise
        )
        super().__init__(gpr.length_scales, search_space, gpr.device)

    def compute_per_sample_log_utility(self, x: "torch.Tensor") -> "torch.Tensor":
        y_post = self._cond_gpr.sample_joint_posterior(x)
        return torch.nn.functional.logsigmoid((y_post - self._threshold) / self._tau)

    def eval_acqf(self, x: "torch.Tensor") -> "torch.Tensor":
        return _mean_max_log_utility(self.compute_per_sample_log_utility(x))


class UCB(BaseAcquisitionFunc):
    def __init__(self, gpr: GPRegressor, search_space: "SearchSpace", beta: float) -> None:
        self._gpr = gpr
        self._beta = beta
        super().__init__(gpr.length_scales, search_space, gpr.device)

    def eval_acqf(self, x: "torch.Tensor") -> "torch.Tensor":
        mean, var = self._gpr.posterior(x)
        return mean + torch.sqrt(self._beta * var)


class LCB(BaseAcquisitionFunc):
    def __init__(self, gpr: GPRegressor, search_space: "SearchSpace", beta: float) -> None:
        self._gpr = gpr
        self._beta = beta
        super().__init__(gpr.length_scales, search_space, gpr.device)

    def eval_acqf(self, x: "torch.Tensor") -> "torch.Tensor":
        mean, var = self._gpr.posterior(x)
        return mean - torch.sqrt(self._beta * var)


class LogCEI(BaseAcquisitionFunc):
    """log(EI × ∏ feasibility-PI of each constraint)."""

    def __init__(
        self,
        gpr: GPRegressor,
        search_space: "SearchSpace",
        threshold: float,
        constraints_gpr_list: list[GPRegressor],
        constraints_threshold_list: list[float],
        stabilizing_noise: float = 1e-12,
    ) -> None:
        assert constraints_gpr_list and len(constraints_gpr_list) == len(
            constraints_threshold_list
        )
        self._acqf = LogEI(gpr, search_space, threshold, stabilizing_noise)
        self._constraints_acqf_list = [
            LogPI(c_gpr, search_space, c_threshold, None, stabilizing_noise)
            for c_gpr, c_threshold in zip(constraints_gpr_list, constraints_threshold_list)
        ]
        super().__init__(gpr.length_scales, search_space, gpr.device)

    def eval_acqf(self, x: "torch.Tensor") -> "torch.Tensor":
        return self._acqf.eval_acqf(x) + sum(
            acqf.eval_acqf(x) for acqf in self._constraints_acqf_list
        )


class qLogCEI(BaseAcquisitionFunc):
    def __init__(
        self,
        gpr: GPRegressor,
        search_space: "SearchSpace",
        threshold: float,
        normalized_params_of_running_trials: np.ndarray,
        n_qmc_samples: int,
        qmc_seed: int,
        constraints_gpr_list: list[GPRegressor],
        constraints_threshold_list: list[float],
        stabilizing_noise: float = 1e-12,
    ) -> None:
        assert constraints_gpr_list and len(constraints_gpr_list) == len(
            constraints_threshold_list
        )
        self._acqf = qLogEI(
            gpr, search_space, threshold,
            normalized_params_of_running_trials, n_qmc_samples, qmc_seed, stabilizing_noise,
        )
        self._constraints_acqf_list = [
            qLogPI(
                gpr=c_gpr,
                search_space=search_space,
                threshold=c_threshold,
                n_qmc_samples=n_qmc_samples,
                qmc_seed=qmc_seed + i + 1,
                normalized_params_of_running_trials=normalized_params_of_running_trials,
                stabilizing_noise=stabilizing_noise,
            )
            for i, (c_gpr, c_threshold) in enumerate(
                zip(constraints_gpr_list, constraints_threshold_list)
            )
        ]
        super().__init__(gpr.length_scales, search_space, gpr.device)

    def eval_acqf(self, x: "torch.Tensor") -> "torch.Tensor":
        log_feasible_improvement = self._acqf.compute_per_sample_log_utility(x) + sum(
            acqf.compute_per_sample_log_utility(x) for acqf in self._constraints_acqf_list
        )
        return _mean_max_log_utility(log_feasible_improvement)


class LogEHVI(BaseAcquisitionFunc):
    def __init__(
        self,
        gpr_list: list[GPRegressor],
        search_space: "SearchSpace",
        Y_train: "torch.Tensor",
        *,
        normalized_params_of_running_trials: np.ndarray | None = None,
        n_qmc_samples: int,
        qmc_seed: int,
        stabilizing_noise: float = 1e-12,
    ) -> None:
        self._stabilizing_noise = stabilizing_noise
        self._gpr_list = gpr_list
        dev = gpr_list[0].device
        if normalized_params_of_running_trials is not None:
            X_running = torch.from_numpy(normalized_params_of_running_trials).to(dev)
            self._gpr_list = [
                g.cloned_with_running(X_running, g.posterior(X_running)[0])
                for g in self._gpr_list
            ]
        self._fixed_samples = sample_from_normal_sobol(
            dim=Y_train.shape[-1], n_samples=n_qmc_samples, seed=qmc_seed, device=dev
        )
        ref_point = _get_reference_point(Y_train)
        box_lower, box_upper = _get_boxes(Y_train, ref_point)
        self._box_lower = box_lower.to(dev)
        self._box_intervals = (box_upper - box_lower).clamp_min_(_EPS).to(dev)
        super().__init__(
            np.mean([gpr.length_scales for gpr in gpr_list], axis=0), search_space, dev
        )

    def eval_acqf(self, x: "torch.Tensor") -> "torch.Tensor":
        Y_post = []
        for i, gpr in enumerate(self._gpr_list):
            mean, var = gpr.posterior(x)
            stdev = torch.sqrt(var + self._stabilizing_noise)
            Y_post.append(mean[..., None] + stdev[..., None] * self._fixed_samples[..., i])
        return logehvi(
            Y_post=torch.stack(Y_post, dim=-1),
            non_dominated_box_lower_bounds=self._box_lower,
            non_dominated_box_intervals=self._box_intervals,
        )


class qLogEHVI(BaseAcquisitionFunc):
    def __init__(
        self,
        gpr_list: list[GPRegressor],
        search_space: "SearchSpace",
        Y_train: "torch.Tensor",
        normalized_params_of_running_trials: np.ndarray,
        n_qmc_samples: int,
        qmc_seed: int,
        stabilizing_noise: float = 1e-12,
    ) -> None:
        self._Y_train = Y_train
        self._cond_gpr_list = [
            _fantasize(
                gpr, normalized_params_of_running_trials, n_qmc_samples, qmc_seed + i, stabilizing_noise
            )
            for i, gpr in enumerate(gpr_list)
        ]
        dev = gpr_list[0].device
        ref_point = _get_reference_point(Y_train)
        lower_list, interval_list = [], []
        for fantasy in torch.stack(
            [cg.get_fantasy_samples().cpu() for cg in self._cond_gpr_list], dim=-1
        ):
            Y_fantasy = torch.cat([Y_train, fantasy], dim=0)
            lower, upper = _get_boxes(Y_fantasy, ref_point)
            lower_list.append(lower)
            interval_list.append((upper - lower).clamp_min_(_EPS))
        self._box_lower = torch.nn.utils.rnn.pad_sequence(lower_list, batch_first=True).to(dev)
        self._box_intervals = torch.nn.utils.rnn.pad_sequence(
            interval_list, batch_first=True, padding_value=_EPS
        ).to(dev)
        super().__init__(
            np.mean([gpr.length_scales for gpr in gpr_list], axis=0), search_space, dev
        )

    def compute_per_sample_log_utility(self, x: "torch.Tensor") -> "torch.Tensor":
        Y_candidate_post = torch.stack(
            [cg.sample_joint_posterior(x, return_fantasy=False) for cg in self._cond_gpr_list],
            dim=-1,
        )
        return _per_sample_log_hvi(
            Y_post=Y_candidate_post,
            non_dominated_box_lower_bounds=self._box_lower,
            non_dominated_box_intervals=self._box_intervals,
        )

    def eval_acqf(self, x: "torch.Tensor") -> "torch.Tensor":
        log_utils = self.compute_per_sample_log_utility(x)
        return torch.special.logsumexp(log_utils, dim=-1) - math.log(log_utils.shape[-1])


class LogCEHVI(BaseAcquisitionFunc):
    """EHVI over feasible observations × feasibility-PI of each constraint."""

    def __init__(
        self,
        gpr_list: list[GPRegressor],
        search_space: "SearchSpace",
        Y_feasible: "torch.Tensor | None",
        *,
        normalized_params_of_running_trials: np.ndarray | None = None,
        n_qmc_samples: int,
        qmc_seed: int,
        constraints_gpr_list: list[GPRegressor],
        constraints_threshold_list: list[float],
        stabilizing_noise: float = 1e-12,
    ) -> None:
        assert constraints_gpr_list and len(constraints_gpr_list) == len(
            constraints_threshold_list
        )
        self._acqf: BaseAcquisitionFunc | None = (
            LogEHVI(
                gpr_list=gpr_list,
                search_space=search_space,
                Y_train=Y_feasible,
                n_qmc_samples=n_qmc_samples,
                qmc_seed=qmc_seed,
                normalized_params_of_running_trials=normalized_params_of_running_trials,
                stabilizing_noise=stabilizing_noise,
            )
            if Y_feasible is not None
            else None
        )
        self._constraints_acqf_list = [
            LogPI(
                c_gpr,
                search_space,
                c_threshold,
                normalized_params_of_running_trials,
                stabilizing_noise,
            )
            for c_gpr, c_threshold in zip(constraints_gpr_list, constraints_threshold_list)
        ]
        super().__init__(
            np.mean([gpr.length_scales for gpr in gpr_list], axis=0),
            search_space,
            gpr_list[0].device,
        )

    def eval_acqf(self, x: "torch.Tensor") -> "torch.Tensor":
        feasibility = sum(acqf.eval_acqf(x) for acqf in self._constraints_acqf_list)
        if self._acqf is None:
            return feasibility  # no feasible observation yet: maximize feasibility
        return self._acqf.eval_acqf(x) + feasibility
