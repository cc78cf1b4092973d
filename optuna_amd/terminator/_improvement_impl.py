"""Improvement evaluators: regret bound (GP-UCB), stagnation, and EMMR.

Parity: reference ``optuna/terminator/improvement/evaluator.py``
(RegretBoundEvaluator :50-85 — UCB−LCB with β from GP-UCB theory;
BestValueStagnationEvaluator) and ``improvement/emmr.py`` (EMMREvaluator —
expected minimum model regret upper bound, Ishibashi et al. AISTATS 2023).
"""
from __future__ import annotations

import abc
import math
import sys
import warnings
from typing import TYPE_CHECKING

import numpy as np

from optuna_amd._gp import acqf as acqf_module
from optuna_amd._gp import gp
from optuna_amd._gp import optim_sample
from optuna_amd._gp import prior
from optuna_amd._gp import search_space as gp_search_space
from optuna_amd.distributions import BaseDistribution
from optuna_amd.samplers._lazy_random_state import LazyRandomState
from optuna_amd.search_space import intersection_search_space
from optuna_amd.study._study_direction import StudyDirection
from optuna_amd.trial import FrozenTrial, TrialState


if TYPE_CHECKING:
    import torch
else:
    from optuna_amd._imports import _LazyImport

    torch = _LazyImport("torch")

DEFAULT_TOP_TRIALS_RATIO = 0.5
DEFAULT_MIN_N_TRIALS = 20
_MARGIN = 0.1


def _get_beta(n_params: int, n_trials: int, delta: float = 0.1) -> float:
    # GP-UCB theory beta, tempered by 1/5 as in the Makarova et al. implementation.
    return 2 * np.log(n_params * n_trials**2 * np.pi**2 / 6 / delta) / 5


def _compute_standardized_regret_bound(
    gpr: gp.GPRegressor,
    search_space: gp_search_space.SearchSpace,
    normalized_top_n_params: np.ndarray,
    standardized_top_n_values: np.ndarray,
    delta: float = 0.1,
    optimize_n_samples: int = 2048,
    rng: np.random.RandomState | None = None,
) -> float:
    """UCB(max over space) − LCB(max over observed) ≥ regret of the incumbent."""
    n_trials, n_params = normalized_top_n_params.shape
    beta = _get_beta(n_params, n_trials, delta)
    ucb = acqf_module.UCB(gpr, search_space, beta)
    ucb_value = max(
        float(ucb.eval_acqf_no_grad(normalized_top_n_params).max()),
        optim_sample.optimize_acqf_sample(ucb, n_samples=optimize_n_samples, rng=rng)[1],
    )
    lcb = acqf_module.LCB(gpr, search_space, beta)
    lcb_value = float(np.max(lcb.eval_acqf_no_grad(normalized_top_n_params)))
    return ucb_value - lcb_value


class BaseImprovementEvaluator(abc.ABC):
    @abc.abstractmethod
    def evaluate(
        self, trials: list[FrozenTrial], study_direction: StudyDirection
    ) -> float:
        raise NotImplementedError


class RegretBoundEvaluator(BaseImprovementEvaluator):
    """Upper bound on the simple regret of the incumbent under a GP model."""

    def __init__(
        self,
        top_trials_ratio: float = DEFAULT_TOP_TRIALS_RATIO,
        min_n_trials: int = DEFAULT_MIN_N_TRIALS,
        seed: int | None = None,
    ) -> None:
        self._top_trials_ratio = top_trials_ratio
        self._min_n_trials = min_n_trials
        self._log_prior = prior.default_log_prior
        self._minimum_noise = prior.DEFAULT_MINIMUM_NOISE_VAR
        self._optimize_n_samples = 2048
        self._rng = LazyRandomState(seed)

    def _get_top_n(
        self, normalized_params: np.ndarray, values: np.ndarray
    ) -> tuple[np.ndarray, np.ndarray]:
        n_trials = len(normalized_params)
        top_n = int(np.clip(int(n_trials * self._top_trials_ratio), self._min_n_trials, n_trials))
        top_n_val = np.partition(values, n_trials - top_n)[n_trials - top_n]
        mask = values >= top_n_val
        return normalized_params[mask], values[mask]

    def evaluate(
        self, trials: list[FrozenTrial], study_direction: StudyDirection
    ) -> float:
        optuna_space = intersection_search_space(trials)
        self._validate_input(trials, optuna_space)
        complete = [t for t in trials if t.state == TrialState.COMPLETE]
        sign = -1 if study_direction == StudyDirection.MINIMIZE else 1
        values = np.array([t.value for t in complete]) * sign
        search_space = gp_search_space.SearchSpace(optuna_space)
        normalized_params = search_space.get_normalized_params(complete)
        top_params, top_values = self._get_top_n(normalized_params, values)
        mean = top_values.mean()
        std = max(1e-10, top_values.std())
        standardized = (top_values - mean) / std
        gpr = gp.fit_kernel_params(
            X=top_params,
            Y=standardized,
            is_categorical=search_space.is_categorical,
            log_prior=self._log_prior,
            minimum_noise=self._minimum_noise,
            deterministic_objective=False,
            gpr_cache=None,
        )
        return (
            _compute_standardized_regret_bound(
                gpr, search_space, top_params, standardized, rng=self._rng.rng
            )
            * std
        )

    @classmethod
    def _validate_input(
        cls, trials: list[FrozenTrial], search_space: dict[str, BaseDistribution]
    ) -> None:
        if len([t for t in trials if t.state == TrialState.COMPLETE]) == 0:
            raise ValueError(
                "Because no trial has been completed yet, the regret bound cannot be "
                "evaluated."
            )
        if len(search_space) == 0:
            raise ValueError(
                "The intersection search space is empty. This condition is not supported "
                f"by {cls.__name__}."
            )


class BestValueStagnationEvaluator(BaseImprovementEvaluator):
    """Remaining trials until the allowed stagnation budget is exhausted."""

    def __init__(self, max_stagnation_trials: int = 30) -> None:
        if max_stagnation_trials < 0:
            raise ValueError("The maximum number of stagnant trials must not be negative.")
        self._max_stagnation_trials = max_stagnation_trials

    def evaluate(
        self, trials: list[FrozenTrial], study_direction: StudyDirection
    ) -> float:
        complete = [t for t in trials if t.state == TrialState.COMPLETE]
        if len(complete) == 0:
            raise ValueError(
                "Because no trial has been completed yet, the improvement cannot be "
                "evaluated."
            )
        maximize = study_direction == StudyDirection.MAXIMIZE
        best_step = 0
        for i, trial in enumerate(complete):
            best_value = complete[best_step].value
            current_value = trial.value
            assert best_value is not None and current_value is not None
            if (maximize and current_value > best_value) or (
                not maximize and current_value < best_value
            ):
                best_step = i
        current_step = len(complete) - 1
        return self._max_stagnation_trials - (current_step - best_step)


class EMMREvaluator(BaseImprovementEvaluator):
    """Upper bound on the gap of expected minimum simple regrets between steps.

    Ishibashi et al., "A stopping criterion for Bayesian optimization by the gap
    of expected minimum simple regrets" (AISTATS 2023).
    """

    def __init__(
        self,
        deterministic_objective: bool = False,
        delta: float = 0.1,
        min_n_trials: int = 2,
        seed: int | None = None,
    ) -> None:
        if min_n_trials <= 1 or not np.isfinite(min_n_trials):
            raise ValueError("`min_n_trials` is expected to be a finite integer more than one.")
        self._deterministic = deterministic_objective
        self._delta = delta
        self.min_n_trials = min_n_trials
        self._rng = LazyRandomState(seed)

    def evaluate(
        self, trials: list[FrozenTrial], study_direction: StudyDirection
    ) -> float:
        from scipy import stats as scipy_stats

        optuna_space = intersection_search_space(trials)
        complete = [t for t in trials if t.state == TrialState.COMPLETE]
        if len(complete) < self.min_n_trials:
            return sys.float_info.max * _MARGIN  # do not terminate yet
        search_space = gp_search_space.SearchSpace(optuna_space)
        if not search_space.dim:
            warnings.warn(
                f"{self.__class__.__name__} cannot consider any search space. "
                "Termination will never occur in this study."
            )
            return sys.float_info.max * _MARGIN
        normalized_params = search_space.get_normalized_params(complete)
        sign = -1 if study_direction == StudyDirection.MINIMIZE else 1
        score_vals = gp.warn_and_convert_inf(
            np.array([t.value for t in complete]) * sign
        )
        standardized = (score_vals - score_vals.mean()) / max(
            sys.float_info.min, score_vals.std()
        )

        gpr_t1 = gp.fit_kernel_params(  # model with observations up to t-1
            X=normalized_params[:-1, :],
            Y=standardized[:-1],
            is_categorical=search_space.is_categorical,
            log_prior=prior.default_log_prior,
            minimum_noise=prior.DEFAULT_MINIMUM_NOISE_VAR,
            gpr_cache=None,
            deterministic_objective=self._deterministic,
        )
        gpr_t = gp.fit_kernel_params(  # model with all observations
            X=normalized_params,
            Y=standardized,
            is_categorical=search_space.is_categorical,
            log_prior=prior.default_log_prior,
            minimum_noise=prior.DEFAULT_MINIMUM_NOISE_VAR,
            gpr_cache=gpr_t1,
            deterministic_objective=self._deterministic,
        )

        i_star_t = int(np.argmax(standardized))
        i_star_t1 = int(np.argmax(standardized[:-1]))
        theta_t_star = normalized_params[i_star_t]
        theta_t1_star = normalized_params[i_star_t1]

        def posterior(x: np.ndarray, gpr: gp.GPRegressor) -> tuple[float, float]:
            mean, var = gpr.posterior(torch.from_numpy(x))
            return mean.item(), var.item()

        if i_star_t == i_star_t1:
            cov_stars = posterior(theta_t_star, gpr_t)[1]
        else:
            _, covar = gpr_t.posterior(
                torch.from_numpy(normalized_params[[i_star_t, i_star_t1]]), joint=True
            )
            cov_stars = covar[0, 1].item()

        mu_t1_last, var_t1_last = posterior(normalized_params[-1], gpr_t1)
        _, var_t_star_t1 = posterior(theta_t1_star, gpr_t)
        mu_t_star_t, var_t_star_t = posterior(theta_t_star, gpr_t)
        mu_t1_star_t1, _ = posterior(theta_t1_star, gpr_t1)
        y_t = float(standardized[-1])

        kappa_t1 = _compute_standardized_regret_bound(
            gpr_t1,
            search_space,
            normalized_params[:-1, :],
            standardized[:-1],
            self._delta,
            rng=self._rng.rng,
        )

        term1 = mu_t1_star_t1 - mu_t_star_t
        v = math.sqrt(max(1e-10, var_t_star_t - 2.0 * cov_stars + var_t_star_t1))
        g = (mu_t_star_t - mu_t1_star_t1) / v
        term2 = v * scipy_stats.norm.pdf(g)
        term3 = v * g * scipy_stats.norm.cdf(g)
        lam = prior.DEFAULT_MINIMUM_NOISE_VAR**-1
        kl1 = 0.5 * math.log(1.0 + lam * var_t1_last)
        kl2 = -0.5 * var_t1_last / (var_t1_last + lam**-1)
        kl3 = 0.5 * var_t1_last * (y_t - mu_t1_last) ** 2 / (var_t1_last + lam**-1) ** 2
        term4 = kappa_t1 * math.sqrt(0.5 * (kl1 + kl2 + kl3))
        return min(sys.float_info.max * 0.5, term1 + term2 + term3 + term4)
