"""CmaEsSampler: generation-based ask/tell over the native CMA-ES core.

Driver protocol parity with reference ``optuna/samplers/_cmaes.py``
(sample_relative :390-465, _init_optimizer :507-600, state pickled → hex →
2045-char system-attr chunks ``{prefix}optimizer:{i}``, per-trial
``{prefix}generation`` tag, per-trial RNG reseed ``randint(1,2^16)+number``):
the optimizer state travels through storage so any distributed worker can resume
the strategy; stale generations are tolerated.

Unlike the reference, the CMA/SepCMA/CMAwM update equations are implemented
natively (``_core.py``) rather than via the external ``cmaes`` package.
``with_margin=True`` uses the native CMAwM (Hamano et al. margin correction)
with the reference's ask/tell protocol: raw samples stored per trial under the
``x_for_tell`` system attr (reference _cmaes.py:450-456) and replayed at tell.
"""
from __future__ import annotations

import copy
import pickle
from typing import TYPE_CHECKING, Any, Sequence

import numpy as np

from optuna_amd import logging as _logging
from optuna_amd._transform import _SearchSpaceTransform
from optuna_amd.distributions import (
    BaseDistribution,
    FloatDistribution,
    IntDistribution,
)
from optuna_amd.samplers._base import BaseSampler
from optuna_amd.samplers._cmaes._core import CMA, CMAwM, SepCMA, get_warm_start_mgd
from optuna_amd.samplers._lazy_random_state import LazyRandomState
from optuna_amd.samplers._random import RandomSampler
from optuna_amd.search_space import IntersectionSearchSpace
from optuna_amd.study._study_direction import StudyDirection
from optuna_amd.trial import FrozenTrial, TrialState


if TYPE_CHECKING:
    from optuna_amd.study import Study

_logger = _logging.get_logger(__name__)

# RDB system-attr values are capped below 2046 chars; state is chunked.
_SYSTEM_ATTR_MAX_LENGTH = 2045


class CmaEsSampler(BaseSampler):
    """CMA-ES sampler (native MI355X build; see module docstring)."""

    def __init__(
        self,
        x0: dict[str, Any] | None = None,
        sigma0: float | None = None,
        n_startup_trials: int = 1,
        independent_sampler: BaseSampler | None = None,
        warn_independent_sampling: bool = True,
        seed: int | None = None,
        *,
        consider_pruned_trials: bool = False,
        popsize: int | None = None,
        use_separable_cma: bool = False,
        with_margin: bool = False,
        lr_adapt: bool = False,
        source_trials: list[FrozenTrial] | None = None,
    ) -> None:
        self._x0 = x0
        self._sigma0 = sigma0
        self._independent_sampler = independent_sampler or RandomSampler(seed=seed)
        self._n_startup_trials = n_startup_trials
        self._warn_independent_sampling = warn_independent_sampling
        self._cma_rng = LazyRandomState(seed)
        self._search_space = IntersectionSearchSpace()
        self._consider_pruned_trials = consider_pruned_trials
        self._popsize = popsize
        self._use_separable_cma = use_separable_cma
        self._with_margin = with_margin
        self._lr_adapt = lr_adapt
        self._source_trials = source_trials

        if use_separable_cma:
            self._attr_prefix = "sepcma:"
        elif with_margin:
            self._attr_prefix = "cmawm:"
        else:
            self._attr_prefix = "cma:"

        from optuna_amd._experimental import warn_experimental_argument

        if consider_pruned_trials:
            warn_experimental_argument("consider_pruned_trials")
        if use_separable_cma:
            warn_experimental_argument("use_separable_cma")
        if source_trials is not None:
            warn_experimental_argument("source_trials")
        if with_margin:
            warn_experimental_argument("with_margin")
        if lr_adapt:
            warn_experimental_argument("lr_adapt")

        if source_trials is not None and (x0 is not None or sigma0 is not None):
            raise ValueError(
                "It is prohibited to pass `source_trials` argument when x0 or sigma0 is "
                "specified."
            )
        if source_trials is not None and use_separable_cma:
            raise ValueError(
                "It is prohibited to pass `source_trials` argument when using separable "
                "CMA-ES."
            )
        if lr_adapt and (use_separable_cma or with_margin):
            raise ValueError(
                "It is prohibited to pass `use_separable_cma` or `with_margin` argument "
                "when using `lr_adapt`."
            )
        if use_separable_cma and with_margin:
            raise ValueError(
                "Currently, we do not support `use_separable_cma=True` and "
                "`with_margin=True`."
            )

    def reseed_rng(self) -> None:
        # The CMA RNG is reseeded per trial inside sample_relative.
        self._independent_sampler.reseed_rng()

    # ---- search space ---------------------------------------------------------------

    def infer_relative_search_space(
        self, study: "Study", trial: FrozenTrial
    ) -> dict[str, BaseDistribution]:
        search_space: dict[str, BaseDistribution] = {}
        for name, distribution in self._search_space.calculate(study).items():
            if distribution.single():
                continue
            if not isinstance(distribution, (FloatDistribution, IntDistribution)):
                continue  # categorical is unsupported by CMA-ES
            search_space[name] = distribution
        return search_space

    # ---- relative sampling ----------------------------------------------------------

    def sample_relative(
        self, study: "Study", trial: FrozenTrial, search_space: dict[str, BaseDistribution]
    ) -> dict[str, Any]:
        self._raise_error_if_multi_objective(study)
        if len(search_space) == 0:
            return {}

        completed_trials = self._get_trials(study)
        if len(completed_trials) < self._n_startup_trials:
            return {}

        # With margin handling the discrete grid survives the transform (steps
        # are passed to CMAwM instead); reference _cmaes.py:405-408.
        trans = _SearchSpaceTransform(
            search_space, transform_step=not self._with_margin, transform_0_1=True
        )

        optimizer = self._restore_optimizer(completed_trials)
        if optimizer is None:
            optimizer = self._init_optimizer(trans, study.direction)

        if optimizer.dim != len(trans.bounds):
            if self._warn_independent_sampling:
                _logger.warning(
                    "`CmaEsSampler` does not support dynamic search space. "
                    f"`{self._independent_sampler.__class__.__name__}` is used instead."
                )
                self._warn_independent_sampling = False
            return {}

        # Complete a generation when population_size members of it finished.
        solution_trials = [
            t
            for t in completed_trials
            if t.system_attrs.get(self._attr_key_generation, -1) == optimizer.generation
        ]
        if len(solution_trials) >= optimizer.population_size:
            sign = 1.0 if study.direction == StudyDirection.MINIMIZE else -1.0
            solutions = []
            for t in solution_trials[: optimizer.population_size]:
                assert t.value is not None, "completed trials must have a value"
                if isinstance(optimizer, CMAwM):
                    # The raw (pre-discretization) sample drives the update
                    # (reference _cmaes.py:432-433).
                    x = np.array(t.system_attrs["x_for_tell"])
                else:
                    x = trans.transform(t.params)
                solutions.append((x, sign * t.value))
            optimizer.tell(solutions)
            self._store_optimizer(study, trial, optimizer)

        # Per-trial reseed keeps parallel asks decorrelated yet reproducible.
        seed = self._cma_rng.rng.randint(1, 2**16) + trial.number
        optimizer._rng.seed(seed)
        if isinstance(optimizer, CMAwM):
            params, x_for_tell = optimizer.ask()
            study._storage.set_trial_system_attr(
                trial._trial_id, "x_for_tell", x_for_tell.tolist()
            )
        else:
            params = optimizer.ask()

        study._storage.set_trial_system_attr(
            trial._trial_id, self._attr_key_generation, optimizer.generation
        )
        return trans.untransform(params)

    # ---- optimizer state via system attrs -------------------------------------------

    @property
    def _attr_key_generation(self) -> str:
        return self._attr_prefix + "generation"

    @property
    def _attr_key_optimizer(self) -> str:
        return self._attr_prefix + "optimizer"

    def _store_optimizer(self, study: "Study", trial: FrozenTrial, optimizer: CMA) -> None:
        optimizer_str = pickle.dumps(optimizer).hex()
        for i in range(0, len(optimizer_str), _SYSTEM_ATTR_MAX_LENGTH):
            study._storage.set_trial_system_attr(
                trial._trial_id,
                f"{self._attr_key_optimizer}:{i // _SYSTEM_ATTR_MAX_LENGTH}",
                optimizer_str[i : i + _SYSTEM_ATTR_MAX_LENGTH],
            )

    _restore_cache: tuple[tuple[int, int], CMA] | None = None

    def _restore_optimizer(self, completed_trials: list[FrozenTrial]) -> CMA | None:
        # Scan backwards: the newest stored state wins. The ~160 KB hex pickle
        # only changes at generation boundaries, so the decoded optimizer is
        # cached keyed by (source trial id, payload length) — every other ask
        # skips hex-decode + unpickle (milliseconds per suggest at 100 dims).
        for trial in reversed(completed_trials):
            chunks = [
                (key, value)
                for key, value in trial.system_attrs.items()
                if key.startswith(self._attr_key_optimizer)
            ]
            if len(chunks) == 0:
                continue
            chunks.sort(key=lambda kv: int(kv[0].rsplit(":", 1)[1]))
            optimizer_str = "".join(v for _, v in chunks)
            cache_key = (trial._trial_id, len(optimizer_str))
            cached = self._restore_cache
            if cached is not None and cached[0] == cache_key:
                return copy.deepcopy(cached[1])
            optimizer = pickle.loads(bytes.fromhex(optimizer_str))
            self._restore_cache = (cache_key, copy.deepcopy(optimizer))
            return optimizer
        return None

    # ---- initialization -------------------------------------------------------------

    def _init_optimizer(
        self, trans: _SearchSpaceTransform, direction: StudyDirection
    ) -> CMA:
        lower_bounds = trans.bounds[:, 0]
        upper_bounds = trans.bounds[:, 1]
        n_dimension = len(trans.bounds)

        if self._source_trials is not None:
            # Warm start from a source task (single-objective, minimize-normalized).
            sign = 1.0 if direction == StudyDirection.MINIMIZE else -1.0
            source_solutions = [
                (trans.transform(t.params), sign * t.value)
                for t in self._source_trials
                if t.state == TrialState.COMPLETE and t.value is not None
            ]
            if len(source_solutions) == 0:
                raise ValueError("No complete trials in `source_trials`.")
            mean, sigma0, cov = get_warm_start_mgd(source_solutions)
            return CMA(
                mean=mean,
                sigma=sigma0,
                cov=cov,
                bounds=trans.bounds,
                seed=self._cma_rng.rng.randint(1, 2**31 - 2),
                n_max_resampling=10 * n_dimension,
                population_size=self._popsize,
                lr_adapt=self._lr_adapt,
            )

        if self._x0 is None:
            mean = lower_bounds + (upper_bounds - lower_bounds) / 2
        else:
            mean = trans.transform(self._x0)

        if self._sigma0 is None:
            sigma0 = float(np.min((upper_bounds - lower_bounds) / 6))
        else:
            sigma0 = self._sigma0
        sigma0 = max(sigma0, 1e-10)

        if self._use_separable_cma:
            return SepCMA(
                mean=mean,
                sigma=sigma0,
                bounds=trans.bounds,
                seed=self._cma_rng.rng.randint(1, 2**31 - 2),
                n_max_resampling=10 * n_dimension,
                population_size=self._popsize,
            )
        if self._with_margin:
            # Normalized steps in the 0-1 transform space; 0.0 marks continuous
            # dims (reference _cmaes.py:568-579).
            steps = np.empty(len(trans._search_space), dtype=float)
            for i, dist in enumerate(trans._search_space.values()):
                assert isinstance(dist, (FloatDistribution, IntDistribution))
                if dist.step is None or dist.log:
                    steps[i] = 0.0
                elif dist.low == dist.high:
                    steps[i] = 1.0
                else:
                    steps[i] = dist.step / (dist.high - dist.low)
            return CMAwM(
                mean=mean,
                sigma=sigma0,
                bounds=trans.bounds,
                steps=steps,
                seed=self._cma_rng.rng.randint(1, 2**31 - 2),
                n_max_resampling=10 * n_dimension,
                population_size=self._popsize,
            )
        return CMA(
            mean=mean,
            sigma=sigma0,
            bounds=trans.bounds,
            seed=self._cma_rng.rng.randint(1, 2**31 - 2),
            n_max_resampling=10 * n_dimension,
            population_size=self._popsize,
            lr_adapt=self._lr_adapt,
        )

    # ---- independent fallback / bookkeeping ------------------------------------------

    def sample_independent(
        self,
        study: "Study",
        trial: FrozenTrial,
        param_name: str,
        param_distribution: BaseDistribution,
    ) -> Any:
        self._raise_error_if_multi_objective(study)
        if self._warn_independent_sampling:
            complete_trials = self._get_trials(study)
            if len(complete_trials) >= self._n_startup_trials:
                _logger.warning(
                    f"The parameter '{param_name}' in trial#{trial.number} is sampled "
                    "independently by using `RandomSampler` instead of `CmaEsSampler` "
                    "(e.g. dynamic search space or categorical distribution)."
                )
        return self._independent_sampler.sample_independent(
            study, trial, param_name, param_distribution
        )

    def _get_trials(self, study: "Study") -> list[FrozenTrial]:
        complete_trials = []
        for t in study._get_trials(deepcopy=False, use_cache=True):
            if t.state == TrialState.COMPLETE:
                complete_trials.append(t)
            elif (
                t.state == TrialState.PRUNED
                and self._consider_pruned_trials
                and len(t.intermediate_values) > 0
            ):
                # Use the last intermediate value as the objective value.
                _, value = max(t.intermediate_values.items())
                copied = t
                import copy as _copy

                copied = _copy.copy(t)
                copied.value = value
                complete_trials.append(copied)
        return complete_trials

    def before_trial(self, study: "Study", trial: FrozenTrial) -> None:
        self._independent_sampler.before_trial(study, trial)

    def after_trial(
        self,
        study: "Study",
        trial: FrozenTrial,
        state: TrialState,
        values: Sequence[float] | None,
    ) -> None:
        self._independent_sampler.after_trial(study, trial, state, values)
