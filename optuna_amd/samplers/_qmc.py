"""Quasi-Monte-Carlo sampler (Sobol / Halton).

A global per-(engine, search-space) sample counter lives in study system attrs,
keyed by a SHA-256 of the QMC configuration, so distributed workers draw from one
shared low-discrepancy sequence.

Parity: reference ``optuna/samplers/_qmc.py`` (QMCSampler :38, _find_sample_id
:329-347, fast_forward :320-326, categorical handling via pseudo-float axes).
"""
from __future__ import annotations

import hashlib
import json
import threading
from typing import TYPE_CHECKING, Any, Sequence

import numpy as np

from optuna_amd import logging as _logging
from optuna_amd._transform import _SearchSpaceTransform
from optuna_amd.distributions import (
    BaseDistribution,
    CategoricalDistribution,
    FloatDistribution,
)
from optuna_amd._experimental import experimental_class
from optuna_amd.samplers._base import (
    _INDEPENDENT_SAMPLING_WARNING_TEMPLATE,
    BaseSampler,
)
from optuna_amd.samplers._random import RandomSampler
from optuna_amd.trial import FrozenTrial, TrialState


if TYPE_CHECKING:
    from optuna_amd.study import Study

_logger = _logging.get_logger(__name__)

_SUGGESTED_STATES = (TrialState.COMPLETE, TrialState.PRUNED)
# scipy's Sobol engine shares internal direction-number state across instances.
_threading_lock = threading.Lock()


@experimental_class("3.0.0")
class QMCSampler(BaseSampler):
    """Low-discrepancy sequence sampler over a fixed (non-conditional) space."""

    def __init__(
        self,
        *,
        qmc_type: str = "sobol",
        scramble: bool = False,
        seed: int | None = None,
        independent_sampler: BaseSampler | None = None,
        warn_asynchronous_seeding: bool = True,
        warn_independent_sampling: bool = True,
    ) -> None:
        self._scramble = scramble
        self._seed = int(np.random.PCG64().random_raw()) if seed is None else seed
        self._independent_sampler = independent_sampler or RandomSampler(seed=seed)
        self._warn_independent_sampling = warn_independent_sampling
        if qmc_type not in ("halton", "sobol"):
            raise ValueError(
                f"The `qmc_type={qmc_type!r}` is invalid. Choose either `halton` or `sobol`."
            )
        self._qmc_type = qmc_type
        if seed is None and scramble and warn_asynchronous_seeding:
            self._log_asynchronous_seeding()

    def reseed_rng(self) -> None:
        self._independent_sampler.reseed_rng()

    def infer_relative_search_space(
        self, study: "Study", trial: FrozenTrial
    ) -> dict[str, BaseDistribution]:
        past_trials = study._get_trials(deepcopy=False, states=_SUGGESTED_STATES, use_cache=True)
        if len(past_trials) == 0:
            pending = study._get_trials(
                deepcopy=False, states=(TrialState.RUNNING,), use_cache=True
            )
            if len(pending) == 0:
                return {}
            union: dict[str, BaseDistribution] = {}
            intersection_keys: set[str] | None = None
            for t in pending:
                space = dict(t.distributions)
                union.update(space)
                if intersection_keys is None:
                    intersection_keys = set(space.keys())
                elif len(space.keys()):
                    intersection_keys &= space.keys()
            if intersection_keys is not None and intersection_keys != set(union.keys()):
                _logger.warning(
                    "`QMCSampler` assumes that the search space does not include any "
                    "conditions. Please make sure the provided search space is "
                    "non-conditional."
                )
            return union
        first_trial = min(past_trials, key=lambda t: t.number)
        return dict(first_trial.distributions)

    def sample_relative(
        self, study: "Study", trial: FrozenTrial, search_space: dict[str, BaseDistribution]
    ) -> dict[str, Any]:
        if search_space == {}:
            return {}
        categorical_space = {
            name: dist
            for name, dist in search_space.items()
            if isinstance(dist, CategoricalDistribution)
        }
        # One QMC axis in [0, C) per categorical; int() gives a uniform index
        # (never reaches C since engines emit [0, 1)).
        pseudo = {
            name: FloatDistribution(0, len(dist.choices))
            for name, dist in categorical_space.items()
        }
        trans = _SearchSpaceTransform({**search_space, **pseudo}, transform_0_1=True)
        sample = trans.untransform(self._sample_qmc(study, search_space)[0])
        return {
            name: (
                categorical_space[name].to_external_repr(int(value))
                if name in categorical_space
                else value
            )
            for name, value in sample.items()
        }

    @staticmethod
    def _log_asynchronous_seeding() -> None:
        _logger.warning(
            "No seed is provided for `QMCSampler` and the seed is set randomly. "
            "If you are running multiple `QMCSampler`s in parallel and/or distributed "
            "environment, the same seed must be used in all samplers to ensure that "
            "the low-discrepancy property holds across the union of their draws."
        )

    def _log_independent_sampling(self, trial: FrozenTrial, param_name: str) -> None:
        _logger.warning(
            _INDEPENDENT_SAMPLING_WARNING_TEMPLATE.format(
                param_name=param_name,
                trial_number=trial.number,
                sampler_name=self.__class__.__name__,
                fallback_name=self._independent_sampler.__class__.__name__,
                reason="dynamic search space is not supported by `QMCSampler`",
            )
        )

    def sample_independent(
        self,
        study: "Study",
        trial: FrozenTrial,
        param_name: str,
        param_distribution: BaseDistribution,
    ) -> Any:
        if len(study._get_trials(deepcopy=False, states=_SUGGESTED_STATES, use_cache=True)):
            if self._warn_independent_sampling:
                self._log_independent_sampling(trial, param_name)
        return self._independent_sampler.sample_independent(
            study, trial, param_name, param_distribution
        )

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

    def _sample_qmc(self, study: "Study", search_space: dict[str, BaseDistribution]) -> np.ndarray:
        from scipy.stats import qmc as qmc_module

        sample_id = self._find_sample_id(study, search_space)
        d = len(search_space)
        if self._qmc_type == "halton":
            engine = qmc_module.Halton(d, seed=self._seed, scramble=self._scramble)
        else:
            with _threading_lock:
                engine = qmc_module.Sobol(d, seed=self._seed, scramble=self._scramble)
        if sample_id > 0:
            engine.fast_forward(sample_id)
        return engine.random(1)

    def _find_sample_id(self, study: "Study", search_space: dict[str, BaseDistribution]) -> int:
        space_str = {name: str(dist) for name, dist in sorted(search_space.items())}
        qmc_vars: dict[str, Any] = {"qmc_type": self._qmc_type, "search_space": space_str}
        if self._scramble:
            qmc_vars.update(scramble=True, seed=self._seed)
        else:
            qmc_vars.update(scramble=False)
        key = "qmc:" + hashlib.sha256(json.dumps(qmc_vars).encode()).hexdigest()
        # Best effort atomicity: every sample_id is drawn at least once.
        sample_id = study._storage.get_study_system_attrs(study._study_id).get(key, -1) + 1
        study._storage.set_study_system_attr(study._study_id, key, sample_id)
        return sample_id
