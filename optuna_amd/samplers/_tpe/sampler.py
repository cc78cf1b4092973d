"""TPESampler — the default sampler and the framework's headline compute path.

Algorithm (parity with reference ``optuna/samplers/_tpe/sampler.py`` :87-880):
split history into below/above by ``gamma(n)``; fit Parzen KDEs to each; draw
``n_ei_candidates`` samples from the below-KDE; rank by
``log l(x) − log g(x)``; return the argmax. Multivariate by default for
single-objective studies; ``group=True`` decomposes conditional spaces;
``constant_liar`` treats RUNNING trials as bad and shares relative params via
chunked system attrs (2045-char chunks, RDB-compatible).

On a GPU box the KDE fit and the S×K×D log-pdf/EI reduction run in the
``optuna_amd._hip`` K1/K2 kernels over the SoA estimator arrays; the numpy path
in ``parzen.py`` is the host fallback and golden reference.
"""
from __future__ import annotations

import json
import math
from typing import TYPE_CHECKING, Any, Callable, Sequence

import numpy as np

from optuna_amd import logging as _logging
from optuna_amd.distributions import BaseDistribution
from optuna_amd.samplers._base import (
    _INDEPENDENT_SAMPLING_WARNING_TEMPLATE,
    BaseSampler,
    _process_constraints_after_trial,
)
from optuna_amd.samplers._lazy_random_state import LazyRandomState
from optuna_amd.samplers._random import RandomSampler
from optuna_amd.samplers._tpe.parzen import _ParzenEstimator, _ParzenEstimatorParameters
from optuna_amd.search_space import IntersectionSearchSpace
from optuna_amd.search_space.group_decomposed import (
    _GroupDecomposedSearchSpace,
    _SearchSpaceGroup,
)
from optuna_amd.study._multi_objective import _fast_non_domination_rank
from optuna_amd.study._study_direction import StudyDirection
from optuna_amd.trial import FrozenTrial, TrialState


if TYPE_CHECKING:
    from optuna_amd.study import Study

EPS = 1e-12
_logger = _logging.get_logger(__name__)

# Sentinel: the batched independent-mode path does not apply; sample per-dim.
_BATCH_MISS = object()

_RELATIVE_PARAMS_KEY = "tpe:relative_params"
# RDB system-attr values must stay under 2046 chars; long payloads are chunked.
_SYSTEM_ATTR_MAX_LENGTH = 2045


def default_gamma(x: int) -> int:
    return min(math.ceil(0.1 * x), 25)


def default_gamma_multiobjective(x: int) -> int:
    return math.ceil(0.1 * x)


def hyperopt_default_gamma(x: int) -> int:
    return min(math.ceil(0.25 * math.sqrt(x)), 25)


def default_weights(x: int) -> np.ndarray:
    """Old trials ramp linearly; the most recent 25 have full weight."""
    if x == 0:
        return np.asarray([])
    if x < 25:
        return np.ones(x)
    ramp = np.linspace(1.0 / x, 1.0, num=x - 25)
    flat = np.ones(25)
    return np.concatenate([ramp, flat], axis=0)


def _count_finished(study: "Study", states: tuple[TrialState, ...]) -> int:
    """Finished-trial count for the startup/refresh gates.

    Normally the storage's O(1) counter. When the trial listing has been
    replaced on the storage INSTANCE (test instrumentation, wrappers), counts
    must come from that same listing or the gates would disagree with the
    data the sampler is about to read.
    """
    storage = study._storage
    if "get_all_trials" in vars(storage):
        return len(storage.get_all_trials(study._study_id, deepcopy=False, states=states))
    return storage.get_n_trials(study._study_id, states)


class TPESampler(BaseSampler):
    """Tree-structured Parzen Estimator sampler (see module docstring)."""

    def __init__(
        self,
        consider_prior: bool = True,
        prior_weight: float = 1.0,
        consider_magic_clip: bool = True,
        consider_endpoints: bool = False,
        n_startup_trials: int = 10,
        n_ei_candidates: int = 24,
        gamma: Callable[[int], int] | None = None,
        weights: Callable[[int], np.ndarray] = default_weights,
        seed: int | None = None,
        *,
        multivariate: bool | None = None,
        group: bool = False,
        warn_independent_sampling: bool = True,
        constant_liar: bool = False,
        constraints_func: Callable[[FrozenTrial], Sequence[float]] | None = None,
    ) -> None:
        self._parzen_estimator_parameters = _ParzenEstimatorParameters(
            prior_weight=prior_weight,
            consider_magic_clip=consider_magic_clip,
            consider_endpoints=consider_endpoints,
            weights=weights,
            multivariate=multivariate if multivariate is not None else False,
            consider_prior=consider_prior,
        )
        self._n_startup_trials = n_startup_trials
        self._n_ei_candidates = n_ei_candidates
        self._gamma = gamma
        self._warn_independent_sampling = warn_independent_sampling
        self._rng = LazyRandomState(seed)
        self._random_sampler = RandomSampler(seed=seed)
        self._multivariate = multivariate
        if group:
            from optuna_amd._experimental import warn_experimental_argument

            warn_experimental_argument("group")
        self._group = group
        self._group_decomposed_search_space: _GroupDecomposedSearchSpace | None = None
        self._search_space_group: _SearchSpaceGroup | None = None
        self._search_space = IntersectionSearchSpace(include_pruned=True)
        self._constant_liar = constant_liar
        if constraints_func is not None:
            import warnings

            warnings.warn(
                "`constraints_func` is deprecated; set constraints via "
                "Trial.set_constraint instead.",
                FutureWarning,
            )
        self._constraints_func = constraints_func
        # Overridable for customization (reference sampler.py keeps the same hook).
        self._parzen_estimator_cls = _ParzenEstimator
        # Per-study incremental history mirrors (see _history.py).
        self._histories: dict[int, Any] = {}
        # Per-trial batched prefetch for independent-mode sampling (the
        # multi-objective default): ONE device round scores all dims at once;
        # later per-dim requests of the same trial hit this cache. Keyed by
        # trial id (bounded) so n_jobs>1 threads interleaving trials don't
        # evict each other's batch.
        self._indep_prefetch: dict[int, dict[str, tuple[Any, Any]]] = {}
        # n_jobs>1 runs suggests from worker threads; the history mirror is
        # stateful (capacity buffers, incrementally sorted indices), so its
        # read-modify-write cycle must be serialized. The reference's samplers
        # are effectively GIL-serialized because they rebuild state per call.
        import threading

        self._history_lock = threading.RLock()

        if group:
            if multivariate is False:
                raise ValueError(
                    "``group`` option can only be enabled when ``multivariate`` is enabled."
                )
            self._group_decomposed_search_space = _GroupDecomposedSearchSpace(True)

    def __getstate__(self) -> dict:
        # Locks and device-resident history caches don't pickle; both are
        # rebuilt lazily on first use after unpickling.
        state = self.__dict__.copy()
        del state["_history_lock"]
        state["_indep_prefetch"] = {}
        state["_histories"] = {}
        return state

    def __setstate__(self, state: dict) -> None:
        import threading

        self.__dict__.update(state)
        self._history_lock = threading.RLock()

    def reseed_rng(self) -> None:
        self._rng.rng.seed()
        self._random_sampler.reseed_rng()

    def _is_multivariate(self, study: "Study") -> bool:
        if self._multivariate is not None:
            return self._multivariate
        if self._group:
            if study._is_multi_objective():
                import warnings

                # group=True forces the multivariate path even though the
                # multi-objective default would be independent sampling.
                warnings.warn(
                    "`group=True` overrides the multi-objective default "
                    "`multivariate=False`; the multivariate TPE is used."
                )
            return True
        # Multivariate for single-objective, independent for multi-objective.
        return not study._is_multi_objective()

    # ---- search-space inference -----------------------------------------------------

    def infer_relative_search_space(
        self, study: "Study", trial: FrozenTrial
    ) -> dict[str, BaseDistribution]:
        if not self._is_multivariate(study):
            return {}

        search_space: dict[str, BaseDistribution] = {}
        if self._group:
            assert self._group_decomposed_search_space is not None
            self._search_space_group = self._group_decomposed_search_space.calculate(study)
            for sub_space in self._search_space_group.search_spaces:
                for name, distribution in sorted(sub_space.items()):
                    if distribution.single():
                        continue
                    search_space[name] = distribution
            return search_space

        for name, distribution in self._search_space.calculate(study).items():
            if distribution.single():
                continue
            search_space[name] = distribution
        return search_space

    # ---- relative sampling ----------------------------------------------------------

    def sample_relative(
        self, study: "Study", trial: FrozenTrial, search_space: dict[str, BaseDistribution]
    ) -> dict[str, Any]:
        if self._group:
            assert self._search_space_group is not None
            params: dict[str, Any] = {}
            for sub_space in self._search_space_group.search_spaces:
                _search_space = {}
                for name, distribution in sorted(sub_space.items()):
                    if distribution.single():
                        continue
                    if name not in search_space:
                        # PartialFixedSampler may shrink the space below inference.
                        continue
                    _search_space[name] = distribution
                params.update(self._sample_relative(study, trial, _search_space))
        else:
            params = self._sample_relative(study, trial, search_space)

        if params != {} and self._constant_liar:
            # Publish the sampled relative params so other workers' liar splits see
            # this RUNNING trial's position in the space.
            params_str = json.dumps(params)
            for i in range(0, len(params_str), _SYSTEM_ATTR_MAX_LENGTH):
                study._storage.set_trial_system_attr(
                    trial._trial_id,
                    f"{_RELATIVE_PARAMS_KEY}:{i // _SYSTEM_ATTR_MAX_LENGTH}",
                    params_str[i : i + _SYSTEM_ATTR_MAX_LENGTH],
                )
        return params

    def _sample_relative(
        self, study: "Study", trial: FrozenTrial, search_space: dict[str, BaseDistribution]
    ) -> dict[str, Any]:
        if search_space == {}:
            return {}
        states = (TrialState.COMPLETE, TrialState.PRUNED)
        n_finished = _count_finished(study, states)
        if n_finished < self._n_startup_trials:
            return {}
        return self._sample(study, trial, search_space)

    def sample_independent(
        self,
        study: "Study",
        trial: FrozenTrial,
        param_name: str,
        param_distribution: BaseDistribution,
    ) -> Any:
        states = (TrialState.COMPLETE, TrialState.PRUNED)
        # O(1) startup check: a full trial fetch here would run once per DIM
        # per suggest (this is the per-dim entry point).
        n_finished = _count_finished(study, states)
        if n_finished < self._n_startup_trials:
            return self._random_sampler.sample_independent(
                study, trial, param_name, param_distribution
            )

        if self._warn_independent_sampling and self._is_multivariate(study):
            trials = study._get_trials(deepcopy=False, states=states, use_cache=True)
            if any(param_name in t.params for t in trials):
                _logger.warning(
                    _INDEPENDENT_SAMPLING_WARNING_TEMPLATE.format(
                        param_name=param_name,
                        trial_number=trial.number,
                        sampler_name=self.__class__.__name__,
                        fallback_name=self._random_sampler.__class__.__name__,
                        reason=(
                            "multivariate=True,group=False does not support dynamic "
                            "search spaces (multivariate=True,group=True does)"
                        ),
                    )
                )

        batched = self._sample_independent_batched(
            study, trial, param_name, param_distribution
        )
        if batched is not _BATCH_MISS:
            return batched
        return self._sample(study, trial, {param_name: param_distribution})[param_name]

    def _sample_independent_batched(
        self,
        study: "Study",
        trial: FrozenTrial,
        param_name: str,
        param_distribution: BaseDistribution,
    ) -> Any:
        """Serve independent-mode requests from a per-trial all-dims batch.

        Independent mode asks one dim at a time, but each ask re-reads the
        history, rebuilds estimators and round-trips the device — D times per
        trial. When the history's intersection space contains the requested
        dim (same distribution) and every finished trial defines every dim of
        that space (so the per-dim observation subsets coincide with the joint
        subsets), all dims are sampled in one per-dim-batched pass and cached
        for the trial. Falls back per-dim otherwise (_BATCH_MISS).
        """
        with self._history_lock:
            pf = self._indep_prefetch.get(trial._trial_id)
            if pf is not None:
                hit = pf.get(param_name)
                if hit is not None and hit[1] == param_distribution:
                    return hit[0]
            space = {
                name: dist
                for name, dist in self._search_space.calculate(study).items()
                if not dist.single()
            }
            if (
                len(space) <= 1
                or param_name not in space
                or space[param_name] != param_distribution
            ):
                return _BATCH_MISS
            ret = self._sample_locked(study, trial, space, per_dim=True)
            if ret is None:
                return _BATCH_MISS
            if len(self._indep_prefetch) >= 64:
                self._indep_prefetch.pop(next(iter(self._indep_prefetch)))
            self._indep_prefetch[trial._trial_id] = {
                n: (v, space[n]) for n, v in ret.items()
            }
            return ret[param_name]

    # ---- the compute core -----------------------------------------------------------

    def _get_params(self, trial: FrozenTrial, study: "Study") -> dict[str, Any]:
        """Params incl. constant-liar shared relative params of RUNNING trials."""
        if trial.state.is_finished() or not self._is_multivariate(study):
            return trial.params
        params_strs = []
        i = 0
        while params_str_i := trial.system_attrs.get(f"{_RELATIVE_PARAMS_KEY}:{i}"):
            params_strs.append(params_str_i)
            i += 1
        if len(params_strs) == 0:
            return trial.params
        try:
            params = json.loads("".join(params_strs))
        except json.JSONDecodeError:
            # Concurrent chunk writes can race; fall back to committed params.
            return trial.params
        params.update(trial.params)
        return params

    def _get_internal_repr(
        self,
        trials: list[FrozenTrial],
        search_space: dict[str, BaseDistribution],
        study: "Study",
    ) -> dict[str, np.ndarray]:
        values: dict[str, list[float]] = {name: [] for name in search_space}
        for trial in trials:
            params = self._get_params(trial, study)
            if search_space.keys() <= params.keys():
                for name, distribution in search_space.items():
                    values[name].append(distribution.to_internal_repr(params[name]))
        return {k: np.asarray(v) for k, v in values.items()}

    def _sample(
        self, study: "Study", trial: FrozenTrial, search_space: dict[str, BaseDistribution]
    ) -> dict[str, Any]:
        with self._history_lock:
            return self._sample_locked(study, trial, search_space)

    def _sample_locked(
        self,
        study: "Study",
        trial: FrozenTrial,
        search_space: dict[str, BaseDistribution],
        per_dim: bool = False,
    ) -> dict[str, Any] | None:
        from optuna_amd.samplers._tpe._history import _TpeHistory

        history = self._histories.get(study._study_id)
        if history is None:
            history = self._histories[study._study_id] = _TpeHistory()
        # Finished trials only accumulate; an O(1) count check skips the full
        # storage read on suggests that arrive between tells.
        n_finished = _count_finished(study, (TrialState.COMPLETE, TrialState.PRUNED))
        if n_finished != len(history):
            # Same coherence rule as _count_finished: an instance-patched
            # listing (test instrumentation) must also feed the mirror, so the
            # storage-native delta API is bypassed in that case.
            delta_read = (
                None
                if "get_all_trials" in vars(study._storage)
                else getattr(study._storage, "get_finished_trials_since", None)
            )
            if delta_read is not None:
                history.update(
                    delta_read(study._study_id, len(history)),
                    len(study.directions),
                    delta=True,
                )
            else:
                finished = study._get_trials(
                    deepcopy=False,
                    states=(TrialState.COMPLETE, TrialState.PRUNED),
                    use_cache=True,
                )
                history.update(finished, len(study.directions))

        if self._gamma is None:
            self._gamma = (
                default_gamma if len(study.directions) <= 1 else default_gamma_multiobjective
            )
        n = len(history)
        below_rows, above_rows = history.split(study, self._gamma(n))

        from optuna_amd.samplers._tpe import _device

        above_sel = history.valid_rows(search_space, above_rows)
        if per_dim:
            # The batch is only equivalent to D independent 1-dim suggests
            # when every trial of both subsets defines every dim — then the
            # per-dim observation sets coincide with the joint ones.
            below_sel_chk = history.valid_rows(search_space, below_rows)
            if len(below_sel_chk) != len(below_rows) or len(above_sel) != len(
                above_rows
            ):
                return None
        obs_below, orders_below = history.observations(search_space, below_rows)

        # Constant-liar rows: other workers' RUNNING trials join the "above"
        # set (params shared via system attrs). As an (L, D) matrix they ride
        # the device path too — merged into the resident-table subsets by the
        # k_merge_extras kernel, so multi-GPU ranks keep single-rank speed.
        liar_extras: np.ndarray | None = None
        if self._constant_liar:
            running = [
                t
                for t in study._get_trials(
                    deepcopy=False, states=(TrialState.RUNNING,), use_cache=False
                )
                if t.number != trial.number
            ]
            if running:
                rows_list = []
                for t in running:
                    params = self._get_params(t, study)
                    if search_space.keys() <= params.keys():
                        rows_list.append(
                            [
                                dist.to_internal_repr(params[name])
                                for name, dist in search_space.items()
                            ]
                        )
                if rows_list:
                    liar_extras = np.asarray(rows_list, dtype=np.float64)
        n_liar = 0 if liar_extras is None else len(liar_extras)

        use_device = (
            self._parzen_estimator_cls is _ParzenEstimator
            and _device.space_is_device_eligible(search_space)
            and _device.device_ready(len(above_sel) + n_liar + 1)
        )

        obs_above: dict[str, np.ndarray] = {}
        orders_above: dict[str, np.ndarray] | None = None
        if not use_device:
            obs_above, orders_above = history.observations(search_space, above_rows)
            if n_liar:
                assert liar_extras is not None
                obs_above = {
                    name: np.concatenate([obs_above[name], liar_extras[:, c]])
                    for c, name in enumerate(search_space)
                }
                orders_above = None  # appended rows invalidate presorted orders

        mpe_below = self._build_mpe(
            study, search_space, obs_below, handle_below=True, orders=orders_below
        )
        if per_dim:
            samples_below = mpe_below.sample_per_dim(self._rng.rng, self._n_ei_candidates)
        else:
            samples_below = mpe_below.sample(self._rng.rng, self._n_ei_candidates)

        # Device path (K1+K2): the big "above" KDE is fit and scored against the
        # HBM-resident parameter table for every distribution type (continuous,
        # discrete cells, categorical); the small "below" estimator stays on
        # host (it also drives candidate sampling). The host path remains for
        # small histories and custom estimator classes.
        if use_device:
            weights_above = self._above_weights(len(above_sel) + n_liar)
            log_g = _device.score_above_resident(
                history.space_cache(search_space),
                above_sel,
                weights_above,
                samples_below,
                self._parzen_estimator_parameters.consider_endpoints,
                self._parzen_estimator_parameters.consider_magic_clip,
                prior_weight=self._parzen_estimator_parameters.prior_weight,
                extras=liar_extras,
                per_dim=per_dim,
            )
            if per_dim:
                # The MO below set is gamma(n) ~ 0.1 n kernels (uncapped) — at
                # thousands of kernels the host per-dim logsumexp costs ~1 ms;
                # score it against the resident table like the above set.
                if len(below_sel_chk) >= 128:
                    log_l = _device.score_above_resident(
                        history.space_cache(search_space),
                        below_sel_chk,
                        mpe_below.weights,
                        samples_below,
                        self._parzen_estimator_parameters.consider_endpoints,
                        self._parzen_estimator_parameters.consider_magic_clip,
                        prior_weight=self._parzen_estimator_parameters.prior_weight,
                        per_dim=True,
                    )
                else:
                    log_l = mpe_below.log_pdf_per_dim(samples_below)
                acq_func_vals = log_l - log_g
            else:
                acq_func_vals = mpe_below.log_pdf(samples_below) - log_g
        else:
            mpe_above = self._build_mpe(
                study, search_space, obs_above, handle_below=False, orders=orders_above
            )
            if per_dim:
                acq_func_vals = mpe_below.log_pdf_per_dim(
                    samples_below
                ) - mpe_above.log_pdf_per_dim(samples_below)
            else:
                acq_func_vals = self._compute_acquisition_func(
                    samples_below, mpe_below, mpe_above
                )
        if per_dim:
            # Per-dim argmax: each dim independently keeps its own best of the
            # S candidates — exactly what D separate 1-dim suggests would do.
            ret = {}
            for c, (param_name, dist) in enumerate(search_space.items()):
                best = int(np.argmax(acq_func_vals[:, c]))
                ret[param_name] = dist.to_external_repr(
                    samples_below[param_name][best].item()
                )
            return ret
        ret = TPESampler._compare(samples_below, acq_func_vals)

        for param_name, dist in search_space.items():
            ret[param_name] = dist.to_external_repr(ret[param_name])
        return ret

    def _above_weights(self, n: int) -> np.ndarray:
        """Mixture weights of an n-observation estimator (+prior), normalized."""
        p = self._parzen_estimator_parameters
        w = _ParzenEstimator._call_weights_func(p.weights, n)
        if n == 0:
            w = np.array([1.0])
        else:
            w = np.append(w, [p.prior_weight])
        return w / w.sum()

    def _build_mpe(
        self,
        study: "Study",
        search_space: dict[str, BaseDistribution],
        observations: dict[str, np.ndarray],
        handle_below: bool,
        orders: dict[str, np.ndarray] | None = None,
    ) -> _ParzenEstimator:
        predetermined = None
        if handle_below and study._is_multi_objective():
            n_below = len(next(iter(observations.values()))) if observations else 0
            predetermined = np.ones(n_below)
        if self._parzen_estimator_cls is _ParzenEstimator:
            mpe = _ParzenEstimator(
                observations,
                search_space,
                self._parzen_estimator_parameters,
                predetermined,
                sorted_orders=orders,
            )
        else:
            # Custom estimator subclasses keep the reference 4-arg signature.
            mpe = self._parzen_estimator_cls(
                observations, search_space, self._parzen_estimator_parameters, predetermined
            )
        if not isinstance(mpe, _ParzenEstimator):
            raise RuntimeError("_parzen_estimator_cls must override _ParzenEstimator.")
        return mpe

    def _build_parzen_estimator(
        self,
        study: "Study",
        search_space: dict[str, BaseDistribution],
        trials: list[FrozenTrial],
        handle_below: bool,
    ) -> _ParzenEstimator:
        observations = self._get_internal_repr(trials, search_space, study)
        if handle_below and study._is_multi_objective():
            n_below = len(next(iter(observations.values()))) if observations else 0
            weights_below = np.ones(n_below)
            mpe = self._parzen_estimator_cls(
                observations, search_space, self._parzen_estimator_parameters, weights_below
            )
        else:
            mpe = self._parzen_estimator_cls(
                observations, search_space, self._parzen_estimator_parameters
            )
        if not isinstance(mpe, _ParzenEstimator):
            raise RuntimeError("_parzen_estimator_cls must override _ParzenEstimator.")
        return mpe

    def _compute_acquisition_func(
        self,
        samples: dict[str, np.ndarray],
        mpe_below: _ParzenEstimator,
        mpe_above: _ParzenEstimator,
    ) -> np.ndarray:
        return mpe_below.log_pdf(samples) - mpe_above.log_pdf(samples)

    @classmethod
    def _compare(
        cls, samples: dict[str, np.ndarray], acquisition_func_vals: np.ndarray
    ) -> dict[str, int | float]:
        sample_size = next(iter(samples.values())).size
        if sample_size == 0:
            raise ValueError(f"The size of `samples` must be positive, but got {sample_size}.")
        if sample_size != acquisition_func_vals.size:
            raise ValueError(
                "The sizes of `samples` and `acquisition_func_vals` must be same, but got "
                f"({sample_size}, {acquisition_func_vals.size})."
            )
        best_idx = np.argmax(acquisition_func_vals)
        return {k: v[best_idx].item() for k, v in samples.items()}

    @staticmethod
    def hyperopt_parameters() -> dict[str, Any]:
        """Default parameters of hyperopt v0.1.2 (reference sampler.py:677-720)."""
        import warnings

        warnings.warn(
            "TPESampler.hyperopt_parameters is deprecated and will be removed in a "
            "future release.",
            FutureWarning,
        )
        return {
            "consider_prior": True,
            "prior_weight": 1.0,
            "consider_magic_clip": True,
            "consider_endpoints": False,
            "n_startup_trials": 20,
            "n_ei_candidates": 24,
            "gamma": hyperopt_default_gamma,
            "weights": default_weights,
        }

    def before_trial(self, study: "Study", trial: FrozenTrial) -> None:
        self._random_sampler.before_trial(study, trial)

    def after_trial(
        self,
        study: "Study",
        trial: FrozenTrial,
        state: TrialState,
        values: Sequence[float] | None,
    ) -> None:
        assert state in [TrialState.COMPLETE, TrialState.FAIL, TrialState.PRUNED]
        if self._constraints_func is not None:
            _process_constraints_after_trial(self._constraints_func, study, trial, state)
        self._random_sampler.after_trial(study, trial, state, values)


# ----------------------------------------------------------------------------------
# History splitting (below/above)
# ----------------------------------------------------------------------------------


def _get_reference_point(loss_vals: np.ndarray) -> np.ndarray:
    worst_point = np.max(loss_vals, axis=0)
    reference_point = np.maximum(1.1 * worst_point, 0.9 * worst_point)
    reference_point[reference_point == 0] = EPS
    return reference_point


def _split_trials(
    study: "Study", trials: list[FrozenTrial], n_below: int
) -> tuple[list[FrozenTrial], list[FrozenTrial]]:
    complete_trials = []
    pruned_trials = []
    running_trials = []
    infeasible_trials = []

    for t in trials:
        if t.state == TrialState.RUNNING:
            # RUNNING first: their constraint attrs are not yet written.
            running_trials.append(t)
        elif _get_infeasible_trial_score(t) > 0:
            infeasible_trials.append(t)
        elif t.state == TrialState.COMPLETE:
            complete_trials.append(t)
        elif t.state == TrialState.PRUNED:
            pruned_trials.append(t)
        else:
            raise AssertionError(f"unexpected state {t.state}")

    below_complete, above_complete = _split_complete_trials(complete_trials, study, n_below)
    n_below = max(0, n_below - len(below_complete))
    below_pruned, above_pruned = _split_pruned_trials(pruned_trials, study, n_below)
    n_below = max(0, n_below - len(below_pruned))
    below_infeasible, above_infeasible = _split_infeasible_trials(infeasible_trials, n_below)

    below = below_complete + below_pruned + below_infeasible
    above = above_complete + above_pruned + above_infeasible + running_trials
    below.sort(key=lambda t: t.number)
    above.sort(key=lambda t: t.number)
    return below, above


def _split_complete_trials(
    trials: Sequence[FrozenTrial], study: "Study", n_below: int
) -> tuple[list[FrozenTrial], list[FrozenTrial]]:
    n_below = min(n_below, len(trials))
    if len(study.directions) <= 1:
        return _split_complete_trials_single_objective(trials, study, n_below)
    return _split_complete_trials_multi_objective(trials, study, n_below)


def _split_complete_trials_single_objective(
    trials: Sequence[FrozenTrial], study: "Study", n_below: int
) -> tuple[list[FrozenTrial], list[FrozenTrial]]:
    reverse = study.direction == StudyDirection.MAXIMIZE
    sorted_trials = sorted(trials, key=lambda t: t.value, reverse=reverse)  # type: ignore[arg-type,return-value]
    return sorted_trials[:n_below], sorted_trials[n_below:]


def _split_complete_trials_multi_objective(
    trials: Sequence[FrozenTrial], study: "Study", n_below: int
) -> tuple[list[FrozenTrial], list[FrozenTrial]]:
    from optuna_amd._hypervolume.hssp import _solve_hssp

    if n_below == 0:
        return [], list(trials)
    if n_below == len(trials):
        return list(trials), []
    assert 0 < n_below < len(trials)

    lvals = np.array([t.values for t in trials])
    lvals *= np.array(
        [-1.0 if d == StudyDirection.MAXIMIZE else 1.0 for d in study.directions]
    )
    nondomination_ranks = _fast_non_domination_rank(lvals, n_below=n_below)
    ranks, rank_counts = np.unique(nondomination_ranks, return_counts=True)
    last_rank_before_tiebreak = int(
        np.max(ranks[np.cumsum(rank_counts) <= n_below], initial=-1)
    )
    indices = np.arange(len(trials))
    indices_below = indices[nondomination_ranks <= last_rank_before_tiebreak]

    if indices_below.size < n_below:
        # Tie-break the boundary front with greedy hypervolume subset selection.
        need_tiebreak = nondomination_ranks == last_rank_before_tiebreak + 1
        rank_i_lvals = lvals[need_tiebreak]
        subset_size = n_below - indices_below.size
        selected = _solve_hssp(
            rank_i_lvals,
            indices[need_tiebreak],
            subset_size,
            _get_reference_point(rank_i_lvals),
        )
        indices_below = np.append(indices_below, selected)

    below_set = set(indices_below.tolist())
    below = [trials[i] for i in range(len(trials)) if i in below_set]
    above = [trials[i] for i in range(len(trials)) if i not in below_set]
    return below, above


def _get_pruned_trial_score(trial: FrozenTrial, study: "Study") -> tuple[float, float]:
    if len(trial.intermediate_values) > 0:
        step, intermediate_value = max(trial.intermediate_values.items())
        if math.isnan(intermediate_value):
            return -step, float("inf")
        if study.direction == StudyDirection.MINIMIZE:
            return -step, intermediate_value
        return -step, -intermediate_value
    return 1, 0.0


def _split_pruned_trials(
    trials: Sequence[FrozenTrial], study: "Study", n_below: int
) -> tuple[list[FrozenTrial], list[FrozenTrial]]:
    n_below = min(n_below, len(trials))
    sorted_trials = sorted(trials, key=lambda t: _get_pruned_trial_score(t, study))
    return sorted_trials[:n_below], sorted_trials[n_below:]


def _get_infeasible_trial_score(trial: FrozenTrial) -> float:
    constraints = trial.system_attrs.get("constraints")
    if constraints is None:
        return 0.0
    return sum(v for v in constraints if v > 0)


def _split_infeasible_trials(
    trials: Sequence[FrozenTrial], n_below: int
) -> tuple[list[FrozenTrial], list[FrozenTrial]]:
    n_below = min(n_below, len(trials))
    sorted_trials = sorted(trials, key=_get_infeasible_trial_score)
    return sorted_trials[:n_below], sorted_trials[n_below:]
