"""Wire-compatible gRPC storage client (reference api.proto protocol).

Implements BaseStorage over the reference's 19 StorageService rpcs with the
same client-side trial cache strategy as the reference client
(reference ``optuna/storages/_grpc/client.py``: finished trials cached by id,
unfinished ids re-requested via ``included_trial_ids``, new trials via
``trial_id_greater_than``). Interoperates with reference servers and with
``run_grpc_wire_proxy_server``.
"""
from __future__ import annotations

import json
import threading
from typing import Any, Container, Sequence

from optuna_amd.exceptions import DuplicatedStudyError, UpdateFinishedTrialError
from optuna_amd.storages._base import BaseStorage
from optuna_amd.storages._grpc import _wire
from optuna_amd.storages._grpc.wire_server import trial_from_wire, trial_to_wire
from optuna_amd.study._frozen import FrozenStudy
from optuna_amd.study._study_direction import StudyDirection
from optuna_amd.trial import FrozenTrial, TrialState


class GrpcWireStorageProxy(BaseStorage):
    """BaseStorage proxy speaking the reference's protobuf wire format."""

    def __init__(self, *, host: str = "localhost", port: int = 13000) -> None:
        self._host = host
        self._port = port
        self._lock = threading.Lock()
        # per-study cache: trial_id -> FrozenTrial (finished immutable)
        self._cache: dict[int, dict[int, FrozenTrial]] = {}
        self._unfinished: dict[int, set[int]] = {}
        self._max_seen: dict[int, int] = {}
        self._setup()

    def _setup(self) -> None:
        import grpc

        self._channel = grpc.insecure_channel(f"{self._host}:{self._port}")
        self._calls = {
            method: self._channel.unary_unary(
                f"/{_wire.SERVICE}/{method}",
                request_serializer=lambda b: b,
                response_deserializer=lambda b: b,
            )
            for method in _wire.METHODS
        }

    def wait_server_ready(self, timeout: float | None = None) -> None:
        import grpc

        grpc.channel_ready_future(self._channel).result(timeout=timeout)

    def close(self) -> None:
        self._channel.close()

    def __getstate__(self) -> dict[str, Any]:
        state = self.__dict__.copy()
        for k in ("_channel", "_calls", "_lock"):
            state.pop(k, None)
        return state

    def __setstate__(self, state: dict[str, Any]) -> None:
        self.__dict__.update(state)
        self._lock = threading.Lock()
        self._setup()

    # ---- rpc plumbing --------------------------------------------------------------

    def _rpc(self, method: str, req: dict[str, Any]) -> dict[str, Any]:
        import grpc

        req_schema, rep_schema = _wire.METHODS[method]
        try:
            payload = self._calls[method](_wire.encode(req_schema, req))
        except grpc.RpcError as e:
            code = e.code()
            detail = e.details() or ""
            if code == grpc.StatusCode.ALREADY_EXISTS:
                raise DuplicatedStudyError(detail) from e
            if code == grpc.StatusCode.NOT_FOUND:
                raise KeyError(detail) from e
            if code == grpc.StatusCode.FAILED_PRECONDITION:
                raise UpdateFinishedTrialError(detail) from e
            if code == grpc.StatusCode.INVALID_ARGUMENT:
                raise ValueError(detail) from e
            raise
        return _wire.decode(rep_schema, payload)

    # ---- studies -------------------------------------------------------------------

    def create_new_study(
        self, directions: Sequence[StudyDirection], study_name: str | None = None
    ) -> int:
        rep = self._rpc(
            "CreateNewStudy",
            {
                "directions": [int(d) for d in directions],
                "study_name": study_name or "",
            },
        )
        return rep["study_id"]

    def delete_study(self, study_id: int) -> None:
        self._rpc("DeleteStudy", {"study_id": study_id})
        with self._lock:
            self._cache.pop(study_id, None)
            self._unfinished.pop(study_id, None)
            self._max_seen.pop(study_id, None)

    def set_study_user_attr(self, study_id: int, key: str, value: Any) -> None:
        self._rpc(
            "SetStudyUserAttribute",
            {"study_id": study_id, "key": key, "value": json.dumps(value)},
        )

    def set_study_system_attr(self, study_id: int, key: str, value: Any) -> None:
        self._rpc(
            "SetStudySystemAttribute",
            {"study_id": study_id, "key": key, "value": json.dumps(value)},
        )

    def get_study_id_from_name(self, study_name: str) -> int:
        return self._rpc("GetStudyIdFromName", {"study_name": study_name})["study_id"]

    def get_study_name_from_id(self, study_id: int) -> str:
        return self._rpc("GetStudyNameFromId", {"study_id": study_id})["study_name"]

    def get_study_directions(self, study_id: int) -> list[StudyDirection]:
        rep = self._rpc("GetStudyDirections", {"study_id": study_id})
        return [StudyDirection(d) for d in rep["directions"]]

    def get_study_user_attrs(self, study_id: int) -> dict[str, Any]:
        rep = self._rpc("GetStudyUserAttributes", {"study_id": study_id})
        return {k: json.loads(v) for k, v in rep["user_attributes"].items()}

    def get_study_system_attrs(self, study_id: int) -> dict[str, Any]:
        rep = self._rpc("GetStudySystemAttributes", {"study_id": study_id})
        return {k: json.loads(v) for k, v in rep["system_attributes"].items()}

    def get_all_studies(self) -> list[FrozenStudy]:
        rep = self._rpc("GetAllStudies", {})
        out = []
        for s in rep["studies"]:
            out.append(
                FrozenStudy(
                    study_name=s["study_name"],
                    direction=None,
                    directions=[StudyDirection(d) for d in s["directions"]],
                    user_attrs={k: json.loads(v) for k, v in s["user_attributes"].items()},
                    system_attrs={
                        k: json.loads(v) for k, v in s["system_attributes"].items()
                    },
                    study_id=s["study_id"],
                )
            )
        return out

    # ---- trials --------------------------------------------------------------------

    def create_new_trial(self, study_id: int, template_trial: FrozenTrial | None = None) -> int:
        req = {
            "study_id": study_id,
            "template_trial_is_none": template_trial is None,
        }
        if template_trial is not None:
            req["template_trial"] = trial_to_wire(template_trial)
        return self._rpc("CreateNewTrial", req)["trial_id"]

    def set_trial_param(
        self,
        trial_id: int,
        param_name: str,
        param_value_internal: float,
        distribution: Any,
    ) -> None:
        from optuna_amd.distributions import distribution_to_json

        self._rpc(
            "SetTrialParameter",
            {
                "trial_id": trial_id,
                "param_name": param_name,
                "param_value_internal": param_value_internal,
                "distribution": distribution_to_json(distribution),
            },
        )

    def get_trial_id_from_study_id_trial_number(self, study_id: int, trial_number: int) -> int:
        return self._rpc(
            "GetTrialIdFromStudyIdTrialNumber",
            {"study_id": study_id, "trial_number": trial_number},
        )["trial_id"]

    def set_trial_state_values(
        self, trial_id: int, state: TrialState, values: Sequence[float] | None = None
    ) -> bool:
        rep = self._rpc(
            "SetTrialStateValues",
            {
                "trial_id": trial_id,
                "state": int(state),
                "values": list(values) if values is not None else [],
            },
        )
        return rep["trial_updated"]

    def set_trial_intermediate_value(
        self, trial_id: int, step: int, intermediate_value: float
    ) -> None:
        self._rpc(
            "SetTrialIntermediateValue",
            {"trial_id": trial_id, "step": step, "intermediate_value": intermediate_value},
        )

    def set_trial_user_attr(self, trial_id: int, key: str, value: Any) -> None:
        self._rpc(
            "SetTrialUserAttribute",
            {"trial_id": trial_id, "key": key, "value": json.dumps(value)},
        )

    def set_trial_system_attr(self, trial_id: int, key: str, value: Any) -> None:
        self._rpc(
            "SetTrialSystemAttribute",
            {"trial_id": trial_id, "key": key, "value": json.dumps(value)},
        )

    def get_trial(self, trial_id: int) -> FrozenTrial:
        rep = self._rpc("GetTrial", {"trial_id": trial_id})
        return trial_from_wire(rep["trial"])

    def get_trial_number_from_id(self, trial_id: int) -> int:
        return self.get_trial(trial_id).number

    def get_trial_param(self, trial_id: int, param_name: str) -> float:
        trial = self.get_trial(trial_id)
        return trial.distributions[param_name].to_internal_repr(trial.params[param_name])

    def get_all_trials(
        self,
        study_id: int,
        deepcopy: bool = True,
        states: Container[TrialState] | None = None,
    ) -> list[FrozenTrial]:
        import copy

        with self._lock:
            cache = self._cache.setdefault(study_id, {})
            unfinished = self._unfinished.setdefault(study_id, set())
            req = {
                "study_id": study_id,
                "included_trial_ids": sorted(unfinished),
                "trial_id_greater_than": self._max_seen.get(study_id, -1),
            }
            rep = self._rpc("GetTrials", req)
            for msg in rep["trials"]:
                t = trial_from_wire(msg)
                cache[t._trial_id] = t
                self._max_seen[study_id] = max(
                    self._max_seen.get(study_id, -1), t._trial_id
                )
                if t.state.is_finished():
                    unfinished.discard(t._trial_id)
                else:
                    unfinished.add(t._trial_id)
            trials = sorted(cache.values(), key=lambda t: t.number)
        if states is not None:
            trials = [t for t in trials if t.state in states]
        return copy.deepcopy(trials) if deepcopy else trials
