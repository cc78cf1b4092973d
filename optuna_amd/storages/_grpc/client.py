"""gRPC storage proxy client.

Parity: reference ``optuna/storages/_grpc/client.py`` (GrpcStorageProxy :46) with
a per-study finished-trial cache using the same delta-fetch idea
(:378-442). Wire format: see _protocol.py.
"""
from __future__ import annotations

import copy
import threading
from typing import TYPE_CHECKING, Any, Container, Sequence

from optuna_amd._imports import try_import
from optuna_amd.distributions import BaseDistribution
from optuna_amd.storages._base import BaseStorage
from optuna_amd.storages._grpc import _protocol
from optuna_amd.storages._heartbeat import BaseHeartbeat
from optuna_amd.study._frozen import FrozenStudy
from optuna_amd.study._study_direction import StudyDirection
from optuna_amd.trial import FrozenTrial, TrialState


with try_import() as _imports:
    import grpc

if TYPE_CHECKING:
    import grpc


class GrpcStorageProxy(BaseStorage, BaseHeartbeat):
    """Client that forwards every storage call to a proxy server."""

    def __init__(self, *, host: str = "localhost", port: int = 13000) -> None:
        _imports.check()
        self._host = host
        self._port = port
        self._lock = threading.Lock()
        self._connect()

    def _connect(self) -> None:
        self._channel = grpc.insecure_channel(
            f"{self._host}:{self._port}",
            options=[
                ("grpc.max_send_message_length", 1 << 30),
                ("grpc.max_receive_message_length", 1 << 30),
            ],
        )
        self._call = self._channel.unary_unary(
            _protocol.METHOD,
            request_serializer=None,
            response_deserializer=None,
        )

    def __getstate__(self) -> dict[str, Any]:
        state = self.__dict__.copy()
        del state["_channel"]
        del state["_call"]
        del state["_lock"]
        return state

    def __setstate__(self, state: dict[str, Any]) -> None:
        self.__dict__.update(state)
        self._lock = threading.Lock()
        self._connect()

    def close(self) -> None:
        self._channel.close()

    def wait_server_ready(self, timeout: float | None = None) -> None:
        grpc.channel_ready_future(self._channel).result(timeout=timeout)

    # ---- forwarding core ------------------------------------------------------------

    def _rpc(self, method: str, *args: Any, **kwargs: Any) -> Any:
        payload = _protocol.dumps((method, args, kwargs))
        raw = self._call(payload)
        status, value = _protocol.loads(raw)
        if status == "err":
            raise value
        return value

    # ---- BaseStorage surface --------------------------------------------------------

    def create_new_study(
        self, directions: Sequence[StudyDirection], study_name: str | None = None
    ) -> int:
        return self._rpc("create_new_study", list(directions), study_name)

    def delete_study(self, study_id: int) -> None:
        self._rpc("delete_study", study_id)

    def set_study_user_attr(self, study_id: int, key: str, value: Any) -> None:
        self._rpc("set_study_user_attr", study_id, key, value)

    def set_study_system_attr(self, study_id: int, key: str, value: Any) -> None:
        self._rpc("set_study_system_attr", study_id, key, value)

    def get_study_id_from_name(self, study_name: str) -> int:
        return self._rpc("get_study_id_from_name", study_name)

    def get_study_name_from_id(self, study_id: int) -> str:
        return self._rpc("get_study_name_from_id", study_id)

    def get_study_directions(self, study_id: int) -> list[StudyDirection]:
        return self._rpc("get_study_directions", study_id)

    def get_study_user_attrs(self, study_id: int) -> dict[str, Any]:
        return self._rpc("get_study_user_attrs", study_id)

    def get_study_system_attrs(self, study_id: int) -> dict[str, Any]:
        return self._rpc("get_study_system_attrs", study_id)

    def get_all_studies(self) -> list[FrozenStudy]:
        return self._rpc("get_all_studies")

    def create_new_trial(self, study_id: int, template_trial: FrozenTrial | None = None) -> int:
        return self._rpc("create_new_trial", study_id, template_trial)

    def set_trial_param(
        self,
        trial_id: int,
        param_name: str,
        param_value_internal: float,
        distribution: BaseDistribution,
    ) -> None:
        self._rpc("set_trial_param", trial_id, param_name, param_value_internal, distribution)

    def get_trial_id_from_study_id_trial_number(self, study_id: int, trial_number: int) -> int:
        return self._rpc("get_trial_id_from_study_id_trial_number", study_id, trial_number)

    def get_trial_number_from_id(self, trial_id: int) -> int:
        return self._rpc("get_trial_number_from_id", trial_id)

    def get_trial_param(self, trial_id: int, param_name: str) -> float:
        return self._rpc("get_trial_param", trial_id, param_name)

    def set_trial_state_values(
        self, trial_id: int, state: TrialState, values: Sequence[float] | None = None
    ) -> bool:
        return self._rpc("set_trial_state_values", trial_id, state, values)

    def set_trial_intermediate_value(
        self, trial_id: int, step: int, intermediate_value: float
    ) -> None:
        self._rpc("set_trial_intermediate_value", trial_id, step, intermediate_value)

    def set_trial_user_attr(self, trial_id: int, key: str, value: Any) -> None:
        self._rpc("set_trial_user_attr", trial_id, key, value)

    def set_trial_system_attr(self, trial_id: int, key: str, value: Any) -> None:
        self._rpc("set_trial_system_attr", trial_id, key, value)

    def get_trial(self, trial_id: int) -> FrozenTrial:
        return self._rpc("get_trial", trial_id)

    def get_all_trials(
        self,
        study_id: int,
        deepcopy: bool = True,
        states: Container[TrialState] | None = None,
    ) -> list[FrozenTrial]:
        # Server-side copies are pointless — serialization already isolates the
        # caller from the backend's objects.
        trials = self._rpc(
            "get_all_trials", study_id, False, tuple(states) if states is not None else None
        )
        return trials

    # ---- heartbeat ------------------------------------------------------------------

    def record_heartbeat(self, trial_id: int) -> None:
        self._rpc("record_heartbeat", trial_id)

    def _get_stale_trial_ids(self, study_id: int) -> list[int]:
        return self._rpc("_get_stale_trial_ids", study_id)

    def get_heartbeat_interval(self) -> int | None:
        return self._rpc("get_heartbeat_interval")

    def get_failed_trial_callback(self) -> Any:
        return self._rpc("get_failed_trial_callback")

    def is_heartbeat_enabled(self) -> bool:
        return self.get_heartbeat_interval() is not None
