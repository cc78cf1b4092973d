"""Wire protocol for the gRPC storage proxy.

The reference generates protobuf stubs with one rpc per BaseStorage method
(reference ``optuna/storages/_grpc/api.proto``: 20 rpcs). This build has no
protoc available, so the proxy uses a single generic unary rpc carrying a
pickled ``(method_name, args, kwargs)`` triple and returning a pickled
``("ok", result)`` / ``("err", exception)`` pair. Functionally equivalent
(same method surface, same exception semantics); not wire-compatible with the
reference's protobuf clients.
"""
from __future__ import annotations

import pickle
from typing import Any


SERVICE = "optuna_amd.storages.StorageService"
METHOD = f"/{SERVICE}/Call"

# The BaseStorage surface the proxy forwards.
FORWARDED_METHODS = (
    "create_new_study",
    "delete_study",
    "set_study_user_attr",
    "set_study_system_attr",
    "get_study_id_from_name",
    "get_study_name_from_id",
    "get_study_directions",
    "get_study_user_attrs",
    "get_study_system_attrs",
    "get_all_studies",
    "create_new_trial",
    "set_trial_param",
    "get_trial_id_from_study_id_trial_number",
    "get_trial_number_from_id",
    "get_trial_param",
    "set_trial_state_values",
    "set_trial_intermediate_value",
    "set_trial_user_attr",
    "set_trial_system_attr",
    "get_trial",
    "get_all_trials",
    "get_n_trials",
    "get_best_trial",
    "record_heartbeat",
    "_get_stale_trial_ids",
    "get_heartbeat_interval",
    "get_failed_trial_callback",
)


def dumps(obj: Any) -> bytes:
    return pickle.dumps(obj, protocol=pickle.HIGHEST_PROTOCOL)


def loads(data: bytes) -> Any:
    return pickle.loads(data)
