"""Wire-compatible gRPC storage server (reference api.proto protocol).

Serves the exact protobuf wire format of the reference's StorageService
(reference ``optuna/storages/_grpc/servicer.py``; codec in ``_wire.py``), so
reference clients can talk to this server and vice versa. Error mapping
follows the reference: DuplicatedStudyError → ALREADY_EXISTS, KeyError →
NOT_FOUND, UpdateFinishedTrialError → FAILED_PRECONDITION, ValueError/
RuntimeError → INVALID_ARGUMENT.
"""
from __future__ import annotations

import json
from concurrent import futures
from datetime import datetime
from typing import Any

from optuna_amd.distributions import distribution_to_json, json_to_distribution
from optuna_amd.exceptions import DuplicatedStudyError, UpdateFinishedTrialError
from optuna_amd.storages._base import BaseStorage
from optuna_amd.storages._grpc import _wire
from optuna_amd.study._study_direction import StudyDirection
from optuna_amd.trial import FrozenTrial, TrialState

DATETIME_FORMAT = "%Y-%m-%d %H:%M:%S.%f"


def trial_to_wire(trial: FrozenTrial) -> dict[str, Any]:
    return {
        "trial_id": trial._trial_id,
        "number": trial.number,
        "state": int(trial.state),
        "values": list(trial.values) if trial.values is not None else [],
        "datetime_start": (
            trial.datetime_start.strftime(DATETIME_FORMAT) if trial.datetime_start else ""
        ),
        "datetime_complete": (
            trial.datetime_complete.strftime(DATETIME_FORMAT)
            if trial.datetime_complete
            else ""
        ),
        "params": {
            k: trial.distributions[k].to_internal_repr(v) for k, v in trial.params.items()
        },
        "distributions": {
            k: distribution_to_json(d) for k, d in trial.distributions.items()
        },
        "user_attributes": {k: json.dumps(v) for k, v in trial.user_attrs.items()},
        "system_attributes": {k: json.dumps(v) for k, v in trial.system_attrs.items()},
        "intermediate_values": dict(trial.intermediate_values),
    }


def trial_from_wire(msg: dict[str, Any]) -> FrozenTrial:
    distributions = {
        k: json_to_distribution(v) for k, v in msg["distributions"].items()
    }
    return FrozenTrial(
        trial_id=msg["trial_id"],
        number=msg["number"],
        state=TrialState(msg["state"]),
        value=None,
        values=list(msg["values"]) if msg["values"] else None,
        datetime_start=(
            datetime.strptime(msg["datetime_start"], DATETIME_FORMAT)
            if msg["datetime_start"]
            else None
        ),
        datetime_complete=(
            datetime.strptime(msg["datetime_complete"], DATETIME_FORMAT)
            if msg["datetime_complete"]
            else None
        ),
        params={
            k: distributions[k].to_external_repr(v) for k, v in msg["params"].items()
        },
        distributions=distributions,
        user_attrs={k: json.loads(v) for k, v in msg["user_attributes"].items()},
        system_attrs={k: json.loads(v) for k, v in msg["system_attributes"].items()},
        intermediate_values={int(k): v for k, v in msg["intermediate_values"].items()},
    )


class _WireServicer:
    def __init__(self, backend: BaseStorage) -> None:
        self._backend = backend

    def handle(self, method: str, req: dict[str, Any], context: Any) -> dict[str, Any]:
        import grpc

        b = self._backend
        try:
            if method == "CreateNewStudy":
                sid = b.create_new_study(
                    [StudyDirection(d) for d in req["directions"]],
                    req["study_name"] or None,
                )
                return {"study_id": sid}
            if method == "DeleteStudy":
                b.delete_study(req["study_id"])
                return {}
            if method == "SetStudyUserAttribute":
                b.set_study_user_attr(req["study_id"], req["key"], json.loads(req["value"]))
                return {}
            if method == "SetStudySystemAttribute":
                b.set_study_system_attr(req["study_id"], req["key"], json.loads(req["value"]))
                return {}
            if method == "GetStudyIdFromName":
                return {"study_id": b.get_study_id_from_name(req["study_name"])}
            if method == "GetStudyNameFromId":
                return {"study_name": b.get_study_name_from_id(req["study_id"])}
            if method == "GetStudyDirections":
                return {
                    "directions": [int(d) for d in b.get_study_directions(req["study_id"])]
                }
            if method == "GetStudyUserAttributes":
                return {
                    "user_attributes": {
                        k: json.dumps(v)
                        for k, v in b.get_study_user_attrs(req["study_id"]).items()
                    }
                }
            if method == "GetStudySystemAttributes":
                return {
                    "system_attributes": {
                        k: json.dumps(v)
                        for k, v in b.get_study_system_attrs(req["study_id"]).items()
                    }
                }
            if method == "GetAllStudies":
                return {
                    "studies": [
                        {
                            "study_id": fs._study_id,
                            "study_name": fs.study_name,
                            "directions": [int(d) for d in fs.directions],
                            "user_attributes": {
                                k: json.dumps(v) for k, v in fs.user_attrs.items()
                            },
                            "system_attributes": {
                                k: json.dumps(v) for k, v in fs.system_attrs.items()
                            },
                        }
                        for fs in b.get_all_studies()
                    ]
                }
            if method == "CreateNewTrial":
                template = (
                    None
                    if req["template_trial_is_none"]
                    else trial_from_wire(req["template_trial"])
                )
                return {"trial_id": b.create_new_trial(req["study_id"], template)}
            if method == "SetTrialParameter":
                b.set_trial_param(
                    req["trial_id"],
                    req["param_name"],
                    req["param_value_internal"],
                    json_to_distribution(req["distribution"]),
                )
                return {}
            if method == "GetTrialIdFromStudyIdTrialNumber":
                return {
                    "trial_id": b.get_trial_id_from_study_id_trial_number(
                        req["study_id"], req["trial_number"]
                    )
                }
            if method == "SetTrialStateValues":
                updated = b.set_trial_state_values(
                    req["trial_id"],
                    TrialState(req["state"]),
                    list(req["values"]) if req["values"] else None,
                )
                return {"trial_updated": bool(updated)}
            if method == "SetTrialIntermediateValue":
                b.set_trial_intermediate_value(
                    req["trial_id"], req["step"], req["intermediate_value"]
                )
                return {}
            if method == "SetTrialUserAttribute":
                b.set_trial_user_attr(req["trial_id"], req["key"], json.loads(req["value"]))
                return {}
            if method == "SetTrialSystemAttribute":
                b.set_trial_system_attr(
                    req["trial_id"], req["key"], json.loads(req["value"])
                )
                return {}
            if method == "GetTrial":
                return {"trial": trial_to_wire(b.get_trial(req["trial_id"]))}
            if method == "GetTrials":
                included = set(req["included_trial_ids"])
                gt = req["trial_id_greater_than"]
                trials = b.get_all_trials(req["study_id"], deepcopy=False)
                return {
                    "trials": [
                        trial_to_wire(t)
                        for t in trials
                        if t._trial_id > gt or t._trial_id in included
                    ]
                }
            context.abort(grpc.StatusCode.UNIMPLEMENTED, f"unknown method {method}")
        except DuplicatedStudyError as e:
            context.abort(grpc.StatusCode.ALREADY_EXISTS, str(e))
        except KeyError as e:
            context.abort(grpc.StatusCode.NOT_FOUND, str(e))
        except UpdateFinishedTrialError as e:
            context.abort(grpc.StatusCode.FAILED_PRECONDITION, str(e))
        except (ValueError, RuntimeError) as e:
            context.abort(grpc.StatusCode.INVALID_ARGUMENT, str(e))
        raise AssertionError("unreachable")


def make_wire_server(storage: BaseStorage, host: str, port: int, n_threads: int = 10):
    """Build (not start) a grpc.Server speaking the reference wire protocol."""
    import grpc

    servicer = _WireServicer(storage)
    handlers = {}
    for method, (req_schema, rep_schema) in _wire.METHODS.items():
        def make(method=method, req_schema=req_schema, rep_schema=rep_schema):
            def unary(request_bytes, context):
                req = _wire.decode(req_schema, request_bytes)
                rep = servicer.handle(method, req, context)
                return _wire.encode(rep_schema, rep)

            return grpc.unary_unary_rpc_method_handler(
                unary,
                request_deserializer=lambda b: b,
                response_serializer=lambda b: b,
            )

        handlers[method] = make()
    generic = grpc.method_handlers_generic_handler(_wire.SERVICE, handlers)
    server = grpc.server(futures.ThreadPoolExecutor(max_workers=n_threads))
    server.add_generic_rpc_handlers((generic,))
    server.add_insecure_port(f"{host}:{port}")
    return server


def run_grpc_wire_proxy_server(
    storage: BaseStorage, *, host: str = "localhost", port: int = 13000, n_threads: int = 10
) -> None:
    server = make_wire_server(storage, host, port, n_threads)
    server.start()
    server.wait_for_termination()
