"""End-to-end wire-protocol proxy tests: our wire client against our wire
server, then cross-compatibility against the REFERENCE's generated stubs
(their client classes driving our server) when the reference is available."""
from __future__ import annotations

import socket
import threading

import pytest

import optuna_amd
from optuna_amd.storages import InMemoryStorage
from optuna_amd.storages._grpc.wire_client import GrpcWireStorageProxy
from optuna_amd.storages._grpc.wire_server import make_wire_server
from optuna_amd.trial import TrialState


def _free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


@pytest.fixture
def wire_pair():
    port = _free_port()
    backend = InMemoryStorage()
    server = make_wire_server(backend, "127.0.0.1", port)
    server.start()
    client = GrpcWireStorageProxy(host="127.0.0.1", port=port)
    client.wait_server_ready(timeout=20)
    yield client, backend
    client.close()
    server.stop(None)


def test_wire_proxy_end_to_end(wire_pair) -> None:
    client, _ = wire_pair
    study = optuna_amd.create_study(storage=client, study_name="wire")
    study.set_user_attr("team", {"a": 1})

    def obj(t):
        x = t.suggest_float("x", -1, 1)
        i = t.suggest_int("i", 1, 8)
        t.report(x, step=0)
        if t.number == 2:
            raise optuna_amd.TrialPruned()
        return x * x + i

    study.optimize(obj, n_trials=6)
    assert len(study.trials) == 6
    assert study.user_attrs == {"team": {"a": 1}}
    states = [t.state for t in study.trials]
    assert states.count(TrialState.PRUNED) == 1
    # reload via a second client
    c2 = GrpcWireStorageProxy(host=client._host, port=client._port)
    s2 = optuna_amd.load_study(study_name="wire", storage=c2)
    assert len(s2.trials) == 6
    assert s2.best_value == study.best_value
    c2.close()


def test_reference_stub_against_our_server(wire_pair) -> None:
    """The reference's generated protobuf classes drive our wire server."""
    import importlib.util
    import os
    import sys

    path = "/root/reference/optuna/storages/_grpc/auto_generated"
    if not os.path.isdir(path):
        pytest.skip("reference checkout not available")
    spec = importlib.util.spec_from_file_location("api_pb2", f"{path}/api_pb2.py")
    pb2 = importlib.util.module_from_spec(spec)
    sys.modules["api_pb2"] = pb2
    try:
        spec.loader.exec_module(pb2)
    except Exception as e:
        pytest.skip(f"reference pb2 not loadable: {e}")
    import grpc

    client, _ = wire_pair
    channel = grpc.insecure_channel(f"127.0.0.1:{client._port}")
    call = channel.unary_unary(
        "/optuna.StorageService/CreateNewStudy",
        request_serializer=lambda m: m.SerializeToString(),
        response_deserializer=pb2.CreateNewStudyReply.FromString,
    )
    rep = call(pb2.CreateNewStudyRequest(directions=[pb2.MINIMIZE], study_name="refstub"))
    sid = rep.study_id

    call2 = channel.unary_unary(
        "/optuna.StorageService/CreateNewTrial",
        request_serializer=lambda m: m.SerializeToString(),
        response_deserializer=pb2.CreateNewTrialReply.FromString,
    )
    tid = call2(pb2.CreateNewTrialRequest(study_id=sid, template_trial_is_none=True)).trial_id

    call3 = channel.unary_unary(
        "/optuna.StorageService/GetTrial",
        request_serializer=lambda m: m.SerializeToString(),
        response_deserializer=pb2.GetTrialReply.FromString,
    )
    trial = call3(pb2.GetTrialRequest(trial_id=tid)).trial
    assert trial.trial_id == tid and trial.state == pb2.RUNNING
    # And our wire client sees the study made through the reference stubs.
    assert client.get_study_id_from_name("refstub") == sid
    channel.close()
