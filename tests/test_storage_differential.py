"""Differential storage fuzz: the same randomized op sequence applied to every
backend must produce identical observable state (trials, params, attrs,
states, numbers). Catches divergence between the dict-based store, the journal
replay, and the TCPStore op log."""
from __future__ import annotations

import datetime

import numpy as np
import pytest

import optuna_amd
from optuna_amd.distributions import FloatDistribution, IntDistribution
from optuna_amd.storages import InMemoryStorage, JournalStorage
from optuna_amd.storages.journal import JournalFileBackend
from optuna_amd.study import StudyDirection
from optuna_amd.trial import TrialState


def _apply_sequence(storage, seed: int) -> list[dict]:
    rng = np.random.RandomState(seed)
    sid = storage.create_new_study([StudyDirection.MINIMIZE], study_name=f"fuzz{seed}")
    trial_ids: list[int] = []
    for step in range(120):
        op = rng.randint(0, 10)
        live = [
            t
            for t in storage.get_all_trials(sid, deepcopy=False)
            if not t.state.is_finished()
        ]
        if op <= 2 or not trial_ids:
            trial_ids.append(storage.create_new_trial(sid))
        elif op <= 5 and live:
            t = live[rng.randint(len(live))]
            name = f"p{rng.randint(3)}"
            if name in t.params:
                continue
            dist = (
                FloatDistribution(-5.0, 5.0)
                if name != "p2"
                else IntDistribution(0, 9)
            )
            storage.set_trial_param(
                t._trial_id, name, float(rng.randint(0, 10)), dist
            )
        elif op == 6 and live:
            t = live[rng.randint(len(live))]
            storage.set_trial_intermediate_value(
                t._trial_id, int(rng.randint(5)), float(rng.rand())
            )
        elif op == 7 and live:
            t = live[rng.randint(len(live))]
            storage.set_trial_user_attr(t._trial_id, f"u{rng.randint(2)}", int(rng.randint(100)))
        elif op == 8 and live:
            t = live[rng.randint(len(live))]
            storage.set_trial_system_attr(t._trial_id, f"s{rng.randint(2)}", int(rng.randint(100)))
        elif live:
            t = live[rng.randint(len(live))]
            if t.state == TrialState.WAITING:
                storage.set_trial_state_values(t._trial_id, TrialState.RUNNING)
            else:
                state = (
                    TrialState.COMPLETE
                    if rng.rand() < 0.7
                    else (TrialState.PRUNED if rng.rand() < 0.5 else TrialState.FAIL)
                )
                values = [float(rng.rand())] if state == TrialState.COMPLETE else None
                storage.set_trial_state_values(t._trial_id, state, values)
    out = []
    for t in storage.get_all_trials(sid, deepcopy=False):
        out.append(
            {
                "number": t.number,
                "state": int(t.state),
                "params": dict(sorted(t.params.items())),
                "values": t.values,
                "user_attrs": dict(sorted(t.user_attrs.items())),
                "system_attrs": dict(sorted(t.system_attrs.items())),
                "intermediate": dict(sorted(t.intermediate_values.items())),
            }
        )
    return out


@pytest.mark.parametrize("seed", [1, 2, 3])
def test_backends_agree_on_random_sequences(tmp_path, seed) -> None:
    import socket

    def free_port() -> int:
        with socket.socket() as s:
            s.bind(("127.0.0.1", 0))
            return s.getsockname()[1]

    from torch.distributed import TCPStore

    from optuna_amd.storages._rccl import RcclStorage

    results = {}
    results["inmemory"] = _apply_sequence(InMemoryStorage(), seed)
    results["journal"] = _apply_sequence(
        JournalStorage(JournalFileBackend(str(tmp_path / f"j{seed}.jsonl"))), seed
    )
    store = TCPStore(
        "127.0.0.1", free_port(), 1, is_master=True,
        timeout=datetime.timedelta(seconds=30),
    )
    results["rccl"] = _apply_sequence(RcclStorage(store), seed)
    from optuna_amd.storages import RDBStorage

    results["rdb"] = _apply_sequence(
        RDBStorage(f"sqlite:///{tmp_path}/f{seed}.db"), seed
    )

    assert results["inmemory"] == results["journal"], "inmemory vs journal"
    assert results["inmemory"] == results["rccl"], "inmemory vs rccl"
    assert results["inmemory"] == results["rdb"], "inmemory vs rdb"
