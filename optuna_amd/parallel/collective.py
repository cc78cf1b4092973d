"""Collective op-exchange plane for lockstep multi-GPU studies (RCCL/xGMI).

The sequencer-backed ``RcclStorage`` is fully general: any worker may write at
any time, ordered by a TCPStore fetch-add. Its cost is a handful of loopback
RPCs per write. A torchrun study job, though, is *bulk-synchronous*: every rank
runs the same ask → suggest → tell loop, so op traffic is naturally organized
into rounds. This plane exploits that: once per round the ranks exchange their
buffered op records as ONE ``all_gather`` of packed byte tensors — device
tensors under the "nccl" (=RCCL) backend, so the records move HBM-to-HBM over
the xGMI mesh and rank 0's loopback socket leaves the steady-state path
entirely.

Total order without a sequencer: records are merged in (round, rank,
local-sequence) order, which every replica computes identically, so trial
numbering, attribute last-writer-wins and replay state converge on all ranks.
Trial ids are rank-partitioned (``_ID_BASE + seq * world + rank``) so a rank
knows its own trial's id *before* the merge.

Contract (lockstep): every rank must reach ``exchange`` the same number of
times — which `RcclStorage` arranges by exchanging inside ``create_new_trial``
(each rank asks once per round). WAITING→RUNNING claims (enqueue_trial) are
not part of this mode; use the sequencer mode for queue-based workflows.

There is no reference counterpart (the reference's storages coordinate through
SQL/files: optuna/storages/journal/_storage.py:41-52); this is the MI355X-native
replacement described in SURVEY §5.8.
"""
from __future__ import annotations

import json
from typing import Any


class CollectiveOpPlane:
    """One all_gather of packed op batches per round over the process group."""

    def __init__(self, group: Any = None, device: Any = None) -> None:
        import torch
        import torch.distributed as dist

        assert dist.is_initialized(), "torch.distributed must be initialized"
        self._dist = dist
        self._group = group
        self.rank = dist.get_rank(group)
        self.world = dist.get_world_size(group)
        if device is not None:
            self._device = device
        else:
            backend = dist.get_backend(group)
            self._device = (
                torch.device("cuda") if "nccl" in str(backend) else torch.device("cpu")
            )
        self.n_rounds = 0

    def exchange_bytes(self, payload: bytes) -> list[bytes]:
        """all_gather one byte payload per rank; returns payloads in rank order."""
        import torch

        dist = self._dist
        n = torch.tensor([len(payload)], dtype=torch.int64, device=self._device)
        sizes = [
            torch.zeros(1, dtype=torch.int64, device=self._device)
            for _ in range(self.world)
        ]
        dist.all_gather(sizes, n, group=self._group)
        lens = [int(s.item()) for s in sizes]
        width = max(lens + [1])
        send = torch.zeros(width, dtype=torch.uint8, device=self._device)
        if payload:
            send[: len(payload)] = torch.frombuffer(
                bytearray(payload), dtype=torch.uint8
            ).to(self._device)
        recv = [
            torch.empty(width, dtype=torch.uint8, device=self._device)
            for _ in range(self.world)
        ]
        dist.all_gather(recv, send, group=self._group)
        self.n_rounds += 1
        return [bytes(recv[r][: lens[r]].cpu().numpy().tobytes()) for r in range(self.world)]

    # Payloads above this size ride compressed (CMA-ES generations carry
    # ~160 KB of hex-pickled optimizer state per rank; JSON+hex deflates ~3x).
    _COMPRESS_MIN = 4096

    def exchange_records(
        self, records: list[dict[str, Any]]
    ) -> list[list[dict[str, Any]]]:
        """Exchange op-record batches; returns one batch per rank, rank order."""
        import zlib

        raw = json.dumps(records, separators=(",", ":")).encode()
        if len(raw) >= self._COMPRESS_MIN:
            raw = b"Z" + zlib.compress(raw, level=1)
        else:
            raw = b"J" + raw
        out: list[list[dict[str, Any]]] = []
        for p in self.exchange_bytes(raw):
            if not p:
                out.append([])
                continue
            body = zlib.decompress(p[1:]) if p[:1] == b"Z" else p[1:]
            out.append(json.loads(body) if body else [])
        return out
