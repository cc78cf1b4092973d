"""ArtifactStore protocol (parity: reference ``optuna/artifacts/_protocol.py``)."""
from __future__ import annotations

from typing import TYPE_CHECKING


if TYPE_CHECKING:
    from typing import BinaryIO, Protocol

    class ArtifactStore(Protocol):
        def open_reader(self, artifact_id: str) -> "BinaryIO":
            """Open a binary reader for the artifact; raises ArtifactNotFound."""
            ...

        def write(self, artifact_id: str, content_body: "BinaryIO") -> None:
            ...

        def remove(self, artifact_id: str) -> None:
            ...

else:
    ArtifactStore = object
