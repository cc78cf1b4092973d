"""Local-filesystem artifact store (parity: reference ``optuna/artifacts/_filesystem.py``)."""
from __future__ import annotations

import os
import shutil
from pathlib import Path
from typing import BinaryIO

from optuna_amd.artifacts.exceptions import ArtifactNotFound


class FileSystemArtifactStore:
    def __init__(self, base_path: str | Path) -> None:
        if isinstance(base_path, str):
            base_path = Path(base_path)
        if not base_path.exists():
            raise FileNotFoundError(f"The base path {base_path} does not exist.")
        self._base_path = base_path

    def open_reader(self, artifact_id: str) -> BinaryIO:
        filepath = os.path.join(self._base_path, artifact_id)
        try:
            f = open(filepath, "rb")
        except FileNotFoundError as e:
            raise ArtifactNotFound("not found") from e
        return f

    def write(self, artifact_id: str, content_body: BinaryIO) -> None:
        filepath = os.path.join(self._base_path, artifact_id)
        with open(filepath, "wb") as f:
            shutil.copyfileobj(content_body, f)

    def remove(self, artifact_id: str) -> None:
        filepath = os.path.join(self._base_path, artifact_id)
        try:
            os.remove(filepath)
        except FileNotFoundError as e:
            raise ArtifactNotFound("not found") from e
