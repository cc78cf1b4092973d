"""Google Cloud Storage artifact store (parity: reference ``optuna/artifacts/_gcs.py``)."""
from __future__ import annotations

import io
from typing import TYPE_CHECKING, BinaryIO

from optuna_amd._imports import try_import
from optuna_amd.artifacts.exceptions import ArtifactNotFound


with try_import() as _imports:
    import google.cloud.storage

if TYPE_CHECKING:
    from google.cloud.storage import Client  # noqa: F401


class GCSArtifactStore:
    def __init__(self, bucket_name: str, client: "Client | None" = None) -> None:
        _imports.check()
        self.bucket_name = bucket_name
        self.client = client or google.cloud.storage.Client()
        self.bucket_obj = self.client.bucket(bucket_name)

    def open_reader(self, artifact_id: str) -> BinaryIO:
        blob = self.bucket_obj.blob(artifact_id)
        if not blob.exists():
            raise ArtifactNotFound(f"Artifact not found with id {artifact_id}")
        return io.BytesIO(blob.download_as_bytes())

    def write(self, artifact_id: str, content_body: BinaryIO) -> None:
        blob = self.bucket_obj.blob(artifact_id)
        blob.upload_from_string(content_body.read())

    def remove(self, artifact_id: str) -> None:
        blob = self.bucket_obj.blob(artifact_id)
        if not blob.exists():
            raise ArtifactNotFound(f"Artifact not found with id {artifact_id}")
        blob.delete()
