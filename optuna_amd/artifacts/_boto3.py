"""S3 artifact store via boto3 (parity: reference ``optuna/artifacts/_boto3.py``).

boto3 is optional; the import error surfaces on construction.
"""
from __future__ import annotations

from typing import TYPE_CHECKING, BinaryIO

from optuna_amd._imports import try_import
from optuna_amd.artifacts.exceptions import ArtifactNotFound


with try_import() as _imports:
    import boto3
    from botocore.exceptions import ClientError

if TYPE_CHECKING:
    from mypy_boto3_s3 import S3Client  # noqa: F401


class Boto3ArtifactStore:
    def __init__(
        self,
        bucket_name: str,
        client: "S3Client | None" = None,
        *,
        avoid_buf_copy: bool = False,
    ) -> None:
        _imports.check()
        self.bucket = bucket_name
        self.client = client or boto3.client("s3")
        # Skip the local buffer copy when the caller guarantees the stream is
        # positioned and re-readable.
        self._avoid_buf_copy = avoid_buf_copy

    def open_reader(self, artifact_id: str) -> BinaryIO:
        try:
            obj = self.client.get_object(Bucket=self.bucket, Key=artifact_id)
        except ClientError as e:
            if _is_not_found_error(e):
                raise ArtifactNotFound(
                    f"Artifact not found with id {artifact_id}"
                ) from e
            raise
        return obj["Body"]  # type: ignore[return-value]

    def write(self, artifact_id: str, content_body: BinaryIO) -> None:
        fsrc: BinaryIO = content_body
        if not self._avoid_buf_copy:
            import io
            import shutil

            buf = io.BytesIO()
            shutil.copyfileobj(content_body, buf)
            buf.seek(0)
            fsrc = buf
        self.client.upload_fileobj(fsrc, self.bucket, artifact_id)

    def remove(self, artifact_id: str) -> None:
        self.client.delete_object(Bucket=self.bucket, Key=artifact_id)


def _is_not_found_error(e: "ClientError") -> bool:
    error_code = e.response.get("Error", {}).get("Code")
    http_status_code = e.response.get("ResponseMetadata", {}).get("HTTPStatusCode")
    return error_code == "NoSuchKey" or http_status_code == 404
