from optuna_amd.artifacts._backoff import Backoff
from optuna_amd.artifacts._boto3 import Boto3ArtifactStore
from optuna_amd.artifacts._filesystem import FileSystemArtifactStore
from optuna_amd.artifacts._gcs import GCSArtifactStore
from optuna_amd.artifacts._protocol import ArtifactStore
from optuna_amd.artifacts._store import (
    ArtifactMeta,
    download_artifact,
    get_all_artifact_meta,
    upload_artifact,
)


__all__ = [
    "ArtifactMeta",
    "ArtifactStore",
    "Backoff",
    "Boto3ArtifactStore",
    "FileSystemArtifactStore",
    "GCSArtifactStore",
    "download_artifact",
    "get_all_artifact_meta",
    "upload_artifact",
]
