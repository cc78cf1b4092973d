"""Artifact upload/download/list built on trial/study system attrs.

Artifact metadata is stored under system-attr keys ``artifacts:{artifact_id}``
(JSON of ArtifactMeta); bytes live only in the artifact store backend.

Parity: reference ``optuna/artifacts/_upload.py`` (upload_artifact :58,
ARTIFACTS_ATTR_PREFIX :21, ArtifactMeta :26-50), ``_download.py`` and
``_list_artifact_meta.py``.
"""
from __future__ import annotations

import json
import mimetypes
import os
import shutil
import uuid
from dataclasses import asdict, dataclass
from typing import TYPE_CHECKING, Union

from optuna_amd.artifacts._protocol import ArtifactStore
from optuna_amd.trial import FrozenTrial, Trial


if TYPE_CHECKING:
    from optuna_amd.storages import BaseStorage
    from optuna_amd.study import Study

ARTIFACTS_ATTR_PREFIX = "artifacts:"
DEFAULT_MIME_TYPE = "application/octet-stream"


@dataclass
class ArtifactMeta:
    artifact_id: str
    filename: str
    mimetype: str
    encoding: str | None


def upload_artifact(
    *,
    artifact_store: ArtifactStore,
    file_path: str,
    study_or_trial: Union[Trial, FrozenTrial, "Study"],
    storage: "BaseStorage | None" = None,
    mimetype: str | None = None,
    encoding: str | None = None,
) -> str:
    """Upload a file; returns the artifact_id recorded in the system attrs."""
    from optuna_amd.study import Study

    filename = os.path.basename(file_path)

    if isinstance(study_or_trial, Trial) and storage is None:
        storage = study_or_trial.storage
    elif isinstance(study_or_trial, Study) and storage is None:
        storage = study_or_trial._storage
    if storage is None:
        raise ValueError("storage is required for FrozenTrial.")

    artifact_id = str(uuid.uuid4())
    guess_mimetype, guess_encoding = mimetypes.guess_type(filename)
    artifact = ArtifactMeta(
        artifact_id=artifact_id,
        filename=filename,
        mimetype=mimetype or guess_mimetype or DEFAULT_MIME_TYPE,
        encoding=encoding or guess_encoding,
    )
    attr_key = ARTIFACTS_ATTR_PREFIX + artifact_id
    if isinstance(study_or_trial, (Trial, FrozenTrial)):
        storage.set_trial_system_attr(
            study_or_trial._trial_id, attr_key, json.dumps(asdict(artifact))
        )
    else:
        storage.set_study_system_attr(
            study_or_trial._study_id, attr_key, json.dumps(asdict(artifact))
        )

    with open(file_path, "rb") as f:
        artifact_store.write(artifact_id, f)
    return artifact_id


def download_artifact(
    *, artifact_store: ArtifactStore, artifact_id: str, file_path: str
) -> None:
    """Download an artifact's bytes to ``file_path``."""
    with artifact_store.open_reader(artifact_id) as reader, open(file_path, "wb") as writer:
        shutil.copyfileobj(reader, writer)


def get_all_artifact_meta(
    study_or_trial: Union[Trial, FrozenTrial, "Study"],
    *,
    storage: "BaseStorage | None" = None,
) -> list[ArtifactMeta]:
    """All artifact metadata linked to a study or trial, oldest first."""
    from optuna_amd.study import Study

    if isinstance(study_or_trial, Trial) and storage is None:
        storage = study_or_trial.storage
    elif isinstance(study_or_trial, Study) and storage is None:
        storage = study_or_trial._storage
    if storage is None:
        raise ValueError("storage is required for FrozenTrial.")

    if isinstance(study_or_trial, (Trial, FrozenTrial)):
        system_attrs = storage.get_trial_system_attrs(study_or_trial._trial_id)
    else:
        system_attrs = storage.get_study_system_attrs(study_or_trial._study_id)

    metas = []
    for key, value in system_attrs.items():
        if not key.startswith(ARTIFACTS_ATTR_PREFIX):
            continue
        payload = json.loads(value)
        metas.append(
            ArtifactMeta(
                artifact_id=payload["artifact_id"],
                filename=payload["filename"],
                mimetype=payload["mimetype"],
                encoding=payload["encoding"],
            )
        )
    return metas
