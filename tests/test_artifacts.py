from __future__ import annotations

import io
import os

import pytest

import optuna_amd
from optuna_amd.artifacts import (
    Backoff,
    FileSystemArtifactStore,
    download_artifact,
    get_all_artifact_meta,
    upload_artifact,
)
from optuna_amd.artifacts.exceptions import ArtifactNotFound


optuna_amd.logging.set_verbosity(optuna_amd.logging.WARNING)


@pytest.fixture
def store(tmp_path):
    base = tmp_path / "artifacts"
    base.mkdir()
    return FileSystemArtifactStore(str(base))


def test_filesystem_roundtrip(store, tmp_path) -> None:
    store.write("abc", io.BytesIO(b"hello"))
    with store.open_reader("abc") as f:
        assert f.read() == b"hello"
    store.remove("abc")
    with pytest.raises(ArtifactNotFound):
        store.open_reader("abc")
    with pytest.raises(ArtifactNotFound):
        store.remove("abc")


def test_filesystem_missing_base_path(tmp_path) -> None:
    with pytest.raises(FileNotFoundError):
        FileSystemArtifactStore(str(tmp_path / "missing"))


def test_upload_download_trial(store, tmp_path) -> None:
    study = optuna_amd.create_study()
    trial = study.ask()
    src = tmp_path / "model.txt"
    src.write_text("weights")
    artifact_id = upload_artifact(
        artifact_store=store, file_path=str(src), study_or_trial=trial
    )
    metas = get_all_artifact_meta(trial)
    assert len(metas) == 1
    assert metas[0].artifact_id == artifact_id
    assert metas[0].filename == "model.txt"
    assert metas[0].mimetype == "text/plain"

    dst = tmp_path / "restored.txt"
    download_artifact(artifact_store=store, artifact_id=artifact_id, file_path=str(dst))
    assert dst.read_text() == "weights"


def test_upload_to_study(store, tmp_path) -> None:
    study = optuna_amd.create_study()
    src = tmp_path / "report.json"
    src.write_text("{}")
    artifact_id = upload_artifact(
        artifact_store=store, file_path=str(src), study_or_trial=study
    )
    metas = get_all_artifact_meta(study)
    assert [m.artifact_id for m in metas] == [artifact_id]
    assert metas[0].mimetype == "application/json"


def test_backoff_retries(tmp_path) -> None:
    class Flaky:
        def __init__(self) -> None:
            self.calls = 0
            self.data: dict[str, bytes] = {}

        def write(self, artifact_id, body):  # type: ignore[no-untyped-def]
            self.calls += 1
            if self.calls < 3:
                raise ConnectionError("transient")
            self.data[artifact_id] = body.read()

        def open_reader(self, artifact_id):  # type: ignore[no-untyped-def]
            return io.BytesIO(self.data[artifact_id])

        def remove(self, artifact_id):  # type: ignore[no-untyped-def]
            del self.data[artifact_id]

    flaky = Flaky()
    wrapped = Backoff(flaky, max_retries=5, min_delay=0.001, max_delay=0.002)
    wrapped.write("x", io.BytesIO(b"payload"))
    assert flaky.calls == 3
    with wrapped.open_reader("x") as f:
        assert f.read() == b"payload"
