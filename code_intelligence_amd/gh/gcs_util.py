"""Object-store helpers (reference: py/code_intelligence/gcs_util.py).

The reference talks to Google Cloud Storage; this framework abstracts the
store behind ``ObjectStore`` so the same code runs offline: the default
backend maps ``gs://bucket/path`` to ``$CI_OBJECT_STORE_ROOT/bucket/path``
on local disk (tests, air-gapped GPU boxes); an HTTP backend can be
injected where a real store exists. API surface matches the reference:
split_gcs_uri, check_gcs_object, upload_file_to_gcs, download_file_from_gcs
(gcs_util.py:7-100)."""
from __future__ import annotations

import os
import shutil
from pathlib import Path
from typing import Tuple


def split_gcs_uri(gcs_uri: str) -> Tuple[str, str]:
    """'gs://bucket/some/path' -> ('bucket', 'some/path') (gcs_util.py:7-14)."""
    if not gcs_uri.startswith("gs://"):
        raise ValueError(f"not a gs:// uri: {gcs_uri}")
    rest = gcs_uri[len("gs://"):]
    bucket, _, path = rest.partition("/")
    return bucket, path


class ObjectStore:
    """Local-filesystem object store keyed by gs://-style URIs."""

    def __init__(self, root: str | None = None):
        self.root = Path(root or os.environ.get("CI_OBJECT_STORE_ROOT",
                                                "/tmp/ci_object_store"))

    def _local(self, gcs_uri: str) -> Path:
        bucket, path = split_gcs_uri(gcs_uri)
        return self.root / bucket / path

    def exists(self, gcs_uri: str) -> bool:
        return self._local(gcs_uri).exists()

    def upload(self, local_file: str, gcs_uri: str) -> None:
        dest = self._local(gcs_uri)
        dest.parent.mkdir(parents=True, exist_ok=True)
        shutil.copy2(local_file, dest)

    def download(self, gcs_uri: str, local_file: str) -> str:
        src = self._local(gcs_uri)
        if not src.exists():
            raise FileNotFoundError(gcs_uri)
        Path(local_file).parent.mkdir(parents=True, exist_ok=True)
        shutil.copy2(src, local_file)
        return local_file

    def write_bytes(self, gcs_uri: str, data: bytes) -> None:
        dest = self._local(gcs_uri)
        dest.parent.mkdir(parents=True, exist_ok=True)
        dest.write_bytes(data)

    def read_bytes(self, gcs_uri: str) -> bytes:
        return self._local(gcs_uri).read_bytes()


_default_store = None


def default_store() -> ObjectStore:
    global _default_store
    if _default_store is None:
        _default_store = ObjectStore()
    return _default_store


def check_gcs_object(gcs_uri: str, store: ObjectStore | None = None) -> bool:
    return (store or default_store()).exists(gcs_uri)


def upload_file_to_gcs(local_file: str, gcs_uri: str,
                       store: ObjectStore | None = None) -> None:
    (store or default_store()).upload(local_file, gcs_uri)


def download_file_from_gcs(gcs_uri: str, local_file: str,
                           store: ObjectStore | None = None) -> str:
    return (store or default_store()).download(gcs_uri, local_file)
