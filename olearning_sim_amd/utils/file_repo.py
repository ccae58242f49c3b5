"""File repositories.

Parity with the reference's FileRepo implementations
(ols_core/ofl_commons/infrastructure/FileRepo/s3_file_repo.py:1-64 and
minio_file_repo.py:1-89): the same upload/download/delete/list/
download_payload surface over a local object store (bucket = directory)
— one node needs no MinIO/S3 service, and the engine reads checkpoints
straight from the filesystem.  The transfer-type enum (S3 / MINIO /
HTTP / FILE) maps every mode onto this repo.
"""

from __future__ import annotations

import os
import shutil
from typing import List, Optional


class LocalFileRepo:
    """Bucketed object store on a local root directory."""

    def __init__(self, root: Optional[str] = None):
        self.root = root or os.environ.get(
            "OLSIM_FILE_ROOT",
            os.path.join(os.path.expanduser("~"), ".olearning_sim_amd",
                         "files"))
        os.makedirs(self.root, exist_ok=True)

    def _path(self, bucket: str, key: str) -> str:
        return os.path.join(self.root, bucket, key)

    def bucket_exists(self, bucket: str) -> bool:
        return os.path.isdir(os.path.join(self.root, bucket))

    def make_bucket(self, bucket: str) -> None:
        os.makedirs(os.path.join(self.root, bucket), exist_ok=True)

    def upload_file(self, local_path: str, bucket: str, key: str) -> bool:
        dst = self._path(bucket, key)
        os.makedirs(os.path.dirname(dst), exist_ok=True)
        shutil.copy2(local_path, dst)
        return True

    def download_file(self, bucket: str, key: str, local_path: str) -> bool:
        src = self._path(bucket, key)
        if not os.path.exists(src):
            return False
        os.makedirs(os.path.dirname(os.path.abspath(local_path)), exist_ok=True)
        shutil.copy2(src, local_path)
        return True

    def download_payload(self, bucket: str, key: str, local_path: str) -> bool:
        """Download then delete (reference S3FileRepo.download_payload)."""
        if not self.download_file(bucket, key, local_path):
            return False
        self.delete_file(bucket, key)
        return True

    def delete_file(self, bucket: str, key: str) -> bool:
        src = self._path(bucket, key)
        if os.path.exists(src):
            os.remove(src)
            return True
        return False

    def list_files(self, bucket: str, prefix: str = "") -> List[str]:
        base = os.path.join(self.root, bucket)
        if not os.path.isdir(base):
            return []
        out = []
        for dirpath, _dirs, files in os.walk(base):
            for f in files:
                rel = os.path.relpath(os.path.join(dirpath, f), base)
                if rel.startswith(prefix):
                    out.append(rel)
        return sorted(out)

    def exists(self, bucket: str, key: str) -> bool:
        return os.path.exists(self._path(bucket, key))
