"""Operator code staging.

Parity with the reference's get_operator_code / generateRunTaskFile
(ols_core/taskMgr/utils/utils_runner.py:674-782): fetch an operator's
code into the task working directory — directory copy (FILE), zip
extraction (FILE/S3/MINIO/HTTP all collapse onto the local object
store, utils/file_repo.py), rename to the operator name, and validate
that the declared entry file exists.  ``builtin:<name>`` operators are
in-process (engine operators) and need no staging.
"""

from __future__ import annotations

import os
import shutil
import zipfile
from typing import Optional

from ..utils.file_repo import LocalFileRepo


class OperatorStagingError(Exception):
    pass


def stage_operator_code(code_path: str, entry_file: str, operator_name: str,
                        work_dir: str,
                        repo: Optional[LocalFileRepo] = None,
                        bucket: str = "operators") -> Optional[str]:
    """Stage one operator's code under ``work_dir/<operator_name>``.

    Returns the staged directory, or None for builtin operators.
    Raises OperatorStagingError when the source is missing or the entry
    file is absent after staging (utils_runner.py:771-779 semantics).
    """
    if not code_path or code_path.startswith("builtin:"):
        return None
    # operator_name comes from submitted task JSON: it must resolve to a
    # direct child of work_dir (never '..', absolute paths or separators
    # — the resolved dst is rmtree'd below)
    if (not operator_name or "/" in operator_name or "\\" in operator_name
            or operator_name in (".", "..")):
        raise OperatorStagingError(
            f"operator name {operator_name!r} must be a plain directory name")
    os.makedirs(work_dir, exist_ok=True)
    dst = os.path.join(work_dir, operator_name)
    if os.path.realpath(os.path.dirname(dst)) != os.path.realpath(work_dir):
        raise OperatorStagingError(
            f"operator name {operator_name!r} escapes the task work dir")
    if os.path.isdir(dst):
        shutil.rmtree(dst)

    src = code_path
    if not os.path.exists(src) and repo is not None:
        # object-store key: download to the working dir first
        local = os.path.join(work_dir, os.path.basename(code_path))
        if not repo.download_file(bucket, code_path, local):
            raise OperatorStagingError(
                f"operator {operator_name}: code {code_path!r} not found "
                f"locally or in the file repo")
        src = local

    if os.path.isdir(src):
        shutil.copytree(src, dst)
    elif src.endswith(".zip"):
        with zipfile.ZipFile(src) as z:
            base = os.path.realpath(dst)
            for m in z.namelist():
                tgt = os.path.realpath(os.path.join(dst, m))
                if tgt != base and not tgt.startswith(base + os.sep):
                    raise OperatorStagingError(
                        f"operator {operator_name}: zip member {m!r} "
                        f"escapes the staging dir")
            z.extractall(dst)
        # a zip that wraps everything in one top-level dir is flattened
        entries = os.listdir(dst)
        if len(entries) == 1 and os.path.isdir(os.path.join(dst, entries[0])) \
                and not os.path.exists(os.path.join(dst, entry_file)):
            inner = os.path.join(dst, entries[0])
            for item in os.listdir(inner):
                shutil.move(os.path.join(inner, item), dst)
            os.rmdir(inner)
    else:
        raise OperatorStagingError(
            f"operator {operator_name}: code path {src!r} must be a "
            f"directory or .zip")

    if entry_file and not os.path.exists(os.path.join(dst, entry_file)):
        raise OperatorStagingError(
            f"operator {operator_name}: entry file {entry_file!r} missing "
            f"after staging {code_path!r}")
    return dst
