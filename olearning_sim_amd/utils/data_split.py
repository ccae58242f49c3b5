"""Hybrid data splitting.

Parity with the reference's HybridDataSplitter
(ols_core/taskMgr/utils/utils_runner.py:178-476): user data is split
between the logical-simulation side and the device-simulation side,
either by row-level train/test splitting of CSV files
(sklearn train_test_split in the reference; deterministic shuffling
here) or by pre-split subfolders.  Transport is the local file repo —
there is no S3/MinIO service to round-trip zips through.
"""

from __future__ import annotations

import os
import random
import shutil
import zipfile
from typing import Tuple


class HybridDataSplitter:
    def __init__(self, seed: int = 0):
        self._rng = random.Random(seed)

    # -- archives ---------------------------------------------------------
    @staticmethod
    def extract(zip_path: str, out_dir: str) -> str:
        os.makedirs(out_dir, exist_ok=True)
        with zipfile.ZipFile(zip_path) as z:
            z.extractall(out_dir)
        return out_dir

    @staticmethod
    def archive(src_dir: str, zip_path: str) -> str:
        os.makedirs(os.path.dirname(os.path.abspath(zip_path)), exist_ok=True)
        with zipfile.ZipFile(zip_path, "w", zipfile.ZIP_DEFLATED) as z:
            for dirpath, _dirs, files in os.walk(src_dir):
                for f in files:
                    full = os.path.join(dirpath, f)
                    z.write(full, os.path.relpath(full, src_dir))
        return zip_path

    # -- row split (reference :195-327) -----------------------------------
    def split_csv(self, csv_path: str, logical_path: str, device_path: str,
                  device_fraction: float) -> Tuple[int, int]:
        """Split one CSV's data rows between the two sides; the header
        is kept on both.  Returns (logical_rows, device_rows)."""
        with open(csv_path) as f:
            lines = f.readlines()
        if not lines:
            open(logical_path, "w").close()
            open(device_path, "w").close()
            return 0, 0
        header, rows = lines[0], lines[1:]
        idx = list(range(len(rows)))
        self._rng.shuffle(idx)
        n_dev = int(round(len(rows) * device_fraction))
        dev_set = set(idx[:n_dev])
        with open(logical_path, "w") as lf, open(device_path, "w") as df:
            lf.write(header)
            df.write(header)
            for i, row in enumerate(rows):
                (df if i in dev_set else lf).write(row)
        return len(rows) - n_dev, n_dev

    def split_dir(self, data_dir: str, logical_dir: str, device_dir: str,
                  device_fraction: float) -> Tuple[int, int]:
        """Row-split every CSV under data_dir; other files are copied to
        both sides (configs, vocabularies, ...)."""
        total = [0, 0]
        for dirpath, _dirs, files in os.walk(data_dir):
            rel = os.path.relpath(dirpath, data_dir)
            for f in files:
                src = os.path.join(dirpath, f)
                ldst = os.path.join(logical_dir, rel, f)
                ddst = os.path.join(device_dir, rel, f)
                os.makedirs(os.path.dirname(ldst), exist_ok=True)
                os.makedirs(os.path.dirname(ddst), exist_ok=True)
                if f.endswith(".csv"):
                    nl, nd = self.split_csv(src, ldst, ddst, device_fraction)
                    total[0] += nl
                    total[1] += nd
                else:
                    shutil.copy2(src, ldst)
                    shutil.copy2(src, ddst)
        return total[0], total[1]

    # -- pre-split subfolders (reference :330-376) -------------------------
    @staticmethod
    def split_presplit(data_dir: str, logical_dir: str, device_dir: str,
                       logical_sub: str = "logical",
                       device_sub: str = "device") -> bool:
        lsrc = os.path.join(data_dir, logical_sub)
        dsrc = os.path.join(data_dir, device_sub)
        if not (os.path.isdir(lsrc) and os.path.isdir(dsrc)):
            return False
        shutil.copytree(lsrc, logical_dir, dirs_exist_ok=True)
        shutil.copytree(dsrc, device_dir, dirs_exist_ok=True)
        return True
