"""Resource manager — the quota ledger.

Parity with the reference's ResourceManager
(ols_core/resourceMgr/resource_manager.py:18-332): total resources are
read once at startup, in-use amounts are the sum of live ledger rows,
requests validate against the remainder and insert a row, releases
delete the row.  Differences by design: totals come from the local node
(GPUs + HBM + host CPU/mem) instead of `ray.cluster_resources()`, the
ledger is SQLite instead of a MySQL `resmgr_table`, and the phone-side
quota (PhoneMgr gRPC fan-out, resource_manager.py:262-332) is served by
a configurable static pool since there is no proprietary phone farm.
"""

from __future__ import annotations

import time
from typing import Any, Dict, Optional

from ..utils.sqlite_repo import SqlTableRepo
from ..utils.logging import Logger

RES_COLUMNS = {
    "rowid_": "INTEGER",
    "task_id": "TEXT",
    "user_id": "TEXT",
    "cpu": "REAL",
    "mem": "REAL",
    "gpu": "REAL",
    "hbm_gb": "REAL",
    "phone_json": "TEXT",   # JSON {tier: count}
    "status": "INTEGER",    # 0 = in use (reference semantics)
    "ts": "REAL",
}


def _detect_totals() -> Dict[str, float]:
    import os
    totals: Dict[str, float] = {}
    totals["cpu"] = float(os.cpu_count() or 1)
    try:
        import psutil
        totals["mem"] = psutil.virtual_memory().total / 2**30
    except Exception:
        totals["mem"] = 64.0
    try:
        import torch
        if torch.cuda.is_available():
            n = torch.cuda.device_count()
            totals["gpu"] = float(n)
            totals["hbm_gb"] = sum(
                torch.cuda.get_device_properties(i).total_memory / 2**30
                for i in range(n))
        else:
            totals["gpu"] = 0.0
            totals["hbm_gb"] = 0.0
    except Exception:
        totals["gpu"] = 0.0
        totals["hbm_gb"] = 0.0
    return totals


class ResourceManager:
    def __init__(self, db_path: str = ":memory:",
                 totals: Optional[Dict[str, float]] = None,
                 phone_pool: Optional[Dict[str, Dict[str, int]]] = None):
        self._repo = SqlTableRepo(db_path, "resmgr_table", RES_COLUMNS)
        self.totals = dict(totals) if totals else _detect_totals()
        # {user_id: {tier: count}} simulated real-device quota pool
        self.phone_pool = {u: dict(p) for u, p in (phone_pool or {}).items()}
        self.log = Logger.shared()

    # -- queries ---------------------------------------------------------
    def get_current_res(self) -> Dict[str, float]:
        used = {"cpu": 0.0, "mem": 0.0, "gpu": 0.0, "hbm_gb": 0.0}
        for row in self._repo.get_rows_where({"status": 0}):
            for k in used:
                used[k] += float(row.get(k) or 0.0)
        return used

    def get_remain_res(self) -> Dict[str, float]:
        used = self.get_current_res()
        return {k: self.totals.get(k, 0.0) - used[k] for k in used}

    def _phone_used(self, user_id: str) -> Dict[str, int]:
        import json
        used: Dict[str, int] = {}
        for row in self._repo.get_rows_where({"status": 0, "user_id": user_id}):
            for tier, n in (json.loads(row.get("phone_json") or "{}")).items():
                used[tier] = used.get(tier, 0) + int(n)
        return used

    def get_resource(self, user_id: str = "") -> Dict[str, Any]:
        """Combined view (reference getResource: cluster + phone)."""
        remain = self.get_remain_res()
        out = {"logical_simulation": {"cpu": remain["cpu"], "mem": remain["mem"],
                                      "gpu": remain["gpu"],
                                      "hbm_gb": remain["hbm_gb"]},
               "device_simulation": {}}
        if user_id and user_id in self.phone_pool:
            used = self._phone_used(user_id)
            out["device_simulation"][user_id] = {
                tier: max(0, total - used.get(tier, 0))
                for tier, total in self.phone_pool[user_id].items()}
        return out

    # -- request / release ----------------------------------------------
    def request_resource(self, task_id: str, user_id: str = "",
                         cpu: float = 0.0, mem: float = 0.0, gpu: float = 0.0,
                         hbm_gb: float = 0.0,
                         phones: Optional[Dict[str, int]] = None) -> bool:
        import json
        if cpu < 0 or mem < 0 or gpu < 0 or hbm_gb < 0:
            return False
        if self._repo.get_rows_where({"task_id": task_id, "status": 0}):
            self.log.warning(task_id, "ResourceMgr", "manager",
                             "request denied: task already holds resources")
            return False
        remain = self.get_remain_res()
        if (cpu > remain["cpu"] or mem > remain["mem"]
                or gpu > remain["gpu"] or hbm_gb > remain["hbm_gb"]):
            return False
        phones = phones or {}
        if phones:
            avail = self.get_resource(user_id)["device_simulation"].get(user_id, {})
            for tier, n in phones.items():
                if n > avail.get(tier, 0):
                    return False
        self._repo.add_item({
            "task_id": task_id, "user_id": user_id, "cpu": cpu, "mem": mem,
            "gpu": gpu, "hbm_gb": hbm_gb, "phone_json": json.dumps(phones),
            "status": 0, "ts": time.time()})
        return True

    def release_resource(self, task_id: str) -> bool:
        return self._repo.delete_item("task_id", task_id)

    def holding(self, task_id: str) -> bool:
        return bool(self._repo.get_rows_where({"task_id": task_id, "status": 0}))

    def orphaned_tasks(self) -> list:
        return [r["task_id"] for r in self._repo.get_rows_where({"status": 0})]
