"""Task-oriented deviceflow registry.

Parity with the reference's TaskOrientedDeviceFlowRegistry
(ols_core/deviceflow/non_grpc/registry.py:27-112): which tasks are
registered with the gradient house and the set of compute resources
(logical_simulation / device_simulation) each expects — persisted so a
restarted service resumes (deviceflow_server.py initiate_from_repo).
"""

from __future__ import annotations

import json
import threading
from typing import Dict, List, Optional

from ..utils.sqlite_repo import SqlTableRepo

_COLUMNS = {"task_id": "TEXT", "total_compute_resources": "TEXT"}


class TaskOrientedDeviceFlowRegistry:
    def __init__(self, db_path: str = ":memory:"):
        self._repo = SqlTableRepo(db_path, "deviceflow_table", _COLUMNS,
                                  primary_key="task_id")
        self._lock = threading.Lock()
        self._cache: Dict[str, List[str]] = {}
        for row in self._repo.get_all_rows():
            self._cache[row["task_id"]] = json.loads(
                row["total_compute_resources"] or "[]")

    def register_task(self, task_id: str, resources: List[str]) -> bool:
        with self._lock:
            if task_id in self._cache:
                return False
            self._cache[task_id] = list(resources)
            self._repo.upsert_item("task_id", {
                "task_id": task_id,
                "total_compute_resources": json.dumps(list(resources))})
            return True

    def unregister_task(self, task_id: str) -> bool:
        with self._lock:
            self._cache.pop(task_id, None)
            return self._repo.delete_item("task_id", task_id)

    def resources(self, task_id: str) -> Optional[List[str]]:
        return self._cache.get(task_id)

    def is_registered(self, task_id: str) -> bool:
        return task_id in self._cache

    def task_ids(self) -> List[str]:
        with self._lock:
            return list(self._cache)
