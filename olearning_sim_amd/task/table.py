"""Task table.

Same logical columns as the reference's MySQL taskmgr_table accessed via
TaskTableRepo (ols_core/taskMgr/utils/utils.py:29-267): task_status,
task_params, total_simulation, logical/device targets, per-round cursors,
results, timestamps, resource_occupied and job_id — but embedded SQLite.
"""

from __future__ import annotations

import time
from typing import Any, Dict, List, Optional

from ..utils.sqlite_repo import SqlTableRepo

TASK_COLUMNS = {
    "task_id": "TEXT",
    "user_id": "TEXT",
    "task_status": "TEXT",
    "task_params": "TEXT",          # canonical task JSON
    "total_simulation": "TEXT",     # JSON: {max_round, operator_name_list,
                                    #        data_name_list, total_simulation}
    "logical_target": "TEXT",       # JSON: {"logical_target": [...]}
    "device_target": "TEXT",
    "logical_task_params": "TEXT",  # per-side assembled task JSON
    "device_task_params": "TEXT",   # lowered device-farm config JSON
    "logical_result": "TEXT",       # JSON: {"logical_result": [...]}
    "device_result": "TEXT",
    "logical_round": "INTEGER",
    "logical_operator": "TEXT",
    "device_round": "INTEGER",
    "device_operator": "TEXT",
    "resource_occupied": "INTEGER",
    "job_id": "TEXT",
    "in_queue_time": "REAL",
    "submit_task_time": "REAL",
    "finish_task_time": "REAL",
    "freeze_time": "REAL",
    "release_time": "REAL",
}


class TaskTableRepo:
    def __init__(self, path: str = ":memory:"):
        self._repo = SqlTableRepo(path, "taskmgr_table", TASK_COLUMNS,
                                  primary_key="task_id")

    # reference-style accessors ------------------------------------------
    def has_task(self, task_id: str) -> bool:
        return self._repo.has_item("task_id", task_id)

    def add_task(self, task_id: str, user_id: str = "",
                 task_status: str = "UNDONE") -> None:
        self._repo.add_item({"task_id": task_id, "user_id": user_id,
                             "task_status": task_status,
                             "resource_occupied": 0})

    def delete_task(self, task_id: str) -> None:
        self._repo.delete_item("task_id", task_id)

    def get_item_value(self, task_id: str, item: str) -> Any:
        return self._repo.get_item_value("task_id", task_id, item)

    def set_item_value(self, task_id: str, item: str, value: Any) -> bool:
        return self._repo.set_item_value("task_id", task_id, item, value)

    def set_items(self, task_id: str, **items: Any) -> None:
        for k, v in items.items():
            self._repo.set_item_value("task_id", task_id, k, v)

    def get_row(self, task_id: str) -> Optional[Dict[str, Any]]:
        rows = self._repo.get_rows_where({"task_id": task_id})
        return rows[0] if rows else None

    def tasks_with_status(self, status: str) -> List[str]:
        return [r["task_id"] for r in
                self._repo.get_rows_where({"task_status": status})]

    def tasks_with_resource_occupied(self) -> List[str]:
        return [r["task_id"] for r in
                self._repo.get_rows_where({"resource_occupied": 1})]

    def all_rows(self) -> List[Dict[str, Any]]:
        return self._repo.get_all_rows()

    def mark_in_queue(self, task_id: str) -> None:
        self.set_items(task_id, task_status="QUEUED", in_queue_time=time.time())

    def close(self) -> None:
        self._repo.close()
