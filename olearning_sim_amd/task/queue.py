"""In-memory FIFO of queued TaskConfigs.

Parity with ols_core/taskMgr/task_queue.py:16-50: append-only list with
accessors, plus the deletion-by-id helper that the reference keeps on
TaskManager (task_manager.py deleteTaskinTaskQueue).  Guarded by a lock —
the scheduler thread and the submit path share it.
"""

from __future__ import annotations

import threading
from typing import List, Optional

from .schema import TaskConfig


class TaskQueue:
    def __init__(self):
        self._queue: List[TaskConfig] = []
        self._lock = threading.RLock()

    def add(self, task: TaskConfig) -> None:
        with self._lock:
            self._queue.append(task)

    def get_task_queue(self) -> List[TaskConfig]:
        with self._lock:
            return list(self._queue)

    def get_task_ids(self) -> List[str]:
        with self._lock:
            return [t.task_id for t in self._queue]

    def contains(self, task_id: str) -> bool:
        with self._lock:
            return any(t.task_id == task_id for t in self._queue)

    def remove(self, task_id: str) -> bool:
        """Delete every queued entry with this id (reference walks the
        queue backwards deleting all matches)."""
        with self._lock:
            before = len(self._queue)
            self._queue = [t for t in self._queue if t.task_id != task_id]
            return len(self._queue) != before

    def pop_by_id(self, task_id: str) -> Optional[TaskConfig]:
        with self._lock:
            for i, t in enumerate(self._queue):
                if t.task_id == task_id:
                    return self._queue.pop(i)
        return None

    def __len__(self) -> int:
        with self._lock:
            return len(self._queue)
