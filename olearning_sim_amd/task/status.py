"""Task status model.

Parity with the reference's TaskStatus enum
(ols_core/taskMgr/task_manager.py:41-49) and the Ray JobStatus values the
reference fuses with (PENDING/RUNNING/SUCCEEDED/FAILED/STOPPED).
"""

from __future__ import annotations

import enum


class TaskStatus(str, enum.Enum):
    UNDONE = "UNDONE"          # known to the table but not yet queued
    QUEUED = "QUEUED"
    RUNNING = "RUNNING"
    SUCCEEDED = "SUCCEEDED"
    FAILED = "FAILED"
    STOPPED = "STOPPED"
    MISSING = "MISSING"        # unknown task_id

    def is_terminal(self) -> bool:
        return self in (TaskStatus.SUCCEEDED, TaskStatus.FAILED, TaskStatus.STOPPED)


class JobStatus(str, enum.Enum):
    """Status of the engine job backing a task's logical simulation
    (the reference reads Ray's JobSubmissionClient.get_job_status)."""
    PENDING = "PENDING"
    RUNNING = "RUNNING"
    SUCCEEDED = "SUCCEEDED"
    FAILED = "FAILED"
    STOPPED = "STOPPED"
