"""Dual-sink structured logger.

Parity with the reference's ols_core/simu_log.py Logger (:24-172): every
event goes to a rotating local file AND a log table keyed by
{task_id, other_number, system_name, module_name, message, log_type};
errors in the sink never propagate to the caller (swallow-all policy, as
the reference does).  The table sink is SQLite here instead of MySQL.
"""

from __future__ import annotations

import logging
import logging.handlers
import os
import threading
import time
from typing import Optional

from .sqlite_repo import SqlTableRepo

_LOG_COLUMNS = {
    "ts": "REAL",
    "task_id": "TEXT",
    "other_number": "INTEGER",
    "system_name": "TEXT",
    "module_name": "TEXT",
    "message": "TEXT",
    "log_type": "TEXT",
}


class Logger:
    """`Logger().info(task_id=..., system_name=..., module_name=..., message=...)`."""

    _shared: Optional["Logger"] = None
    _shared_lock = threading.Lock()

    def __init__(self, log_dir: Optional[str] = None,
                 db_path: Optional[str] = None, to_table: bool = True):
        log_dir = log_dir or os.environ.get("OLSIM_LOG_DIR", os.path.join(
            os.path.expanduser("~"), ".olearning_sim_amd", "log"))
        os.makedirs(log_dir, exist_ok=True)
        self._pylog = logging.getLogger("olearning_sim_amd")
        if not self._pylog.handlers:
            handler = logging.handlers.RotatingFileHandler(
                os.path.join(log_dir, "log_simulation.log"),
                maxBytes=50 * 1024 * 1024, backupCount=3)
            handler.setFormatter(logging.Formatter(
                "%(asctime)s %(levelname)s %(message)s"))
            self._pylog.addHandler(handler)
            self._pylog.setLevel(logging.INFO)
        self._table = None
        if to_table:
            try:
                db_path = db_path or os.path.join(log_dir, "log_table.sqlite")
                self._table = SqlTableRepo(db_path, "log_table", _LOG_COLUMNS)
            except Exception:
                self._table = None

    @classmethod
    def shared(cls) -> "Logger":
        with cls._shared_lock:
            if cls._shared is None:
                cls._shared = cls()
            return cls._shared

    def _emit(self, log_type: str, task_id: str, system_name: str,
              module_name: str, message: str, other_number: int = 0) -> None:
        line = f"[{system_name}/{module_name}] task={task_id} {message}"
        try:
            getattr(self._pylog, log_type, self._pylog.info)(line)
        except Exception:
            pass
        if self._table is not None:
            try:
                self._table.add_item({
                    "ts": time.time(), "task_id": task_id,
                    "other_number": other_number, "system_name": system_name,
                    "module_name": module_name, "message": message,
                    "log_type": log_type})
            except Exception:
                pass

    def info(self, task_id: str, system_name: str, module_name: str,
             message: str, other_number: int = 0) -> None:
        self._emit("info", task_id, system_name, module_name, message, other_number)

    def warning(self, task_id: str, system_name: str, module_name: str,
                message: str, other_number: int = 0) -> None:
        self._emit("warning", task_id, system_name, module_name, message, other_number)

    def error(self, task_id: str, system_name: str, module_name: str,
              message: str, other_number: int = 0) -> None:
        self._emit("error", task_id, system_name, module_name, message, other_number)
