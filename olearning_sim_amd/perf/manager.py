"""Performance manager.

The reference declares a PerformanceMgr gRPC service
(ols_core/proto/performanceService.proto:4-6) mounted by
SimulatorSession (simu_session.py:44-46) but its implementation is
absent from the open-source drop.  This build provides a
working one: per-task and per-round timing/throughput metrics collected
from the engine and the task table, queryable in-process and over the
JSON API.
"""

from __future__ import annotations

import json
import threading
import time
from typing import Any, Dict, List, Optional

from ..utils.sqlite_repo import SqlTableRepo

_COLUMNS = {
    "ts": "REAL",
    "task_id": "TEXT",
    "round": "INTEGER",
    "metric": "TEXT",
    "value": "REAL",
    "detail": "TEXT",
}


class PerformanceManager:
    def __init__(self, db_path: str = ":memory:"):
        self._repo = SqlTableRepo(db_path, "perf_table", _COLUMNS)
        self._lock = threading.Lock()

    # -- recording --------------------------------------------------------
    def record(self, task_id: str, metric: str, value: float,
               round_idx: Optional[int] = None,
               detail: Optional[Dict[str, Any]] = None) -> None:
        with self._lock:
            self._repo.add_item({
                "ts": time.time(), "task_id": task_id,
                "round": round_idx if round_idx is not None else -1,
                "metric": metric, "value": float(value),
                "detail": json.dumps(detail or {})})

    def record_round(self, task_id: str, round_idx: int, elapsed_s: float,
                     clients: int, loss: Optional[float] = None) -> None:
        self.record(task_id, "round_time_s", elapsed_s, round_idx)
        if elapsed_s > 0:
            self.record(task_id, "clients_per_s", clients / elapsed_s,
                        round_idx)
        if loss is not None:
            self.record(task_id, "loss", loss, round_idx)

    # -- queries ----------------------------------------------------------
    def metrics(self, task_id: str,
                metric: Optional[str] = None) -> List[Dict[str, Any]]:
        cond = {"task_id": task_id}
        if metric:
            cond["metric"] = metric
        rows = self._repo.get_rows_where(cond)
        rows.sort(key=lambda r: (r["round"], r["ts"]))
        return rows

    def summary(self, task_id: str) -> Dict[str, Any]:
        rows = self.metrics(task_id)
        by_metric: Dict[str, List[float]] = {}
        for r in rows:
            by_metric.setdefault(r["metric"], []).append(r["value"])
        out: Dict[str, Any] = {"task_id": task_id, "metrics": {}}
        for m, vals in by_metric.items():
            s = sorted(vals)
            n = len(s)
            out["metrics"][m] = {
                "count": n, "last": vals[-1],
                "mean": sum(vals) / n,
                "min": s[0], "max": s[-1],
                "p50": s[n // 2],
                "p95": s[min(n - 1, (n * 95) // 100)]}
        return out
