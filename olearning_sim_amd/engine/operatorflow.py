"""Operator-flow round gates.

Parity with the reference's OperatorFlow (ols_core/taskMgr/utils/
operatorflow.py:47-352): each round is bracketed by a START gate and a
STOP gate whose strategies synchronise the simulation with an external
aggregation/selection service.  The reference ships three strategies;
their semantics are kept, with the external transports replaced by
pluggable callables/files:

- "" (empty): no gating — the round runs immediately.
- "waiting_for_global_aggregation" (reference :135-237): poll a
  selection service for the current round index; the stop gate waits
  until the service's round has advanced past this one.  The service
  is a callable returning an int (the reference polls a WebSocket).
- "sample_and_aggregation" (reference :240-291): flag-file handshake
  with an external aggregator — the stop gate writes
  ``simulation_finished.txt`` and waits for
  ``aggregation_finished.txt`` (deleting it once seen).
- custom callables: any ``fn(round_idx) -> bool`` polled until true.

Every wait honours (wait_interval, total_timeout) from the task's
flow_setting (schema.StrategyCondition), like the reference.
"""

from __future__ import annotations

import os
import time
from typing import Callable, Dict, Optional

from ..utils.logging import Logger


class GateTimeout(Exception):
    pass


class OperatorFlow:
    def __init__(self, task_id: str,
                 start_strategy: str = "", stop_strategy: str = "",
                 wait_interval: float = 1.0, total_timeout: float = 0.0,
                 work_dir: str = "",
                 selection_round_fn: Optional[Callable[[], int]] = None,
                 custom_gates: Optional[Dict[str, Callable[[int], bool]]] = None,
                 selection_ws_url: str = ""):
        self.task_id = task_id
        self.start_strategy = start_strategy
        self.stop_strategy = stop_strategy
        self.wait_interval = max(0.01, wait_interval)
        self.total_timeout = total_timeout
        self.work_dir = work_dir or "."
        # waiting_for_global_aggregation polls a selection service for
        # the current round index; the reference polls it over
        # WebSocket (operatorflow.py:158-237).  selection_ws_url builds
        # the poll over utils/ws.py; selection_round_fn injects it
        # directly (tests, in-process services).
        if selection_round_fn is None and selection_ws_url:
            selection_round_fn = self._ws_selection_fn(selection_ws_url)
        self.selection_round_fn = selection_round_fn
        self.custom_gates = custom_gates or {}
        self.log = Logger.shared()
        self._seen_round = -1
        self._ws_conn = None
        self._ws_url = selection_ws_url

    def _ws_selection_fn(self, url: str) -> Callable[[], int]:
        import json as _json

        def query() -> int:
            from ..utils import ws
            try:
                if self._ws_conn is None:
                    self._ws_conn = ws.connect(url, timeout=5.0)
                self._ws_conn.send_text(_json.dumps(
                    {"query": "round_idx", "task_id": self.task_id}))
                reply = self._ws_conn.recv_text(timeout=5.0)
                if reply is None:
                    self._ws_conn = None
                    return self._seen_round
                self._seen_round = int(_json.loads(reply)["round_idx"])
            except (OSError, ConnectionError, ValueError, KeyError):
                try:
                    if self._ws_conn is not None:
                        self._ws_conn.close()
                except Exception:
                    pass
                self._ws_conn = None
            return self._seen_round

        return query

    # ------------------------------------------------------------------
    def _poll(self, cond: Callable[[], bool], what: str) -> None:
        deadline = (time.time() + self.total_timeout
                    if self.total_timeout > 0 else None)
        while not cond():
            if deadline is not None and time.time() > deadline:
                raise GateTimeout(
                    f"task {self.task_id}: {what} gate timed out after "
                    f"{self.total_timeout}s")
            time.sleep(self.wait_interval)

    def _gate(self, strategy: str, round_idx: int, phase: str) -> None:
        if not strategy:
            return
        if strategy in self.custom_gates:
            self._poll(lambda: self.custom_gates[strategy](round_idx),
                       f"{phase}:{strategy}")
            return
        if strategy == "waiting_for_global_aggregation":
            if self.selection_round_fn is None:
                return
            if phase == "start":
                # start once the service reaches this round
                self._poll(lambda: self.selection_round_fn() >= round_idx,
                           "start:selection")
            else:
                # stop waits until the service advances past this round
                self._poll(lambda: self.selection_round_fn() > round_idx,
                           "stop:selection")
            return
        if strategy in ("sample_and_aggregation",
                        "sample_dc_and_aggregation"):
            if phase == "stop":
                os.makedirs(self.work_dir, exist_ok=True)
                flag = os.path.join(self.work_dir, "simulation_finished.txt")
                with open(flag, "w") as f:
                    f.write(f"{self.task_id} round {round_idx}\n")
                done = os.path.join(self.work_dir, "aggregation_finished.txt")
                self._poll(lambda: os.path.exists(done), "stop:aggregation")
                try:
                    os.remove(done)
                except OSError:
                    pass
            return
        self.log.warning(self.task_id, "Engine", "operatorflow",
                         f"unknown flow strategy {strategy!r}; not gating")

    # reference OperatorFlow.start / .stop
    def start(self, round_idx: int) -> None:
        self._gate(self.start_strategy, round_idx, "start")

    def stop(self, round_idx: int) -> None:
        self._gate(self.stop_strategy, round_idx, "stop")
