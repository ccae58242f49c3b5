"""Deviceflow service: the gradient house's lifecycle and message plane.

Parity with the reference's DeviceFlowGrpcService + DeviceFlow state
machine (ols_core/deviceflow/grpc_service/deviceflow_server.py:43-473,
non_grpc/deviceflow.py:29-197):

- RegisterTask / UnRegisterTask: registry of tasks + their compute
  resources (both simulation sides must register before flows open).
- NotifyStart(task, operator, round, compute_resource, strategy,
  outbound): creates/merges the flow's params; a flow_id is
  f"{task}_{operator}_{round}"; real-time strategies get a running
  Dispatcher immediately, flow strategies at completion.
- NotifyComplete: when every registered compute resource has completed,
  the dispatcher is released (flow schedules start draining).
- CheckDeviceflowDispatchFinished: poll used by the task manager before
  releasing a task's resources.
- Sorter thread: admits inbound messages into their flow's shelf only
  between NotifyStart and NotifyComplete, else discards
  (non_grpc/sorter.py:56-91).
"""

from __future__ import annotations

import json
import random
import threading
from typing import Dict, List, Optional, Tuple

from ..utils.logging import Logger
from ..utils.sqlite_repo import SqlTableRepo
from .dispatcher import Dispatcher
from .registry import TaskOrientedDeviceFlowRegistry
from .rooms import InboundRoom, Message, OutboundRoom, ShelfRoom

_FLOW_COLUMNS = {
    "flow_id": "TEXT", "task_id": "TEXT", "strategy": "TEXT",
    "outbound_service": "TEXT", "resources": "TEXT",
    "notify_start_called": "TEXT", "notify_complete_called": "TEXT",
    "is_finished": "INTEGER",
}


class FlowState:
    """flow_params of the reference (deviceflow.py:29-121)."""

    def __init__(self, task_id: str, flow_id: str, strategy: str,
                 outbound_service: str, resources: List[str]):
        self.task_id = task_id
        self.flow_id = flow_id
        self.strategy = strategy
        self.outbound_service = outbound_service
        self.resources = list(resources)
        self.notify_start_called = {r: False for r in resources}
        self.notify_complete_called = {r: False for r in resources}
        self.is_finished = False
        self.to_sort = True
        self.to_dispatch = False

    def all_started(self) -> bool:
        return all(self.notify_start_called.values())

    def all_completed(self) -> bool:
        return all(self.notify_complete_called.values())

    def to_row(self) -> Dict[str, object]:
        return {
            "flow_id": self.flow_id, "task_id": self.task_id,
            "strategy": self.strategy,
            "outbound_service": self.outbound_service,
            "resources": json.dumps(self.resources),
            "notify_start_called": json.dumps(self.notify_start_called),
            "notify_complete_called": json.dumps(self.notify_complete_called),
            "is_finished": int(self.is_finished),
        }

    @classmethod
    def from_row(cls, row: Dict[str, object]) -> "FlowState":
        f = cls(str(row["task_id"]), str(row["flow_id"]),
                str(row["strategy"] or ""),
                str(row["outbound_service"] or ""),
                json.loads(str(row["resources"] or "[]")))
        f.notify_start_called.update(
            json.loads(str(row["notify_start_called"] or "{}")))
        f.notify_complete_called.update(
            json.loads(str(row["notify_complete_called"] or "{}")))
        f.is_finished = bool(row.get("is_finished"))
        return f


class DeviceFlowService:
    def __init__(self, db_path: str = ":memory:", time_scale: float = 1.0,
                 seed: Optional[int] = None, auto_start: bool = True):
        self.registry = TaskOrientedDeviceFlowRegistry(db_path)
        self.inbound = InboundRoom()
        self.shelf = ShelfRoom()
        self.outbound = OutboundRoom()
        self.flows: Dict[str, FlowState] = {}
        self.dispatchers: Dict[str, Dispatcher] = {}
        # released flows keep their dispatch curves for dashboards
        # (reference operation/accumulated_amount_table demo rows)
        self.dispatch_history: Dict[str, List[tuple]] = {}
        self.time_scale = time_scale
        self._ws_producers: Dict[str, object] = {}
        self._rng = random.Random(seed)
        self._lock = threading.RLock()
        self._stop = threading.Event()
        self.log = Logger.shared()
        self._sorter_thread: Optional[threading.Thread] = None
        self._flow_repo = SqlTableRepo(db_path, "deviceflow_flow_table",
                                       _FLOW_COLUMNS, primary_key="flow_id")
        self._initiate_from_repo()
        if auto_start:
            self.start()

    def _initiate_from_repo(self) -> None:
        """Crash recovery: rebuild unfinished flows (and their
        dispatchers) from the persisted flow table, like the reference's
        initiate_from_repo (deviceflow_server.py:83-164).  Only tasks
        still present in the registry are revived — a released task's
        rows are stale and are dropped."""
        for row in self._flow_repo.get_all_rows():
            flow = FlowState.from_row(row)
            if flow.is_finished or not self.registry.is_registered(flow.task_id):
                self._flow_repo.delete_item("flow_id", flow.flow_id)
                continue
            self.flows[flow.flow_id] = flow
            self.shelf.ensure_shelf(flow.flow_id)
            if flow.strategy:
                d = Dispatcher(flow.flow_id, flow.strategy, self.shelf,
                               self.outbound, self.time_scale, self._rng)
                self.dispatchers[flow.flow_id] = d
                d.start()
                if flow.all_completed():
                    flow.to_sort = False
                    d.release_event.set()

    def _persist_flow(self, flow: FlowState) -> None:
        self._flow_repo.upsert_item("flow_id", flow.to_row())

    # -- lifecycle --------------------------------------------------------
    def start(self) -> None:
        self._sorter_thread = threading.Thread(target=self._sort_loop,
                                               daemon=True)
        self._sorter_thread.start()

    def shutdown(self) -> None:
        self._stop.set()
        for d in self.dispatchers.values():
            d.stop_event.set()
            d.release_event.set()
        for prod in self._ws_producers.values():
            try:
                prod.close()
            except Exception:
                pass

    # -- RPC surface ------------------------------------------------------
    def register_task(self, task_id: str,
                      total_compute_resources: List[str]) -> bool:
        return self.registry.register_task(task_id, total_compute_resources)

    def unregister_task(self, task_id: str) -> bool:
        with self._lock:
            for fid in [f for f in self.flows
                        if self.flows[f].task_id == task_id]:
                self._release_flow(fid)
            return self.registry.unregister_task(task_id)

    def notify_start(self, task_id: str, operator_name: str, round_idx: int,
                     compute_resource: str, strategy: str = "",
                     outbound_service: str = "") -> Optional[str]:
        """Returns the flow_id, or None when the task is unregistered
        (deviceflow_server.py:182-289)."""
        resources = self.registry.resources(task_id)
        if resources is None:
            return None
        flow_id = f"{task_id}_{operator_name}_{round_idx}"
        with self._lock:
            flow = self.flows.get(flow_id)
            if flow is None:
                flow = FlowState(task_id, flow_id, strategy,
                                 outbound_service, resources)
                self.flows[flow_id] = flow
                self.shelf.ensure_shelf(flow_id)
                # a ws:// outbound_service gets a live producer (the
                # reference builds a WebsocketProducer from the flow's
                # outbound_service, deviceflow_server.py:182-289)
                if outbound_service.startswith("ws://"):
                    self.attach_websocket_outbound(outbound_service)
            if strategy:
                flow.strategy = strategy
            if compute_resource not in flow.notify_start_called:
                return None
            flow.notify_start_called[compute_resource] = True
            if flow_id not in self.dispatchers and flow.strategy:
                d = Dispatcher(flow_id, flow.strategy, self.shelf,
                               self.outbound, self.time_scale, self._rng)
                self.dispatchers[flow_id] = d
                d.start()
            self._persist_flow(flow)
        return flow_id

    def notify_complete(self, task_id: str, operator_name: str,
                        round_idx: int, compute_resource: str) -> bool:
        flow_id = f"{task_id}_{operator_name}_{round_idx}"
        with self._lock:
            flow = self.flows.get(flow_id)
            if flow is None or compute_resource not in flow.notify_complete_called:
                return False
            flow.notify_complete_called[compute_resource] = True
            if flow.all_completed():
                flow.to_sort = False
                d = self.dispatchers.get(flow_id)
                if d is not None:
                    d.release_event.set()
            self._persist_flow(flow)
        return True

    def check_dispatch_finished(self, task_id: str) -> bool:
        """True when every one of the task's flows has drained
        (deviceflow_server.py:403-427)."""
        with self._lock:
            flows = [f for f in self.flows.values() if f.task_id == task_id]
            if not flows:
                return True
            for f in flows:
                d = self.dispatchers.get(f.flow_id)
                if d is not None and not d.finished.is_set():
                    return False
                if self.shelf.depth(f.flow_id) > 0:
                    return False
            return True

    def flow_release_step(self) -> List[str]:
        """Reap finished dispatchers (flow_release thread,
        deviceflow_server.py:453-473)."""
        released = []
        with self._lock:
            for fid, d in list(self.dispatchers.items()):
                if d.finished.is_set():
                    self._release_flow(fid)
                    released.append(fid)
        return released

    def _release_flow(self, flow_id: str) -> None:
        d = self.dispatchers.pop(flow_id, None)
        if d is not None:
            d.stop_event.set()
            d.release_event.set()
            if d.history:
                self.dispatch_history[flow_id] = list(d.history)
        flow = self.flows.pop(flow_id, None)
        if flow is not None:
            flow.is_finished = True
        self._flow_repo.delete_item("flow_id", flow_id)
        self.shelf.remove_shelf(flow_id)

    def dispatch_curve(self, task_id: str) -> Dict[str, List[Dict[str, float]]]:
        """Per-flow dispatch curves: the per-slot (virtual time, sent,
        dropped, cumulative) rows the reference wrote to its demo
        operation_amount / accumulated_amount tables
        (dispatcher.py:254-395)."""
        out: Dict[str, List[Dict[str, float]]] = {}
        with self._lock:
            sources: List[Tuple[str, List[tuple]]] = [
                (fid, list(d.history)) for fid, d in self.dispatchers.items()]
            sources += list(self.dispatch_history.items())
        for fid, hist in sources:
            if fid.startswith(task_id + "_"):
                out[fid] = [{"t": t, "sent": s, "dropped": dr,
                             "accumulated": acc} for t, s, dr, acc in hist]
        return out

    # -- connection info (GetDeviceflowPulsarClient /
    # GetDeviceflowWebsocket, deviceflow_server.py:167-179) --------------
    def inbound_info(self) -> Dict[str, object]:
        """How producers reach the gradient house.  The reference hands
        out a Pulsar url+topic; here the inbound is the in-process
        queue, published to via the /deviceflow/publish route."""
        return {"kind": "inproc", "endpoint": "/deviceflow/publish",
                "queue_depth": self.inbound.qsize()}

    def outbound_info(self) -> Dict[str, object]:
        """Where aggregated fragments leave the gradient house (the
        reference's websocket/Pulsar outbound)."""
        return {"kind": "inproc", "endpoint": "/deviceflow/outbound",
                "queue_depth": self.outbound.qsize()}

    def attach_websocket_outbound(self, url: str):
        """Forward every outbound message to a ws:// consumer (the
        reference's WebsocketProducer, message_producer.py:59-78 — the
        external aggregation service).  Returns the producer; one per
        URL, reused."""
        prod = self._ws_producers.get(url)
        if prod is None:
            from .producers import WebSocketProducer
            prod = WebSocketProducer(url)
            self._ws_producers[url] = prod
            self.outbound.subscribe(prod)
        return prod

    # -- data plane -------------------------------------------------------
    def publish(self, routing_key: str, compute_resource: str,
                payload=None) -> None:
        self.inbound.publish(Message(routing_key, compute_resource, payload))

    def drain_inbound(self, timeout: float = 5.0) -> bool:
        """Wait until the sorter has absorbed the inbound queue — a
        producer calls this before NotifyComplete so its own messages
        are not discarded by the completion flip (the reference relies
        on Pulsar consume ordering for the same guarantee)."""
        import time as _t
        t0 = _t.time()
        while self.inbound.qsize() > 0:
            if _t.time() - t0 > timeout:
                return False
            _t.sleep(0.002)
        return True

    def _should_put(self, msg: Message) -> bool:
        """Admission: flow exists, resource known, between start and
        complete (sorter.py:56-69)."""
        flow = self.flows.get(msg.routing_key)
        if flow is None or not flow.to_sort:
            return False
        started = flow.notify_start_called.get(msg.compute_resource)
        completed = flow.notify_complete_called.get(msg.compute_resource)
        return bool(started) and not bool(completed)

    def _sort_loop(self) -> None:
        while not self._stop.is_set():
            msg = self.inbound.receive(timeout=0.05)
            if msg is None:
                continue
            with self._lock:
                admit = self._should_put(msg)
            if admit:
                self.shelf.put_on_shelf(msg.routing_key, msg)
            # else: discarded, like the reference's sorter
