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

import random
import threading
from typing import Dict, List, Optional

from ..utils.logging import Logger
from .dispatcher import Dispatcher
from .registry import TaskOrientedDeviceFlowRegistry
from .rooms import InboundRoom, Message, OutboundRoom, ShelfRoom


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


class DeviceFlowService:
    def __init__(self, db_path: str = ":memory:", time_scale: float = 1.0,
                 seed: Optional[int] = None, auto_start: bool = True):
        self.registry = TaskOrientedDeviceFlowRegistry(db_path)
        self.inbound = InboundRoom()
        self.shelf = ShelfRoom()
        self.outbound = OutboundRoom()
        self.flows: Dict[str, FlowState] = {}
        self.dispatchers: Dict[str, Dispatcher] = {}
        self.time_scale = time_scale
        self._rng = random.Random(seed)
        self._lock = threading.RLock()
        self._stop = threading.Event()
        self.log = Logger.shared()
        self._sorter_thread: Optional[threading.Thread] = None
        if auto_start:
            self.start()

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
        flow = self.flows.pop(flow_id, None)
        if flow is not None:
            flow.is_finished = True
        self.shelf.remove_shelf(flow_id)

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
