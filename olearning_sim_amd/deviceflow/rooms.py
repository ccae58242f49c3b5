"""Message rooms: the gradient house's staging queues.

Parity with the reference's Pulsar wrappers
(ols_core/deviceflow/non_grpc/bound_room.py:29-112 InboundRoom /
OutboundRoom and shelf_room.py:8-131 ShelfRoom): the inbound room is
the shared topic every simulated device publishes gradient messages to;
the shelf room holds one staging queue per flow_id; the outbound room
is where the dispatcher forwards to the aggregation consumer.  Pulsar
topics become bounded in-process queues — same at-least-once semantics
within the node, no broker.
"""

from __future__ import annotations

import queue
import threading
from typing import Any, Dict, List, Optional


class Message:
    """A gradient-house message (deviceflow/utils/message.py fields)."""

    __slots__ = ("routing_key", "compute_resource", "payload")

    def __init__(self, routing_key: str, compute_resource: str,
                 payload: Any = None):
        self.routing_key = routing_key          # f"{task}_{operator}_{round}"
        self.compute_resource = compute_resource  # logical_/device_simulation
        self.payload = payload

    def __repr__(self):
        return f"Message({self.routing_key}, {self.compute_resource})"


class InboundRoom:
    """The shared inbound topic (bound_room.py InboundRoom)."""

    def __init__(self, maxsize: int = 1_000_000):
        self._q: "queue.Queue[Message]" = queue.Queue(maxsize)

    def publish(self, msg: Message) -> None:
        self._q.put(msg)

    def receive(self, timeout: float = 1.0) -> Optional[Message]:
        try:
            return self._q.get(timeout=timeout)
        except queue.Empty:
            return None

    def qsize(self) -> int:
        return self._q.qsize()


class ShelfRoom:
    """Per-flow staging shelves (shelf_room.py Shelf/ShelfRoom)."""

    def __init__(self):
        self._shelves: Dict[str, "queue.Queue[Message]"] = {}
        self._lock = threading.Lock()

    def ensure_shelf(self, flow_id: str) -> None:
        with self._lock:
            self._shelves.setdefault(flow_id, queue.Queue())

    def put_on_shelf(self, flow_id: str, msg: Message) -> None:
        self.ensure_shelf(flow_id)
        self._shelves[flow_id].put(msg)

    def take(self, flow_id: str, max_items: int,
             timeout: float = 0.0) -> List[Message]:
        shelf = self._shelves.get(flow_id)
        if shelf is None:
            return []
        out: List[Message] = []
        for _ in range(max_items):
            try:
                out.append(shelf.get(timeout=timeout) if timeout
                           else shelf.get_nowait())
            except queue.Empty:
                break
        return out

    def depth(self, flow_id: str) -> int:
        shelf = self._shelves.get(flow_id)
        return shelf.qsize() if shelf else 0

    def remove_shelf(self, flow_id: str) -> None:
        with self._lock:
            self._shelves.pop(flow_id, None)

    def flow_ids(self) -> List[str]:
        with self._lock:
            return list(self._shelves)


class OutboundRoom:
    """The outbound delivery queue (bound_room.py OutboundRoom /
    message_producer.py PulsarClientProducer,WebsocketProducer).

    ``subscribe`` registers a consumer callback (the in-process
    aggregation service); without one, messages collect in a queue that
    tests and external pollers drain."""

    def __init__(self):
        self._q: "queue.Queue[Message]" = queue.Queue()
        self._consumers: List[Any] = []

    def subscribe(self, callback) -> None:
        self._consumers.append(callback)

    def send(self, msg: Message) -> None:
        if self._consumers:
            for cb in self._consumers:
                cb(msg)
        else:
            self._q.put(msg)

    def drain(self, max_items: int = 1_000_000) -> List[Message]:
        out = []
        for _ in range(max_items):
            try:
                out.append(self._q.get_nowait())
            except queue.Empty:
                break
        return out

    def qsize(self) -> int:
        return self._q.qsize()
