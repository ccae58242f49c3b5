"""Outbound message producers.

The reference forwards dispatched gradient-house messages to external
aggregation services through Pulsar or WebSocket producers
(ols_core/deviceflow/non_grpc/message_producer.py:42-78).  In-process
subscribers and the HTTP outbound drain cover local consumers
(api/server.py); this module adds the WebSocket producer for consumers
OUTSIDE the process: each outbound message is pushed as the reference's
wire shape — a JSON object whose ``message`` field is the
base64-encoded payload (message_producer.py:59-78)."""

from __future__ import annotations

import base64
import json
import threading
from typing import Optional

from ..utils import ws
from ..utils.logging import Logger


class WebSocketProducer:
    """Forwards OutboundRoom messages to a ws:// endpoint.

    Connection is lazy and re-established on failure; a send that
    cannot reach the endpoint is dropped with a log line (the reference
    producer raises into the dispatcher thread; dropping keeps the
    dispatch loop alive, matching its swallow-all logging policy)."""

    def __init__(self, url: str):
        self.url = url
        self._conn: Optional[ws.WSConnection] = None
        self._lock = threading.Lock()
        self.log = Logger.shared()
        self.sent = 0
        self.dropped = 0

    def _ensure(self) -> Optional[ws.WSConnection]:
        if self._conn is None:
            try:
                self._conn = ws.connect(self.url, timeout=2.0)
            except (OSError, ConnectionError) as e:
                self.log.warning("", "DeviceFlow", "ws_producer",
                                 f"connect {self.url} failed: {e}")
                return None
        return self._conn

    def __call__(self, msg) -> None:
        payload = base64.b64encode(
            json.dumps(msg.payload).encode()).decode()
        body = json.dumps({"routing_key": msg.routing_key,
                           "compute_resource": msg.compute_resource,
                           "message": payload})
        with self._lock:
            conn = self._ensure()
            if conn is None:
                self.dropped += 1
                return
            try:
                conn.send_text(body)
                self.sent += 1
            except (OSError, ConnectionError) as e:
                self.log.warning("", "DeviceFlow", "ws_producer",
                                 f"send to {self.url} failed: {e}")
                self._conn = None
                self.dropped += 1

    def close(self) -> None:
        with self._lock:
            if self._conn is not None:
                self._conn.close()
                self._conn = None
