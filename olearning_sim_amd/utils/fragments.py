"""Fragment repositories: consuming training-result fragments.

Parity with the reference's FragmentRepo implementations
(ols_core/ofl_commons/infrastructure/FragmentRepo/json_fragment_repo.py
and proto_fragment_repo.py — Pulsar consumers yielding per-device
training-result "fragments" to an aggregation service; their abstract
base is missing from the open-source drop, SURVEY.md top notes).  Here
a fragment repo consumes the deviceflow OutboundRoom: the aggregation
side iterates fragments as dicts (JSON flavour) or raw payloads.
"""

from __future__ import annotations

import base64
import json
from typing import Any, Dict, Iterator, Optional

from ..deviceflow.rooms import Message, OutboundRoom


class FragmentRepoBase:
    """The missing abstract base, reconstructed: a pull-iterator of
    training-result fragments with explicit ack semantics."""

    def __init__(self, outbound: OutboundRoom):
        self._outbound = outbound

    def receive(self, timeout: float = 1.0) -> Optional[Message]:
        msgs = self._outbound.drain(1)
        return msgs[0] if msgs else None

    def decode(self, msg: Message) -> Any:
        raise NotImplementedError

    def fragments(self, max_items: int = 1_000_000) -> Iterator[Any]:
        for _ in range(max_items):
            msg = self.receive()
            if msg is None:
                return
            yield self.decode(msg)


class JsonFragmentRepo(FragmentRepoBase):
    """JSON fragments; tolerates the reference's base64-wrapped payloads
    (message_producer.py:59-78 wraps outbound JSON in base64)."""

    def decode(self, msg: Message) -> Dict[str, Any]:
        payload = msg.payload
        if isinstance(payload, (bytes, str)):
            try:
                raw = (base64.b64decode(payload)
                       if not str(payload).lstrip().startswith("{")
                       else payload)
                payload = json.loads(raw)
            except Exception:
                payload = {"raw": payload}
        return {"routing_key": msg.routing_key,
                "compute_resource": msg.compute_resource,
                "payload": payload}


class TensorFragmentRepo(FragmentRepoBase):
    """Binary fragments carrying tensors (the proto flavour's role):
    payloads are torch tensors or state dicts, passed through as-is —
    in-process transport needs no serialisation."""

    def decode(self, msg: Message) -> Dict[str, Any]:
        return {"routing_key": msg.routing_key,
                "compute_resource": msg.compute_resource,
                "payload": msg.payload}
