"""Minimal RFC 6455 WebSocket server + client (text frames).

The reference pushes aggregation traffic to external services over
WebSocket (ols_core/deviceflow/non_grpc/message_producer.py:59-78) and
polls a selection service over WebSocket for round gating
(ols_core/taskMgr/utils/operatorflow.py:158-237).  The image ships no
websocket library, so the handshake (HTTP Upgrade + Sec-WebSocket
SHA-1/base64 accept) and framing (FIN text/close/ping frames, client
masking) are implemented here directly over sockets — enough for the
gradient-house outbound producer and the selection-service poll, and
for tests to stand up a real socket pair.
"""

from __future__ import annotations

import base64
import hashlib
import os
import socket
import struct
import threading
from typing import Callable, Optional, Tuple

_GUID = "258EAFA5-E914-47DA-95CA-C5AB0DC85B11"

OP_TEXT = 0x1
OP_CLOSE = 0x8
OP_PING = 0x9
OP_PONG = 0xA


def _accept_key(key: str) -> str:
    digest = hashlib.sha1((key + _GUID).encode()).digest()
    return base64.b64encode(digest).decode()


def _encode_frame(payload: bytes, opcode: int = OP_TEXT,
                  mask: bool = False) -> bytes:
    head = bytearray([0x80 | opcode])
    n = len(payload)
    mbit = 0x80 if mask else 0
    if n < 126:
        head.append(mbit | n)
    elif n < 65536:
        head.append(mbit | 126)
        head += struct.pack(">H", n)
    else:
        head.append(mbit | 127)
        head += struct.pack(">Q", n)
    if mask:
        key = os.urandom(4)
        head += key
        payload = bytes(b ^ key[i % 4] for i, b in enumerate(payload))
    return bytes(head) + payload


def _read_exact(sock: socket.socket, n: int) -> bytes:
    buf = b""
    while len(buf) < n:
        chunk = sock.recv(n - len(buf))
        if not chunk:
            raise ConnectionError("websocket peer closed")
        buf += chunk
    return buf


def _decode_frame(sock: socket.socket) -> Tuple[int, bytes]:
    b0, b1 = _read_exact(sock, 2)
    opcode = b0 & 0x0F
    masked = bool(b1 & 0x80)
    n = b1 & 0x7F
    if n == 126:
        n = struct.unpack(">H", _read_exact(sock, 2))[0]
    elif n == 127:
        n = struct.unpack(">Q", _read_exact(sock, 8))[0]
    key = _read_exact(sock, 4) if masked else None
    payload = _read_exact(sock, n) if n else b""
    if key:
        payload = bytes(b ^ key[i % 4] for i, b in enumerate(payload))
    return opcode, payload


class WSConnection:
    """One established connection (either side).  ``mask`` per RFC:
    client->server frames are masked."""

    def __init__(self, sock: socket.socket, mask: bool):
        self.sock = sock
        self.mask = mask
        self._lock = threading.Lock()

    def send_text(self, text: str) -> None:
        with self._lock:
            self.sock.sendall(_encode_frame(text.encode(), OP_TEXT,
                                            self.mask))

    def recv_text(self, timeout: Optional[float] = None) -> Optional[str]:
        """Next text payload; None on clean close.  Answers pings."""
        self.sock.settimeout(timeout)
        while True:
            op, payload = _decode_frame(self.sock)
            if op == OP_TEXT:
                return payload.decode()
            if op == OP_PING:
                with self._lock:
                    self.sock.sendall(_encode_frame(payload, OP_PONG,
                                                    self.mask))
                continue
            if op == OP_CLOSE:
                try:
                    with self._lock:
                        self.sock.sendall(_encode_frame(b"", OP_CLOSE,
                                                        self.mask))
                except OSError:
                    pass
                return None

    def close(self) -> None:
        try:
            with self._lock:
                self.sock.sendall(_encode_frame(b"", OP_CLOSE, self.mask))
        except OSError:
            pass
        try:
            self.sock.close()
        except OSError:
            pass


class WebSocketServer:
    """Accepts ws:// connections; ``handler(conn)`` runs per connection
    in a daemon thread."""

    def __init__(self, handler: Callable[[WSConnection], None],
                 host: str = "127.0.0.1", port: int = 0):
        self.handler = handler
        self._srv = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        self._srv.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        self._srv.bind((host, port))
        self._srv.listen(8)
        self.host, self.port = self._srv.getsockname()
        self._stop = threading.Event()
        self._thread = threading.Thread(target=self._accept_loop,
                                        daemon=True)
        self._thread.start()

    @property
    def url(self) -> str:
        return f"ws://{self.host}:{self.port}"

    def _accept_loop(self) -> None:
        while not self._stop.is_set():
            try:
                self._srv.settimeout(0.2)
                sock, _ = self._srv.accept()
            except socket.timeout:
                continue
            except OSError:
                return
            threading.Thread(target=self._handshake_and_run,
                             args=(sock,), daemon=True).start()

    def _handshake_and_run(self, sock: socket.socket) -> None:
        try:
            req = b""
            while b"\r\n\r\n" not in req:
                chunk = sock.recv(4096)
                if not chunk:
                    return
                req += chunk
            headers = {}
            for line in req.split(b"\r\n")[1:]:
                if b":" in line:
                    k, v = line.split(b":", 1)
                    headers[k.strip().lower()] = v.strip()
            key = headers.get(b"sec-websocket-key", b"").decode()
            resp = ("HTTP/1.1 101 Switching Protocols\r\n"
                    "Upgrade: websocket\r\n"
                    "Connection: Upgrade\r\n"
                    f"Sec-WebSocket-Accept: {_accept_key(key)}\r\n\r\n")
            sock.sendall(resp.encode())
            self.handler(WSConnection(sock, mask=False))
        except (ConnectionError, OSError):
            pass
        finally:
            try:
                sock.close()
            except OSError:
                pass

    def shutdown(self) -> None:
        self._stop.set()
        try:
            self._srv.close()
        except OSError:
            pass


def connect(url: str, timeout: float = 5.0) -> WSConnection:
    """Open a ws://host:port[/path] connection (client side)."""
    assert url.startswith("ws://"), f"only ws:// supported: {url!r}"
    rest = url[5:]
    path = "/"
    if "/" in rest:
        hostport, path = rest.split("/", 1)
        path = "/" + path
    else:
        hostport = rest
    host, _, port = hostport.partition(":")
    sock = socket.create_connection((host, int(port or 80)),
                                    timeout=timeout)
    key = base64.b64encode(os.urandom(16)).decode()
    req = (f"GET {path} HTTP/1.1\r\n"
           f"Host: {hostport}\r\n"
           "Upgrade: websocket\r\n"
           "Connection: Upgrade\r\n"
           f"Sec-WebSocket-Key: {key}\r\n"
           "Sec-WebSocket-Version: 13\r\n\r\n")
    sock.sendall(req.encode())
    resp = b""
    while b"\r\n\r\n" not in resp:
        chunk = sock.recv(4096)
        if not chunk:
            raise ConnectionError("websocket handshake failed")
        resp += chunk
    status = resp.split(b"\r\n", 1)[0]
    if b"101" not in status:
        raise ConnectionError(f"websocket upgrade rejected: {status!r}")
    expect = _accept_key(key).encode()
    if expect not in resp:
        raise ConnectionError("websocket accept key mismatch")
    return WSConnection(sock, mask=True)
