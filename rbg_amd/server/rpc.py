"""Minimal length-prefixed JSON RPC over localhost TCP.

The dataplane addressing of the reference is headless-service DNS + HTTP
(SURVEY §1); on one node that becomes 127.0.0.1:port sockets carried in the
discovery config.  This module is the transport used between router,
prefill and decode engine processes for CONTROL messages only — KV bytes
move over RCCL/xGMI (parallel/kv_transfer.py), never through these sockets.
"""
from __future__ import annotations

import json
import socket
import socketserver
import struct
import threading
from typing import Any, Callable, Dict, Optional

_LEN = struct.Struct("!I")
# control messages are small (KV bytes never cross these sockets); a frame
# beyond this is a corrupt/hostile peer, not a real request
MAX_FRAME = 1 << 28     # 256 MB


def _send_msg(sock: socket.socket, obj: Any) -> None:
    data = json.dumps(obj).encode()
    sock.sendall(_LEN.pack(len(data)) + data)


def _recv_msg(sock: socket.socket) -> Any:
    hdr = _recv_exact(sock, _LEN.size)
    (n,) = _LEN.unpack(hdr)
    if n > MAX_FRAME:
        raise ConnectionError(f"oversized rpc frame ({n} bytes)")
    try:
        return json.loads(_recv_exact(sock, n).decode())
    except (ValueError, UnicodeDecodeError) as e:
        # corrupt frame: the stream is unrecoverable — drop the connection
        raise ConnectionError(f"malformed rpc frame: {e}") from e


def _recv_exact(sock: socket.socket, n: int) -> bytes:
    buf = b""
    while len(buf) < n:
        chunk = sock.recv(n - len(buf))
        if not chunk:
            raise ConnectionError("peer closed")
        buf += chunk
    return buf


class RpcServer:
    """Threaded server dispatching {"method": ..., "params": {...}} requests
    to registered handlers; replies {"ok": true, "result": ...} or
    {"ok": false, "error": ...}."""

    def __init__(self, host: str = "127.0.0.1", port: int = 0):
        self.handlers: Dict[str, Callable[..., Any]] = {}
        outer = self

        class Handler(socketserver.BaseRequestHandler):
            def handle(self):
                while True:
                    try:
                        req = _recv_msg(self.request)
                    except (ConnectionError, OSError):
                        return
                    try:
                        fn = outer.handlers[req["method"]]
                        result = fn(**req.get("params", {}))
                        _send_msg(self.request, {"ok": True, "result": result})
                    except Exception as e:  # noqa: BLE001
                        _send_msg(self.request,
                                  {"ok": False, "error": repr(e)})

        class Server(socketserver.ThreadingTCPServer):
            allow_reuse_address = True
            daemon_threads = True

        self._server = Server((host, port), Handler)
        self.port = self._server.server_address[1]
        self._thread: Optional[threading.Thread] = None

    def register(self, name: str, fn: Callable[..., Any]) -> None:
        self.handlers[name] = fn

    def start(self) -> None:
        self._thread = threading.Thread(target=self._server.serve_forever,
                                        daemon=True)
        self._thread.start()

    def stop(self) -> None:
        self._server.shutdown()
        self._server.server_close()


class RpcClient:
    def __init__(self, host: str, port: int, timeout: float = 60.0):
        self.addr = (host, port)
        self.timeout = timeout
        self._lock = threading.Lock()
        self._sock: Optional[socket.socket] = None

    def _connect(self) -> socket.socket:
        if self._sock is None:
            s = socket.create_connection(self.addr, timeout=self.timeout)
            s.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
            self._sock = s
        return self._sock

    def call(self, method: str, **params: Any) -> Any:
        with self._lock:
            for attempt in (0, 1):
                try:
                    sock = self._connect()
                    _send_msg(sock, {"method": method, "params": params})
                    resp = _recv_msg(sock)
                    break
                except (ConnectionError, OSError):
                    self.close_locked()
                    if attempt:
                        raise
            if not resp["ok"]:
                raise RuntimeError(f"rpc {method} failed: {resp['error']}")
            return resp["result"]

    def close_locked(self) -> None:
        if self._sock is not None:
            try:
                self._sock.close()
            except OSError:
                pass
            self._sock = None

    def close(self) -> None:
        with self._lock:
            self.close_locked()
