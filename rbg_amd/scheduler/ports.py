"""Local port allocator — single-node realization of the reference's
cluster port pool (reference pkg/port-allocator/{port_allocator.go:40-130,
manager.go:36-240, parser.go:27-205, random.go:43-87}).

Keeps the annotation API shape: the request rides in the
`rbg.workloads.x-k8s.io/port-allocation` annotation as JSON
`[{"name": "http", "scope": "RoleScoped"|"PodScoped", "count": 1}, ...]`;
RoleScoped ports are allocated once per RoleInstanceSet, PodScoped per engine
process.  Allocated ports are injected as PORT_{NAME} env and recorded so
restarts reuse them.  Bind-probing avoids colliding with unrelated listeners.
"""
from __future__ import annotations

import json
import socket
import threading
from dataclasses import dataclass
from typing import Dict, List

from ..api import constants as C

SCOPE_ROLE = "RoleScoped"
SCOPE_POD = "PodScoped"


class PortExhausted(RuntimeError):
    pass


@dataclass
class PortRequest:
    name: str
    scope: str = SCOPE_ROLE
    count: int = 1


def parse_requests(annotations: Dict[str, str]) -> List[PortRequest]:
    raw = annotations.get(C.ANNO_PORT_ALLOCATION, "")
    if not raw:
        return []
    try:
        items = json.loads(raw)
    except json.JSONDecodeError as e:
        raise ValueError(f"bad {C.ANNO_PORT_ALLOCATION} annotation: {e}") from e
    out = []
    for item in items:
        scope = item.get("scope", SCOPE_ROLE)
        if scope not in (SCOPE_ROLE, SCOPE_POD):
            raise ValueError(f"port request {item.get('name')!r}: bad scope {scope!r}")
        out.append(PortRequest(name=item["name"], scope=scope,
                               count=int(item.get("count", 1))))
    return out


def _port_is_free(port: int) -> bool:
    with socket.socket(socket.AF_INET, socket.SOCK_STREAM) as s:
        s.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        try:
            s.bind(("127.0.0.1", port))
            return True
        except OSError:
            return False


class PortAllocator:
    def __init__(self, start: int = 30000, end: int = 40000,
                 probe: bool = True):
        self.start, self.end = start, end
        self.probe = probe
        self._lock = threading.Lock()
        self._next = start
        self._in_use: Dict[int, str] = {}
        self._by_key: Dict[str, List[int]] = {}

    def allocate(self, key: str, count: int = 1) -> List[int]:
        """Idempotent per key: re-requesting returns the same ports."""
        with self._lock:
            if key in self._by_key and len(self._by_key[key]) >= count:
                return self._by_key[key][:count]
            got = list(self._by_key.get(key, []))
            scanned = 0
            span = self.end - self.start
            while len(got) < count:
                if scanned >= span:
                    raise PortExhausted(f"no free port in [{self.start},{self.end})")
                p = self._next
                self._next += 1
                if self._next >= self.end:
                    self._next = self.start
                scanned += 1
                if p in self._in_use:
                    continue
                if self.probe and not _port_is_free(p):
                    continue
                self._in_use[p] = key
                got.append(p)
            self._by_key[key] = got
            return got

    def release(self, key: str) -> None:
        with self._lock:
            for p in self._by_key.pop(key, []):
                self._in_use.pop(p, None)

    def env_for(self, ports: Dict[str, List[int]]) -> Dict[str, str]:
        env = {}
        for name, plist in ports.items():
            env[f"PORT_{name.upper().replace('-', '_')}"] = ",".join(map(str, plist))
        return env
