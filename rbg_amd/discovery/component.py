"""Intra-role component discovery + start ordering.

Mirrors reference pkg/component-discovery/{component_discovery.go:33-204,
component_lifecycle.go:66-262}: two JSON annotations on a RoleInstance (or
its components) drive
  - `rbg.workloads.x-k8s.io/component-discovery`:
      {"env": [{"name": "LEADER_ADDR", "component": "leader"}]}
    -> inject sibling-component address/port env at worker spawn,
  - `rbg.workloads.x-k8s.io/component-depends-on`:
      {"worker": ["leader"]}
    -> create a component's workers only once its dependency components are
      Ready (start ordering); cycles fall back to parallel start
      (reference behavior).
"""
from __future__ import annotations

import json
import logging
from typing import Dict, List

from ..api import constants as C

log = logging.getLogger(__name__)


def parse_depends_on(annotations: Dict[str, str]) -> Dict[str, List[str]]:
    raw = annotations.get(C.ANNO_COMPONENT_DEPENDS_ON, "")
    if not raw:
        return {}
    try:
        data = json.loads(raw)
        return {str(k): [str(d) for d in v] for k, v in data.items()}
    except (json.JSONDecodeError, TypeError, AttributeError) as e:
        log.warning("bad %s annotation: %s", C.ANNO_COMPONENT_DEPENDS_ON, e)
        return {}


def parse_discovery(annotations: Dict[str, str]) -> List[Dict[str, str]]:
    raw = annotations.get(C.ANNO_COMPONENT_DISCOVERY, "")
    if not raw:
        return []
    try:
        return list(json.loads(raw).get("env", []))
    except (json.JSONDecodeError, TypeError, AttributeError) as e:
        log.warning("bad %s annotation: %s", C.ANNO_COMPONENT_DISCOVERY, e)
        return []


def has_cycle(deps: Dict[str, List[str]]) -> bool:
    seen: Dict[str, int] = {}

    def visit(node: str) -> bool:
        state = seen.get(node, 0)
        if state == 1:
            return True
        if state == 2:
            return False
        seen[node] = 1
        for d in deps.get(node, []):
            if visit(d):
                return True
        seen[node] = 2
        return False

    return any(visit(n) for n in deps)


def start_gate(component: str, deps: Dict[str, List[str]],
               component_ready: Dict[str, bool]) -> bool:
    """True when `component` may start. A dependency graph with a cycle
    degrades to parallel start (reference component_lifecycle.go fallback)."""
    if not deps or has_cycle(deps):
        return True
    return all(component_ready.get(d, False) for d in deps.get(component, []))


def delete_waves(components: List[str],
                 deps: Dict[str, List[str]]) -> List[List[str]]:
    """Reverse deletion order (reference component_lifecycle.go:66-262):
    a component stops only after every component that depends on it has
    stopped — i.e. reverse-topological waves of the depends-on graph.
    Cycles (or no deps) degrade to one parallel wave, like start_gate."""
    if not deps or has_cycle(deps):
        return [list(components)]
    remaining = set(components)
    dependents: Dict[str, List[str]] = {}
    for comp, ds in deps.items():
        for d in ds:
            dependents.setdefault(d, []).append(comp)
    waves: List[List[str]] = []
    while remaining:
        # stop everything nothing-still-running depends on
        wave = sorted(c for c in remaining
                      if not any(d in remaining
                                 for d in dependents.get(c, [])))
        if not wave:            # defensive: shouldn't happen (acyclic)
            wave = sorted(remaining)
        waves.append(wave)
        remaining -= set(wave)
    return waves


def sibling_env(component: str, specs: List[Dict[str, str]],
                sibling_ports: Dict[str, List[int]]) -> Dict[str, str]:
    """Env for `component` from its discovery spec: each entry names a
    sibling component; value is `127.0.0.1:port` (first reported port) or
    `127.0.0.1` when the sibling has not reported ports."""
    env: Dict[str, str] = {}
    for item in specs:
        target = item.get("component", "")
        name = item.get("name") or f"RBG_COMPONENT_{target.upper()}_ADDR"
        ports = sibling_ports.get(target, [])
        env[name] = f"127.0.0.1:{ports[0]}" if ports else "127.0.0.1"
    return env
