"""Role dependency manager — DFS topological sort into parallel waves.

Mirrors reference pkg/dependency/dependency.go:43-205: roles with no
unresolved dependencies form wave 0; each later wave depends only on earlier
waves; cycles are detected and reported.  The reconciler creates each wave's
workloads only after every upstream role reports Ready (the readiness gate of
dependency.go:94-117), which on MI355X is what sequences e.g.
router -> prefill -> decode engine startup and RCCL group formation.
"""
from __future__ import annotations

from typing import Dict, List, Sequence


class DependencyCycle(ValueError):
    pass


def sort_roles(roles: Sequence) -> List[List]:
    """Group RoleSpecs into dependency waves (list of lists, wave order).

    Kahn-style layering; deterministic (preserves spec order within a wave).
    Raises DependencyCycle naming the cycle members.
    """
    by_name = {r.name: r for r in roles}
    indeg: Dict[str, int] = {}
    for r in roles:
        deps = [d for d in r.dependencies if d in by_name]
        indeg[r.name] = len(deps)
    remaining = [r.name for r in roles]
    waves: List[List] = []
    while remaining:
        wave = [n for n in remaining if indeg[n] == 0]
        if not wave:
            raise DependencyCycle(
                f"dependency cycle among roles: {sorted(remaining)}")
        waves.append([by_name[n] for n in wave])
        wave_set = set(wave)
        remaining = [n for n in remaining if n not in wave_set]
        for n in remaining:
            indeg[n] -= sum(1 for d in by_name[n].dependencies if d in wave_set)
    return waves


def downstream_roles(roles: Sequence, failed: str) -> List[str]:
    """Transitive closure of roles depending on ``failed`` — the linked-failover
    blast radius (used by the health monitor to gang-restart dependents)."""
    out: List[str] = []
    frontier = {failed}
    changed = True
    while changed:
        changed = False
        for r in roles:
            if r.name in frontier or r.name in out:
                continue
            if any(d in frontier or d in out for d in r.dependencies):
                out.append(r.name)
                changed = True
    return out
