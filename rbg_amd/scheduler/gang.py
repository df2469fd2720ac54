"""Gang GPU allocator — all-or-nothing device + HBM reservation.

The MI355X realization of the reference's PodGroup gang scheduling
(reference pkg/scheduler/podgroup_manager.go:62-92 and the scheduler-plugins
/ Volcano backends): before any engine of a role group starts, the whole
group's GPU set and HBM budget is reserved atomically, with a timeout; a
partial reservation is rolled back (no orphaned engines holding HBM —
the fail-fast of reference instance_scale.go:123-138).

Placement is xGMI-aware: gang members are packed onto the
highest-adjacency GPU set, and sticky bindings (scheduler/placement.py) let a
restarted instance reclaim its previous GPUs (warm HBM/caches).
"""
from __future__ import annotations

import threading
import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Sequence, Tuple

from .topology import NodeTopology, fully_connected


class GangUnschedulable(RuntimeError):
    pass


@dataclass
class GpuClaim:
    """One engine's device request: n exclusive GPUs or an HBM slice of one."""
    gpus: int = 1
    hbm_bytes: int = 0          # 0 => exclusive whole-GPU claim
    prefer: Tuple[int, ...] = ()  # sticky hint: previous device ids
    # in-place-scheduling mode (reference node_binding.go Required
    # affinity): require_prefer=True makes the sticky devices a HARD
    # constraint — the reservation waits for exactly those GPUs
    require_prefer: bool = False
    # exclusive-topology (reference pod_reconciler.go:160-241, consumed via
    # the rbg.workloads.x-k8s.io/exclusive-topology annotation): the claim's
    # GROUP packs onto a shared GPU set that no OTHER group may share
    group: str = ""
    exclusive: bool = False


@dataclass
class Reservation:
    gang_id: str
    assignments: List[List[int]] = field(default_factory=list)  # per claim


class GangAllocator:
    def __init__(self, topo: Optional[NodeTopology] = None):
        self.topo = topo or fully_connected()
        self._lock = threading.Condition()
        # free HBM per GPU; exclusive claims take the whole budget
        self._free_hbm: Dict[int, int] = {
            g: self.topo.hbm_bytes for g in range(self.topo.num_gpus)}
        self._exclusive: Dict[int, str] = {}     # gpu -> gang holding exclusively
        self._shared: Dict[int, Dict[str, int]] = {g: {} for g in self._free_hbm}
        self._gangs: Dict[str, Reservation] = {}
        self._gang_group: Dict[str, str] = {}    # gang -> exclusive group

    # -- queries ------------------------------------------------------------

    def free_gpus(self) -> List[int]:
        return [g for g in self._free_hbm
                if g not in self._exclusive and not self._shared[g]]

    def usage(self) -> Dict[int, Dict[str, int]]:
        with self._lock:
            return {g: dict(v) for g, v in self._shared.items()}

    # -- allocation ---------------------------------------------------------

    def _pick_gpus(self, count: int, prefer: Sequence[int]) -> Optional[List[int]]:
        """Choose `count` free GPUs maximizing xGMI adjacency, honoring sticky
        preferences first (in-place-scheduling analog)."""
        free = self.free_gpus()
        if len(free) < count:
            return None
        chosen: List[int] = [g for g in prefer if g in free][:count]
        remaining = [g for g in free if g not in chosen]
        while len(chosen) < count:
            if not remaining:
                return None
            # greedy: next GPU with best adjacency to what we have
            best = max(remaining,
                       key=lambda g: (self.topo.adjacency_score(chosen + [g]), -g))
            chosen.append(best)
            remaining.remove(best)
        return chosen

    def _try_reserve(self, gang_id: str,
                     claims: Sequence[GpuClaim]) -> Optional[Reservation]:
        """One atomic attempt under the lock; returns None if it cannot be
        satisfied *in full* right now."""
        # snapshot for rollback
        snap_excl = dict(self._exclusive)
        snap_hbm = dict(self._free_hbm)
        snap_shared = {g: dict(v) for g, v in self._shared.items()}
        res = Reservation(gang_id=gang_id)
        ok = True
        snap_group = dict(self._gang_group)
        for claim in claims:
            if claim.hbm_bytes and claim.gpus == 1:
                # shared slice of one GPU.  With exclusive topology a GPU
                # qualifies only if every current sharer belongs to the
                # same group (1 group per device — the "Disaggregated
                # Inference" same-domain use case), and group GPUs are
                # preferred so the group's roles PACK together (affinity
                # half of the reference semantics).
                def allowed(g: int) -> bool:
                    if g in self._exclusive or \
                            self._free_hbm[g] < claim.hbm_bytes:
                        return False
                    if not claim.exclusive:
                        # a non-exclusive claim must not intrude on a GPU
                        # claimed exclusively-by-topology by another group
                        return all(
                            self._gang_group.get(o, "") == "" or
                            self._gang_group.get(o) == claim.group
                            for o in self._shared[g])
                    return all(self._gang_group.get(o) == claim.group
                               for o in self._shared[g])

                def group_share(g: int) -> int:
                    return sum(v for o, v in self._shared[g].items()
                               if claim.group and
                               self._gang_group.get(o) == claim.group)

                cand = [g for g in claim.prefer if g in self._free_hbm
                        and allowed(g)]
                if not cand and not (claim.require_prefer and claim.prefer):
                    cand = sorted(
                        (g for g in self._free_hbm if allowed(g)),
                        key=lambda g: (-group_share(g)
                                       if claim.exclusive else 0,
                                       -self._free_hbm[g]))
                if not cand:
                    ok = False
                    break
                g = cand[0]
                self._free_hbm[g] -= claim.hbm_bytes
                self._shared[g][gang_id] = \
                    self._shared[g].get(gang_id, 0) + claim.hbm_bytes
                res.assignments.append([g])
            else:
                gpus = self._pick_gpus(claim.gpus, claim.prefer)
                if gpus is not None and claim.require_prefer and \
                        claim.prefer and \
                        not set(gpus) <= set(claim.prefer):
                    gpus = None      # required stickiness: wait for them
                if gpus is None:
                    ok = False
                    break
                for g in gpus:
                    self._exclusive[g] = gang_id
                    self._free_hbm[g] = 0
                res.assignments.append(gpus)
        if not ok:
            self._exclusive = snap_excl
            self._free_hbm = snap_hbm
            self._shared = snap_shared
            self._gang_group = snap_group
            return None
        grp = next((c.group for c in claims if c.exclusive and c.group), "")
        if grp:
            self._gang_group[gang_id] = grp
        self._gangs[gang_id] = res
        return res

    def reserve(self, gang_id: str, claims: Sequence[GpuClaim],
                timeout: float = 0.0) -> Reservation:
        """Atomically reserve every claim or raise GangUnschedulable after
        ``timeout`` seconds (the gang-scheduling timeout annotation)."""
        deadline = time.monotonic() + timeout
        with self._lock:
            if gang_id in self._gangs:
                return self._gangs[gang_id]
            while True:
                res = self._try_reserve(gang_id, claims)
                if res is not None:
                    return res
                remaining = deadline - time.monotonic()
                if remaining <= 0:
                    raise GangUnschedulable(
                        f"gang {gang_id!r}: cannot reserve "
                        f"{sum(c.gpus for c in claims)} GPUs "
                        f"(free: {self.free_gpus()})")
                self._lock.wait(timeout=min(remaining, 0.5))

    def release(self, gang_id: str) -> None:
        with self._lock:
            res = self._gangs.pop(gang_id, None)
            if res is None:
                return
            for g, owner in list(self._exclusive.items()):
                if owner == gang_id:
                    del self._exclusive[g]
                    self._free_hbm[g] = self.topo.hbm_bytes
            for g, owners in self._shared.items():
                taken = owners.pop(gang_id, 0)
                if taken:
                    self._free_hbm[g] += taken
            self._gang_group.pop(gang_id, None)
            self._lock.notify_all()

    def holding(self, gang_id: str) -> Optional[Reservation]:
        with self._lock:
            return self._gangs.get(gang_id)
