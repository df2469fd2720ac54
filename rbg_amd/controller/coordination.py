"""CoordinatedPolicy math — cross-role scaling and rolling-update skew caps.

Hardware-independent algebra ported per SURVEY §7 phase 5 from the reference:
- scaling: pkg/coordination/coordinationscaling/scaler.go:29-270
  (CalculateTargetReplicas: cap each role's next replica target so pairwise
  progress skew <= maxSkew percentage points; progression gate
  OrderScheduled|OrderReady decides which count is "progress"),
- rolling update: internal/controller/workloads/rolebasedgroup_controller.go
  :1266-1428 calculateNextRollingTarget with the skew bound
  a/b - x/d <= s/100 (:1493-1510), including the forward-progress corner
  cases fixed in the reference (:1403-1426): when the bound would freeze
  every role, the minimum-progress role still advances by one.

On MI355X these caps drive engine-process counts per role (e.g. a decode
pool scaling in step with prefill) and bound how many TP rank-groups are torn
down simultaneously during a weight rollout.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, List, Optional

from ..api import constants as C
from ..api.types import CoordinationRule


@dataclass
class RoleScaleState:
    """Observed state of one role, as the scaler sees it."""
    name: str
    desired: int            # spec replicas (the scale target)
    current: int            # created instances (OrderScheduled progress)
    ready: int              # ready instances (OrderReady progress)


def _progress_pct(progress: int, desired: int) -> float:
    if desired <= 0:
        return 100.0
    return 100.0 * min(progress, desired) / desired


def calculate_scaling_targets(rule: CoordinationRule,
                              states: Dict[str, RoleScaleState]) -> Dict[str, int]:
    """Next per-role replica target capped by the rule's scaling maxSkew.

    Returns the number of instances each role may have *created* now; always
    >= current and <= desired, and guarantees forward progress.
    """
    scaling = rule.strategy.scaling
    members = [states[r] for r in rule.roles if r in states]
    if scaling is None or not members:
        return {s.name: s.desired for s in members}
    skew = float(scaling.max_skew)
    use_ready = scaling.progression == C.PROGRESSION_ORDER_READY

    def progress_of(s: RoleScaleState) -> int:
        return s.ready if use_ready else s.current

    min_pct = min(_progress_pct(progress_of(s), s.desired) for s in members)
    targets: Dict[str, int] = {}
    for s in members:
        allowed_pct = min(100.0, min_pct + skew)
        allowed = int(allowed_pct / 100.0 * s.desired + 1e-9)
        targets[s.name] = max(s.current, min(s.desired, allowed))
    # Forward-progress guarantee: if nobody may create another instance yet
    # the whole rule deadlocks (reference scaler.go handles the same corner):
    # let every minimum-progress role advance by one.
    if all(targets[s.name] <= s.current for s in members) and \
            any(s.current < s.desired for s in members):
        for s in members:
            if s.current < s.desired and \
                    _progress_pct(progress_of(s), s.desired) <= min_pct + 1e-9:
                targets[s.name] = s.current + 1
    return targets


@dataclass
class RoleUpdateState:
    """Observed rollout state of one role."""
    name: str
    total: int              # replicas
    updated: int            # instances already on the update revision


def calculate_rolling_partitions(rule: CoordinationRule,
                                 states: Dict[str, RoleUpdateState]) -> Dict[str, int]:
    """Per-role update-strategy ``partition`` override enforcing the rolling
    maxSkew bound: for any two roles a (leader) and x (follower),
    updated_a/total_a - updated_x/total_x <= maxSkew/100
    (reference rolebasedgroup_controller.go:1493-1510).

    partition = instances that must REMAIN on the old revision, so
    partition_i = total_i - allowed_updated_i.
    """
    ru = rule.strategy.rolling_update
    members = [states[r] for r in rule.roles if r in states]
    if ru is None or not members:
        return {s.name: 0 for s in members}
    skew = float(ru.max_skew) / 100.0

    def frac(s: RoleUpdateState) -> float:
        return (s.updated / s.total) if s.total > 0 else 1.0

    min_frac = min(frac(s) for s in members)
    allowed: Dict[str, int] = {}
    for s in members:
        a = int((min_frac + skew) * s.total + 1e-9)
        allowed[s.name] = max(s.updated, min(s.total, a))
    # Forward-progress corner (reference :1403-1426): with small maxSkew and
    # uneven replica counts the floor can freeze every role; the
    # minimum-fraction role(s) must still advance by one.
    if all(allowed[s.name] <= s.updated for s in members) and \
            any(s.updated < s.total for s in members):
        for s in members:
            if s.updated < s.total and frac(s) <= min_frac + 1e-9:
                allowed[s.name] = s.updated + 1
    out: Dict[str, int] = {}
    for s in members:
        part = s.total - allowed[s.name]
        # rule-level floor: never update past the rule's own partition
        part = max(part, ru.partition)
        out[s.name] = max(0, min(s.total, part))
    return out


def max_pairwise_skew(states: List[RoleUpdateState]) -> float:
    """Observed skew in percentage points — exposed as a gauge and asserted
    in tests against the configured bound."""
    if len(states) < 2:
        return 0.0
    fracs = [100.0 * s.updated / s.total if s.total else 100.0 for s in states]
    return max(fracs) - min(fracs)


def find_rule(policy, role_name: str) -> Optional[CoordinationRule]:
    if policy is None:
        return None
    for rule in policy.spec.rules:
        if role_name in rule.roles:
            return rule
    return None
