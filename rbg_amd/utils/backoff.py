"""Restart backoff algebra — exact semantics of the reference's failure
handling (reference internal/.../roleinstance/sync/instance_scale.go:390-509
and doc/features/failure-handling.md):

  delay(n)        = min(base * 2^(n-1), max_delay)       for the n-th restart
  stability window = max(2 * max_delay, 10 min): a restart followed by that
                     much uninterrupted healthy time resets the counter.

The health monitor uses this to pace gang-recreation of engine groups after
HIP errors / heartbeat loss, and the group-recovery-time metric brackets the
whole sequence.
"""
from __future__ import annotations

import time
from dataclasses import dataclass, field
from typing import Dict, Optional

DEFAULT_BASE_SECONDS = 10.0
DEFAULT_MAX_SECONDS = 300.0
MIN_STABILITY_WINDOW = 600.0    # 10 min


def backoff_delay(restart_count: int,
                  base: float = DEFAULT_BASE_SECONDS,
                  max_delay: float = DEFAULT_MAX_SECONDS) -> float:
    """Delay before restart number ``restart_count`` (1-based). 0 -> no delay."""
    if restart_count <= 0:
        return 0.0
    return min(base * (2.0 ** (restart_count - 1)), max_delay)


def stability_window(max_delay: float = DEFAULT_MAX_SECONDS) -> float:
    return max(2.0 * max_delay, MIN_STABILITY_WINDOW)


@dataclass
class RestartTracker:
    """Per-instance restart bookkeeping with the stability-window reset."""
    base: float = DEFAULT_BASE_SECONDS
    max_delay: float = DEFAULT_MAX_SECONDS
    restart_count: int = 0
    last_restart_time: float = 0.0

    def observe_healthy(self, now: Optional[float] = None) -> None:
        """Call while the instance is healthy; resets the counter once the
        stability window has elapsed since the last restart."""
        now = time.time() if now is None else now
        if self.restart_count and \
                now - self.last_restart_time >= stability_window(self.max_delay):
            self.restart_count = 0

    def next_allowed_at(self) -> float:
        """Earliest wall time the next restart may begin."""
        if self.restart_count == 0:
            return 0.0
        return self.last_restart_time + backoff_delay(
            self.restart_count, self.base, self.max_delay)

    def may_restart(self, now: Optional[float] = None) -> bool:
        now = time.time() if now is None else now
        return now >= self.next_allowed_at()

    def record_restart(self, now: Optional[float] = None) -> float:
        """Record a restart; returns the delay that applied to it."""
        now = time.time() if now is None else now
        self.observe_healthy(now)
        self.restart_count += 1
        self.last_restart_time = now
        return backoff_delay(self.restart_count, self.base, self.max_delay)


@dataclass
class RestartRegistry:
    """Keyed trackers (one per role instance) — the in-memory LRU-ish cache
    that prevents restart cascades (reference instance_scale.go:511-540)."""
    base: float = DEFAULT_BASE_SECONDS
    max_delay: float = DEFAULT_MAX_SECONDS
    trackers: Dict[str, RestartTracker] = field(default_factory=dict)

    def for_key(self, key: str) -> RestartTracker:
        t = self.trackers.get(key)
        if t is None:
            t = RestartTracker(base=self.base, max_delay=self.max_delay)
            self.trackers[key] = t
        return t

    def evict(self, key: str) -> None:
        self.trackers.pop(key, None)
