"""Pool utilization monitor with warning/critical/exhausted thresholds
and short-lease mode under pressure (ref pkg/resilience/pool_monitor.go,
types.go:69-100)."""
from __future__ import annotations

import threading
from typing import Callable, List, Optional

LEVEL_OK = "ok"
LEVEL_WARNING = "warning"
LEVEL_CRITICAL = "critical"
LEVEL_EXHAUSTED = "exhausted"


class PoolMonitor:
    def __init__(self, utilization_fn: Callable[[], float],
                 warning: float = 0.8, critical: float = 0.9,
                 exhausted: float = 0.98,
                 normal_lease: int = 3600, short_lease: int = 300):
        self.utilization_fn = utilization_fn
        self.warning = warning
        self.critical = critical
        self.exhausted = exhausted
        self.normal_lease = normal_lease
        self.short_lease = short_lease
        self.level = LEVEL_OK
        self._listeners: List[Callable[[str, str, float], None]] = []
        self._lock = threading.Lock()

    def on_level_change(self, cb: Callable[[str, str, float], None]):
        self._listeners.append(cb)

    def check(self) -> str:
        u = self.utilization_fn()
        new = LEVEL_OK
        if u >= self.exhausted:
            new = LEVEL_EXHAUSTED
        elif u >= self.critical:
            new = LEVEL_CRITICAL
        elif u >= self.warning:
            new = LEVEL_WARNING
        with self._lock:
            old, self.level = self.level, new
        if new != old:
            for cb in self._listeners:
                try:
                    cb(old, new, u)
                except Exception:
                    pass
        return new

    def effective_lease_time(self) -> int:
        """Short-lease mode on pool pressure (ref types.go:69-100):
        under critical utilization leases shrink so churned addresses
        return to the pool faster."""
        return self.short_lease if self.level in (
            LEVEL_CRITICAL, LEVEL_EXHAUSTED) else self.normal_lease
