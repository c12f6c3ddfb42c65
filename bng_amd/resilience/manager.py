"""Resilience — network-partition tolerance
(ref pkg/resilience/manager.go, types.go:12-100).

Partition state machine Online -> Partitioned -> Recovering -> Online,
driven by a pluggable upstream health check; while partitioned the BNG
keeps serving with degraded-mode policies (RADIUS cached/allow modes,
short leases on pool pressure, queued deferred operations)."""
from __future__ import annotations

import threading
import time
from typing import Callable, List, Optional

STATE_ONLINE = "online"
STATE_PARTITIONED = "partitioned"
STATE_RECOVERING = "recovering"


class Manager:
    def __init__(self, health_check: Callable[[], bool],
                 check_interval: float = 5.0, failure_threshold: int = 3,
                 recovery_checks: int = 2, recovery_hold: float = 0.0):
        self.health_check = health_check
        self.check_interval = check_interval
        self.failure_threshold = failure_threshold
        self.recovery_checks = recovery_checks
        self.recovery_hold = recovery_hold
        self.state = STATE_ONLINE
        self.partitioned_at: Optional[float] = None
        self._fails = 0
        self._oks = 0
        self._listeners: List[Callable[[str, str], None]] = []
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self._lock = threading.RLock()

    def on_transition(self, cb: Callable[[str, str], None]):
        self._listeners.append(cb)

    def start(self):
        self._thread = threading.Thread(target=self._loop, daemon=True)
        self._thread.start()
        return self

    def stop(self):
        self._stop.set()

    def _loop(self):
        while not self._stop.wait(self.check_interval):
            self.check_once()

    def check_once(self) -> str:
        """One health check -> state transition (ref manager.go:22-110)."""
        try:
            ok = bool(self.health_check())
        except Exception:
            ok = False
        with self._lock:
            old = self.state
            if ok:
                self._fails = 0
                self._oks += 1
                if self.state == STATE_PARTITIONED and \
                        self._oks >= self.recovery_checks:
                    self._transition(STATE_RECOVERING)
                elif self.state == STATE_RECOVERING:
                    self._transition(STATE_ONLINE)
            else:
                self._oks = 0
                self._fails += 1
                if self.state in (STATE_ONLINE, STATE_RECOVERING) and \
                        self._fails >= self.failure_threshold:
                    self._transition(STATE_PARTITIONED)
            return self.state

    def _transition(self, new: str):
        old = self.state
        self.state = new
        if new == STATE_PARTITIONED:
            self.partitioned_at = time.time()
        elif new == STATE_ONLINE:
            self.partitioned_at = None
        for cb in self._listeners:
            try:
                cb(old, new)
            except Exception:
                pass

    @property
    def is_partitioned(self) -> bool:
        return self.state == STATE_PARTITIONED

    def partition_duration(self) -> float:
        with self._lock:
            return time.time() - self.partitioned_at \
                if self.partitioned_at else 0.0
