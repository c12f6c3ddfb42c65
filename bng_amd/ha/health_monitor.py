"""HA health monitor — HTTP polling of the partner with failure/recovery
thresholds driving health events (ref pkg/ha/health_monitor.go:16-417)."""
from __future__ import annotations

import threading
import time
from dataclasses import dataclass
from typing import Callable, List, Optional

EVENT_PARTNER_DOWN = "partner_down"
EVENT_PARTNER_UP = "partner_up"


@dataclass
class HealthEvent:
    type: str
    node_id: str
    timestamp: float
    consecutive: int


class HealthMonitor:
    def __init__(self, partner_url: str, interval: float = 1.0,
                 timeout: float = 2.0, failure_threshold: int = 3,
                 recovery_threshold: int = 2):
        self.partner_url = partner_url.rstrip("/")
        self.interval = interval
        self.timeout = timeout
        self.failure_threshold = failure_threshold
        self.recovery_threshold = recovery_threshold
        self.partner_healthy = True
        self.consecutive_failures = 0
        self.consecutive_successes = 0
        self.last_check: float = 0.0
        self.last_response_time: float = 0.0   # seconds (ref
        # health_monitor.go ResponseTime tracking)
        self._listeners: List[Callable[[HealthEvent], None]] = []
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None

    def on_event(self, cb: Callable[[HealthEvent], None]):
        self._listeners.append(cb)

    def start(self):
        self._thread = threading.Thread(target=self._loop, daemon=True)
        self._thread.start()
        return self

    def stop(self):
        self._stop.set()

    def _loop(self):
        while not self._stop.wait(self.interval):
            self.check_once()

    def check_once(self) -> bool:
        """One health probe (ref health_monitor.go:232 performCheck)."""
        import requests
        self.last_check = time.time()
        t0 = time.perf_counter()
        ok = False
        try:
            r = requests.get(f"{self.partner_url}/health",
                             timeout=self.timeout)
            ok = r.status_code == 200
        except Exception:
            ok = False
        self.last_response_time = time.perf_counter() - t0
        if ok:
            self.consecutive_failures = 0
            self.consecutive_successes += 1
            if (not self.partner_healthy and
                    self.consecutive_successes >= self.recovery_threshold):
                self.partner_healthy = True
                self._emit(EVENT_PARTNER_UP)
        else:
            self.consecutive_successes = 0
            self.consecutive_failures += 1
            if (self.partner_healthy and
                    self.consecutive_failures >= self.failure_threshold):
                self.partner_healthy = False
                self._emit(EVENT_PARTNER_DOWN)
        return ok

    def _emit(self, typ: str):
        ev = HealthEvent(typ, self.partner_url, time.time(),
                         self.consecutive_failures or
                         self.consecutive_successes)
        for cb in self._listeners:
            try:
                cb(ev)
            except Exception:
                pass
