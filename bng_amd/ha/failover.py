"""HA failover controller — state machine NORMAL -> FAILING_OVER ->
FAILED_OVER -> FAILING_BACK with role-change callback and forced
operations (ref pkg/ha/failover.go:52-637)."""
from __future__ import annotations

import threading
import time
from typing import Callable, List, Optional

from .health_monitor import (EVENT_PARTNER_DOWN, EVENT_PARTNER_UP,
                             HealthEvent, HealthMonitor)
from .protocol import ROLE_ACTIVE, ROLE_STANDBY

STATE_NORMAL = "normal"
STATE_FAILING_OVER = "failing_over"
STATE_FAILED_OVER = "failed_over"
STATE_FAILING_BACK = "failing_back"


class FailoverController:
    def __init__(self, node_id: str, initial_role: str,
                 monitor: Optional[HealthMonitor] = None,
                 failover_delay: float = 0.0,
                 failback_delay: float = 0.0,
                 auto_failback: bool = True,
                 enabled: bool = True,
                 role_change_callback: Optional[Callable[[str], None]] = None):
        self.node_id = node_id
        self.role = initial_role
        self.initial_role = initial_role
        self.state = STATE_NORMAL
        self.monitor = monitor
        self.failover_delay = failover_delay
        self.failback_delay = failback_delay
        self.auto_failback = auto_failback
        self.enabled = enabled
        self.role_change_callback = role_change_callback
        self._handlers: List[Callable[[str], None]] = []
        self._lock = threading.RLock()
        self.history: List[dict] = []
        self.stats = {"failovers": 0, "failbacks": 0, "canceled": 0,
                      "forced": 0}
        if monitor is not None:
            monitor.on_event(self.handle_health_event)

    def on_role_change(self, cb: Callable[[str], None]):
        """Additional role-change handlers (ref MultipleHandlers)."""
        self._handlers.append(cb)

    def _notify(self, role: str):
        if self.role_change_callback:
            self.role_change_callback(role)
        for cb in self._handlers:
            try:
                cb(role)
            except Exception:
                pass

    # ------------------------------------------------------------ events
    def handle_health_event(self, ev: HealthEvent):
        """ref failover.go:322 handleHealthEvent."""
        if not self.enabled:
            return
        if ev.type == EVENT_PARTNER_DOWN:
            if self.role == ROLE_STANDBY and self.state == STATE_NORMAL:
                self.initiate_failover(reason="partner_down")
        elif ev.type == EVENT_PARTNER_UP:
            if self.role == ROLE_ACTIVE and self.state == STATE_FAILED_OVER \
                    and self.auto_failback:
                self.initiate_failback(reason="partner_recovered")

    # ---------------------------------------------------------- failover
    def initiate_failover(self, reason: str = "", forced: bool = False):
        """ref failover.go:404 initiateFailover -> :428 executeFailover."""
        with self._lock:
            if self.role == ROLE_ACTIVE:
                return False
            self.state = STATE_FAILING_OVER
        if self.failover_delay and not forced:
            time.sleep(self.failover_delay)
            # re-check: partner may have recovered during the delay
            if self.monitor is not None and self.monitor.partner_healthy:
                with self._lock:
                    self.state = STATE_NORMAL
                    self.stats["canceled"] += 1
                return False
        return self._execute_failover(reason, forced)

    def _execute_failover(self, reason: str, forced: bool) -> bool:
        with self._lock:
            self.role = ROLE_ACTIVE
            self.state = STATE_FAILED_OVER
            self.stats["failovers"] += 1
            if forced:
                self.stats["forced"] += 1
            self.history.append({"event": "failover", "reason": reason,
                                 "forced": forced, "at": time.time()})
        self._notify(ROLE_ACTIVE)
        return True

    def initiate_failback(self, reason: str = "", forced: bool = False):
        """ref failover.go:502 initiateFailback."""
        with self._lock:
            if self.role != ROLE_ACTIVE or (
                    self.state != STATE_FAILED_OVER and not forced):
                return False
            self.state = STATE_FAILING_BACK
        if self.failback_delay and not forced:
            time.sleep(self.failback_delay)
        with self._lock:
            self.role = ROLE_STANDBY
            self.state = STATE_NORMAL
            self.stats["failbacks"] += 1
            self.history.append({"event": "failback", "reason": reason,
                                 "forced": forced, "at": time.time()})
        self._notify(ROLE_STANDBY)
        return True

    def force_failover(self):
        return self.initiate_failover(reason="forced", forced=True)

    def force_failback(self):
        return self.initiate_failback(reason="forced", forced=True)

    def status(self) -> dict:
        with self._lock:
            return {"node_id": self.node_id, "role": self.role,
                    "state": self.state, "enabled": self.enabled,
                    "at_original_role": self.role == self.initial_role,
                    "history_len": len(self.history), **self.stats}
