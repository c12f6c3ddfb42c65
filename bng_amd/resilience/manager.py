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


class Reconciler:
    """Partition-heal pipeline (ref manager.go performReconciliation
    :342-427): (1) detect allocation conflicts against the remote view
    and resolve them with the site policy, (2) queue re-auths for every
    session admitted on a degraded answer, (3) process them rate-
    limited, (4) replay buffered accounting, (5) drain the deferred
    request queue.  Returns the ReconciliationResult dict."""

    def __init__(self, detector=None, radius=None, request_queue=None,
                 reauth_rate_limit: int = 0,
                 on_conflict: Optional[Callable] = None):
        self.detector = detector
        self.radius = radius
        self.request_queue = request_queue
        self.reauth_rate_limit = reauth_rate_limit
        self.on_conflict = on_conflict

    def reconcile(self, remote_allocations=None,
                  credentials=None) -> dict:
        t0 = time.time()
        result = {"started_at": t0, "conflicts_found": 0,
                  "conflicts_resolved": 0, "reauths_queued": 0,
                  "reauths_completed": 0, "reauths_failed": 0,
                  "acct_records_synced": 0, "requests_drained": 0,
                  "errors": []}
        if self.detector is not None and remote_allocations is not None:
            conflicts = self.detector.detect(remote_allocations)
            result["conflicts_found"] = len(conflicts)
            for c in conflicts:
                try:
                    self.detector.resolve(c)
                    result["conflicts_resolved"] += 1
                    if self.on_conflict:
                        self.on_conflict(c)
                except Exception as e:
                    result["errors"].append(f"resolve {c.ip}: {e}")
            self.detector.clear_partition_flags()
        if self.radius is not None:
            degraded = self.radius.degraded_sessions()
            result["reauths_queued"] = len(degraded)
            for user in degraded:
                self.radius.queue_reauth(user)
            done, failed = self.radius.process_reauths(
                self.reauth_rate_limit, credentials)
            result["reauths_completed"] = done
            result["reauths_failed"] = failed
            result["acct_records_synced"] = self.radius.replay_buffered()
        if self.request_queue is not None:
            result["requests_drained"] = self.request_queue.drain()
        result["duration"] = time.time() - t0
        return result


class ShortLeasePolicy:
    """While partitioned, hand out short DHCP leases so address churn
    stays reconcilable after heal (ref manager.go ShouldUseShortLease
    :620-641 — driven by partition state + pool pressure)."""

    def __init__(self, manager: Manager, short_lease: float = 300.0,
                 normal_lease: float = 86400.0,
                 pool_monitor=None):
        self.manager = manager
        self.short_lease = short_lease
        self.normal_lease = normal_lease
        self.pool_monitor = pool_monitor
        self.short_leases_issued = 0

    def should_use_short_lease(self) -> bool:
        if self.manager.is_partitioned:
            return True
        if self.pool_monitor is not None:
            return self.pool_monitor.check() in ("warning", "critical")
        return False

    def lease_time(self) -> float:
        if self.should_use_short_lease():
            self.short_leases_issued += 1
            return self.short_lease
        return self.normal_lease
