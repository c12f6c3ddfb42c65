"""Walled garden — captive-portal quarantine state machine
(ref pkg/walledgarden/manager.go): states walledgarden -> active |
blocked (:16-44), allowed destinations (DNS/portal), TTL expiry checker
(:347-396).  Like the reference (which has no walled_garden.c — the
redirect exists only in docs), enforcement hooks are optional: the
manager keeps authoritative state and can push allow-lists into the
dataplane via set_dataplane_hooks."""
from __future__ import annotations

import threading
import time
from dataclasses import dataclass, field
from typing import Callable, Dict, List, Optional

STATE_WALLED = "walledgarden"
STATE_ACTIVE = "active"
STATE_BLOCKED = "blocked"


@dataclass
class Entry:
    mac: str
    ip: str
    state: str = STATE_WALLED
    added_at: float = field(default_factory=time.time)
    ttl: float = 3600.0
    reason: str = ""


class Manager:
    def __init__(self, portal_ip: str = "", dns_servers: Optional[List[str]] = None,
                 default_ttl: float = 3600.0):
        self.portal_ip = portal_ip
        self.dns_servers = dns_servers or []
        self.default_ttl = default_ttl
        self.entries: Dict[str, Entry] = {}
        self.allowed_destinations: List[str] = (
            ([portal_ip] if portal_ip else []) + self.dns_servers)
        self._lock = threading.RLock()
        self._hooks: List[Callable[[Entry], None]] = []
        self._stop = threading.Event()
        self._checker: Optional[threading.Thread] = None
        self.stats = {"added": 0, "activated": 0, "blocked": 0,
                      "expired": 0}

    def set_dataplane_hooks(self, on_change: Callable[[Entry], None]):
        """Optional enforcement hook (ref SetEBPFMaps manager.go:173-180)."""
        self._hooks.append(on_change)

    def start(self, check_interval: float = 1.0):
        self._checker = threading.Thread(
            target=self._check_loop, args=(check_interval,), daemon=True)
        self._checker.start()
        return self

    def stop(self):
        self._stop.set()

    # ------------------------------------------------------------- state
    def add(self, mac: str, ip: str, ttl: Optional[float] = None,
            reason: str = "unknown_subscriber") -> Entry:
        """ref manager.go:285 AddToWalledGarden."""
        e = Entry(mac=mac.lower(), ip=ip, ttl=ttl or self.default_ttl,
                  reason=reason)
        with self._lock:
            self.entries[e.mac] = e
        self.stats["added"] += 1
        self._notify(e)
        return e

    def activate(self, mac: str) -> bool:
        """Subscriber provisioned/paid: walledgarden -> active."""
        with self._lock:
            e = self.entries.get(mac.lower())
            if e is None or e.state == STATE_BLOCKED:
                return False
            e.state = STATE_ACTIVE
        self.stats["activated"] += 1
        self._notify(e)
        return True

    def block(self, mac: str, reason: str = "") -> bool:
        with self._lock:
            e = self.entries.get(mac.lower())
            if e is None:
                return False
            e.state = STATE_BLOCKED
            e.reason = reason or e.reason
        self.stats["blocked"] += 1
        self._notify(e)
        return True

    def remove(self, mac: str):
        with self._lock:
            e = self.entries.pop(mac.lower(), None)
        if e is not None:
            self._notify(e)

    def state_of(self, mac: str) -> Optional[str]:
        with self._lock:
            e = self.entries.get(mac.lower())
            return e.state if e else None

    def is_quarantined(self, mac: str) -> bool:
        return self.state_of(mac) == STATE_WALLED

    def is_destination_allowed(self, ip: str) -> bool:
        """For quarantined clients only DNS + portal are reachable."""
        return ip in self.allowed_destinations

    # ------------------------------------------------------------ expiry
    def _check_loop(self, interval: float):
        while not self._stop.wait(interval):
            self.expire_stale()

    def expire_stale(self, now: Optional[float] = None) -> int:
        """TTL expiry (ref manager.go:347-396): stale walled-garden
        entries are dropped so the next DHCP attempt re-evaluates."""
        now = now or time.time()
        dead = []
        with self._lock:
            for mac, e in list(self.entries.items()):
                if e.state == STATE_WALLED and now - e.added_at >= e.ttl:
                    dead.append(self.entries.pop(mac))
        for e in dead:
            self.stats["expired"] += 1
            self._notify(e)
        return len(dead)

    def _notify(self, e: Entry):
        for h in self._hooks:
            try:
                h(e)
            except Exception:
                pass

    def list_macs(self, state: Optional[str] = None):
        """MACs currently tracked, optionally filtered by state (ref
        manager.go ListWalledGardenMACs)."""
        with self._lock:
            return sorted(mac for mac, e in self.entries.items()
                          if state is None or e.state == state)

    def get_stats(self):
        with self._lock:
            by_state: Dict[str, int] = {}
            for e in self.entries.values():
                by_state[e.state] = by_state.get(e.state, 0) + 1
            return {**self.stats, "tracked": len(self.entries),
                    **{f"state_{k}": v for k, v in by_state.items()}}
