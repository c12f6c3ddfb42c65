"""Standalone WiFi/guest-network gateway mode (ref pkg/wifi/gateway.go:
35-470): short-lease session management over the epoch allocator, a
captive-portal acceptance step, and periodic epoch advancing — the
'lease' pool-mode deployment of SURVEY §2.4."""
from __future__ import annotations

import threading
import time
from dataclasses import dataclass, field
from typing import Dict, Optional

from ..allocator.epoch_bitmap import EpochBitmapAllocator, PoolExhaustedError


@dataclass
class GuestSession:
    mac: str
    ip: str
    accepted_terms: bool = False
    started: float = field(default_factory=time.time)
    last_seen: float = field(default_factory=time.time)
    bytes_in: int = 0
    bytes_out: int = 0


class Gateway:
    def __init__(self, network: str = "192.168.100.0/24",
                 lease_epochs: int = 1, epoch_seconds: float = 300.0,
                 portal_url: str = "http://portal.local"):
        self.alloc = EpochBitmapAllocator(network, 32,
                                         grace_period=lease_epochs)
        self.epoch_seconds = epoch_seconds
        self.portal_url = portal_url
        self.sessions: Dict[str, GuestSession] = {}
        self._lock = threading.RLock()
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self.stats = {"joined": 0, "accepted": 0, "expired": 0,
                      "exhausted": 0}

    def start(self):
        self._thread = threading.Thread(target=self._epoch_loop, daemon=True)
        self._thread.start()
        return self

    def stop(self):
        self._stop.set()

    def _epoch_loop(self):
        while not self._stop.wait(self.epoch_seconds):
            self.advance_epoch()

    # -------------------------------------------------------------- API
    def join(self, mac: str) -> GuestSession:
        """Guest connects: short-lease IP, quarantined until portal
        acceptance."""
        mac = mac.lower()
        with self._lock:
            s = self.sessions.get(mac)
            if s is not None:
                self.alloc.renew(mac)
                s.last_seen = time.time()
                return s
            try:
                ip = self.alloc.allocate(mac)
            except PoolExhaustedError:
                self.stats["exhausted"] += 1
                raise
            s = GuestSession(mac, ip)
            self.sessions[mac] = s
        self.stats["joined"] += 1
        return s

    def accept_terms(self, mac: str) -> bool:
        with self._lock:
            s = self.sessions.get(mac.lower())
            if s is None:
                return False
            s.accepted_terms = True
        self.stats["accepted"] += 1
        return True

    def is_quarantined(self, mac: str) -> bool:
        with self._lock:
            s = self.sessions.get(mac.lower())
            return s is not None and not s.accepted_terms

    def touch(self, mac: str):
        """Traffic seen: renew the short lease."""
        with self._lock:
            s = self.sessions.get(mac.lower())
            if s is not None:
                s.last_seen = time.time()
                try:
                    self.alloc.renew(mac.lower())
                except Exception:
                    pass

    def advance_epoch(self) -> int:
        """Idle guests age out after grace_period epochs without renew."""
        before = set(self.alloc.subscribers)
        self.alloc.advance_epoch()
        gone = before - set(self.alloc.subscribers)
        with self._lock:
            for mac in gone:
                self.sessions.pop(mac, None)
        self.stats["expired"] += len(gone)
        return len(gone)

    def update_traffic(self, mac: str, bytes_in: int = 0,
                       bytes_out: int = 0) -> bool:
        """Accumulate guest usage + keep the session warm (ref
        wifi/gateway_test.go UpdateTrafficStats)."""
        with self._lock:
            s = self.sessions.get(mac)
            if s is None:
                return False
            s.bytes_in += bytes_in
            s.bytes_out += bytes_out
            s.last_seen = time.time()
            return True

    def get_stats(self):
        with self._lock:
            return {**self.stats,
                    "sessions": len(self.sessions),
                    "accepted": sum(1 for s in self.sessions.values()
                                    if s.accepted_terms),
                    "bytes_in": sum(s.bytes_in
                                    for s in self.sessions.values()),
                    "bytes_out": sum(s.bytes_out
                                     for s in self.sessions.values())}

    def session_count(self) -> int:
        with self._lock:
            return len(self.sessions)
