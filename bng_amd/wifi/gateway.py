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


# ----------------------------------------------------------------------
# Full session manager with captive-portal grace periods and dual
# operating modes (ref gateway.go:27-553)

MODE_WIFI = "wifi_gateway"           # DHCP-first allocation
MODE_OLT_BNG = "olt_bng"             # RADIUS-first allocation

S_NEW = "new"
S_GRACE = "grace_period"
S_AUTHENTICATED = "authenticated"
S_ACTIVE = "active"
S_EXPIRED = "expired"


@dataclass
class WifiConfig:
    """ref Config gateway.go:35-100.  The two presets encode who drives
    the address lifecycle: the WiFi mode allocates on DHCP DISCOVER and
    releases on lease expiry; OLT-BNG allocates after RADIUS auth and
    releases on session termination."""
    mode: str = MODE_WIFI
    allocation_trigger: str = "dhcp_discover"
    deallocation_trigger: str = "lease_expiry"
    lease_duration: float = 1800.0
    captive_portal_enabled: bool = True
    captive_portal_url: str = ""
    grace_period: float = 300.0

    @classmethod
    def olt_bng(cls):
        return cls(mode=MODE_OLT_BNG, allocation_trigger="radius_auth",
                   deallocation_trigger="session_termination",
                   lease_duration=86400.0,
                   captive_portal_enabled=False)


@dataclass
class WifiSession:
    """ref Session gateway.go:102-135."""
    id: str
    mac: str
    ip: str = ""
    hostname: str = ""
    pool_id: int = 0
    state: str = S_NEW
    authenticated: bool = False
    auth_method: str = ""
    user_identity: str = ""
    created_at: float = 0.0
    lease_expiry: float = 0.0
    authenticated_at: float = 0.0
    grace_period_ends: float = 0.0
    last_renewal: float = 0.0
    bytes_in: int = 0
    bytes_out: int = 0
    packets_in: int = 0
    packets_out: int = 0
    vendor_class: str = ""
    user_class: str = ""


class Manager:
    """WiFi gateway session manager (ref gateway.go Manager :151-553):
    sessions created at DHCP time enter a captive-portal grace period;
    authentication via the portal promotes them; unauthenticated
    sessions die when the grace period lapses, everything dies at lease
    expiry."""

    def __init__(self, config: Optional[WifiConfig] = None):
        self.config = config or WifiConfig()
        self.sessions: Dict[str, WifiSession] = {}
        self.by_ip: Dict[str, str] = {}
        self._lock = threading.RLock()
        self.on_create = None
        self.on_auth = None
        self.on_expire = None

    def create_session(self, mac: str, hostname: str = "",
                       pool_id: int = 0, ip: str = "") -> WifiSession:
        import uuid
        mac = mac.lower()
        now = time.time()
        with self._lock:
            s = self.sessions.get(mac)
            if s is not None:
                # re-create on a known MAC renews the lease (ref
                # gateway_test.go SessionRenewalOnCreate)
                s.last_renewal = now
                s.lease_expiry = now + self.config.lease_duration
                return s
            s = WifiSession(id=uuid.uuid4().hex[:12], mac=mac, ip=ip,
                            hostname=hostname, pool_id=pool_id,
                            created_at=now, last_renewal=now,
                            lease_expiry=now + self.config.lease_duration)
            if self.config.captive_portal_enabled:
                s.state = S_GRACE
                s.grace_period_ends = now + self.config.grace_period
            else:
                s.state = S_ACTIVE
                s.authenticated = True
            self.sessions[mac] = s
            if ip:
                self.by_ip[ip] = mac
        if self.on_create:
            self.on_create(s)
        return s

    def renew_session(self, mac: str) -> bool:
        """DHCP renewal extends the lease (ref RenewSession
        :280-301)."""
        now = time.time()
        with self._lock:
            s = self.sessions.get(mac.lower())
            if s is None:
                return False
            s.last_renewal = now
            s.lease_expiry = now + self.config.lease_duration
        return True

    def authenticate_session(self, mac: str, auth_method: str = "portal",
                             user_identity: str = "") -> bool:
        """Captive-portal success promotes grace -> authenticated (ref
        AuthenticateSession :303-333)."""
        with self._lock:
            s = self.sessions.get(mac.lower())
            if s is None:
                return False
            s.authenticated = True
            s.auth_method = auth_method
            s.user_identity = user_identity
            s.authenticated_at = time.time()
            s.state = S_AUTHENTICATED
        if self.on_auth:
            self.on_auth(s)
        return True

    def release_session(self, mac: str) -> bool:
        with self._lock:
            s = self.sessions.pop(mac.lower(), None)
            if s is None:
                return False
            if s.ip:
                self.by_ip.pop(s.ip, None)
        return True

    def get_session(self, mac: str) -> Optional[WifiSession]:
        with self._lock:
            return self.sessions.get(mac.lower())

    def get_session_by_ip(self, ip: str) -> Optional[WifiSession]:
        with self._lock:
            mac = self.by_ip.get(ip)
            return self.sessions.get(mac) if mac else None

    def update_traffic(self, mac: str, bytes_in: int = 0,
                       bytes_out: int = 0, packets_in: int = 0,
                       packets_out: int = 0):
        with self._lock:
            s = self.sessions.get(mac.lower())
            if s is None:
                return
            s.bytes_in += bytes_in
            s.bytes_out += bytes_out
            s.packets_in += packets_in
            s.packets_out += packets_out
            if s.state == S_AUTHENTICATED:
                s.state = S_ACTIVE

    def is_in_grace_period(self, mac: str) -> bool:
        """ref IsInGracePeriod :416-427."""
        with self._lock:
            s = self.sessions.get(mac.lower())
            return (s is not None and s.state == S_GRACE and
                    time.time() < s.grace_period_ends)

    def needs_authentication(self, mac: str) -> bool:
        """ref NeedsAuthentication :429-444: unknown MAC needs auth;
        portal disabled means nobody does."""
        if not self.config.captive_portal_enabled:
            return False
        with self._lock:
            s = self.sessions.get(mac.lower())
            return s is None or not s.authenticated

    def cleanup_expired(self, now: Optional[float] = None) -> int:
        """Lease expiry kills everything; grace-period lapse kills the
        unauthenticated (ref cleanupExpiredSessions :496-545)."""
        now = now if now is not None else time.time()
        expired = []
        with self._lock:
            for mac, s in list(self.sessions.items()):
                dead = now > s.lease_expiry or \
                    (s.state == S_GRACE and not s.authenticated and
                     now > s.grace_period_ends)
                if dead:
                    s.state = S_EXPIRED
                    expired.append(s)
                    del self.sessions[mac]
                    if s.ip:
                        self.by_ip.pop(s.ip, None)
        for s in expired:
            if self.on_expire:
                try:
                    self.on_expire(s)
                except Exception:
                    pass
        return len(expired)

    def manager_stats(self) -> Dict[str, int]:
        now = time.time()
        with self._lock:
            return {
                "active_sessions": len(self.sessions),
                "authenticated_sessions":
                    sum(1 for s in self.sessions.values()
                        if s.authenticated),
                "grace_period_sessions":
                    sum(1 for s in self.sessions.values()
                        if s.state == S_GRACE and
                        now < s.grace_period_ends),
                "total_bytes_in": sum(s.bytes_in
                                      for s in self.sessions.values()),
                "total_bytes_out": sum(s.bytes_out
                                       for s in self.sessions.values())}
