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


# allowed-destination reason codes (ref AllowedDestination.Reason
# manager.go:56-63 / initAllowedDestinations :187-235)
REASON_DNS = 1
REASON_PORTAL = 2
REASON_CUSTOM = 3

# classification verdicts for quarantined traffic
V_FORWARD = "forward"
V_REDIRECT = "redirect"
V_DROP = "drop"


class Manager:
    def __init__(self, portal_ip: str = "", dns_servers: Optional[List[str]] = None,
                 default_ttl: float = 3600.0, portal_port: int = 8080):
        self.portal_ip = portal_ip
        self.portal_port = portal_port
        self.dns_servers = dns_servers or []
        self.default_ttl = default_ttl
        self.entries: Dict[str, Entry] = {}
        self.allowed_destinations: List[str] = (
            ([portal_ip] if portal_ip else []) + self.dns_servers)
        # keyed (ip, port, proto) -> reason (ref allowedDestKey :237-242)
        self.allowed_dests: Dict[tuple, int] = {}
        for dns in self.dns_servers:
            self.allowed_dests[(dns, 53, 17)] = REASON_DNS
            self.allowed_dests[(dns, 53, 6)] = REASON_DNS
        if portal_ip:
            self.allowed_dests[(portal_ip, portal_port, 6)] = REASON_PORTAL
        self._lock = threading.RLock()
        self._hooks: List[Callable[[Entry], None]] = []
        self._redirect_cbs: List[Callable[[str, str], None]] = []
        self._stop = threading.Event()
        self._checker: Optional[threading.Thread] = None
        self.stats = {"added": 0, "activated": 0, "blocked": 0,
                      "expired": 0, "redirects": 0, "dropped": 0,
                      "allowed": 0}

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

    def is_destination_allowed(self, ip: str, port: int = 0,
                               proto: int = 0) -> bool:
        """For quarantined clients only DNS + portal (+ configured
        extras) are reachable.  Bare-IP calls keep the legacy
        any-port semantics; (ip, port, proto) checks the keyed table
        (ref allowedDestKey manager.go:237-242)."""
        if port == 0 and proto == 0:
            return ip in self.allowed_destinations
        return (ip, port, proto) in self.allowed_dests

    def allow_destination(self, ip: str, port: int, proto: int,
                          reason: int = REASON_CUSTOM):
        """ref Config.AllowedDestinations / initAllowedDestinations
        :216-230."""
        self.allowed_dests[(ip, port, proto)] = reason
        if ip not in self.allowed_destinations:
            self.allowed_destinations.append(ip)

    def on_redirect(self, cb: Callable[[str, str], None]):
        """cb(mac, dst_ip) whenever quarantined HTTP gets redirected
        to the portal (ref OnRedirect manager.go:182-185)."""
        self._redirect_cbs.append(cb)

    def classify(self, mac: str, dst_ip: str, dst_port: int = 0,
                 proto: int = 6) -> str:
        """Quarantine-time verdict for one flow: active subscribers
        and unknown MACs forward; blocked MACs drop; quarantined
        traffic forwards to allowed destinations, HTTP redirects to
        the captive portal (firing redirect callbacks), everything
        else drops — the walled-garden decision table the reference
        encodes in its eBPF program."""
        st = self.state_of(mac)
        if st == STATE_BLOCKED:
            self.stats["dropped"] += 1
            return V_DROP
        if st != STATE_WALLED:
            return V_FORWARD
        if self.is_destination_allowed(dst_ip, dst_port, proto) or \
                (dst_port == 0 and dst_ip in self.allowed_destinations):
            self.stats["allowed"] += 1
            return V_FORWARD
        if proto == 6 and dst_port in (80, 8080):
            self.stats["redirects"] += 1
            for cb in self._redirect_cbs:
                try:
                    cb(mac.lower(), dst_ip)
                except Exception:
                    pass
            return V_REDIRECT
        self.stats["dropped"] += 1
        return V_DROP

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


def attach_dns(manager: Manager, resolver, portal_ips: List[str]):
    """Keep the DNS resolver's walled-client registry in sync with the
    garden (the reference couples pkg/dns WalledGardenClients to the
    walled-garden state the same way): entries in STATE_WALLED resolve
    everything to the portal; activation/removal/expiry releases them.
    Returns the hook (already registered) for tests."""
    resolver.set_intercept_all(portal_ips)

    def hook(entry: Entry):
        tracked = manager.state_of(entry.mac) is not None
        if tracked and entry.state == STATE_WALLED:
            if entry.ip:
                resolver.add_walled_client(entry.ip)
        elif entry.ip:
            resolver.remove_walled_client(entry.ip)

    manager.set_dataplane_hooks(hook)
    # adopt anything already quarantined
    with manager._lock:
        for e in manager.entries.values():
            if e.state == STATE_WALLED and e.ip:
                resolver.add_walled_client(e.ip)
    return hook
