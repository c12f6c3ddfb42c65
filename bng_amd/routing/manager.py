"""Policy routing + subscriber routes (ref pkg/routing/manager.go:159-573,
subscriber_routes.go:183-671, netlink_stub.go:12-50, health.go:36-174).

RoutingPlatform abstracts the kernel (netlink in production, in-memory
here and in tests, the same seam the reference uses for non-Linux);
Manager builds per-ISP route tables + `from <subscriberIP> lookup
<table>` rules; SubscriberRouteManager injects per-subscriber /32 BGP
routes with a retry queue and periodic reconcile; HealthChecker probes
next-hops with hysteresis."""
from __future__ import annotations

import socket
import threading
import time
from dataclasses import dataclass, field
from typing import Callable, Dict, List, Optional, Set, Tuple


@dataclass
class Route:
    prefix: str
    next_hop: str = ""
    table: int = 254
    metric: int = 0
    device: str = ""                  # output interface (RTA_OIF)


@dataclass
class Rule:
    src: str
    table: int
    priority: int = 1000
    fwmark: int = 0                   # FRA_FWMARK match


class RoutingPlatform:
    """Kernel seam (ref RoutingPlatform manager.go:159)."""

    def add_route(self, r: Route) -> None: ...
    def del_route(self, r: Route) -> None: ...
    def add_rule(self, r: Rule) -> None: ...
    def del_rule(self, r: Rule) -> None: ...
    def routes(self, table: int) -> List[Route]: ...


class MemoryPlatform(RoutingPlatform):
    """In-memory platform (ref netlink_stub.go:12-50)."""

    def __init__(self):
        self._routes: Dict[int, Dict[str, Route]] = {}
        self._rules: List[Rule] = []
        self._lock = threading.RLock()
        self.fail = False

    def add_route(self, r: Route):
        if self.fail:
            raise OSError("netlink failure (simulated)")
        with self._lock:
            self._routes.setdefault(r.table, {})[r.prefix] = r

    def del_route(self, r: Route):
        with self._lock:
            self._routes.get(r.table, {}).pop(r.prefix, None)

    def add_rule(self, r: Rule):
        if self.fail:
            raise OSError("netlink failure (simulated)")
        with self._lock:
            self._rules.append(r)

    def del_rule(self, r: Rule):
        with self._lock:
            self._rules = [x for x in self._rules
                           if (x.src, x.table) != (r.src, r.table)]

    def routes(self, table: int) -> List[Route]:
        with self._lock:
            return list(self._routes.get(table, {}).values())

    def rules(self) -> List[Rule]:
        with self._lock:
            return list(self._rules)


class Manager:
    """Per-ISP policy routing (ref manager.go:258-573): each ISP gets its
    own route table; subscriber IPs get `from <ip> lookup <table>`
    rules so different ISPs' traffic exits via different upstreams."""

    ISP_TABLE_BASE = 100

    def __init__(self, platform: Optional[RoutingPlatform] = None):
        self.platform = platform or MemoryPlatform()
        self.isp_tables: Dict[str, int] = {}
        self._next_table = self.ISP_TABLE_BASE
        self._lock = threading.RLock()

    def create_isp_table(self, isp_id: str, default_next_hop: str,
                         device: str = "") -> int:
        """ref manager.go:521 CreateISPTable; the default route needs
        a gateway or an output device (a gatewayless, deviceless
        default is invalid on a real kernel)."""
        with self._lock:
            if isp_id in self.isp_tables:
                return self.isp_tables[isp_id]
            table = self._next_table
            self._next_table += 1
            self.isp_tables[isp_id] = table
        if default_next_hop or device:
            self.platform.add_route(Route("0.0.0.0/0", default_next_hop,
                                          table, device=device))
        return table

    def remove_isp_table(self, isp_id: str):
        with self._lock:
            table = self.isp_tables.pop(isp_id, None)
        if table is not None:
            for r in self.platform.routes(table):
                self.platform.del_route(r)

    def add_subscriber_rule(self, subscriber_ip: str, isp_id: str):
        with self._lock:
            table = self.isp_tables.get(isp_id)
        if table is None:
            raise KeyError(f"no table for ISP {isp_id}")
        self.platform.add_rule(Rule(subscriber_ip, table))

    def remove_subscriber_rule(self, subscriber_ip: str, isp_id: str):
        with self._lock:
            table = self.isp_tables.get(isp_id)
        if table is not None:
            self.platform.del_rule(Rule(subscriber_ip, table))


class SubscriberRouteManager:
    """Per-subscriber /32 route injection into BGP with retry + reconcile
    (ref subscriber_routes.go:183-671)."""

    def __init__(self, bgp, retry_interval: float = 5.0,
                 max_retries: int = 5):
        self.bgp = bgp
        self.retry_interval = retry_interval
        self.max_retries = max_retries
        self.desired: Set[str] = set()        # /32 prefixes
        self.installed: Set[str] = set()
        self.retry_queue: Dict[str, int] = {}  # prefix -> attempts
        self._lock = threading.RLock()
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self.stats = {"installed": 0, "withdrawn": 0, "retries": 0,
                      "gave_up": 0, "reconciled": 0}

    def start(self):
        self._thread = threading.Thread(target=self._loop, daemon=True)
        self._thread.start()
        return self

    def stop(self):
        self._stop.set()

    def add_subscriber_route(self, ip: str):
        prefix = f"{ip}/32"
        with self._lock:
            self.desired.add(prefix)
        self._try_install(prefix)

    def remove_subscriber_route(self, ip: str):
        prefix = f"{ip}/32"
        with self._lock:
            self.desired.discard(prefix)
            self.retry_queue.pop(prefix, None)
        try:
            self.bgp.withdraw_prefix(prefix)
            with self._lock:
                self.installed.discard(prefix)
            self.stats["withdrawn"] += 1
        except Exception:
            pass

    def _try_install(self, prefix: str) -> bool:
        try:
            self.bgp.announce_prefix(prefix)
            with self._lock:
                self.installed.add(prefix)
                self.retry_queue.pop(prefix, None)
            self.stats["installed"] += 1
            return True
        except Exception:
            with self._lock:
                self.retry_queue[prefix] = \
                    self.retry_queue.get(prefix, 0) + 1
            return False

    def _loop(self):
        while not self._stop.wait(self.retry_interval):
            self.retry_pending()
            self.reconcile()

    def retry_pending(self) -> int:
        with self._lock:
            pending = dict(self.retry_queue)
        done = 0
        for prefix, attempts in pending.items():
            if attempts > self.max_retries:
                with self._lock:
                    self.retry_queue.pop(prefix, None)
                self.stats["gave_up"] += 1
                continue
            self.stats["retries"] += 1
            if self._try_install(prefix):
                done += 1
        return done

    def reconcile(self) -> int:
        """Desired-vs-installed drift repair (ref :reconcile)."""
        with self._lock:
            missing = self.desired - self.installed - \
                set(self.retry_queue)
            stale = self.installed - self.desired
        fixed = 0
        for p in missing:
            if self._try_install(p):
                fixed += 1
        for p in stale:
            try:
                self.bgp.withdraw_prefix(p)
                with self._lock:
                    self.installed.discard(p)
                fixed += 1
            except Exception:
                pass
        if fixed:
            self.stats["reconciled"] += fixed
        return fixed


class HealthChecker:
    """Next-hop health with hysteresis (ref health.go:36-174): TCP
    connect (or pluggable probe); up after N successes, down after M
    failures."""

    def __init__(self, target: str, port: int = 179, interval: float = 1.0,
                 timeout: float = 1.0, up_threshold: int = 2,
                 down_threshold: int = 3,
                 probe: Optional[Callable[[], bool]] = None):
        self.target = target
        self.port = port
        self.interval = interval
        self.timeout = timeout
        self.up_threshold = up_threshold
        self.down_threshold = down_threshold
        self.probe = probe or self._tcp_probe
        self.healthy = True
        self._succ = 0
        self._fail = 0
        self._listeners: List[Callable[[bool], None]] = []
        self._stop = threading.Event()

    def _tcp_probe(self) -> bool:
        try:
            with socket.create_connection((self.target, self.port),
                                          timeout=self.timeout):
                return True
        except OSError:
            return False

    def on_change(self, cb: Callable[[bool], None]):
        self._listeners.append(cb)

    def check_once(self) -> bool:
        ok = False
        try:
            ok = bool(self.probe())
        except Exception:
            ok = False
        if ok:
            self._succ += 1
            self._fail = 0
            if not self.healthy and self._succ >= self.up_threshold:
                self._set(True)
        else:
            self._fail += 1
            self._succ = 0
            if self.healthy and self._fail >= self.down_threshold:
                self._set(False)
        return ok

    def _set(self, healthy: bool):
        self.healthy = healthy
        for cb in self._listeners:
            try:
                cb(healthy)
            except Exception:
                pass

    def start(self):
        def loop():
            while not self._stop.wait(self.interval):
                self.check_once()
        threading.Thread(target=loop, daemon=True).start()
        return self

    def stop(self):
        self._stop.set()


@dataclass
class SessionRouteConfig:
    """ref session_integration.go:35-61 SessionRouteConfig."""
    enable_injection: bool = True
    enable_withdrawal: bool = True
    injection_timeout: float = 5.0
    withdrawal_timeout: float = 5.0
    default_subscriber_class: str = ""


@dataclass
class TrackedSession:
    session_id: str
    subscriber_id: str
    subscriber_class: str = ""
    ipv4: str = ""
    ipv6: str = ""
    route_injected: bool = False
    injected_at: float = 0.0


class SessionRouteIntegration:
    """Session lifecycle -> per-subscriber route plumbing (ref
    session_integration.go:16-353): activate injects the /32, terminate
    withdraws it, intermediate states are ignored, and RecoverRoutes
    re-injects everything after an FRR restart."""

    def __init__(self, route_manager: SubscriberRouteManager,
                 config: Optional[SessionRouteConfig] = None):
        self.rm = route_manager
        self.config = config or SessionRouteConfig()
        self.sessions: Dict[str, TrackedSession] = {}
        self._lock = threading.RLock()
        self.stats = {"activations": 0, "terminations": 0,
                      "routes_injected": 0, "routes_withdrawn": 0,
                      "recoveries": 0}

    def on_session_activate(self, session_id: str, subscriber_id: str,
                            ipv4: str = "", ipv6: str = "",
                            subscriber_class: str = "") -> bool:
        """ref :113-186; returns whether a route was injected."""
        if not self.config.enable_injection:
            return False
        with self._lock:
            ex = self.sessions.get(session_id)
            if ex is not None and ex.route_injected:
                return False                       # already injected
            t = TrackedSession(
                session_id, subscriber_id,
                subscriber_class or self.config.default_subscriber_class,
                ipv4, ipv6)
            self.sessions[session_id] = t
        self.stats["activations"] += 1
        if ipv4:
            self.rm.add_subscriber_route(ipv4)
            with self._lock:
                t.route_injected = True
                t.injected_at = time.time()
            self.stats["routes_injected"] += 1
            return True
        return False

    def on_session_terminate(self, session_id: str,
                             reason: str = "") -> bool:
        """ref :188-237; returns whether a route was withdrawn."""
        if not self.config.enable_withdrawal:
            return False
        with self._lock:
            t = self.sessions.pop(session_id, None)
        if t is None:
            return False
        self.stats["terminations"] += 1
        if t.ipv4 and t.route_injected:
            self.rm.remove_subscriber_route(t.ipv4)
            self.stats["routes_withdrawn"] += 1
            return True
        return False

    def on_session_state_change(self, session_id: str, subscriber_id: str,
                                old_state: str, new_state: str,
                                ipv4: str = "", ipv6: str = "",
                                subscriber_class: str = "",
                                reason: str = "") -> bool:
        """ref :239-256: active -> inject, terminal states -> withdraw,
        intermediate states -> no route change."""
        if new_state == "active":
            return self.on_session_activate(session_id, subscriber_id,
                                            ipv4, ipv6, subscriber_class)
        if new_state in ("terminated", "error", "timeout"):
            return self.on_session_terminate(session_id, reason)
        return False

    def recover_routes(self) -> int:
        """Re-inject every tracked session's route after an FRR restart
        (ref :258-270)."""
        with self._lock:
            tracked = list(self.sessions.values())
        n = 0
        for t in tracked:
            if t.ipv4:
                self.rm.add_subscriber_route(t.ipv4)
                with self._lock:
                    t.route_injected = True
                n += 1
        self.stats["recoveries"] += 1
        return n

    def tracked_sessions(self) -> List[TrackedSession]:
        with self._lock:
            return list(self.sessions.values())
