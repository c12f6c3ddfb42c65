"""Typed in-memory operational store (ref pkg/state/store.go:15-1049).

The reference keeps a mutex-guarded runtime database of subscribers,
pools, leases, sessions and NAT bindings with secondary indexes
(MAC / NTE / IP), capacity limits, pool selection by priority/ISP/class,
and periodic cleanup loops (expired leases, idle or hard-timed-out
sessions, expired NAT bindings).  This is that database; durable JSON
persistence stays in state.store.StateStore."""
from __future__ import annotations

import threading
import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional


class StateError(Exception):
    pass


class LimitExceeded(StateError):
    pass


@dataclass
class Subscriber:
    id: str
    mac: str = ""
    nte_id: str = ""
    isp_id: str = ""
    klass: str = ""
    enabled: bool = True


@dataclass
class Pool:
    id: str
    name: str = ""
    version: int = 4
    enabled: bool = True
    priority: int = 0
    total_addresses: int = 0
    allocated_addresses: int = 0
    reserved_addresses: int = 0
    isp_ids: List[str] = field(default_factory=list)
    subscriber_class: List[str] = field(default_factory=list)


@dataclass
class Lease:
    id: str
    subscriber_id: str = ""
    mac: str = ""
    ipv4: str = ""
    ipv6: str = ""
    expires_at: float = 0.0
    state: str = "active"


@dataclass
class Session:
    id: str
    subscriber_id: str = ""
    mac: str = ""
    ipv4: str = ""
    idle_timeout: float = 0.0      # 0 = no idle limit
    session_timeout: float = 0.0   # 0 = no hard limit
    started_at: float = field(default_factory=time.time)
    last_activity: float = field(default_factory=time.time)
    bytes_in: int = 0
    bytes_out: int = 0


@dataclass
class NATBinding:
    id: str
    private_ip: str = ""
    private_port: int = 0
    public_ip: str = ""
    public_port: int = 0
    protocol: int = 17
    expires_at: float = 0.0


@dataclass
class Config:
    max_subscribers: int = 1_000_000
    max_leases: int = 2_000_000
    max_sessions: int = 1_000_000
    max_nat_bindings: int = 8_000_000
    lease_cleanup_interval: float = 60.0
    session_cleanup_interval: float = 60.0
    nat_cleanup_interval: float = 60.0


class RuntimeStore:
    def __init__(self, config: Optional[Config] = None):
        self.cfg = config or Config()
        self._lock = threading.RLock()
        self.subscribers: Dict[str, Subscriber] = {}
        self._sub_by_mac: Dict[str, str] = {}
        self._sub_by_nte: Dict[str, str] = {}
        self.pools: Dict[str, Pool] = {}
        self.leases: Dict[str, Lease] = {}
        self._lease_by_ip: Dict[str, str] = {}
        self._lease_by_mac: Dict[str, str] = {}
        self.sessions: Dict[str, Session] = {}
        self._sess_by_mac: Dict[str, str] = {}
        self._sess_by_ip: Dict[str, str] = {}
        self.nat_bindings: Dict[str, NATBinding] = {}
        self._nat_by_priv: Dict[tuple, str] = {}
        self._stop = threading.Event()
        self._threads: List[threading.Thread] = []
        self.cleanup_stats = {"leases_expired": 0, "sessions_reaped": 0,
                              "nat_expired": 0}

    # ------------------------------------------------------- lifecycle
    def start(self):
        for iv, fn in ((self.cfg.lease_cleanup_interval,
                        self.cleanup_expired_leases),
                       (self.cfg.session_cleanup_interval,
                        self.cleanup_idle_sessions),
                       (self.cfg.nat_cleanup_interval,
                        self.cleanup_expired_nat)):
            t = threading.Thread(target=self._loop, args=(iv, fn),
                                 daemon=True)
            t.start()
            self._threads.append(t)
        return self

    def _loop(self, interval, fn):
        while not self._stop.wait(interval):
            fn()

    def stop(self):
        self._stop.set()

    def stats(self) -> Dict[str, int]:
        with self._lock:
            return {"subscribers": len(self.subscribers),
                    "pools": len(self.pools),
                    "leases": len(self.leases),
                    "sessions": len(self.sessions),
                    "nat_bindings": len(self.nat_bindings),
                    **self.cleanup_stats}

    # ----------------------------------------------------- subscribers
    def create_subscriber(self, sub: Subscriber):
        with self._lock:
            if len(self.subscribers) >= self.cfg.max_subscribers:
                raise LimitExceeded("max subscribers reached")
            if sub.id in self.subscribers:
                raise StateError(f"subscriber {sub.id} exists")
            if sub.mac and sub.mac in self._sub_by_mac:
                raise StateError(f"MAC {sub.mac} already registered")
            self.subscribers[sub.id] = sub
            if sub.mac:
                self._sub_by_mac[sub.mac] = sub.id
            if sub.nte_id:
                self._sub_by_nte[sub.nte_id] = sub.id

    def get_subscriber(self, sid: str) -> Optional[Subscriber]:
        with self._lock:
            return self.subscribers.get(sid)

    def get_subscriber_by_mac(self, mac: str) -> Optional[Subscriber]:
        with self._lock:
            sid = self._sub_by_mac.get(mac)
            return self.subscribers.get(sid) if sid else None

    def get_subscriber_by_nte(self, nte: str) -> Optional[Subscriber]:
        with self._lock:
            sid = self._sub_by_nte.get(nte)
            return self.subscribers.get(sid) if sid else None

    def delete_subscriber(self, sid: str):
        with self._lock:
            sub = self.subscribers.pop(sid, None)
            if sub is None:
                raise StateError(f"subscriber {sid} not found")
            self._sub_by_mac.pop(sub.mac, None)
            self._sub_by_nte.pop(sub.nte_id, None)

    def list_subscribers(self) -> List[Subscriber]:
        with self._lock:
            return list(self.subscribers.values())

    # ----------------------------------------------------------- pools
    def create_pool(self, pool: Pool):
        with self._lock:
            if pool.id in self.pools:
                raise StateError(f"pool {pool.id} exists")
            self.pools[pool.id] = pool

    def get_pool(self, pid: str) -> Optional[Pool]:
        with self._lock:
            return self.pools.get(pid)

    def get_pool_by_name(self, name: str) -> Optional[Pool]:
        with self._lock:
            for p in self.pools.values():
                if p.name == name:
                    return p
            return None

    def find_pool_for_subscriber(self, sub: Subscriber,
                                 version: int = 4) -> Pool:
        """Highest-priority enabled pool with free capacity matching the
        subscriber's ISP and class (ref store.go:356-415)."""
        with self._lock:
            best, best_prio = None, -1
            for p in self.pools.values():
                if not p.enabled or p.version != version:
                    continue
                if p.allocated_addresses >= \
                        p.total_addresses - p.reserved_addresses:
                    continue
                if p.isp_ids and sub.isp_id not in p.isp_ids:
                    continue
                if p.subscriber_class and \
                        sub.klass not in p.subscriber_class:
                    continue
                if p.priority > best_prio:
                    best, best_prio = p, p.priority
            if best is None:
                raise StateError("no suitable pool found")
            return best

    def delete_pool(self, pid: str):
        with self._lock:
            if self.pools.pop(pid, None) is None:
                raise StateError(f"pool {pid} not found")

    # ---------------------------------------------------------- leases
    def create_lease(self, lease: Lease):
        with self._lock:
            if len(self.leases) >= self.cfg.max_leases:
                raise LimitExceeded("max leases reached")
            if lease.id in self.leases:
                raise StateError(f"lease {lease.id} exists")
            self.leases[lease.id] = lease
            for ip in (lease.ipv4, lease.ipv6):
                if ip:
                    self._lease_by_ip[ip] = lease.id
            if lease.mac:
                self._lease_by_mac[lease.mac] = lease.id

    def get_lease(self, lid: str) -> Optional[Lease]:
        with self._lock:
            return self.leases.get(lid)

    def get_lease_by_ip(self, ip: str) -> Optional[Lease]:
        with self._lock:
            lid = self._lease_by_ip.get(ip)
            return self.leases.get(lid) if lid else None

    def get_lease_by_mac(self, mac: str) -> Optional[Lease]:
        with self._lock:
            lid = self._lease_by_mac.get(mac)
            return self.leases.get(lid) if lid else None

    def renew_lease(self, lid: str, duration: float):
        with self._lock:
            lease = self.leases.get(lid)
            if lease is None:
                raise StateError(f"lease {lid} not found")
            lease.expires_at = time.time() + duration

    def delete_lease(self, lid: str):
        with self._lock:
            self._drop_lease_unlocked(lid)

    def _drop_lease_unlocked(self, lid: str):
        lease = self.leases.pop(lid, None)
        if lease is None:
            raise StateError(f"lease {lid} not found")
        for ip in (lease.ipv4, lease.ipv6):
            if ip:
                self._lease_by_ip.pop(ip, None)
        if lease.mac:
            self._lease_by_mac.pop(lease.mac, None)

    def cleanup_expired_leases(self, now: Optional[float] = None) -> int:
        now = now if now is not None else time.time()
        with self._lock:
            dead = [l.id for l in self.leases.values()
                    if l.expires_at and now > l.expires_at]
            for lid in dead:
                self.leases[lid].state = "expired"
                self._drop_lease_unlocked(lid)
            self.cleanup_stats["leases_expired"] += len(dead)
            return len(dead)

    # -------------------------------------------------------- sessions
    def create_session(self, sess: Session):
        with self._lock:
            if len(self.sessions) >= self.cfg.max_sessions:
                raise LimitExceeded("max sessions reached")
            if sess.id in self.sessions:
                raise StateError(f"session {sess.id} exists")
            self.sessions[sess.id] = sess
            if sess.mac:
                self._sess_by_mac[sess.mac] = sess.id
            if sess.ipv4:
                self._sess_by_ip[sess.ipv4] = sess.id

    def get_session(self, sid: str) -> Optional[Session]:
        with self._lock:
            return self.sessions.get(sid)

    def get_session_by_mac(self, mac: str) -> Optional[Session]:
        with self._lock:
            sid = self._sess_by_mac.get(mac)
            return self.sessions.get(sid) if sid else None

    def get_session_by_ip(self, ip: str) -> Optional[Session]:
        with self._lock:
            sid = self._sess_by_ip.get(ip)
            return self.sessions.get(sid) if sid else None

    def update_session_activity(self, sid: str, bytes_in: int = 0,
                                bytes_out: int = 0):
        with self._lock:
            s = self.sessions.get(sid)
            if s is None:
                raise StateError(f"session {sid} not found")
            s.bytes_in += bytes_in
            s.bytes_out += bytes_out
            s.last_activity = time.time()

    def delete_session(self, sid: str):
        with self._lock:
            self._drop_session_unlocked(sid)

    def _drop_session_unlocked(self, sid: str):
        s = self.sessions.pop(sid, None)
        if s is None:
            raise StateError(f"session {sid} not found")
        if s.mac:
            self._sess_by_mac.pop(s.mac, None)
        if s.ipv4:
            self._sess_by_ip.pop(s.ipv4, None)

    def cleanup_idle_sessions(self, now: Optional[float] = None) -> int:
        """Idle-timeout AND hard session-timeout reaping (ref
        store.go cleanupIdleSessions)."""
        now = now if now is not None else time.time()
        with self._lock:
            dead = []
            for s in self.sessions.values():
                if s.idle_timeout and now - s.last_activity > s.idle_timeout:
                    dead.append(s.id)
                elif s.session_timeout and \
                        now - s.started_at > s.session_timeout:
                    dead.append(s.id)
            for sid in dead:
                self._drop_session_unlocked(sid)
            self.cleanup_stats["sessions_reaped"] += len(dead)
            return len(dead)

    # ---------------------------------------------------- NAT bindings
    def create_nat_binding(self, b: NATBinding):
        with self._lock:
            if len(self.nat_bindings) >= self.cfg.max_nat_bindings:
                raise LimitExceeded("max NAT bindings reached")
            if b.id in self.nat_bindings:
                raise StateError(f"binding {b.id} exists")
            self.nat_bindings[b.id] = b
            self._nat_by_priv[(b.private_ip, b.private_port,
                               b.protocol)] = b.id

    def get_nat_binding_by_private(self, ip: str, port: int,
                                   protocol: int) -> Optional[NATBinding]:
        with self._lock:
            bid = self._nat_by_priv.get((ip, port, protocol))
            return self.nat_bindings.get(bid) if bid else None

    def delete_nat_binding(self, bid: str):
        with self._lock:
            b = self.nat_bindings.pop(bid, None)
            if b is None:
                raise StateError(f"binding {bid} not found")
            self._nat_by_priv.pop((b.private_ip, b.private_port,
                                   b.protocol), None)

    def cleanup_expired_nat(self, now: Optional[float] = None) -> int:
        now = now if now is not None else time.time()
        with self._lock:
            dead = [b.id for b in self.nat_bindings.values()
                    if b.expires_at and now > b.expires_at]
            for bid in dead:
                b = self.nat_bindings.pop(bid)
                self._nat_by_priv.pop((b.private_ip, b.private_port,
                                       b.protocol), None)
            self.cleanup_stats["nat_expired"] += len(dead)
            return len(dead)
