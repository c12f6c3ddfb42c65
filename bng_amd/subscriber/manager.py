"""Generic subscriber session manager over pluggable Authenticator and
AddressAllocator interfaces (ref pkg/subscriber/manager.go:15-56,
types.go:64-238): session lifecycle + events, shared by DHCP and PPPoE
access methods."""
from __future__ import annotations

import threading
import time
import uuid
from dataclasses import dataclass, field
from typing import Callable, Dict, List, Optional, Protocol

S_CREATED = "created"
S_AUTHENTICATING = "authenticating"
S_AUTHENTICATED = "authenticated"
S_ADDRESSED = "addressed"
S_ACTIVE = "active"
S_WALLED = "walled_garden"
S_TERMINATED = "terminated"


@dataclass
class Session:
    id: str
    subscriber_id: str
    mac: str = ""
    ip: str = ""
    access_type: str = "dhcp"       # dhcp | pppoe | static
    state: str = S_AUTHENTICATING
    policy_name: str = ""
    isp_id: str = ""
    started_at: float = field(default_factory=time.time)
    last_activity: float = field(default_factory=time.time)
    terminated_at: float = 0.0
    input_octets: int = 0
    output_octets: int = 0
    attributes: dict = field(default_factory=dict)


class Authenticator(Protocol):
    def authenticate(self, subscriber_id: str, credentials: dict) -> bool: ...


class AddressAllocator(Protocol):
    def allocate(self, subscriber_id: str) -> str: ...
    def release(self, subscriber_id: str) -> None: ...


class AllowAllAuthenticator:
    def authenticate(self, subscriber_id, credentials):
        return True


class Manager:
    def __init__(self, authenticator: Optional[Authenticator] = None,
                 allocator: Optional[AddressAllocator] = None,
                 max_sessions: int = 1_000_000,
                 idle_timeout: float = 0.0,
                 session_timeout: float = 0.0):
        self.auth = authenticator or AllowAllAuthenticator()
        self.allocator = allocator
        self.max_sessions = max_sessions
        self.idle_timeout = idle_timeout
        self.session_timeout = session_timeout
        self.sessions: Dict[str, Session] = {}
        self.by_subscriber: Dict[str, str] = {}
        self.by_ip: Dict[str, str] = {}
        self.by_mac: Dict[str, str] = {}
        self._lock = threading.RLock()
        self._listeners: List[Callable[[str, Session], None]] = []
        self.stats = {"created": 0, "auth_failed": 0, "terminated": 0,
                      "rejected_capacity": 0, "walled": 0,
                      "cleaned_up": 0}

    def on_event(self, cb: Callable[[str, Session], None]):
        self._listeners.append(cb)

    def _emit(self, event: str, s: Session):
        for cb in self._listeners:
            try:
                cb(event, s)
            except Exception:
                pass

    def create_session(self, subscriber_id: str, credentials: dict = None,
                       mac: str = "", access_type: str = "dhcp",
                       ip: str = "") -> Optional[Session]:
        with self._lock:
            if len(self.sessions) >= self.max_sessions:
                self.stats["rejected_capacity"] += 1
                return None
            existing = self.by_subscriber.get(subscriber_id)
            if existing:
                return self.sessions[existing]
        if not self.auth.authenticate(subscriber_id, credentials or {}):
            self.stats["auth_failed"] += 1
            return None
        if not ip and self.allocator is not None:
            ip = self.allocator.allocate(subscriber_id)
        s = Session(id=uuid.uuid4().hex[:12], subscriber_id=subscriber_id,
                    mac=mac, ip=ip, access_type=access_type,
                    state=S_ACTIVE)
        with self._lock:
            self.sessions[s.id] = s
            self.by_subscriber[subscriber_id] = s.id
            if ip:
                self.by_ip[ip] = s.id
            if mac:
                self.by_mac[mac] = s.id
        self.stats["created"] += 1
        self._emit("session_start", s)
        return s

    # --------------------------- staged lifecycle (ref manager.go
    # CreateSession -> Authenticate -> AssignAddress -> ActivateSession)
    def open_session(self, subscriber_id: str, mac: str = "",
                     access_type: str = "dhcp", isp_id: str = "",
                     metadata: Optional[dict] = None) -> Optional[Session]:
        with self._lock:
            if len(self.sessions) >= self.max_sessions:
                self.stats["rejected_capacity"] += 1
                return None
            existing = self.by_subscriber.get(subscriber_id)
            if existing:
                return self.sessions[existing]
            s = Session(id=uuid.uuid4().hex[:12],
                        subscriber_id=subscriber_id, mac=mac,
                        access_type=access_type, state=S_CREATED,
                        isp_id=isp_id,
                        attributes=dict(metadata or {}))
            self.sessions[s.id] = s
            self.by_subscriber[subscriber_id] = s.id
            if mac:
                self.by_mac[mac] = s.id
        self.stats["created"] += 1
        self._emit("session_created", s)
        return s

    def authenticate(self, session_id: str,
                     credentials: Optional[dict] = None) -> bool:
        s = self.get(session_id)
        if s is None:
            return False
        if not self.auth.authenticate(s.subscriber_id, credentials or {}):
            self.stats["auth_failed"] += 1
            return False
        s.state = S_AUTHENTICATED
        return True

    def assign_address(self, session_id: str) -> Optional[str]:
        s = self.get(session_id)
        if s is None or self.allocator is None:
            return None
        ip = self.allocator.allocate(s.subscriber_id)
        with self._lock:
            s.ip = ip
            self.by_ip[ip] = s.id
        s.state = S_ADDRESSED
        return ip

    def activate_session(self, session_id: str,
                         walled: bool = False) -> bool:
        s = self.get(session_id)
        if s is None:
            return False
        s.state = S_WALLED if walled else S_ACTIVE
        if walled:
            self.stats["walled"] += 1
        self._emit("session_start", s)
        return True

    def set_walled_garden(self, session_id: str) -> bool:
        s = self.get(session_id)
        if s is None or s.state == S_WALLED:
            return False
        s.state = S_WALLED
        self.stats["walled"] += 1
        self._emit("session_walled", s)
        return True

    def clear_walled_garden(self, session_id: str) -> bool:
        s = self.get(session_id)
        if s is None or s.state != S_WALLED:
            return False
        s.state = S_ACTIVE
        self._emit("session_released", s)
        return True

    def terminate_session(self, session_id: str,
                          reason: str = "") -> bool:
        with self._lock:
            s = self.sessions.pop(session_id, None)
            if s is None:
                return False
            self.by_subscriber.pop(s.subscriber_id, None)
            if s.ip:
                self.by_ip.pop(s.ip, None)
            if s.mac:
                self.by_mac.pop(s.mac, None)
        s.state = S_TERMINATED
        s.terminated_at = time.time()
        s.attributes["terminate_reason"] = reason
        if self.allocator is not None:
            try:
                self.allocator.release(s.subscriber_id)
            except Exception:
                pass
        self.stats["terminated"] += 1
        self._emit("session_stop", s)
        return True

    def get(self, session_id: str) -> Optional[Session]:
        with self._lock:
            return self.sessions.get(session_id)

    def get_by_subscriber(self, subscriber_id: str) -> Optional[Session]:
        with self._lock:
            sid = self.by_subscriber.get(subscriber_id)
            return self.sessions.get(sid) if sid else None

    def get_by_ip(self, ip: str) -> Optional[Session]:
        with self._lock:
            sid = self.by_ip.get(ip)
            return self.sessions.get(sid) if sid else None

    def get_by_mac(self, mac: str) -> Optional[Session]:
        with self._lock:
            sid = self.by_mac.get(mac)
            return self.sessions.get(sid) if sid else None

    def list_sessions(self) -> List[Session]:
        with self._lock:
            return list(self.sessions.values())

    def list_by_isp(self, isp_id: str) -> List[Session]:
        with self._lock:
            return [s for s in self.sessions.values()
                    if s.isp_id == isp_id]

    def cleanup(self, now: Optional[float] = None) -> int:
        """Idle/session-timeout reaping (ref manager.go cleanup loop)."""
        now = now if now is not None else time.time()
        with self._lock:
            dead = []
            for s in self.sessions.values():
                if self.idle_timeout and \
                        now - s.last_activity > self.idle_timeout:
                    dead.append((s.id, "idle-timeout"))
                elif self.session_timeout and \
                        now - s.started_at > self.session_timeout:
                    dead.append((s.id, "session-timeout"))
        for sid, why in dead:
            self.terminate_session(sid, reason=why)
        self.stats["cleaned_up"] += len(dead)
        return len(dead)

    def count(self) -> int:
        with self._lock:
            return len(self.sessions)

    def update_counters(self, session_id: str, input_octets: int,
                        output_octets: int):
        with self._lock:
            s = self.sessions.get(session_id)
            if s:
                s.input_octets = input_octets
                s.output_octets = output_octets
                s.last_activity = time.time()
