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

S_AUTHENTICATING = "authenticating"
S_ACTIVE = "active"
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
    started_at: float = field(default_factory=time.time)
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
                 max_sessions: int = 1_000_000):
        self.auth = authenticator or AllowAllAuthenticator()
        self.allocator = allocator
        self.max_sessions = max_sessions
        self.sessions: Dict[str, Session] = {}
        self.by_subscriber: Dict[str, str] = {}
        self.by_ip: Dict[str, str] = {}
        self._lock = threading.RLock()
        self._listeners: List[Callable[[str, Session], None]] = []
        self.stats = {"created": 0, "auth_failed": 0, "terminated": 0,
                      "rejected_capacity": 0}

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
        self.stats["created"] += 1
        self._emit("session_start", s)
        return s

    def terminate_session(self, session_id: str,
                          reason: str = "") -> bool:
        with self._lock:
            s = self.sessions.pop(session_id, None)
            if s is None:
                return False
            self.by_subscriber.pop(s.subscriber_id, None)
            if s.ip:
                self.by_ip.pop(s.ip, None)
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
