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
    # access context (ref types.go SessionRequest :9-39 / Session
    # :64-159): QinQ tags, NTE/PON identity, DHCP option-82
    s_tag: int = 0
    c_tag: int = 0
    nte_id: str = ""
    circuit_id: str = ""
    remote_id: str = ""
    # RADIUS-applied service attributes (ref Authenticate :229-259)
    radius_session_id: str = ""
    session_timeout: float = 0.0     # per-session override (s)
    idle_timeout: float = 0.0
    download_rate_bps: int = 0
    upload_rate_bps: int = 0
    qos_policy_id: str = ""
    walled_reason: str = ""
    state_reason: str = ""
    packets_in: int = 0
    packets_out: int = 0


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

    def authenticate_full(self, session_id: str,
                          credentials: Optional[dict] = None) -> dict:
        """Rich authentication applying RADIUS-style service attributes
        to the session (ref Authenticate manager.go:179-296).  Uses the
        authenticator's `authenticate_session(session, credentials) ->
        dict` when available (keys: success, subscriber_id, isp_id,
        radius_session_id, session_timeout, idle_timeout,
        download_rate_bps, upload_rate_bps, qos_policy_id,
        walled_garden, walled_reason, error), else falls back to the
        boolean protocol.  A walled-garden grant lands the session in
        StateWalledGarden instead of failing it."""
        s = self.get(session_id)
        if s is None:
            return {"success": False, "error": "session not found"}
        old_state = s.state
        s.state = S_AUTHENTICATING
        try:
            if hasattr(self.auth, "authenticate_session"):
                result = self.auth.authenticate_session(
                    s, credentials or {})
            else:
                ok = self.auth.authenticate(s.subscriber_id,
                                            credentials or {})
                result = {"success": bool(ok)}
        except Exception as e:
            s.state = old_state
            s.state_reason = f"auth error: {e}"
            self.stats["auth_failed"] += 1
            self._emit("session_auth_fail", s)
            return {"success": False, "error": str(e)}
        if not result.get("success"):
            s.state = old_state
            s.state_reason = result.get("error", "rejected")
            self.stats["auth_failed"] += 1
            self._emit("session_auth_fail", s)
            return result
        if result.get("subscriber_id"):
            with self._lock:
                self.by_subscriber.pop(s.subscriber_id, None)
                s.subscriber_id = result["subscriber_id"]
                self.by_subscriber[s.subscriber_id] = s.id
        s.isp_id = result.get("isp_id", s.isp_id)
        s.radius_session_id = result.get("radius_session_id", "")
        # only positive attributes override (ref :234-249)
        for key in ("session_timeout", "idle_timeout",
                    "download_rate_bps", "upload_rate_bps"):
            if result.get(key, 0) > 0:
                setattr(s, key, result[key])
        if result.get("qos_policy_id"):
            s.qos_policy_id = result["qos_policy_id"]
        if result.get("walled_garden"):
            s.state = S_WALLED
            s.walled_reason = result.get("walled_reason", "")
            self.stats["walled"] += 1
        else:
            s.state = S_AUTHENTICATED
        self.stats["auth_ok"] = self.stats.get("auth_ok", 0) + 1
        self._emit("session_auth", s)
        return result

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
        """Idle/session-timeout reaping; per-session RADIUS-applied
        timeouts override the manager defaults (ref manager.go
        cleanupExpiredSessions :648-690)."""
        now = now if now is not None else time.time()
        with self._lock:
            dead = []
            for s in self.sessions.values():
                sess_to = s.session_timeout or self.session_timeout
                idle_to = s.idle_timeout or self.idle_timeout
                if sess_to and now - s.started_at > sess_to:
                    dead.append((s.id, "session-timeout"))
                elif idle_to and now - s.last_activity > idle_to:
                    dead.append((s.id, "idle-timeout"))
        for sid, why in dead:
            self.terminate_session(sid, reason=why)
        self.stats["cleaned_up"] += len(dead)
        return len(dead)

    def update_activity(self, session_id: str, bytes_in: int = 0,
                        bytes_out: int = 0, packets_in: int = 0,
                        packets_out: int = 0) -> bool:
        """Accumulate traffic counters + touch last_activity (ref
        UpdateActivity manager.go:535-552)."""
        with self._lock:
            s = self.sessions.get(session_id)
            if s is None:
                return False
            s.input_octets += bytes_in
            s.output_octets += bytes_out
            s.packets_in += packets_in
            s.packets_out += packets_out
            s.last_activity = time.time()
        return True

    def manager_stats(self) -> dict:
        """ref ManagerStats types.go:278-288."""
        with self._lock:
            walled = sum(1 for s in self.sessions.values()
                         if s.state == S_WALLED)
            bytes_in = sum(s.input_octets for s in self.sessions.values())
            bytes_out = sum(s.output_octets
                            for s in self.sessions.values())
            return {"active_sessions": len(self.sessions),
                    "walled_garden_sessions": walled,
                    "total_sessions_created": self.stats["created"],
                    "total_sessions_ended": self.stats["terminated"],
                    "auth_successes": self.stats.get("auth_ok", 0),
                    "auth_failures": self.stats["auth_failed"],
                    "total_bytes_in": bytes_in,
                    "total_bytes_out": bytes_out}

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
