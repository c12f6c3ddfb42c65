"""HA protocol: roles, sync message schema, session store interface
(ref pkg/ha/protocol.go:18-177)."""
from __future__ import annotations

import threading
import time
from dataclasses import asdict, dataclass, field
from typing import Dict, List, Optional

ROLE_ACTIVE = "active"
ROLE_STANDBY = "standby"
ROLE_UNKNOWN = "unknown"

SYNC_FULL = "full"
SYNC_ADD = "add"
SYNC_UPDATE = "update"
SYNC_DELETE = "delete"
SYNC_HEARTBEAT = "heartbeat"
SYNC_FULL_REQUEST = "full_request"


@dataclass
class SessionState:
    """Replicated session record (ref protocol.go:76-111)."""
    session_id: str
    subscriber_id: str = ""
    mac: str = ""
    ip: str = ""
    ipv6: str = ""
    gateway: str = ""
    vlan: int = 0
    s_tag: int = 0
    c_tag: int = 0
    access_type: str = "dhcp"
    policy_name: str = ""
    lease_expiry: float = 0.0
    nat_public_ip: str = ""
    nat_port_start: int = 0
    nat_port_end: int = 0
    created_at: float = field(default_factory=time.time)
    updated_at: float = field(default_factory=time.time)

    def to_dict(self):
        return asdict(self)

    @classmethod
    def from_dict(cls, d):
        return cls(**{k: v for k, v in d.items()
                      if k in cls.__dataclass_fields__})


@dataclass
class SyncMessage:
    type: str
    sessions: List[dict] = field(default_factory=list)
    timestamp: float = field(default_factory=time.time)
    seq: int = 0
    node_id: str = ""

    def to_dict(self):
        return asdict(self)

    @classmethod
    def from_dict(cls, d):
        return cls(**{k: v for k, v in d.items()
                      if k in cls.__dataclass_fields__})


class InMemorySessionStore:
    """ref pkg/ha/store.go InMemorySessionStore — the SessionStore both
    sides of the pair use; the standby's copy becomes authoritative at
    failover."""

    def __init__(self):
        self._sessions: Dict[str, SessionState] = {}
        self._lock = threading.RLock()

    def put(self, s: SessionState):
        with self._lock:
            s.updated_at = time.time()
            self._sessions[s.session_id] = s

    def get(self, session_id: str) -> Optional[SessionState]:
        with self._lock:
            return self._sessions.get(session_id)

    def delete(self, session_id: str):
        with self._lock:
            self._sessions.pop(session_id, None)

    def all(self) -> List[SessionState]:
        with self._lock:
            return list(self._sessions.values())

    def count(self) -> int:
        with self._lock:
            return len(self._sessions)

    def replace_all(self, sessions: List[SessionState]):
        with self._lock:
            self._sessions = {s.session_id: s for s in sessions}
