"""Audit severity model and retention management (ref pkg/audit
types.go:254-403, retention.go:9-357).

Severity is syslog-graded; every event action maps to a severity and a
category through dict tables (the reference's switch ladders,
types.go:291-366).  RetentionManager resolves how long an event class
must be kept — per-category days with per-action overrides — and
tracks LegalHold objects whose criteria (subscriber / IP / MAC /
session / action / time-window, every SET criterion must match,
retention.go:174-265) exempt matching events from expiry."""
from __future__ import annotations

import time
import uuid
from dataclasses import dataclass, field
from typing import Dict, List, Optional

# syslog-graded severities (ref types.go:254-288)
DEBUG, INFO, NOTICE, WARNING, ERROR, CRITICAL, ALERT, EMERGENCY = range(8)

SEVERITY_NAMES = ["DEBUG", "INFO", "NOTICE", "WARNING", "ERROR",
                  "CRITICAL", "ALERT", "EMERGENCY"]


def severity_name(sev: int) -> str:
    if 0 <= sev < len(SEVERITY_NAMES):
        return SEVERITY_NAMES[sev]
    return "UNKNOWN"


# action -> severity (ref types.go GetSeverity :291-366); anything
# unlisted is INFO
ACTION_SEVERITY: Dict[str, int] = {
    "auth_failure": WARNING, "auth_reject": WARNING,
    "policy_violation": WARNING,
    "system_error": ERROR,
    "nat_mapping": DEBUG, "nat_expiry": DEBUG,
    "device_registration_failure": WARNING,
    "device_deregistration": NOTICE,
    "api_auth_failure": WARNING, "api_access_denied": WARNING,
    "api_rate_limited": WARNING,
    "suspicious_activity": WARNING,
    "brute_force_detected": ALERT, "unauthorized_access": ALERT,
    "dhcp_starvation_attempt": ALERT,
    "mac_spoof": CRITICAL, "ip_spoof": CRITICAL,
    "resource_exhausted": WARNING,
    "tls_handshake_failure": WARNING, "mtls_auth_failure": WARNING,
    "certificate_expiring": WARNING,
    "certificate_expired": ERROR, "certificate_invalid": ERROR,
    "certificate_pin_failed": CRITICAL, "certificate_revoked": CRITICAL,
    "ztp_bootstrap_failure": WARNING, "ztp_config_rejected": WARNING,
}


def action_severity(action: str) -> int:
    return ACTION_SEVERITY.get(action, INFO)


# action prefix -> category (ref types.go Category :367-403); used when
# the caller does not pass an explicit category
ACTION_CATEGORY_PREFIX = [
    ("session_", "session"), ("auth_", "auth"), ("dhcp_", "dhcp"),
    ("nat_", "nat"), ("policy_", "policy"),
    ("walledgarden_", "walledgarden"), ("config_", "admin"),
    ("admin_", "admin"), ("system_", "system"), ("device_", "device"),
    ("api_", "api"), ("resource_", "resource"), ("tls_", "tls"),
    ("certificate_", "tls"), ("mtls_", "tls"), ("ztp_", "ztp"),
]
SECURITY_ACTIONS = {"suspicious_activity", "brute_force_detected",
                    "unauthorized_access", "mac_spoof", "ip_spoof",
                    "dhcp_starvation_attempt"}


def action_category(action: str) -> str:
    if action in SECURITY_ACTIONS:
        return "security"
    for prefix, cat in ACTION_CATEGORY_PREFIX:
        if action.startswith(prefix):
            return cat
    return "other"


def standard_retention_policies() -> Dict[str, int]:
    """Days per category (ref retention.go:305-345 — the common legal
    data-retention floor for an ISP audit trail)."""
    return {"session": 365, "nat": 365, "auth": 365, "dhcp": 90,
            "admin": 730, "policy": 365, "walledgarden": 90,
            "system": 30, "device": 365, "api": 365, "security": 730,
            "resource": 365}


@dataclass
class LegalHold:
    """Preservation order: matching events are exempt from retention
    expiry (ref retention.go:26-41).  Every criterion that is set must
    match; empty criteria match everything."""
    id: str = ""
    description: str = ""
    created_at: float = 0.0
    expires_at: float = 0.0              # 0 => never expires
    subscribers: List[str] = field(default_factory=list)
    ips: List[str] = field(default_factory=list)
    macs: List[str] = field(default_factory=list)
    sessions: List[str] = field(default_factory=list)
    actions: List[str] = field(default_factory=list)
    start_time: float = 0.0              # event-timestamp window
    end_time: float = 0.0

    def __post_init__(self):
        if not self.id:
            self.id = uuid.uuid4().hex[:12]
        if not self.created_at:
            self.created_at = time.time()

    def matches(self, ev) -> bool:
        """ref eventMatchesHold retention.go:174-265."""
        ts = getattr(ev, "timestamp", 0.0)
        if self.start_time and ts < self.start_time:
            return False
        if self.end_time and ts > self.end_time:
            return False
        if self.subscribers and \
                getattr(ev, "subscriber", "") not in self.subscribers:
            return False
        if self.ips and getattr(ev, "ip", "") not in self.ips:
            return False
        d = getattr(ev, "details", {}) or {}
        if self.macs and d.get("mac", "") not in self.macs:
            return False
        if self.sessions and d.get("session_id", "") not in self.sessions:
            return False
        if self.actions and getattr(ev, "action", "") not in self.actions:
            return False
        return True


class RetentionManager:
    """Resolve retention per event and track legal holds (ref
    retention.go:9-124)."""

    def __init__(self, default_days: int = 365,
                 category_days: Optional[Dict[str, int]] = None):
        self.default_days = default_days
        self.category_days = dict(category_days or
                                  standard_retention_policies())
        self.action_days: Dict[str, int] = {}
        self.holds: Dict[str, LegalHold] = {}

    # ------------------------------------------------- policy lookup
    def get_retention(self, category: str) -> int:
        return self.category_days.get(category, self.default_days)

    def get_retention_for_action(self, action: str) -> int:
        """Action override beats category (ref GetRetentionForEvent
        retention.go:80-97)."""
        if action in self.action_days:
            return self.action_days[action]
        return self.get_retention(action_category(action))

    def set_category_retention(self, category: str, days: int):
        self.category_days[category] = days

    def set_action_retention(self, action: str, days: int):
        self.action_days[action] = days

    def policy_summary(self) -> Dict[str, int]:
        out = dict(self.category_days)
        out["__default__"] = self.default_days
        return out

    # --------------------------------------------------- legal holds
    def add_hold(self, hold: LegalHold) -> str:
        self.holds[hold.id] = hold
        return hold.id

    def remove_hold(self, hold_id: str) -> bool:
        return self.holds.pop(hold_id, None) is not None

    def get_holds(self) -> List[LegalHold]:
        return list(self.holds.values())

    def is_under_hold(self, ev) -> bool:
        now = time.time()
        for h in self.holds.values():
            if h.expires_at and h.expires_at < now:
                continue
            if h.matches(ev):
                return True
        return False

    def cleanup_expired_holds(self) -> int:
        now = time.time()
        expired = [hid for hid, h in self.holds.items()
                   if h.expires_at and h.expires_at < now]
        for hid in expired:
            del self.holds[hid]
        return len(expired)

    # ------------------------------------------------------- expiry
    def expired(self, ev, now: Optional[float] = None) -> bool:
        """True iff the event is past its retention AND not held."""
        now = now if now is not None else time.time()
        days = self.get_retention_for_action(getattr(ev, "action", ""))
        if getattr(ev, "timestamp", now) + days * 86400.0 > now:
            return False
        return not self.is_under_hold(ev)
