"""Direct (RADIUS-less) authentication (ref pkg/direct/authenticator.go:
40-165): subscriber identity derived from the physical path — the ONT/
VLAN the frame arrived on — validated against Nexus and an optional
BSS (business support system) client."""
from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, Optional, Protocol

from ..nexus.client import Client as NexusClient
from ..nexus.model import Subscriber


class BindingEvent:
    """DHCP binding notification pushed to the BSS (ref
    authenticator.go:129-141 ReportBinding)."""

    def __init__(self, subscriber_id: str, mac: str, ip: str,
                 event: str = "bind"):
        self.subscriber_id = subscriber_id
        self.mac = mac
        self.ip = ip
        self.event = event


class BSSClient(Protocol):
    """Billing/CRM lookup: is this subscriber in good standing?"""

    def subscriber_status(self, subscriber_id: str) -> str: ...


class StubBSS:
    """ref BSS stub: everything active unless listed."""

    def __init__(self, statuses: Optional[Dict[str, str]] = None):
        self.statuses = statuses or {}

    def subscriber_status(self, subscriber_id: str) -> str:
        return self.statuses.get(subscriber_id, "active")


@dataclass
class DirectAuthResult:
    success: bool
    subscriber_id: str = ""
    isp_id: str = ""
    reason: str = ""


class Authenticator:
    def __init__(self, nexus: NexusClient, bss: Optional[BSSClient] = None,
                 cache_ttl: float = 60.0):
        self.nexus = nexus
        self.bss = bss or StubBSS()
        self.cache_ttl = cache_ttl
        self._cache: Dict[str, tuple] = {}   # key -> (result, expires)
        self.stats = {"ok": 0, "unknown": 0, "suspended": 0,
                      "cache_hits": 0, "binding_events": 0,
                      "synced": 0}

    def _cached(self, key: str) -> Optional[DirectAuthResult]:
        import time
        hit = self._cache.get(key)
        if hit and hit[1] > time.time():
            self.stats["cache_hits"] += 1
            return hit[0]
        return None

    def _remember(self, key: str, res: DirectAuthResult):
        import time
        if res.success:   # only positive results cached (ref :266-280)
            self._cache[key] = (res, time.time() + self.cache_ttl)

    def invalidate_cache(self, key: Optional[str] = None):
        if key is None:
            self._cache.clear()
        else:
            self._cache.pop(key, None)

    def report_binding_event(self, subscriber_id: str, mac: str,
                             ip: str, event: str = "bind") -> bool:
        """Notify the BSS of a DHCP bind/unbind (ref ReportBinding)."""
        self.stats["binding_events"] += 1
        fn = getattr(self.bss, "report_binding", None)
        if fn is None:
            return False
        try:
            fn(BindingEvent(subscriber_id, mac, ip, event))
            return True
        except Exception:
            return False

    def sync_from_bss(self) -> int:
        """Pull the BSS's full mapping list into the cache (ref
        SyncMappings cache population)."""
        fn = getattr(self.bss, "sync_mappings", None)
        if fn is None:
            return 0
        n = 0
        for m in fn():
            res = DirectAuthResult(True, m.get("subscriber_id", ""),
                                   m.get("isp_id", ""))
            for k in ("mac", "vlan"):
                if m.get(k):
                    self._remember(f"{k}:{m[k]}", res)
                    n += 1
        self.stats["synced"] += n
        return n

    def authenticate_by_vlan(self, s_tag: int, c_tag: int) -> DirectAuthResult:
        """Identity from the QinQ pair the NTE was provisioned with."""
        hit = self._cached(f"vlan:{s_tag}.{c_tag}")
        if hit is not None:
            return hit
        for sub_id, d in self.nexus.subscribers.list().items():
            sub = Subscriber.from_dict(d)
            if sub.s_tag == s_tag and sub.c_tag == c_tag:
                res = self._check(sub)
                self._remember(f"vlan:{s_tag}.{c_tag}", res)
                return res
        self.stats["unknown"] += 1
        return DirectAuthResult(False, reason="unknown vlan pair")

    def authenticate_by_mac(self, mac: str) -> DirectAuthResult:
        hit = self._cached(f"mac:{mac}")
        if hit is not None:
            return hit
        sub = self.nexus.get_subscriber_by_mac(mac)
        if sub is None:
            self.stats["unknown"] += 1
            return DirectAuthResult(False, reason="unknown mac")
        res = self._check(sub)
        self._remember(f"mac:{mac}", res)
        return res

    def _check(self, sub: Subscriber) -> DirectAuthResult:
        status = self.bss.subscriber_status(sub.id)
        if status != "active":
            self.stats["suspended"] += 1
            return DirectAuthResult(False, sub.id, sub.isp_id,
                                    reason=f"bss status {status}")
        self.stats["ok"] += 1
        return DirectAuthResult(True, sub.id, sub.isp_id)


# ----------------------------------------------------------------------
# ONT-mapping-driven authentication (ref authenticator.go:93-470): the
# BSS answers "which subscriber owns this ONT / circuit-id", and the
# authenticator turns that into the subscriber.Manager's rich auth
# protocol (authenticate_session), including walled-garden placement
# for suspended accounts.

@dataclass
class ONTMapping:
    """ref ONTMapping authenticator.go:93-125."""
    ont_serial: str
    subscriber_id: str
    circuit_id: str = ""
    pon_port: str = ""
    isp_id: str = ""
    service_class: str = ""
    qos_policy: str = ""
    download_bps: int = 0
    upload_bps: int = 0
    ipv4_pool: str = ""
    ipv4_addr: str = ""
    ipv6_prefix: str = ""
    s_tag: int = 0
    c_tag: int = 0
    status: str = "active"            # active | suspended | disconnected


EV_ASSIGN = "assign"
EV_RENEW = "renew"
EV_RELEASE = "release"
EV_EXPIRE = "expire"


class MappingBSS:
    """In-memory BSS with ONT mappings + binding-event capture (ref
    bss_stub.go)."""

    def __init__(self):
        self.by_serial: Dict[str, ONTMapping] = {}
        self.by_circuit: Dict[str, ONTMapping] = {}
        self.bindings: list = []

    def add_mapping(self, m: ONTMapping):
        self.by_serial[m.ont_serial] = m
        if m.circuit_id:
            self.by_circuit[m.circuit_id] = m

    def get_ont_mapping(self, ont_serial: str) -> Optional[ONTMapping]:
        return self.by_serial.get(ont_serial)

    def get_ont_mapping_by_circuit_id(self, circuit_id: str):
        return self.by_circuit.get(circuit_id)

    def report_binding(self, event: dict):
        self.bindings.append(event)

    def sync_mappings(self):
        return list(self.by_serial.values())


class ONTAuthenticator:
    """Physical-path authenticator implementing the subscriber
    manager's rich protocol (ref Authenticate authenticator.go:182-263;
    lookup order circuit-id then ONT serial, :265-351)."""

    def __init__(self, bss: MappingBSS, default_isp: str = "",
                 default_qos: str = "", session_timeout: float = 0.0,
                 idle_timeout: float = 0.0, cache_ttl: float = 300.0):
        import time as _t
        self._t = _t
        self.bss = bss
        self.default_isp = default_isp
        self.default_qos = default_qos
        self.session_timeout = session_timeout
        self.idle_timeout = idle_timeout
        self.cache_ttl = cache_ttl
        self._cache: Dict[str, tuple] = {}
        self.stats = {"ok": 0, "not_found": 0, "suspended": 0,
                      "disconnected": 0, "cache_hits": 0, "synced": 0,
                      "binding_events": 0}

    # ------------------------------------------------------- cache
    def _cached(self, key: str) -> Optional[ONTMapping]:
        hit = self._cache.get(key)
        if hit and hit[1] > self._t.time():
            self.stats["cache_hits"] += 1
            return hit[0]
        return None

    def _remember(self, m: ONTMapping):
        # snapshot, not reference: a remote BSS returns copies, and the
        # cache must stay stale until invalidated even against the
        # in-memory stub (which hands out live objects)
        import dataclasses
        m = dataclasses.replace(m)
        exp = self._t.time() + self.cache_ttl
        self._cache[f"serial:{m.ont_serial}"] = (m, exp)
        if m.circuit_id:
            self._cache[f"circuit:{m.circuit_id}"] = (m, exp)

    def invalidate_cache(self, ont_serial: str = "",
                         circuit_id: str = ""):
        """ref InvalidateCache :380-391."""
        if ont_serial:
            self._cache.pop(f"serial:{ont_serial}", None)
        if circuit_id:
            self._cache.pop(f"circuit:{circuit_id}", None)
        if not ont_serial and not circuit_id:
            self._cache.clear()

    def sync_from_bss(self) -> int:
        """Pre-warm the cache with every mapping (ref SyncFromBSS
        :393-425)."""
        n = 0
        for m in self.bss.sync_mappings():
            self._remember(m)
            n += 1
        self.stats["synced"] += n
        return n

    # ---------------------------------------------------- protocol
    def _lookup(self, circuit_id: str, ont_serial: str):
        if circuit_id:
            m = self._cached(f"circuit:{circuit_id}") or \
                self.bss.get_ont_mapping_by_circuit_id(circuit_id)
            if m is not None:
                self._remember(m)
                return m
        if ont_serial:
            m = self._cached(f"serial:{ont_serial}") or \
                self.bss.get_ont_mapping(ont_serial)
            if m is not None:
                self._remember(m)
                return m
        return None

    def authenticate_session(self, session, credentials) -> dict:
        """subscriber.Manager.authenticate_full protocol: identity from
        the session's option-82 circuit id or NTE serial."""
        m = self._lookup(getattr(session, "circuit_id", ""),
                         getattr(session, "nte_id", ""))
        if m is None:
            self.stats["not_found"] += 1
            return {"success": False, "error": "ONT not found"}
        if m.status == "suspended":
            # suspended accounts land in the walled garden, not a
            # hard reject (ref :204-215)
            self.stats["suspended"] += 1
            return {"success": True, "subscriber_id": m.subscriber_id,
                    "isp_id": m.isp_id or self.default_isp,
                    "walled_garden": True,
                    "walled_reason": "Account suspended"}
        if m.status not in ("", "active"):
            self.stats["disconnected"] += 1
            return {"success": False,
                    "error": f"subscriber not active ({m.status})"}
        self.stats["ok"] += 1
        return {"success": True, "subscriber_id": m.subscriber_id,
                "isp_id": m.isp_id or self.default_isp,
                "qos_policy_id": m.qos_policy or self.default_qos,
                "download_rate_bps": m.download_bps,
                "upload_rate_bps": m.upload_bps,
                "session_timeout": self.session_timeout,
                "idle_timeout": self.idle_timeout,
                "ipv4_pool": m.ipv4_pool, "ipv4_addr": m.ipv4_addr}

    def report_binding(self, event_type: str, ont_serial: str,
                       subscriber_id: str, mac: str = "",
                       ipv4: str = "", session_id: str = "",
                       lease_expiry: float = 0.0):
        """Typed DHCP binding notification to the BSS (ref
        ReportBindingEvent :427-451; types assign/renew/release/
        expire)."""
        if event_type not in (EV_ASSIGN, EV_RENEW, EV_RELEASE,
                              EV_EXPIRE):
            raise ValueError(f"unknown binding event {event_type}")
        self.bss.report_binding({
            "event_type": event_type, "timestamp": self._t.time(),
            "ont_serial": ont_serial, "subscriber_id": subscriber_id,
            "mac": mac, "ipv4_addr": ipv4, "session_id": session_id,
            "lease_expiry": lease_expiry})
        self.stats["binding_events"] += 1
