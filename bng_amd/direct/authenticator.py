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
