"""Direct (RADIUS-less) authentication (ref pkg/direct/authenticator.go:
40-165): subscriber identity derived from the physical path — the ONT/
VLAN the frame arrived on — validated against Nexus and an optional
BSS (business support system) client."""
from __future__ import annotations

import threading
from dataclasses import dataclass
from typing import Dict, Optional, Protocol

from ..nexus.client import Client as NexusClient
from ..nexus.model import Subscriber


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
    def __init__(self, nexus: NexusClient, bss: Optional[BSSClient] = None):
        self.nexus = nexus
        self.bss = bss or StubBSS()
        self.stats = {"ok": 0, "unknown": 0, "suspended": 0}

    def authenticate_by_vlan(self, s_tag: int, c_tag: int) -> DirectAuthResult:
        """Identity from the QinQ pair the NTE was provisioned with."""
        for sub_id, d in self.nexus.subscribers.list().items():
            sub = Subscriber.from_dict(d)
            if sub.s_tag == s_tag and sub.c_tag == c_tag:
                return self._check(sub)
        self.stats["unknown"] += 1
        return DirectAuthResult(False, reason="unknown vlan pair")

    def authenticate_by_mac(self, mac: str) -> DirectAuthResult:
        sub = self.nexus.get_subscriber_by_mac(mac)
        if sub is None:
            self.stats["unknown"] += 1
            return DirectAuthResult(False, reason="unknown mac")
        return self._check(sub)

    def _check(self, sub: Subscriber) -> DirectAuthResult:
        status = self.bss.subscriber_status(sub.id)
        if status != "active":
            self.stats["suspended"] += 1
            return DirectAuthResult(False, sub.id, sub.isp_id,
                                    reason=f"bss status {status}")
        self.stats["ok"] += 1
        return DirectAuthResult(True, sub.id, sub.isp_id)
