"""Policy manager: named bandwidth policies (ref pkg/radius/policy.go).

RADIUS Access-Accept carries a policy NAME (Filter-Id); the manager maps
it to concrete rate/burst/priority numbers which the QoS manager pushes
into the GPU token-bucket tables."""
from __future__ import annotations

import threading
from dataclasses import dataclass
from typing import Dict, List, Optional


@dataclass
class Policy:
    name: str
    download_rate_bps: int
    upload_rate_bps: int
    download_burst: int = 0
    upload_burst: int = 0
    priority: int = 0

    def __post_init__(self):
        # default burst: 1 second of traffic, min 64KB (sane TBF default)
        if not self.download_burst:
            self.download_burst = max(self.download_rate_bps // 8, 65536)
        if not self.upload_burst:
            self.upload_burst = max(self.upload_rate_bps // 8, 65536)


class PolicyManager:
    def __init__(self, default_policy: Optional[Policy] = None):
        self._policies: Dict[str, Policy] = {}
        self._lock = threading.RLock()
        self.default_policy = default_policy
        self._listeners: List = []

    def add_policy(self, policy: Policy):
        with self._lock:
            self._policies[policy.name] = policy
        for cb in self._listeners:
            try:
                cb(policy)
            except Exception:
                pass

    def remove_policy(self, name: str):
        with self._lock:
            self._policies.pop(name, None)

    def get(self, name: str) -> Optional[Policy]:
        with self._lock:
            return self._policies.get(name) or self.default_policy

    def names(self) -> List[str]:
        with self._lock:
            return sorted(self._policies)

    def on_change(self, cb):
        """Notify (e.g. re-push to GPU tables) when a policy changes."""
        self._listeners.append(cb)

    @classmethod
    def from_config(cls, entries: List[dict],
                    default: Optional[str] = None) -> "PolicyManager":
        """entries: [{name, download_mbps, upload_mbps, priority}]."""
        pm = cls()
        for e in entries:
            pm.add_policy(Policy(
                e["name"],
                int(e.get("download_mbps", 0) * 1e6) or
                int(e.get("download_rate_bps", 0)),
                int(e.get("upload_mbps", 0) * 1e6) or
                int(e.get("upload_rate_bps", 0)),
                priority=e.get("priority", 0)))
        if default:
            pm.default_policy = pm._policies.get(default)
        return pm
