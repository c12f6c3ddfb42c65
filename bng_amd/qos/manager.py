"""QoS manager — userspace side of the GPU token-bucket kernel
(ref pkg/qos/manager.go:167-296): converts named PolicyManager policies
into per-subscriber token buckets in the device tables.

Direction mapping (ref qos_ratelimit.c): egress table keys the
subscriber's IP as DESTINATION (download shaping), ingress keys it as
SOURCE (upload shaping)."""
from __future__ import annotations

import threading
from typing import Dict, Optional

from ..radius.policy import Policy, PolicyManager


class Manager:
    def __init__(self, launcher=None, policy_manager: Optional[PolicyManager] = None):
        self.launcher = launcher
        self.policies = policy_manager or PolicyManager()
        self.active: Dict[int, str] = {}     # subscriber ip -> policy name
        self._lock = threading.RLock()
        # re-push buckets when a policy definition changes
        self.policies.on_change(self._on_policy_change)

    def apply_policy(self, ip: int, policy_name: str = "") -> bool:
        """ref manager.go:248 SetSubscriberPolicy."""
        pol = self.policies.get(policy_name) if policy_name else \
            self.policies.default_policy
        if pol is None:
            return False
        with self._lock:
            self.active[ip] = pol.name
        if self.launcher is not None:
            self.launcher.set_qos_policy(ip, pol.download_rate_bps,
                                         pol.download_burst, pol.priority,
                                         direction="egress")
            self.launcher.set_qos_policy(ip, pol.upload_rate_bps,
                                         pol.upload_burst, pol.priority,
                                         direction="ingress")
        return True

    def remove_policy(self, ip: int):
        with self._lock:
            self.active.pop(ip, None)
        if self.launcher is not None:
            self.launcher.remove_qos_policy(ip, direction="egress")
            self.launcher.remove_qos_policy(ip, direction="ingress")

    def update_subscriber_policy(self, ip: int, policy_name: str) -> bool:
        """CoA hook: re-apply a (possibly new) named policy
        (ref coa_handler.go:61 -> qos updater)."""
        return self.apply_policy(ip, policy_name)

    def _on_policy_change(self, pol: Policy):
        with self._lock:
            targets = [ip for ip, name in self.active.items()
                       if name == pol.name]
        for ip in targets:
            self.apply_policy(ip, pol.name)

    def get_stats(self) -> Dict[str, int]:
        if self.launcher is not None:
            return self.launcher.qos_get_stats()
        return {}
