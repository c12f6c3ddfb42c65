"""Antispoof manager — userspace side of the GPU uRPF kernel
(ref pkg/antispoof/manager.go): binding CRUD (:200-303), allowed ranges
(:304), mode switching (:362), violation-event drain."""
from __future__ import annotations

import ipaddress
import threading
from typing import Dict, List, Optional, Tuple

from ..dataplane import abi
from ..dataplane.abi import mac_to_u64
from ..dataplane.packets import ip2u32, mac_bytes

MODES = {"disabled": abi.AS_DISABLED, "strict": abi.AS_STRICT,
         "loose": abi.AS_LOOSE, "log_only": abi.AS_LOG_ONLY}


class Manager:
    def __init__(self, launcher=None, default_mode: str = "strict",
                 log_violations: bool = True):
        self.launcher = launcher
        self.default_mode = MODES[default_mode]
        self.log_violations = log_violations
        self.bindings: Dict[int, dict] = {}
        self.allowed_ranges: List[Tuple[int, int]] = []
        self._lock = threading.RLock()
        self._push_config()

    def _push_config(self):
        if self.launcher is not None:
            self.launcher.set_antispoof_config(
                default_mode=self.default_mode,
                log_violations=self.log_violations,
                allowed_ranges=self.allowed_ranges)

    # ----------------------------------------------------------- bindings
    def add_binding(self, mac, ipv4: str = "", ipv6: bytes = b"",
                    mode: str = "strict"):
        """ref manager.go:200 AddBinding."""
        key = mac_to_u64(mac_bytes(mac))
        b = {"ipv4": ip2u32(ipv4) if ipv4 else 0, "ipv6": ipv6,
             "mode": MODES[mode]}
        with self._lock:
            self.bindings[key] = b
        if self.launcher is not None:
            self.launcher.add_binding(key, ipv4=b["ipv4"], ipv6=ipv6,
                                      mode=b["mode"])

    def update_binding(self, mac, **kw):
        self.add_binding(mac, **kw)

    def remove_binding(self, mac):
        key = mac_to_u64(mac_bytes(mac))
        with self._lock:
            self.bindings.pop(key, None)
        if self.launcher is not None:
            self.launcher.remove_binding(key)

    def get_binding(self, mac) -> Optional[dict]:
        with self._lock:
            return self.bindings.get(mac_to_u64(mac_bytes(mac)))

    # ------------------------------------------------------------- ranges
    def add_allowed_range(self, cidr: str):
        """Loose-mode ranges (ref manager.go:304)."""
        net = ipaddress.IPv4Network(cidr, strict=False)
        with self._lock:
            self.allowed_ranges.append(
                (int(net.network_address), int(net.netmask)))
        self._push_config()

    def set_mode(self, mode: str):
        """ref manager.go:362."""
        self.default_mode = MODES[mode]
        self._push_config()

    def get_stats(self) -> Dict[str, int]:
        if self.launcher is not None:
            return self.launcher.antispoof_get_stats()
        return {}

    def drain_violations(self) -> List[dict]:
        if self.launcher is not None:
            return self.launcher.drain_spoof_events()
        return []
