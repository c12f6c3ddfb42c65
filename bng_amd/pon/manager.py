"""PON / ONT management (ref pkg/pon/manager.go:41-124): ONT/ONU
discovery + provisioning over Nexus, QoS profiles, discovery events."""
from __future__ import annotations

import threading
import time
from dataclasses import dataclass, field
from typing import Callable, Dict, List, Optional

from ..nexus.model import NTE
from ..nexus.store import Store, TypedStore


@dataclass
class QoSProfile:
    name: str
    downstream_mbps: int
    upstream_mbps: int
    tconts: int = 4


class Manager:
    def __init__(self, store: Store, vlan_mapper=None):
        self.store = store
        self.ntes = TypedStore(store, "nexus/ntes")
        self.vlan_mapper = vlan_mapper       # qinq.Mapper
        self.profiles: Dict[str, QoSProfile] = {}
        self._listeners: List[Callable[[str, NTE], None]] = []
        self._lock = threading.RLock()
        self.stats = {"discovered": 0, "provisioned": 0, "offline": 0}

    def add_profile(self, p: QoSProfile):
        self.profiles[p.name] = p

    def on_event(self, cb: Callable[[str, NTE], None]):
        self._listeners.append(cb)

    def _emit(self, ev: str, nte: NTE):
        for cb in self._listeners:
            try:
                cb(ev, nte)
            except Exception:
                pass

    # ---------------------------------------------------------- lifecycle
    def ont_discovered(self, serial: str, pon_port: str,
                       device_id: str = "") -> NTE:
        """New ONT seen on a PON port (ref discovery events)."""
        nte_id = f"nte-{serial}"
        existing = self.ntes.get(nte_id)
        if existing:
            nte = NTE.from_dict(existing)
            nte.last_seen = time.time()
            nte.state = "discovered" if not nte.provisioned else nte.state
        else:
            nte = NTE(id=nte_id, device_id=device_id, serial_number=serial,
                      pon_port=pon_port)
            self.stats["discovered"] += 1
        self.ntes.put(nte_id, nte.to_dict())
        self._emit("discovered", nte)
        return nte

    def provision(self, nte_id: str, profile: str = "",
                  s_tag: int = 0, c_tag: int = 0) -> NTE:
        """Assign VLANs + QoS profile; ONT goes active."""
        d = self.ntes.get(nte_id)
        if d is None:
            raise KeyError(nte_id)
        nte = NTE.from_dict(d)
        if s_tag and c_tag:
            nte.s_tag, nte.c_tag = s_tag, c_tag
        elif self.vlan_mapper is not None:
            nte.s_tag, nte.c_tag = self.vlan_mapper.auto_assign(nte_id)
        if profile and profile not in self.profiles:
            raise KeyError(f"unknown QoS profile {profile}")
        nte.provisioned = True
        nte.state = "active"
        self.ntes.put(nte_id, nte.to_dict())
        self.stats["provisioned"] += 1
        self._emit("provisioned", nte)
        return nte

    def ont_offline(self, nte_id: str):
        d = self.ntes.get(nte_id)
        if d is None:
            return
        nte = NTE.from_dict(d)
        nte.state = "offline"
        self.ntes.put(nte_id, nte.to_dict())
        self.stats["offline"] += 1
        self._emit("offline", nte)

    def list_ntes(self) -> List[NTE]:
        return [NTE.from_dict(d) for d in self.ntes.list().values()]


# NTE reachability states (ref manager.go:16-39)
ST_UNKNOWN = "UNKNOWN"
ST_CONNECTED = "CONNECTED"
ST_DISCONNECTED = "DISCONNECTED"
ST_UNCONFIGURED = "UNCONFIGURED"


class AutoProvisioner:
    """Discovery -> provision pipeline with retries (ref
    handleDiscoveryEvent manager.go:216-277): a discovered NTE goes
    UNCONFIGURED + pending, provisioning is retried `retries` times
    with `retry_delay` between attempts, success promotes it to
    CONNECTED and fires the provisioned callback with the assigned
    VLANs; an NTE that was already provisioned just reconnects with its
    existing tags (provisionNTE :279-330)."""

    def __init__(self, manager: Manager, retries: int = 3,
                 retry_delay: float = 5.0, default_profile: str = ""):
        self.manager = manager
        self.retries = retries
        self.retry_delay = retry_delay
        self.default_profile = default_profile
        self.states: Dict[str, str] = {}
        self.pending: Dict[str, dict] = {}
        self.on_discovered = None
        self.on_provisioned = None
        self.on_disconnected = None
        self.stats = {"discovered": 0, "provisioned": 0, "failed": 0,
                      "reconnected": 0, "disconnected": 0}

    def handle_discovery(self, serial: str, pon_port: str,
                         device_id: str = "") -> dict:
        """Returns the ProvisioningResult-shaped dict."""
        t0 = time.time()
        self.stats["discovered"] += 1
        if self.on_discovered:
            self.on_discovered({"serial": serial, "pon_port": pon_port})
        nte = self.manager.ont_discovered(serial, pon_port, device_id)
        # reconnection of an already-provisioned NTE keeps its tags
        if nte.provisioned:
            self.states[serial] = ST_CONNECTED
            self.pending.pop(serial, None)
            self.stats["reconnected"] += 1
            result = {"nte_id": nte.id, "success": True,
                      "s_tag": nte.s_tag, "c_tag": nte.c_tag,
                      "duration": time.time() - t0}
            if self.on_provisioned:
                self.on_provisioned(result)
            return result
        self.states[serial] = ST_UNCONFIGURED
        self.pending[serial] = {"serial": serial, "pon_port": pon_port,
                                "ts": t0}
        last_err = None
        for attempt in range(self.retries + 1):
            if attempt and self.retry_delay:
                time.sleep(self.retry_delay)
            try:
                nte = self.manager.provision(
                    nte.id, profile=self.default_profile)
            except Exception as e:
                last_err = e
                continue
            self.states[serial] = ST_CONNECTED
            self.pending.pop(serial, None)
            self.stats["provisioned"] += 1
            result = {"nte_id": nte.id, "success": True,
                      "s_tag": nte.s_tag, "c_tag": nte.c_tag,
                      "duration": time.time() - t0}
            if self.on_provisioned:
                self.on_provisioned(result)
            return result
        self.stats["failed"] += 1
        result = {"nte_id": f"nte-{serial}", "success": False,
                  "error": str(last_err),
                  "duration": time.time() - t0}
        if self.on_provisioned:
            self.on_provisioned(result)
        return result

    def handle_disconnect(self, serial: str):
        """ref HandleDisconnect manager.go:398-427."""
        self.states[serial] = ST_DISCONNECTED
        self.pending.pop(serial, None)
        self.stats["disconnected"] += 1
        self.manager.ont_offline(f"nte-{serial}")
        if self.on_disconnected:
            self.on_disconnected(serial)

    def nte_state(self, serial: str) -> str:
        return self.states.get(serial, ST_UNKNOWN)

    def list_connected(self) -> List[str]:
        return sorted(s for s, st in self.states.items()
                      if st == ST_CONNECTED)

    def list_pending(self) -> List[dict]:
        return list(self.pending.values())
