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
