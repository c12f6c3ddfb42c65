"""Nexus agent (ref pkg/agent/agent.go:41-77, types.go:42-231): node
state machine bootstrap -> connected -> partitioned -> recovering,
config watcher over the store, heartbeat."""
from __future__ import annotations

import json
import threading
import time
from dataclasses import dataclass, field
from typing import Callable, Dict, List, Optional

from ..nexus.store import Store, TypedStore

S_BOOTSTRAP = "bootstrap"
S_CONNECTED = "connected"
S_PARTITIONED = "partitioned"
S_RECOVERING = "recovering"


@dataclass
class Subscriber:
    """Wholesale subscriber: stable physical layer (NetCo) + mutable
    service layer (ISPCo) + session state (ref agent/types.go
    Subscriber :156-180)."""
    subscriber_id: str
    nte_id: str = ""
    device_id: str = ""
    vlan: str = ""                    # "s-tag:c-tag"
    netco_id: str = ""
    isp_id: str = ""
    radius_realm: str = ""
    service_tier: str = ""
    qos_policy: str = ""
    mac: str = ""
    ipv4: str = ""
    ipv6_prefix: str = ""
    authenticated: bool = False
    session_start: float = 0.0


@dataclass
class NTE:
    """Discovered network-terminating equipment (ONU/ONT) (ref
    types.go NTE :182-192)."""
    serial: str
    device_id: str = ""
    port: int = 0
    status: str = "discovered"        # discovered|provisioned|active
    vendor: str = ""
    model: str = ""
    firmware: str = ""
    discovered_at: float = 0.0


class Agent:
    def __init__(self, store: Store, node_id: str,
                 heartbeat_interval: float = 5.0,
                 partition_after: float = 15.0):
        self.store = store
        self.node_id = node_id
        self.heartbeat_interval = heartbeat_interval
        self.partition_after = partition_after
        self.state = S_BOOTSTRAP
        self.config: Dict = {}
        self.config_store = TypedStore(store, "nexus/device_configs")
        self._listeners: List[Callable[[str, str], None]] = []
        self._config_listeners: List[Callable[[Dict], None]] = []
        self._last_ok = 0.0
        self._stop = threading.Event()
        self._threads: List[threading.Thread] = []
        self._watch_cancel = None
        self._started_at = 0.0
        # wholesale registries (ref agent.go:315-440)
        self._subs: Dict[str, Subscriber] = {}
        self._subs_by_mac: Dict[str, str] = {}
        self._subs_by_nte: Dict[str, str] = {}
        self._ntes: Dict[str, NTE] = {}
        self._churn_listeners: List[Callable[[Dict], None]] = []
        self._reg_lock = threading.RLock()

    def on_state_change(self, cb):
        self._listeners.append(cb)

    def on_config_change(self, cb):
        self._config_listeners.append(cb)

    def on_isp_churn(self, cb):
        """cb(event_dict) when a subscriber's ISP assignment changes
        (ref OnISPChurn agent.go:186-191)."""
        self._churn_listeners.append(cb)

    def start(self):
        self._started_at = time.time()
        cfg = self.config_store.get(self.node_id)
        if cfg:
            self.config = cfg
        self._watch_cancel = self.config_store.watch(self._on_cfg)
        self._transition(S_CONNECTED)
        self._last_ok = time.time()
        t = threading.Thread(target=self._hb_loop, daemon=True)
        t.start()
        self._threads.append(t)
        return self

    def stop(self):
        self._stop.set()
        if self._watch_cancel:
            self._watch_cancel()

    def _on_cfg(self, typ, key, obj):
        if key != self.node_id:
            return
        self.config = obj or {}
        for cb in self._config_listeners:
            try:
                cb(self.config)
            except Exception:
                pass

    def _transition(self, new: str):
        old, self.state = self.state, new
        if old != new:
            for cb in self._listeners:
                try:
                    cb(old, new)
                except Exception:
                    pass

    # ------------------------------------- subscriber registry (NetCo)
    def set_subscriber(self, sub: Subscriber):
        """Store/update; an ISP change on an existing subscriber fires
        the churn handlers (ref SetSubscriber agent.go:348-361 +
        handleISPChurn :389-412)."""
        with self._reg_lock:
            old = self._subs.get(sub.subscriber_id)
            self._subs[sub.subscriber_id] = sub
            if sub.mac:
                self._subs_by_mac[sub.mac.lower()] = sub.subscriber_id
            if sub.nte_id:
                self._subs_by_nte[sub.nte_id] = sub.subscriber_id
        if old is not None and old.isp_id and sub.isp_id and \
                old.isp_id != sub.isp_id:
            event = {"subscriber_id": sub.subscriber_id,
                     "old_isp_id": old.isp_id,
                     "new_isp_id": sub.isp_id,
                     "timestamp": time.time()}
            for cb in self._churn_listeners:
                try:
                    cb(event)
                except Exception:
                    pass

    def get_subscriber(self, subscriber_id: str) -> Optional[Subscriber]:
        with self._reg_lock:
            return self._subs.get(subscriber_id)

    def get_subscriber_by_mac(self, mac: str) -> Optional[Subscriber]:
        with self._reg_lock:
            sid = self._subs_by_mac.get(mac.lower())
            return self._subs.get(sid) if sid else None

    def get_subscriber_by_nte(self, nte_serial: str) -> Optional[Subscriber]:
        with self._reg_lock:
            sid = self._subs_by_nte.get(nte_serial)
            return self._subs.get(sid) if sid else None

    def remove_subscriber(self, subscriber_id: str):
        with self._reg_lock:
            sub = self._subs.pop(subscriber_id, None)
            if sub:
                self._subs_by_mac.pop(sub.mac.lower(), None)
                self._subs_by_nte.pop(sub.nte_id, None)

    def subscriber_count(self) -> int:
        with self._reg_lock:
            return len(self._subs)

    def subscriber_count_by_isp(self) -> Dict[str, int]:
        """ref GetSubscriberCountByISP agent.go:377-387."""
        out: Dict[str, int] = {}
        with self._reg_lock:
            for s in self._subs.values():
                out[s.isp_id or "unassigned"] = \
                    out.get(s.isp_id or "unassigned", 0) + 1
        return out

    # ------------------------------------------------- NTE registry
    def set_nte(self, nte: NTE):
        with self._reg_lock:
            self._ntes[nte.serial] = nte

    def get_nte(self, serial: str) -> Optional[NTE]:
        with self._reg_lock:
            return self._ntes.get(serial)

    def remove_nte(self, serial: str):
        with self._reg_lock:
            self._ntes.pop(serial, None)

    def nte_count(self) -> int:
        with self._reg_lock:
            return len(self._ntes)

    # ------------------------------------------------------ queries
    def get_isp_config(self, isp_id: str) -> Optional[Dict]:
        """ISP block from the device config (ref GetISPConfig
        agent.go:442-455)."""
        for isp in self.config.get("isps", []):
            if isp.get("isp_id") == isp_id:
                return isp
        return None

    def uptime(self) -> float:
        return time.time() - self._started_at if self._started_at else 0.0

    def is_online(self) -> bool:
        return self.state == S_CONNECTED

    def health(self) -> Dict:
        """ref Health agent.go:457-467."""
        return {"status": self.state, "device_id": self.node_id,
                "uptime_seconds": int(self.uptime()),
                "subscribers": self.subscriber_count(),
                "ntes": self.nte_count(), "online": self.is_online()}

    def heartbeat_once(self) -> bool:
        try:
            self.store.put(f"nexus/heartbeats/{self.node_id}",
                           json.dumps({"ts": time.time(),
                                       "state": self.state,
                                       "uptime": int(self.uptime()),
                                       "subscribers":
                                       self.subscriber_count(),
                                       "ntes": self.nte_count()}).encode())
            self._last_ok = time.time()
            if self.state == S_PARTITIONED:
                self._transition(S_RECOVERING)
            elif self.state == S_RECOVERING:
                self._transition(S_CONNECTED)
            return True
        except Exception:
            if self.state in (S_CONNECTED, S_RECOVERING) and \
                    time.time() - self._last_ok > self.partition_after:
                self._transition(S_PARTITIONED)
            return False

    def _hb_loop(self):
        while not self._stop.wait(self.heartbeat_interval):
            self.heartbeat_once()
