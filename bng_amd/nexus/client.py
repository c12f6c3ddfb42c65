"""Nexus client: cached subscriber/ISP/pool views with watchers, MAC
lookup, heartbeat, and the deterministic hashring IP allocation that is
the reference's core design invariant — IPs are allocated at
RADIUS-auth time by FNV(subscriberID) mod pool-hosts, so DHCP is a pure
read (ref pkg/nexus/client.go:47-575, README.md:19-33).
"""
from __future__ import annotations

import ipaddress
import threading
import time
from typing import Callable, Dict, List, Optional, Tuple

from ..dataplane.abi import fnv1a64
from .model import Device, IPPool, ISPConfig, NTE, Subscriber
from .store import Store, TypedStore


class NexusError(Exception):
    pass


class Client:
    """Cached Nexus client over a Store (ref client.go:47-485)."""

    def __init__(self, store: Store, node_id: str = "bng-1",
                 heartbeat_interval: float = 30.0):
        self.store = store
        self.node_id = node_id
        self.subscribers = TypedStore(store, "nexus/subscribers")
        self.ntes = TypedStore(store, "nexus/ntes")
        self.isps = TypedStore(store, "nexus/isps")
        self.pools = TypedStore(store, "nexus/pools")
        self.devices = TypedStore(store, "nexus/devices")
        self._cache: Dict[str, Subscriber] = {}
        self._mac_index: Dict[str, str] = {}
        self._lock = threading.RLock()
        self._hb_interval = heartbeat_interval
        self._hb_stop = threading.Event()
        self._hb_thread: Optional[threading.Thread] = None
        self._watch_cancel = None
        self._change_cbs: List[Callable[[str, Optional[Subscriber]], None]] = []

    # ------------------------------------------------------------ lifecycle
    def start(self):
        self._refresh_cache()
        self._watch_cancel = self.subscribers.watch(self._on_change)
        self._hb_thread = threading.Thread(target=self._hb_loop, daemon=True)
        self._hb_thread.start()

    def stop(self):
        self._hb_stop.set()
        if self._watch_cancel:
            self._watch_cancel()
        if self._hb_thread:
            self._hb_thread.join(timeout=2)

    def _hb_loop(self):
        while not self._hb_stop.wait(self._hb_interval):
            try:
                self.heartbeat()
            except Exception:
                pass

    def heartbeat(self):
        self.store.put(f"nexus/heartbeats/{self.node_id}",
                       str(time.time()).encode())

    def on_subscriber_change(self, cb):
        self._change_cbs.append(cb)

    def _on_change(self, typ, key, obj):
        with self._lock:
            if typ == "delete":
                old = self._cache.pop(key, None)
                if old and old.mac:
                    self._mac_index.pop(old.mac.lower(), None)
                sub = None
            else:
                sub = Subscriber.from_dict(obj)
                self._cache[key] = sub
                if sub.mac:
                    self._mac_index[sub.mac.lower()] = key
        for cb in self._change_cbs:
            try:
                cb(key, sub)
            except Exception:
                pass

    def _refresh_cache(self):
        with self._lock:
            self._cache.clear()
            self._mac_index.clear()
            for k, d in self.subscribers.list().items():
                sub = Subscriber.from_dict(d)
                self._cache[k] = sub
                if sub.mac:
                    self._mac_index[sub.mac.lower()] = k

    # ------------------------------------------------------------- lookups
    def get_subscriber(self, sub_id: str) -> Optional[Subscriber]:
        with self._lock:
            if sub_id in self._cache:
                return self._cache[sub_id]
        d = self.subscribers.get(sub_id)
        if d is None:
            return None
        sub = Subscriber.from_dict(d)
        with self._lock:
            self._cache[sub_id] = sub
            if sub.mac:
                self._mac_index[sub.mac.lower()] = sub_id
        return sub

    def get_subscriber_by_mac(self, mac: str) -> Optional[Subscriber]:
        """ref client.go MAC lookup."""
        with self._lock:
            sub_id = self._mac_index.get(mac.lower())
        return self.get_subscriber(sub_id) if sub_id else None

    def save_subscriber(self, sub: Subscriber):
        sub.updated_at = time.time()
        self.subscribers.put(sub.id, sub.to_dict())
        with self._lock:
            self._cache[sub.id] = sub
            if sub.mac:
                self._mac_index[sub.mac.lower()] = sub.id

    def get_isp(self, isp_id: str) -> Optional[ISPConfig]:
        d = self.isps.get(isp_id)
        return ISPConfig.from_dict(d) if d else None

    def get_pool(self, pool_id: str) -> Optional[IPPool]:
        d = self.pools.get(pool_id)
        return IPPool.from_dict(d) if d else None

    # ------------------------------------------- hashring IP allocation
    def allocate_ip_for_subscriber(self, sub_id: str) -> str:
        """Deterministic allocation at auth time (ref client.go:487-539):
        existing IP wins; else FNV(subscriberID) mod pool hosts."""
        sub = self.get_subscriber(sub_id)
        if sub is None:
            raise NexusError(f"subscriber {sub_id} not found")
        if sub.ipv4_addr:
            return sub.ipv4_addr
        pool_id = sub.ipv4_pool
        if not pool_id:
            isp = self.get_isp(sub.isp_id) if sub.isp_id else None
            if isp and isp.ipv4_pools:
                pool_id = isp.ipv4_pools[0]
        if not pool_id:
            raise NexusError(f"no IPv4 pool configured for {sub_id}")
        pool = self.get_pool(pool_id)
        if pool is None:
            raise NexusError(f"pool {pool_id} not found")
        ip = self.allocate_from_pool(pool.cidr, sub_id)
        sub.ipv4_addr = ip
        sub.ipv4_pool = pool_id
        self.save_subscriber(sub)
        return ip

    @staticmethod
    def allocate_from_pool(cidr: str, sub_id: str) -> str:
        """FNV hash -> deterministic host offset (ref client.go:542-575):
        offset = hash % (2^hostbits - 2) + 1, skipping network/broadcast."""
        net = ipaddress.IPv4Network(cidr, strict=False)
        num_hosts = net.num_addresses - 2
        if num_hosts <= 0:
            raise NexusError(f"pool {cidr} has no usable addresses")
        offset = fnv1a64(sub_id.encode()) % num_hosts + 1
        return str(net.network_address + offset)

    def lookup_subscriber_ip(self, sub_id: str) -> Optional[str]:
        """The read-only DHCP-time operation (ref client.go LookupSubscriberIP)."""
        sub = self.get_subscriber(sub_id)
        return sub.ipv4_addr or None if sub else None


class VLANAllocator:
    """S-TAG/C-TAG allocation (ref pkg/nexus/vlan.go): sequential C-TAG
    assignment within an S-TAG, with persistence through the store."""

    def __init__(self, store: Store, s_tag: int, c_tag_range=(2, 4094)):
        self.store = store
        self.s_tag = s_tag
        self.lo, self.hi = c_tag_range
        self.typed = TypedStore(store, f"nexus/vlans/{s_tag}")
        self._lock = threading.Lock()

    def allocate(self, sub_id: str) -> Tuple[int, int]:
        with self._lock:
            existing = self.typed.list()
            for c_tag_s, owner in existing.items():
                if owner == sub_id:
                    return self.s_tag, int(c_tag_s)
            used = {int(k) for k in existing}
            for c in range(self.lo, self.hi + 1):
                if c not in used:
                    self.typed.put(str(c), sub_id)
                    return self.s_tag, c
        raise NexusError(f"no free C-TAG under S-TAG {self.s_tag}")

    def release(self, sub_id: str):
        with self._lock:
            for c_tag_s, owner in self.typed.list().items():
                if owner == sub_id:
                    self.typed.delete(c_tag_s)
