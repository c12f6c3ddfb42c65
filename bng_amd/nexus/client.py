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
from .model import IPPool, ISPConfig, NTE, Subscriber
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


class VLANAllocation:
    """One NTE's QinQ pair (ref pkg/nexus/vlan.go:33-43)."""

    def __init__(self, s_tag: int, c_tag: int, nte_id: str):
        self.s_tag, self.c_tag, self.nte_id = s_tag, c_tag, nte_id


class VLANAllocator:
    """S-TAG/C-TAG allocation for QinQ NTEs (ref pkg/nexus/vlan.go).

    Sequential C-TAG assignment inside the current S-TAG with rollover
    to the next S-TAG when full (findAvailable vlan.go:170-190), sticky
    per NTE, ISP-assigned S-TAG override (AllocateWithSTag :103), stats
    (:212-227), and warm-start from persisted NTEs (LoadFromStore :239).
    Optionally persists each allocation through the nexus store."""

    def __init__(self, store: Optional[Store] = None, s_tag: int = None,
                 c_tag_range=(2, 4094), s_tag_range=None):
        if s_tag_range is None:
            # legacy single-S-TAG form: VLANAllocator(store, s_tag=100)
            s_tag_range = (s_tag, s_tag) if s_tag is not None else (100, 199)
        self.s_lo, self.s_hi = s_tag_range
        self.lo, self.hi = c_tag_range
        self.store = store
        self.typed = (TypedStore(store, "nexus/vlans")
                      if store is not None else None)
        self._alloc: Dict[str, VLANAllocation] = {}
        self._usage: Dict[int, Dict[int, str]] = {}
        self._cur_s = self.s_lo
        self._lock = threading.Lock()

    def _find_c(self, s_tag: int) -> Optional[int]:
        used = self._usage.get(s_tag)
        if not used:
            return self.lo
        for c in range(self.lo, self.hi + 1):
            if c not in used:
                return c
        return None

    def _record(self, nte_id: str, s: int, c: int) -> "VLANAllocation":
        a = VLANAllocation(s, c, nte_id)
        self._alloc[nte_id] = a
        self._usage.setdefault(s, {})[c] = nte_id
        if self.typed is not None:
            self.typed.put(nte_id, f"{s}:{c}")
        return a

    def allocate(self, nte_id: str) -> Tuple[int, int]:
        with self._lock:
            if nte_id in self._alloc:
                a = self._alloc[nte_id]
                return a.s_tag, a.c_tag
            for s in list(range(self._cur_s, self.s_hi + 1)) +                     list(range(self.s_lo, self._cur_s)):
                c = self._find_c(s)
                if c is not None:
                    self._cur_s = s
                    a = self._record(nte_id, s, c)
                    return a.s_tag, a.c_tag
        raise NexusError("VLAN space exhausted")

    def allocate_with_s_tag(self, nte_id: str, s_tag: int) -> Tuple[int, int]:
        """ISP-assigned S-TAG (ref AllocateWithSTag vlan.go:103-138)."""
        with self._lock:
            a = self._alloc.get(nte_id)
            if a is not None:
                if a.s_tag == s_tag:
                    return a.s_tag, a.c_tag
                self._release_unlocked(nte_id)
            c = self._find_c(s_tag)
            if c is None:
                raise NexusError(f"no free C-TAG under S-TAG {s_tag}")
            a = self._record(nte_id, s_tag, c)
            return a.s_tag, a.c_tag

    def _release_unlocked(self, nte_id: str):
        a = self._alloc.pop(nte_id, None)
        if a is None:
            return
        u = self._usage.get(a.s_tag)
        if u is not None:
            u.pop(a.c_tag, None)
            if not u:
                del self._usage[a.s_tag]
        if self.typed is not None:
            self.typed.delete(nte_id)

    def release(self, nte_id: str):
        with self._lock:
            self._release_unlocked(nte_id)

    def get(self, nte_id: str) -> Optional["VLANAllocation"]:
        with self._lock:
            return self._alloc.get(nte_id)

    def stats(self) -> Dict[str, int]:
        with self._lock:
            s_cap = self.s_hi - self.s_lo + 1
            c_cap = self.hi - self.lo + 1
            return {"total_allocations": len(self._alloc),
                    "s_tags_in_use": len(self._usage),
                    "s_tag_capacity": s_cap, "c_tag_capacity": c_cap,
                    "total_capacity": s_cap * c_cap}

    def load_from_store(self, ntes) -> None:
        """Warm-start from persisted NTE records: iterable of
        (nte_id, s_tag, c_tag) or dicts (ref LoadFromStore vlan.go:239)."""
        with self._lock:
            for n in ntes:
                if isinstance(n, dict):
                    nid, s, c = n["id"], n["s_tag"], n["c_tag"]
                else:
                    nid, s, c = n
                a = VLANAllocation(s, c, nid)
                self._alloc[nid] = a
                self._usage.setdefault(s, {})[c] = nid

    def sync_to_nte(self, nte: dict) -> dict:
        """Push this NTE's pair into its record (ref SyncToNTE :265)."""
        a = self.get(nte["id"])
        if a is None:
            raise NexusError(f"no VLAN allocation for {nte['id']}")
        nte["s_tag"], nte["c_tag"] = a.s_tag, a.c_tag
        return nte
