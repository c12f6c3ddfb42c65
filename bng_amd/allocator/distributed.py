"""Distributed allocator over a nexus.Store
(ref pkg/allocator/distributed.go:57-560).

Pool modes (ref distributed.go:57-92):
  * session — allocation at RADIUS time, no expiry; DHCP renewals are
    pure reads (work through partitions with a read-only store).
  * lease — allocation at DHCP time with epoch-based expiry; requires a
    writable store during partitions.

Allocation records live in the store under alloc/{pool}/{subscriber};
each node also keeps a local EpochBitmapAllocator view, reconciled by a
store watcher, so two allocators over one store converge
(tested like the reference's distributed_integration_test.go:52-333).
"""
from __future__ import annotations

import json
import threading
import time
from dataclasses import asdict, dataclass, field
from typing import Dict, List, Optional

from ..nexus.store import Store
from .epoch_bitmap import EpochBitmapAllocator, NotFoundError, \
    PoolExhaustedError

MODE_SESSION = "session"
MODE_LEASE = "lease"

# allocator framework modes (ref modes.go:14-30)
MODE_STANDALONE = "standalone"
MODE_WIFI_GATEWAY = "wifi_gateway"
MODE_NEXUS = "nexus"
MODE_HYBRID = "hybrid"


@dataclass
class AllocationRecord:
    """ref allocator/store.go:33-60."""
    subscriber_id: str
    pool_id: str
    prefix: str
    mac: str = ""
    duid: str = ""
    iaid: int = 0
    epoch: int = 0
    allocated_at: float = field(default_factory=time.time)
    expires_at: Optional[float] = None
    metadata: Dict[str, str] = field(default_factory=dict)

    def to_json(self) -> bytes:
        return json.dumps(asdict(self)).encode()

    @classmethod
    def from_json(cls, raw: bytes) -> "AllocationRecord":
        d = json.loads(raw)
        return cls(**{k: v for k, v in d.items()
                      if k in cls.__dataclass_fields__})


class DistributedAllocator:
    def __init__(self, store: Store, pool_id: str, cidr: str,
                 mode: str = MODE_SESSION, prefix_length: int = 0,
                 grace_period: int = 1, epoch_interval: float = 0.0,
                 node_id: str = "node-1"):
        self.store = store
        self.pool_id = pool_id
        self.mode = mode
        self.node_id = node_id
        if prefix_length == 0:
            # host route for v4, /64-per-subscriber style /56 PD default
            # left to the caller for v6 — here: smallest sane default
            import ipaddress as _ip
            n = _ip.ip_network(cidr, strict=False)
            prefix_length = 32 if n.version == 4 else \
                min(128, n.prefixlen + 16)
        self.local = EpochBitmapAllocator(cidr, prefix_length, grace_period)
        self._lock = threading.RLock()
        self._prefix = f"alloc/{pool_id}/"
        self._stop = threading.Event()
        self._epoch_thread = None
        self._watch_cancel = self.store.watch(self._prefix, self._on_remote)
        self._load_existing()
        if epoch_interval > 0 and mode == MODE_LEASE:
            self._epoch_thread = threading.Thread(
                target=self._epoch_loop, args=(epoch_interval,), daemon=True)
            self._epoch_thread.start()

    # ------------------------------------------------------------ intern
    def _key(self, subscriber_id: str) -> str:
        return self._prefix + subscriber_id

    def _load_existing(self):
        for k, raw in self.store.list(self._prefix).items():
            try:
                rec = AllocationRecord.from_json(raw)
                ip = rec.prefix.split("/")[0]
                idx = self.local._ip_to_idx(ip)
                self.local.subscribers[rec.subscriber_id] = idx
                self.local.ip_to_sub[idx] = rec.subscriber_id
                self.local._set_gen(idx, self.local._cur_gen())
            except Exception:
                continue

    def _on_remote(self, ev):
        """Remote-change watcher (ref distributed.go:522): adopt other
        nodes' allocations into the local bitmap view."""
        sub = ev.key[len(self._prefix):]
        with self._lock:
            if ev.type == "delete":
                self.local.release(sub)
                return
            try:
                rec = AllocationRecord.from_json(ev.value)
                ip = rec.prefix.split("/")[0]
                idx = self.local._ip_to_idx(ip)
                old = self.local.subscribers.get(sub)
                if old is not None and old != idx:
                    self.local.release(sub)
                self.local.subscribers[sub] = idx
                self.local.ip_to_sub[idx] = sub
                self.local._set_gen(idx, self.local._cur_gen())
            except Exception:
                pass

    def _epoch_loop(self, interval: float):
        """Periodic epoch advance + store cleanup of expired records
        (ref distributed.go:374-421)."""
        while not self._stop.wait(interval):
            self.advance_epoch()

    # --------------------------------------------------------------- API
    def allocate(self, subscriber_id: str, mac: str = "") -> str:
        with self._lock:
            existing = self.store.get(self._key(subscriber_id))
            if existing is not None:
                rec = AllocationRecord.from_json(existing)
                if self.mode == MODE_LEASE:
                    rec.epoch = self.local.current_epoch
                    self.store.put(self._key(subscriber_id), rec.to_json())
                    try:
                        self.local.renew(subscriber_id)
                    except NotFoundError:
                        pass
                return rec.prefix
            ip = self.local.allocate(subscriber_id)
            prefix = f"{ip}/{self.local.prefix_length}"
            rec = AllocationRecord(
                subscriber_id=subscriber_id, pool_id=self.pool_id,
                prefix=prefix, mac=mac, epoch=self.local.current_epoch)
            self.store.put(self._key(subscriber_id), rec.to_json())
            return prefix

    def renew(self, subscriber_id: str) -> str:
        """Session mode: pure read.  Lease mode: bump epoch."""
        raw = self.store.get(self._key(subscriber_id))
        if raw is None:
            raise NotFoundError(subscriber_id)
        rec = AllocationRecord.from_json(raw)
        if self.mode == MODE_LEASE:
            with self._lock:
                self.local.renew(subscriber_id)
                rec.epoch = self.local.current_epoch
                self.store.put(self._key(subscriber_id), rec.to_json())
        return rec.prefix

    def release(self, subscriber_id: str) -> None:
        with self._lock:
            self.local.release(subscriber_id)
        self.store.delete(self._key(subscriber_id))

    def lookup(self, subscriber_id: str) -> Optional[str]:
        raw = self.store.get(self._key(subscriber_id))
        if raw is None:
            return None
        return AllocationRecord.from_json(raw).prefix

    def lookup_by_ip(self, ip: str) -> Optional[str]:
        return self.local.lookup_by_ip(ip)

    def advance_epoch(self) -> int:
        """Lease mode: advance local epoch and clean expired records out
        of the store (ref :374-421)."""
        with self._lock:
            before = set(self.local.subscribers)
            epoch = self.local.advance_epoch()
            expired = before - set(self.local.subscribers)
        for sub in expired:
            self.store.delete(self._key(sub))
        return epoch

    def stats(self):
        return self.local.stats()

    def close(self):
        self._stop.set()
        if self._watch_cancel:
            self._watch_cancel()
        if self._epoch_thread:
            self._epoch_thread.join(timeout=2)
