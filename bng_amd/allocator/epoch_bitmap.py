"""Epoch bitmap allocator — O(1) epoch-based lease expiry
(ref pkg/allocator/epoch_bitmap.go:11-427).

Each IP slot carries a 2-bit generation; advancing the epoch is O(1)
(increment + lazy map cleanup), and expiry is checked lazily at
allocate/lookup.  Memory: 2 bits/IP = 16 KB per /16.  The GPU dataplane
mirrors the same idea with lease_expiry tags beside the HBM tables.
"""
from __future__ import annotations

import ipaddress
import json
import threading
from typing import Dict, Optional, Tuple


class PoolExhaustedError(Exception):
    pass


class NotFoundError(Exception):
    pass


class EpochBitmapAllocator:
    def __init__(self, base_network: str, prefix_length: int = 32,
                 grace_period: int = 1):
        net = ipaddress.ip_network(base_network, strict=False)
        max_plen = 32 if net.version == 4 else 128
        if not (net.prefixlen <= prefix_length <= max_plen):
            raise ValueError("prefix length out of range")
        if prefix_length - net.prefixlen > 24:
            raise ValueError("pool too large for the epoch bitmap "
                             "(use allocator.bitmap sparse mode)")
        self.net = net
        self._shift = max_plen - prefix_length   # v6 delegation stride
        self.prefix_length = prefix_length
        self.total = 1 << (prefix_length - net.prefixlen)
        self.generations = bytearray((self.total + 3) // 4)  # 2 bits each
        self.subscribers: Dict[str, int] = {}
        self.ip_to_sub: Dict[int, str] = {}
        # generation 0 is reserved as "never allocated"; live slots
        # cycle through gens 1..3, so at most 2 epochs of grace fit in
        # the 2-bit field (current + two previous) (ref :92)
        self.current_epoch = 2
        self.grace_period = min(2, max(1, grace_period))
        self.next_free_hint = 1
        self._lock = threading.RLock()

    # ------------------------------------------------------- generations
    def _cur_gen(self) -> int:
        return (self.current_epoch % 3) + 1

    def _active_gens(self) -> set:
        return {((self.current_epoch - k) % 3) + 1
                for k in range(self.grace_period + 1)}

    def _get_gen(self, idx: int) -> int:
        return (self.generations[idx >> 2] >> ((idx & 3) * 2)) & 3

    def _set_gen(self, idx: int, gen: int):
        shift = (idx & 3) * 2
        b = self.generations[idx >> 2]
        self.generations[idx >> 2] = (b & ~(3 << shift) & 0xFF) | (gen << shift)

    def _idx_to_ip(self, idx: int) -> str:
        return str(self.net.network_address + (idx << self._shift))

    def _ip_to_idx(self, ip: str) -> int:
        off = (int(ipaddress.ip_address(ip)) -
               int(self.net.network_address)) >> self._shift
        if not (0 <= off < self.total):
            raise NotFoundError(ip)
        return off

    # --------------------------------------------------------------- API
    def allocate(self, subscriber_id: str) -> str:
        with self._lock:
            idx = self.subscribers.get(subscriber_id)
            if idx is not None:
                self._set_gen(idx, self._cur_gen())     # renew
                return self._idx_to_ip(idx)
            active = self._active_gens()
            for i in range(self.total):
                idx = (self.next_free_hint + i) % self.total
                if idx == 0 or idx == self.total - 1:   # network/broadcast
                    continue
                if self._get_gen(idx) in active:
                    continue
                # reclaim: drop a stale owner of this slot lazily
                old = self.ip_to_sub.pop(idx, None)
                if old is not None:
                    self.subscribers.pop(old, None)
                self._set_gen(idx, self._cur_gen())
                self.subscribers[subscriber_id] = idx
                self.ip_to_sub[idx] = subscriber_id
                self.next_free_hint = (idx + 1) % self.total
                return self._idx_to_ip(idx)
            raise PoolExhaustedError(str(self.net))

    def renew(self, subscriber_id: str) -> None:
        with self._lock:
            idx = self.subscribers.get(subscriber_id)
            if idx is None:
                raise NotFoundError(subscriber_id)
            self._set_gen(idx, self._cur_gen())

    def release(self, subscriber_id: str) -> None:
        with self._lock:
            idx = self.subscribers.pop(subscriber_id, None)
            if idx is None:
                return
            self._set_gen(idx, 0)                 # back to never-used
            self.ip_to_sub.pop(idx, None)
            if idx < self.next_free_hint:
                self.next_free_hint = idx

    def lookup(self, subscriber_id: str) -> Optional[str]:
        with self._lock:
            idx = self.subscribers.get(subscriber_id)
            if idx is None or self._get_gen(idx) not in self._active_gens():
                return None
            return self._idx_to_ip(idx)

    def lookup_by_ip(self, ip: str) -> Optional[str]:
        with self._lock:
            try:
                idx = self._ip_to_idx(ip)
            except NotFoundError:
                return None
            sub = self.ip_to_sub.get(idx)
            if sub is None or self._get_gen(idx) not in self._active_gens():
                return None
            return sub

    def advance_epoch(self) -> int:
        """O(1) epoch advance + lazy map cleanup (ref :225-244)."""
        with self._lock:
            self.current_epoch += 1
            active = self._active_gens()
            dead = [s for s, i in self.subscribers.items()
                    if self._get_gen(i) not in active]
            for s in dead:
                idx = self.subscribers.pop(s)
                self.ip_to_sub.pop(idx, None)
            return self.current_epoch

    def stats(self) -> Tuple[int, int, float]:
        with self._lock:
            active = self._active_gens()
            n = sum(1 for i in self.subscribers.values()
                    if self._get_gen(i) in active)
            usable = self.total - 2
            return n, usable, n / usable if usable else 0.0

    # ------------------------------------------------------ persistence
    def to_json(self) -> str:
        """ref epoch_bitmap.go:361-427."""
        import base64
        with self._lock:
            return json.dumps({
                "base_network": str(self.net),
                "prefix_length": self.prefix_length,
                "grace_period": self.grace_period,
                "current_epoch": self.current_epoch,
                "next_free_hint": self.next_free_hint,
                "generations": base64.b64encode(bytes(self.generations)).decode(),
                "subscribers": self.subscribers,
            })

    @classmethod
    def from_json(cls, data: str) -> "EpochBitmapAllocator":
        import base64
        d = json.loads(data)
        a = cls(d["base_network"], d["prefix_length"], d["grace_period"])
        a.current_epoch = d["current_epoch"]
        a.next_free_hint = d["next_free_hint"]
        a.generations = bytearray(base64.b64decode(d["generations"]))
        a.subscribers = {s: int(i) for s, i in d["subscribers"].items()}
        a.ip_to_sub = {i: s for s, i in a.subscribers.items()}
        return a
