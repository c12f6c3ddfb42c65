"""Bitmap IP/prefix allocator (ref pkg/allocator/bitmap.go:46-560).

Supports IPv4 address pools and IPv6 address/prefix-delegation pools via
big-int indexing; reserved head/tail, explicit reservations, JSON
(de)serialization for restart survival.
"""
from __future__ import annotations

import ipaddress
import json
from typing import Dict, Optional, Tuple, Union


class PoolExhaustedError(Exception):
    pass


class NotFoundError(Exception):
    pass


class BitmapAllocator:
    """Allocates fixed-size sub-prefixes of a base network.

    base_cidr="10.0.0.0/16", alloc_prefix=32  -> individual IPv4 addresses
    base_cidr="2001:db8::/48", alloc_prefix=56 -> IPv6 delegated prefixes
    """

    def __init__(self, base_cidr: str, alloc_prefix: Optional[int] = None,
                 reserve_head: int = 0, reserve_tail: int = 0):
        self.net = ipaddress.ip_network(base_cidr, strict=False)
        self.v6 = self.net.version == 6
        bits = 128 if self.v6 else 32
        self.alloc_prefix = alloc_prefix if alloc_prefix is not None \
            else bits
        if not (self.net.prefixlen <= self.alloc_prefix <= bits):
            raise ValueError(
                f"alloc prefix {self.alloc_prefix} out of range "
                f"[{self.net.prefixlen}, {bits}]")
        self.total = 1 << (self.alloc_prefix - self.net.prefixlen)
        self.unit = 1 << (bits - self.alloc_prefix)
        self.reserve_head = reserve_head
        self.reserve_tail = reserve_tail
        # dense bitmap for small pools; sparse set + rotor for huge ones
        # (an IPv6 /64 address pool has 2^64 slots)
        self.sparse = self.total > (1 << 22)
        self._bitmap = bytearray(0 if self.sparse
                                 else (self.total + 7) // 8)
        self._used: set = set()
        self._by_sub: Dict[str, int] = {}
        self._by_idx: Dict[int, str] = {}
        self._hint = 0

    # ----------------------------------------------------------- bit ops
    def _test(self, idx: int) -> bool:
        if self.sparse:
            return idx in self._used
        return bool(self._bitmap[idx >> 3] & (1 << (idx & 7)))

    def _set(self, idx: int, v: bool):
        if self.sparse:
            (self._used.add if v else self._used.discard)(idx)
            return
        if v:
            self._bitmap[idx >> 3] |= 1 << (idx & 7)
        else:
            self._bitmap[idx >> 3] &= ~(1 << (idx & 7)) & 0xFF

    def _idx_to_prefix(self, idx: int) -> str:
        base = int(self.net.network_address) + idx * self.unit
        addr = ipaddress.ip_address(base)
        return f"{addr}/{self.alloc_prefix}"

    def _prefix_to_idx(self, prefix: str) -> int:
        p = ipaddress.ip_network(prefix, strict=False)
        off = int(p.network_address) - int(self.net.network_address)
        if off < 0 or off % self.unit or off // self.unit >= self.total:
            raise NotFoundError(f"{prefix} not in {self.net}")
        return off // self.unit

    def _is_reserved(self, idx: int) -> bool:
        if idx < self.reserve_head or idx >= self.total - self.reserve_tail:
            return True
        # for whole-network v4 address pools, skip network & broadcast
        if not self.v6 and self.alloc_prefix == 32 and \
                self.net.prefixlen < 31:
            if idx == 0 or idx == self.total - 1:
                return True
        return False

    # -------------------------------------------------------------- API
    def allocate(self, subscriber_id: str) -> str:
        """Idempotent per subscriber; first-free with rotating hint
        (ref bitmap.go Allocate)."""
        if subscriber_id in self._by_sub:
            return self._idx_to_prefix(self._by_sub[subscriber_id])
        scan = self.total if not self.sparse else \
            min(self.total, len(self._used) + self.reserve_head +
                self.reserve_tail + 16)
        for i in range(scan):
            idx = (self._hint + i) % self.total
            if self._is_reserved(idx) or self._test(idx):
                continue
            self._set(idx, True)
            self._by_sub[subscriber_id] = idx
            self._by_idx[idx] = subscriber_id
            self._hint = (idx + 1) % self.total
            return self._idx_to_prefix(idx)
        raise PoolExhaustedError(str(self.net))

    def allocate_specific(self, subscriber_id: str, prefix: str) -> str:
        idx = self._prefix_to_idx(prefix)
        if self._test(idx):
            owner = self._by_idx.get(idx)
            if owner == subscriber_id:
                return self._idx_to_prefix(idx)
            raise PoolExhaustedError(f"{prefix} already allocated")
        if self._is_reserved(idx):
            raise PoolExhaustedError(f"{prefix} reserved")
        self._set(idx, True)
        self._by_sub[subscriber_id] = idx
        self._by_idx[idx] = subscriber_id
        return self._idx_to_prefix(idx)

    def release(self, subscriber_id: str) -> None:
        idx = self._by_sub.pop(subscriber_id, None)
        if idx is None:
            return
        self._by_idx.pop(idx, None)
        self._set(idx, False)
        if idx < self._hint:
            self._hint = idx

    def release_prefix(self, prefix: str) -> None:
        idx = self._prefix_to_idx(prefix)
        sub = self._by_idx.pop(idx, None)
        if sub is not None:
            self._by_sub.pop(sub, None)
        self._set(idx, False)

    def lookup(self, subscriber_id: str) -> Optional[str]:
        idx = self._by_sub.get(subscriber_id)
        return None if idx is None else self._idx_to_prefix(idx)

    def lookup_by_prefix(self, prefix: str) -> Optional[str]:
        try:
            return self._by_idx.get(self._prefix_to_idx(prefix))
        except NotFoundError:
            return None

    def stats(self) -> Tuple[int, int, float]:
        allocated = len(self._by_sub)
        usable = self.total - self.reserve_head - self.reserve_tail
        return allocated, usable, allocated / usable if usable else 0.0

    # ----------------------------------------------------- persistence
    def to_json(self) -> str:
        """ref bitmap.go:428-496 JSON marshal."""
        return json.dumps({
            "base": str(self.net),
            "alloc_prefix": self.alloc_prefix,
            "reserve_head": self.reserve_head,
            "reserve_tail": self.reserve_tail,
            "allocations": {s: self._idx_to_prefix(i)
                            for s, i in self._by_sub.items()},
        })

    @classmethod
    def from_json(cls, data: str) -> "BitmapAllocator":
        d = json.loads(data)
        a = cls(d["base"], d["alloc_prefix"], d.get("reserve_head", 0),
                d.get("reserve_tail", 0))
        for sub, prefix in d.get("allocations", {}).items():
            a.allocate_specific(sub, prefix)
        return a
