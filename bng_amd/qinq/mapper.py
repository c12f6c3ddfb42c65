"""QinQ S-TAG/C-TAG <-> subscriber mapping with VLAN ranges
(ref pkg/qinq/qinq.go:100-213)."""
from __future__ import annotations

import threading
from dataclasses import dataclass
from typing import Dict, Optional, Tuple


class QinQError(Exception):
    pass


@dataclass
class VLANRange:
    s_tag: int
    c_tag_start: int
    c_tag_end: int


class Mapper:
    def __init__(self):
        self._by_vlan: Dict[Tuple[int, int], str] = {}
        self._by_sub: Dict[str, Tuple[int, int]] = {}
        self._ranges: list = []
        self._lock = threading.RLock()

    def add_range(self, s_tag: int, c_start: int = 2, c_end: int = 4094):
        if not (1 <= s_tag <= 4094 and 1 <= c_start <= c_end <= 4094):
            raise QinQError("invalid VLAN range")
        with self._lock:
            self._ranges.append(VLANRange(s_tag, c_start, c_end))

    def register(self, subscriber_id: str, s_tag: int,
                 c_tag: int) -> Tuple[int, int]:
        """Explicit registration (ref qinq.go:121 Register)."""
        if not (1 <= s_tag <= 4094 and 1 <= c_tag <= 4094):
            raise QinQError("invalid VLAN tag")
        key = (s_tag, c_tag)
        with self._lock:
            owner = self._by_vlan.get(key)
            if owner is not None and owner != subscriber_id:
                raise QinQError(f"{key} already mapped to {owner}")
            old = self._by_sub.get(subscriber_id)
            if old is not None and old != key:
                self._by_vlan.pop(old, None)
            self._by_vlan[key] = subscriber_id
            self._by_sub[subscriber_id] = key
        return key

    def auto_assign(self, subscriber_id: str) -> Tuple[int, int]:
        """Next free (s,c) from the configured ranges."""
        with self._lock:
            if subscriber_id in self._by_sub:
                return self._by_sub[subscriber_id]
            for r in self._ranges:
                for c in range(r.c_tag_start, r.c_tag_end + 1):
                    key = (r.s_tag, c)
                    if key not in self._by_vlan:
                        self._by_vlan[key] = subscriber_id
                        self._by_sub[subscriber_id] = key
                        return key
        raise QinQError("no free VLAN combination")

    def lookup(self, s_tag: int, c_tag: int) -> Optional[str]:
        with self._lock:
            return self._by_vlan.get((s_tag, c_tag))

    def lookup_subscriber(self, subscriber_id: str) -> Optional[Tuple[int, int]]:
        with self._lock:
            return self._by_sub.get(subscriber_id)

    def unregister(self, subscriber_id: str):
        with self._lock:
            key = self._by_sub.pop(subscriber_id, None)
            if key is not None:
                self._by_vlan.pop(key, None)

    def unregister_by_vlan(self, s_tag: int, c_tag: int) -> Optional[str]:
        """Remove a mapping keyed by the VLAN pair (ref qinq.go
        UnregisterByVLAN); returns the former owner."""
        with self._lock:
            owner = self._by_vlan.pop((s_tag, c_tag), None)
            if owner is not None:
                self._by_sub.pop(owner, None)
            return owner

    def stats(self) -> Dict[str, int]:
        """ref qinq.go Stats: mappings + capacity across ranges."""
        with self._lock:
            cap = sum(r.c_tag_end - r.c_tag_start + 1
                      for r in self._ranges)
            return {"mappings": len(self._by_vlan),
                    "ranges": len(self._ranges),
                    "capacity": cap,
                    "free": max(0, cap - len(self._by_vlan))}
