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


# ----------------------------------------------------------------------
# VLAN-pair semantics + configured mapper (ref qinq.go:18-212)

@dataclass(frozen=True)
class VLANPair:
    """(S-TAG, C-TAG); s=0 means single-tagged, both 0 untagged (ref
    VLANPair qinq.go:18-45)."""
    s_tag: int = 0
    c_tag: int = 0

    def __str__(self):
        if self.s_tag == 0:
            return f"c{self.c_tag}"
        return f"s{self.s_tag}.c{self.c_tag}"

    @property
    def is_double_tagged(self):
        return self.s_tag > 0 and self.c_tag > 0

    @property
    def is_single_tagged(self):
        return self.s_tag == 0 and self.c_tag > 0

    @property
    def is_untagged(self):
        return self.s_tag == 0 and self.c_tag == 0


@dataclass
class QinQConfig:
    """ref Config qinq.go:47-98.  lookup_priority: vlan_first |
    mac_first | vlan_only."""
    enabled: bool = False
    s_tag_ranges: list = None       # [(start, end, name)]
    c_tag_range: tuple = (100, 4094)
    default_s_tag: int = 0
    lookup_priority: str = "vlan_first"

    def __post_init__(self):
        if self.s_tag_ranges is None:
            self.s_tag_ranges = [(100, 999, "default")]
        if self.lookup_priority not in ("vlan_first", "mac_first",
                                        "vlan_only"):
            raise QinQError(
                f"invalid lookup_priority {self.lookup_priority}")

    def s_tag_valid(self, s_tag: int) -> bool:
        return any(lo <= s_tag <= hi
                   for lo, hi, *_ in self.s_tag_ranges)

    def c_tag_valid(self, c_tag: int) -> bool:
        lo, hi = self.c_tag_range
        return lo <= c_tag <= hi


class ConfiguredMapper:
    """Range-validated VLAN pair <-> subscriber mapper (ref Mapper
    qinq.go:100-212): S-TAG must fall in a configured (named) range,
    C-TAG in the subscriber range; re-registering a subscriber moves
    them (old pair freed); a pair owned by another subscriber is
    refused."""

    def __init__(self, config: Optional[QinQConfig] = None):
        self.config = config or QinQConfig()
        self._by_vlan: Dict[VLANPair, str] = {}
        self._by_sub: Dict[str, VLANPair] = {}
        self._lock = threading.RLock()

    def register(self, pair: VLANPair, subscriber_id: str):
        if pair.s_tag > 0 and not self.config.s_tag_valid(pair.s_tag):
            raise QinQError(f"S-TAG {pair.s_tag} not in allowed ranges")
        if pair.c_tag > 0 and not self.config.c_tag_valid(pair.c_tag):
            lo, hi = self.config.c_tag_range
            raise QinQError(
                f"C-TAG {pair.c_tag} not in allowed range [{lo}-{hi}]")
        with self._lock:
            owner = self._by_vlan.get(pair)
            if owner is not None and owner != subscriber_id:
                raise QinQError(
                    f"VLAN pair {pair} already mapped to {owner}")
            old = self._by_sub.get(subscriber_id)
            if old is not None and old != pair:
                self._by_vlan.pop(old, None)
            self._by_vlan[pair] = subscriber_id
            self._by_sub[subscriber_id] = pair

    def unregister(self, pair: VLANPair):
        with self._lock:
            sub = self._by_vlan.pop(pair, None)
            if sub is not None:
                self._by_sub.pop(sub, None)

    def unregister_subscriber(self, subscriber_id: str):
        with self._lock:
            pair = self._by_sub.pop(subscriber_id, None)
            if pair is not None:
                self._by_vlan.pop(pair, None)

    def get_subscriber(self, pair: VLANPair) -> Optional[str]:
        with self._lock:
            return self._by_vlan.get(pair)

    def get_vlan(self, subscriber_id: str) -> Optional[VLANPair]:
        with self._lock:
            return self._by_sub.get(subscriber_id)

    def lookup(self, pair: VLANPair, mac_lookup=None,
               mac: str = "") -> Optional[str]:
        """Subscriber resolution honoring lookup_priority (ref Config
        LookupPriority qinq.go:62-66): vlan_only never consults the
        MAC path; mac_first tries it before the VLAN table."""
        by_vlan = lambda: self.get_subscriber(pair)   # noqa: E731
        by_mac = (lambda: mac_lookup(mac)) if mac_lookup and mac \
            else (lambda: None)
        order = {"vlan_first": (by_vlan, by_mac),
                 "mac_first": (by_mac, by_vlan),
                 "vlan_only": (by_vlan,)}[self.config.lookup_priority]
        for fn in order:
            got = fn()
            if got:
                return got
        return None

    def stats(self) -> Dict[str, int]:
        with self._lock:
            return {"total_mappings": len(self._by_vlan),
                    "double_tagged": sum(1 for p in self._by_vlan
                                         if p.is_double_tagged),
                    "single_tagged": sum(1 for p in self._by_vlan
                                         if p.is_single_tagged)}
