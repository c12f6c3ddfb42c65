"""Split-brain allocation conflict detector
(ref pkg/resilience/conflict_detector.go:25-262): after a partition
heals, the same IP may have been handed to different subscribers by the
two sides.  Detect by scanning allocation records; resolve by
keep-oldest (the newer claimant must re-allocate)."""
from __future__ import annotations

import threading
import time
from dataclasses import dataclass, field
from typing import Callable, Dict, List, Optional, Tuple


@dataclass
class Allocation:
    ip: str
    subscriber_id: str
    node_id: str
    allocated_at: float


@dataclass
class Conflict:
    ip: str
    keeper: Allocation
    evicted: List[Allocation]
    detected_at: float = field(default_factory=time.time)


class ConflictDetector:
    def __init__(self,
                 on_conflict: Optional[Callable[[Conflict], None]] = None):
        self.on_conflict = on_conflict
        self.conflicts: List[Conflict] = []
        self._lock = threading.Lock()

    def scan(self, allocations: List[Allocation]) -> List[Conflict]:
        """Find IPs claimed by more than one subscriber; resolution is
        keep-oldest (first allocation wins, ref conflict_detector.go)."""
        by_ip: Dict[str, List[Allocation]] = {}
        for a in allocations:
            by_ip.setdefault(a.ip, []).append(a)
        found = []
        for ip, allocs in by_ip.items():
            subs = {a.subscriber_id for a in allocs}
            if len(subs) <= 1:
                continue
            allocs.sort(key=lambda a: a.allocated_at)
            keeper = allocs[0]
            evicted = [a for a in allocs[1:]
                       if a.subscriber_id != keeper.subscriber_id]
            c = Conflict(ip, keeper, evicted)
            found.append(c)
            if self.on_conflict:
                try:
                    self.on_conflict(c)
                except Exception:
                    pass
        with self._lock:
            self.conflicts.extend(found)
        return found


# ----------------------------------------------------------------------
# Stateful site-aware detector with the reference's resolution policy
# (ref conflict_detector.go:25-330, manager.go resolveConflict
# :430-527)

R_PENDING = "pending"
R_LOCAL_WINS = "local_wins"
R_REMOTE_WINS = "remote_wins"


@dataclass
class IPAllocation:
    """ref types.go IPAllocation: site + partition-era provenance."""
    ip: str
    mac: str
    subscriber_id: str
    site_id: str
    allocated_at: float
    is_partition: bool = False       # allocated while partitioned


@dataclass
class AllocationConflict:
    ip: str
    local: IPAllocation
    remote: IPAllocation
    detected_at: float = field(default_factory=time.time)
    resolution: str = R_PENDING
    affected_mac: str = ""           # the loser needing re-allocation


class ConflictError(Exception):
    def __init__(self, ip: str, holder_mac: str):
        super().__init__(f"IP {ip} already allocated to {holder_mac}")
        self.ip = ip
        self.holder_mac = holder_mac


class SiteConflictDetector:
    """Registry of this site's live allocations; DetectConflicts
    compares against a remote (Nexus) view — the same subscriber on
    both sides is not a conflict, different sites claiming one IP for
    different subscribers is (ref DetectConflicts :121-232)."""

    def __init__(self, site_id: str):
        self.site_id = site_id
        self.local: Dict[str, IPAllocation] = {}
        self.conflicts: List[AllocationConflict] = []
        self._lock = threading.RLock()

    # --------------------------------------------------- registry
    def record(self, ip: str, mac: str, subscriber_id: str,
               allocated_at: Optional[float] = None,
               is_partition: bool = False):
        with self._lock:
            self.local[ip] = IPAllocation(
                ip, mac, subscriber_id, self.site_id,
                allocated_at if allocated_at is not None else time.time(),
                is_partition)

    def remove(self, ip: str):
        with self._lock:
            self.local.pop(ip, None)

    def get(self, ip: str) -> Optional[IPAllocation]:
        with self._lock:
            return self.local.get(ip)

    def partition_allocations(self) -> List[IPAllocation]:
        with self._lock:
            return [a for a in self.local.values() if a.is_partition]

    def clear_partition_flags(self):
        """After reconciliation the era distinction is spent."""
        with self._lock:
            for a in self.local.values():
                a.is_partition = False

    def validate(self, ip: str, mac: str):
        """Pre-allocation guard (ref ValidateAllocation :244-260):
        raises ConflictError if the IP is held by a different MAC."""
        with self._lock:
            a = self.local.get(ip)
        if a is not None and a.mac.lower() != mac.lower():
            raise ConflictError(ip, a.mac)

    def export_allocations(self) -> List[IPAllocation]:
        with self._lock:
            return list(self.local.values())

    def import_allocations(self, allocations: List[IPAllocation]):
        with self._lock:
            for a in allocations:
                self.local[a.ip] = a

    # -------------------------------------------------- detection
    def detect(self, remote: List[IPAllocation]) -> List[AllocationConflict]:
        remote_by_ip = {a.ip: a for a in remote}
        found = []
        with self._lock:
            for ip, la in self.local.items():
                ra = remote_by_ip.get(ip)
                if ra is None:
                    continue
                if la.mac.lower() == ra.mac.lower() and \
                        la.subscriber_id == ra.subscriber_id:
                    continue                       # same subscriber
                if la.site_id != ra.site_id:
                    found.append(AllocationConflict(ip, la, ra))
            self.conflicts.extend(found)
        return found

    @staticmethod
    def resolve(c: AllocationConflict) -> AllocationConflict:
        """The reference's three-step policy (resolveConflict
        manager.go:462-489): same MAC -> most recent wins; a
        pre-partition allocation beats a during-partition one (Nexus
        is the source of truth); both-partition -> most recent wins.
        The losing MAC is recorded for forced re-allocation."""
        la, ra = c.local, c.remote
        if la.mac.lower() == ra.mac.lower():
            c.resolution = R_LOCAL_WINS if \
                la.allocated_at > ra.allocated_at else R_REMOTE_WINS
        elif not la.is_partition and ra.is_partition:
            c.resolution = R_LOCAL_WINS
            c.affected_mac = ra.mac
        elif la.is_partition and not ra.is_partition:
            c.resolution = R_REMOTE_WINS
            c.affected_mac = la.mac
        elif la.allocated_at > ra.allocated_at:
            c.resolution = R_LOCAL_WINS
            c.affected_mac = ra.mac
        else:
            c.resolution = R_REMOTE_WINS
            c.affected_mac = la.mac
        return c

    def unresolved(self) -> List[AllocationConflict]:
        with self._lock:
            return [c for c in self.conflicts
                    if c.resolution == R_PENDING]

    def mark_resolved(self, ip: str, resolution: str):
        with self._lock:
            for c in self.conflicts:
                if c.ip == ip and c.resolution == R_PENDING:
                    c.resolution = resolution

    def detector_stats(self):
        with self._lock:
            return {"local": len(self.local),
                    "partition_era": sum(1 for a in self.local.values()
                                         if a.is_partition),
                    "conflicts": len(self.conflicts)}
