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
