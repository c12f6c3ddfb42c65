"""Deferred-operation queue for partitions
(ref pkg/resilience/request_queue.go): operations that need the
unreachable upstream (Nexus writes, route announcements, ...) are queued
and drained on recovery, with bounded size and per-item retry budget."""
from __future__ import annotations

import threading
import time
from dataclasses import dataclass, field
from typing import Callable, List, Optional


@dataclass
class QueuedRequest:
    op: str
    fn: Callable[[], bool]
    queued_at: float = field(default_factory=time.time)
    attempts: int = 0
    max_attempts: int = 5


class RequestQueue:
    def __init__(self, max_size: int = 10000):
        self.max_size = max_size
        self._q: List[QueuedRequest] = []
        self._lock = threading.Lock()
        self.stats = {"queued": 0, "drained": 0, "dropped": 0,
                      "gave_up": 0}

    def enqueue(self, op: str, fn: Callable[[], bool],
                max_attempts: int = 5) -> bool:
        with self._lock:
            if len(self._q) >= self.max_size:
                self.stats["dropped"] += 1
                return False
            self._q.append(QueuedRequest(op, fn, max_attempts=max_attempts))
            self.stats["queued"] += 1
            return True

    def __len__(self):
        with self._lock:
            return len(self._q)

    def drain(self) -> int:
        """Run queued ops; re-queue failures up to their budget."""
        with self._lock:
            todo, self._q = self._q, []
        done = 0
        for req in todo:
            req.attempts += 1
            ok = False
            try:
                ok = bool(req.fn())
            except Exception:
                ok = False
            if ok:
                done += 1
            elif req.attempts < req.max_attempts:
                with self._lock:
                    self._q.append(req)
            else:
                self.stats["gave_up"] += 1
        self.stats["drained"] += done
        return done
