"""Lawful intercept (ref pkg/intercept): warrant management, intercept
records, ETSI X1/X2-style + JSON exporters (types.go:40-238,
exporter.go:17-442)."""
from __future__ import annotations

import json
import threading
import time
import uuid
from dataclasses import asdict, dataclass, field
from typing import Dict, List, Optional


@dataclass
class Warrant:
    id: str
    target_subscriber: str
    authority: str = ""
    case_reference: str = ""
    start_time: float = 0.0
    end_time: float = 0.0            # 0 = open-ended
    intercept_type: str = "iri"      # iri (metadata) | cc (content)
    active: bool = True


@dataclass
class InterceptRecord:
    warrant_id: str
    record_id: str
    timestamp: float
    record_type: str                 # session_start/session_stop/flow/...
    subscriber: str
    ip: str = ""
    details: Dict[str, str] = field(default_factory=dict)


class X1Exporter:
    """ETSI X1-style administrative records (warrant lifecycle)."""

    def __init__(self):
        self.records: List[str] = []

    def export_admin(self, action: str, w: Warrant):
        self.records.append(
            f"X1 {action} warrant={w.id} target={w.target_subscriber} "
            f"authority={w.authority} case={w.case_reference}")


class X2Exporter:
    """ETSI X2-style IRI (intercept-related information) records."""

    def __init__(self):
        self.records: List[str] = []

    def export(self, rec: InterceptRecord):
        self.records.append(
            f"X2 {rec.record_type} warrant={rec.warrant_id} "
            f"target={rec.subscriber} ip={rec.ip} ts={rec.timestamp:.3f}")


class JSONExporter:
    def __init__(self, sink=None):
        self.lines: List[str] = []
        self.sink = sink

    def export(self, rec: InterceptRecord):
        line = json.dumps(asdict(rec))
        self.lines.append(line)
        if self.sink:
            self.sink(line)


class Manager:
    def __init__(self, exporters: Optional[list] = None,
                 admin_exporter: Optional[X1Exporter] = None):
        self.warrants: Dict[str, Warrant] = {}
        self.by_target: Dict[str, List[str]] = {}
        self.records: List[InterceptRecord] = []
        self.exporters = exporters or []
        self.admin_exporter = admin_exporter
        self._lock = threading.RLock()

    # ----------------------------------------------------------- warrants
    def add_warrant(self, target_subscriber: str, authority: str = "",
                    case_reference: str = "", duration: float = 0.0,
                    intercept_type: str = "iri",
                    valid_from: float = 0.0) -> Warrant:
        """ref AddWarrant manager.go:141-180: a future valid_from makes
        the warrant PENDING (not matching traffic) until the sweep or a
        lookup crosses the start time; intercept_type per ETSI:
        iri (metadata) | cc (content) | iri+cc."""
        if not target_subscriber:
            raise ValueError("warrant needs a target subscriber")
        if intercept_type not in ("iri", "cc", "iri+cc"):
            raise ValueError(f"unknown intercept type {intercept_type}")
        start = valid_from or time.time()
        w = Warrant(id=uuid.uuid4().hex[:12],
                    target_subscriber=target_subscriber,
                    authority=authority, case_reference=case_reference,
                    start_time=start,
                    end_time=start + duration if duration else 0.0,
                    intercept_type=intercept_type)
        with self._lock:
            self.warrants[w.id] = w
            self.by_target.setdefault(target_subscriber, []).append(w.id)
        if self.admin_exporter:
            self.admin_exporter.export_admin("activate", w)
        return w

    def revoke_warrant(self, warrant_id: str) -> bool:
        with self._lock:
            w = self.warrants.get(warrant_id)
            if w is None:
                return False
            w.active = False
        if self.admin_exporter:
            self.admin_exporter.export_admin("deactivate", w)
        return True

    def _active_warrants(self, subscriber: str) -> List[Warrant]:
        now = time.time()
        with self._lock:
            out = []
            for wid in self.by_target.get(subscriber, []):
                w = self.warrants.get(wid)
                if w and w.active and w.start_time <= now and \
                        (w.end_time == 0 or w.end_time > now):
                    out.append(w)
            return out

    def warrant_status(self, warrant_id: str) -> str:
        """pending | active | expired | revoked (ref WarrantStatus)."""
        now = time.time()
        with self._lock:
            w = self.warrants.get(warrant_id)
        if w is None:
            return "unknown"
        if not w.active:
            return "revoked"
        if now < w.start_time:
            return "pending"
        if w.end_time and now >= w.end_time:
            return "expired"
        return "active"

    def is_target(self, subscriber: str) -> bool:
        return bool(self._active_warrants(subscriber))

    # ------------------------------------------------------------ records
    def _record(self, subscriber: str, record_type: str, ip: str = "",
                **details):
        for w in self._active_warrants(subscriber):
            rec = InterceptRecord(
                warrant_id=w.id, record_id=uuid.uuid4().hex[:12],
                timestamp=time.time(), record_type=record_type,
                subscriber=subscriber, ip=ip,
                details={k: str(v) for k, v in details.items()})
            with self._lock:
                self.records.append(rec)
            for ex in self.exporters:
                try:
                    ex.export(rec)
                except Exception:
                    pass

    def on_session_start(self, subscriber: str, ip: str, **details):
        self._record(subscriber, "session_start", ip, **details)

    def on_session_stop(self, subscriber: str, ip: str = "", **details):
        self._record(subscriber, "session_stop", ip, **details)

    def on_nat_event(self, subscriber: str, ip: str = "", **details):
        self._record(subscriber, "nat_mapping", ip, **details)
