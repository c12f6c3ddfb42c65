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
    # LEA-assigned Lawful Interception ID + extra target criteria (ref
    # types.go Warrant :40-83; any set criterion matches a session)
    liid: str = ""
    target_mac: str = ""
    target_ipv4: str = ""
    target_ipv6: str = ""
    target_username: str = ""
    # delivery configuration
    delivery_method: str = "json"    # etsi | json | syslog
    mediation_address: str = ""
    mediation_port: int = 0
    # CC filters: empty list = no filtering on that axis
    filter_source_ports: List[int] = field(default_factory=list)
    filter_dest_ports: List[int] = field(default_factory=list)
    filter_protocols: List[int] = field(default_factory=list)
    filter_dest_ips: List[str] = field(default_factory=list)
    # per-warrant accounting
    sessions_matched: int = 0
    bytes_intercepted: int = 0
    last_activity: float = 0.0


@dataclass
class InterceptSession:
    """A live interception of one subscriber session (ref types.go
    InterceptSession :207-226)."""
    session_id: str
    warrant_id: str
    liid: str
    subscriber: str
    start_time: float
    mac: str = ""
    ipv4: str = ""
    ipv6: str = ""
    iri_records: int = 0
    cc_records: int = 0
    bytes_captured: int = 0
    last_activity: float = 0.0


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
        # secondary target indexes (ref manager.go byMAC/byIPv4/byIPv6)
        self.by_mac: Dict[str, List[str]] = {}
        self.by_ipv4: Dict[str, List[str]] = {}
        self.by_username: Dict[str, List[str]] = {}
        self.records: List[InterceptRecord] = []
        self.sessions: Dict[str, InterceptSession] = {}
        self.exporters = exporters or []
        self.admin_exporter = admin_exporter
        self._lock = threading.RLock()
        self.delivery_errors = 0
        self.total_bytes_delivered = 0

    # ----------------------------------------------------------- warrants
    def add_warrant(self, target_subscriber: str = "",
                    authority: str = "",
                    case_reference: str = "", duration: float = 0.0,
                    intercept_type: str = "iri",
                    valid_from: float = 0.0, **kw) -> Warrant:
        """ref AddWarrant manager.go:141-180: a future valid_from makes
        the warrant PENDING (not matching traffic) until the sweep or a
        lookup crosses the start time; intercept_type per ETSI:
        iri (metadata) | cc (content) | iri+cc."""
        if not target_subscriber and not kw.get("target_mac") and \
                not kw.get("target_ipv4") and \
                not kw.get("target_username"):
            raise ValueError("warrant needs at least one target criterion")
        if intercept_type not in ("iri", "cc", "iri+cc"):
            raise ValueError(f"unknown intercept type {intercept_type}")
        start = valid_from or time.time()
        w = Warrant(id=uuid.uuid4().hex[:12],
                    target_subscriber=target_subscriber,
                    authority=authority, case_reference=case_reference,
                    start_time=start,
                    end_time=start + duration if duration else 0.0,
                    intercept_type=intercept_type,
                    **{k: v for k, v in kw.items()
                       if k in Warrant.__dataclass_fields__})
        if not w.liid:
            w.liid = "LIID-" + w.id
        with self._lock:
            self.warrants[w.id] = w
            if target_subscriber:
                self.by_target.setdefault(
                    target_subscriber, []).append(w.id)
            if w.target_mac:
                self.by_mac.setdefault(
                    w.target_mac.lower(), []).append(w.id)
            if w.target_ipv4:
                self.by_ipv4.setdefault(w.target_ipv4, []).append(w.id)
            if w.target_username:
                self.by_username.setdefault(
                    w.target_username, []).append(w.id)
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

    # ----------------------------------------- multi-criteria matching
    def _is_active(self, w: Warrant, now: float) -> bool:
        return (w.active and w.start_time <= now and
                (w.end_time == 0 or w.end_time > now))

    def match_session(self, subscriber: str = "", mac: str = "",
                      ipv4: str = "", username: str = "") -> List[Warrant]:
        """Active warrants matching ANY provided criterion, deduped
        (ref MatchSession manager.go:260-301)."""
        now = time.time()
        seen, out = set(), []
        with self._lock:
            wids = []
            if subscriber:
                wids += self.by_target.get(subscriber, [])
            if mac:
                wids += self.by_mac.get(mac.lower(), [])
            if ipv4:
                wids += self.by_ipv4.get(ipv4, [])
            if username:
                wids += self.by_username.get(username, [])
            for wid in wids:
                w = self.warrants.get(wid)
                if w and wid not in seen and self._is_active(w, now):
                    seen.add(wid)
                    out.append(w)
        return out

    # --------------------------------------- intercept session lifecycle
    def start_intercept(self, warrant: Warrant, session_id: str,
                        subscriber: str, mac: str = "", ipv4: str = "",
                        ipv6: str = "") -> InterceptSession:
        """Begin intercepting one subscriber session under a warrant
        (ref StartInterceptSession manager.go:381-416); emits the
        session_start IRI."""
        now = time.time()
        s = InterceptSession(session_id=session_id,
                             warrant_id=warrant.id, liid=warrant.liid,
                             subscriber=subscriber, start_time=now,
                             mac=mac, ipv4=ipv4, ipv6=ipv6,
                             last_activity=now)
        with self._lock:
            self.sessions[session_id] = s
            warrant.sessions_matched += 1
            warrant.last_activity = now
        self._record(subscriber, "session_start", ipv4,
                     session_id=session_id, liid=warrant.liid)
        s.iri_records += 1
        return s

    def stop_intercept(self, session_id: str):
        """ref StopInterceptSession manager.go:418-449."""
        with self._lock:
            s = self.sessions.pop(session_id, None)
        if s is not None:
            self._record(s.subscriber, "session_stop", s.ipv4,
                         session_id=session_id, liid=s.liid)
        return s

    def get_intercept(self, session_id: str) -> Optional[InterceptSession]:
        with self._lock:
            return self.sessions.get(session_id)

    # -------------------------------------------- CC (content) capture
    @staticmethod
    def _cc_passes_filters(w: Warrant, src_port: int, dst_port: int,
                           protocol: int, dst_ip: str) -> bool:
        """Every configured filter axis must admit the flow (ref
        RecordCC manager.go:337-379)."""
        if w.filter_source_ports and src_port not in w.filter_source_ports:
            return False
        if w.filter_dest_ports and dst_port not in w.filter_dest_ports:
            return False
        if w.filter_protocols and protocol not in w.filter_protocols:
            return False
        if w.filter_dest_ips and dst_ip not in w.filter_dest_ips:
            return False
        return True

    def record_cc(self, session_id: str, direction: str, src_ip: str,
                  dst_ip: str, src_port: int, dst_port: int,
                  protocol: int, payload: bytes) -> bool:
        """Deliver one content packet for a live interception; returns
        False when the warrant's filters exclude the flow or the
        warrant isn't CC-typed."""
        with self._lock:
            s = self.sessions.get(session_id)
            w = self.warrants.get(s.warrant_id) if s else None
        if s is None or w is None:
            return False
        if "cc" not in w.intercept_type:
            return False
        if not self._cc_passes_filters(w, src_port, dst_port, protocol,
                                       dst_ip):
            return False
        now = time.time()
        rec = InterceptRecord(
            warrant_id=w.id, record_id=uuid.uuid4().hex[:12],
            timestamp=now, record_type="cc",
            subscriber=s.subscriber, ip=src_ip,
            details={"direction": direction, "src_ip": src_ip,
                     "dst_ip": dst_ip, "src_port": str(src_port),
                     "dst_port": str(dst_port),
                     "protocol": str(protocol),
                     "session_id": session_id, "liid": s.liid,
                     "payload_len": str(len(payload))})
        with self._lock:
            self.records.append(rec)
            s.cc_records += 1
            s.bytes_captured += len(payload)
            s.last_activity = now
            w.bytes_intercepted += len(payload)
            w.last_activity = now
            self.total_bytes_delivered += len(payload)
        for ex in self.exporters:
            try:
                if hasattr(ex, "export_cc"):
                    ex.export_cc(rec, payload)
                else:
                    ex.export(rec)
            except Exception:
                self.delivery_errors += 1
        return True

    def stats(self) -> Dict[str, int]:
        """ref ManagerStats types.go:228-236."""
        now = time.time()
        with self._lock:
            active = sum(1 for w in self.warrants.values()
                         if self._is_active(w, now))
            iri = sum(1 for r in self.records if r.record_type != "cc")
            cc = sum(1 for r in self.records if r.record_type == "cc")
            return {"active_warrants": active,
                    "active_interceptions": len(self.sessions),
                    "total_iri_records": iri, "total_cc_records": cc,
                    "total_bytes_delivered": self.total_bytes_delivered,
                    "delivery_errors": self.delivery_errors}
