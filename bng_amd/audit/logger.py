"""Audit subsystem (ref pkg/audit): structured events (types.go:101-238),
async logger over a storage interface (logger.go:15-161), in-memory
storage with query, rotating file exporter with compression
(rotation.go:19-402), syslog/JSON-lines exporters (export.go:17-486),
retention + legal hold (retention.go:9-347)."""
from __future__ import annotations

import gzip
import json
import os
import queue
import threading
import time
import uuid
from dataclasses import asdict, dataclass, field
from typing import Dict, List, Optional

# event categories (ref types.go:101-238)
CAT_SESSION = "session"
CAT_NAT = "nat"
CAT_AUTH = "auth"
CAT_TLS = "tls"
CAT_ZTP = "ztp"
CAT_CONFIG = "config"
CAT_SYSTEM = "system"


@dataclass
class Event:
    id: str
    category: str
    action: str
    timestamp: float
    subscriber: str = ""
    ip: str = ""
    outcome: str = "success"
    details: Dict[str, str] = field(default_factory=dict)
    legal_hold: bool = False
    severity: int = 1                    # retention.INFO


class MemoryStorage:
    def __init__(self, max_events: int = 100_000):
        self.max_events = max_events
        self._events: List[Event] = []
        self._lock = threading.Lock()

    def store(self, ev: Event):
        with self._lock:
            self._events.append(ev)
            if len(self._events) > self.max_events:
                # never evict events under legal hold (ref retention.go)
                keep = [e for e in self._events[:len(self._events) // 2]
                        if e.legal_hold]
                self._events = keep + self._events[len(self._events) // 2:]

    def query(self, category: str = "", subscriber: str = "",
              action: str = "", since: float = 0.0,
              until: float = 0.0, min_severity: int = 0,
              limit: int = 0) -> List[Event]:
        with self._lock:
            out = []
            for e in self._events:
                if category and e.category != category:
                    continue
                if subscriber and e.subscriber != subscriber:
                    continue
                if action and e.action != action:
                    continue
                if since and e.timestamp < since:
                    continue
                if until and e.timestamp > until:
                    continue
                if min_severity and e.severity < min_severity:
                    continue
                out.append(e)
                if limit and len(out) >= limit:
                    break
            return out

    def delete(self, ids) -> int:
        """ref storage.go Delete :198-230."""
        ids = set(ids)
        with self._lock:
            before = len(self._events)
            self._events = [e for e in self._events if e.id not in ids]
            return before - len(self._events)

    def count(self) -> int:
        with self._lock:
            return len(self._events)

    def all(self) -> List[Event]:
        with self._lock:
            return list(self._events)

    def apply_retention(self, max_age: float) -> int:
        cutoff = time.time() - max_age
        with self._lock:
            before = len(self._events)
            self._events = [e for e in self._events
                            if e.legal_hold or e.timestamp >= cutoff]
            return before - len(self._events)

    def apply_retention_manager(self, rm, now: float = 0.0) -> int:
        """Per-category/action expiry honoring legal holds (ref
        logger.go cleanupExpired :588-610 over
        storage.DeleteExpired)."""
        now = now or time.time()
        with self._lock:
            before = len(self._events)
            self._events = [e for e in self._events
                            if e.legal_hold or not rm.expired(e, now)]
            return before - len(self._events)


class FileExporter:
    """Rotating JSON-lines file exporter w/ gzip (ref rotation.go)."""

    def __init__(self, path: str, rotate_bytes: int = 10 << 20,
                 compress: bool = True, retention: int = 10):
        self.path = path
        self.rotate_bytes = rotate_bytes
        self.compress = compress
        self.retention = retention
        self._fh = open(path, "a")
        self._written = os.path.getsize(path)
        self._lock = threading.Lock()

    def export(self, ev: Event):
        line = json.dumps(asdict(ev))
        with self._lock:
            self._fh.write(line + "\n")
            self._written += len(line) + 1
            if self._written >= self.rotate_bytes:
                self._rotate()

    def _rotate(self):
        self._fh.close()
        rotated = f"{self.path}.{time.strftime('%Y%m%d-%H%M%S')}-" \
                  f"{uuid.uuid4().hex[:6]}"
        os.rename(self.path, rotated)
        if self.compress:
            with open(rotated, "rb") as src, \
                    gzip.open(rotated + ".gz", "wb") as dst:
                dst.write(src.read())
            os.unlink(rotated)
        self._fh = open(self.path, "a")
        self._written = 0
        d = os.path.dirname(self.path) or "."
        base = os.path.basename(self.path) + "."
        rotated_files = sorted(f for f in os.listdir(d)
                               if f.startswith(base))
        while len(rotated_files) > self.retention:
            os.unlink(os.path.join(d, rotated_files.pop(0)))

    def close(self):
        with self._lock:
            self._fh.close()


class SyslogExporter:
    """RFC5424-ish lines to a callable/file (network syslog plugs in)."""

    def __init__(self, sink):
        self.sink = sink

    FACILITY = 13                        # log audit (RFC5424)

    def export(self, ev: Event):
        # severity -> syslog level: DEBUG=7 .. EMERGENCY=0 (ref
        # export.go formatMessage :131-141)
        level = max(0, 7 - ev.severity)
        pri = self.FACILITY * 8 + level
        self.sink(f"<{pri}>1 - bng audit - - - [{ev.category}] "
                  f"{ev.action} sub={ev.subscriber} ip={ev.ip} "
                  f"outcome={ev.outcome}")


class Logger:
    """Async audit logger (ref logger.go:15-161)."""

    def __init__(self, storage: Optional[MemoryStorage] = None,
                 exporters: Optional[list] = None, queue_size: int = 10000,
                 min_severity: int = 0,
                 disabled_categories: Optional[set] = None,
                 retention=None):
        self.storage = storage or MemoryStorage()
        self.exporters = exporters or []
        self._q: queue.Queue = queue.Queue(maxsize=queue_size)
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self.dropped = 0
        self.legal_holds: set = set()       # subscriber ids under hold
        # severity floor + category kill-switch (ref logger.go
        # shouldLog :411-434)
        self.min_severity = min_severity
        self.disabled_categories = disabled_categories or set()
        # optional RetentionManager (retention.py) for per-category
        # expiry + criteria-matched legal holds
        self.retention = retention
        self.filtered = 0
        self.logged = 0
        self.exported = 0
        self.export_errors = 0

    def start(self):
        self._thread = threading.Thread(target=self._loop, daemon=True)
        self._thread.start()
        return self

    def stop(self):
        self._stop.set()
        self.flush()

    def set_legal_hold(self, subscriber: str, held: bool = True):
        """ref retention.go legal holds."""
        if held:
            self.legal_holds.add(subscriber)
        else:
            self.legal_holds.discard(subscriber)

    def log(self, action: str, category: str = CAT_SESSION,
            subscriber: str = "", ip: str = "", outcome: str = "success",
            severity: Optional[int] = None, **details):
        from .retention import action_severity
        sev = severity if severity is not None else \
            action_severity(action)
        if sev < self.min_severity or category in self.disabled_categories:
            self.filtered += 1
            return
        ev = Event(id=uuid.uuid4().hex[:12], category=category,
                   action=action, timestamp=time.time(),
                   subscriber=subscriber, ip=ip, outcome=outcome,
                   details={k: str(v) for k, v in details.items()},
                   legal_hold=subscriber in self.legal_holds,
                   severity=sev)
        if self.retention is not None and not ev.legal_hold:
            ev.legal_hold = self.retention.is_under_hold(ev)
        try:
            self._q.put_nowait(ev)
            self.logged += 1
        except queue.Full:
            self.dropped += 1

    def stats(self) -> Dict[str, int]:
        """ref LoggerStats logger.go:150-159."""
        return {"logged": self.logged, "dropped": self.dropped,
                "filtered": self.filtered,
                "stored": self.storage.count()
                if hasattr(self.storage, "count") else 0,
                "exported": self.exported,
                "export_errors": self.export_errors}

    def cleanup_expired(self, now: float = 0.0) -> int:
        """Apply the retention manager to storage (ref logger.go
        retentionLoop/cleanupExpired :562-610)."""
        if self.retention is None or \
                not hasattr(self.storage, "apply_retention_manager"):
            return 0
        self.retention.cleanup_expired_holds()
        return self.storage.apply_retention_manager(self.retention, now)

    def _loop(self):
        while not self._stop.is_set():
            try:
                ev = self._q.get(timeout=0.2)
            except queue.Empty:
                continue
            self._process(ev)

    def _process(self, ev: Event):
        self.storage.store(ev)
        for ex in self.exporters:
            try:
                ex.export(ev)
                self.exported += 1
            except Exception:
                self.export_errors += 1

    def flush(self):
        while True:
            try:
                ev = self._q.get_nowait()
            except queue.Empty:
                return
            self._process(ev)


class SecurityAuditor:
    """TLS / certificate / mTLS security events (ref
    pkg/audit/security.go + security_test.go): typed wrappers over the
    async logger so operators get a uniform bng-security event stream."""

    def __init__(self, logger: "Logger"):
        self.logger = logger

    def log_tls_handshake(self, peer: str, success: bool,
                          tls_version: str = "", cipher: str = "",
                          error: str = ""):
        self.logger.log(
            "tls_handshake", category=CAT_TLS,
            outcome="success" if success else "failure",
            peer=peer, tls_version=tls_version, cipher=cipher,
            error=error)

    def log_certificate_expiring(self, subject: str, not_after: str,
                                 days_left: int):
        self.logger.log("certificate_expiring", category=CAT_TLS,
                        outcome="warning", subject=subject,
                        not_after=not_after, days_left=days_left)

    def log_certificate_expired(self, subject: str, not_after: str):
        self.logger.log("certificate_expired", category=CAT_TLS,
                        outcome="failure", subject=subject,
                        not_after=not_after)

    def log_certificate_invalid(self, subject: str, reason: str):
        self.logger.log("certificate_invalid", category=CAT_TLS,
                        outcome="failure", subject=subject,
                        reason=reason)

    def log_certificate_pin_failed(self, peer: str, fingerprint: str):
        self.logger.log("certificate_pin_failed", category=CAT_TLS,
                        outcome="failure", peer=peer,
                        fingerprint=fingerprint)

    def log_certificate_renewed(self, subject: str, not_after: str):
        self.logger.log("certificate_renewed", category=CAT_TLS,
                        outcome="success", subject=subject,
                        not_after=not_after)

    def log_mtls_auth(self, device_id: str, success: bool,
                      subject: str = "", error: str = ""):
        self.logger.log(
            "mtls_auth", category=CAT_AUTH,
            outcome="success" if success else "failure",
            subscriber=device_id, subject=subject, error=error)


class IPFIXExporter:
    """RFC 7011 IPFIX export of NAT audit events (ref
    pkg/audit/export.go:143-315).  Emits a template set (id 256:
    sourceIPv4Address, postNATSourceIPv4Address, sourceTransportPort,
    postNAPTSourceTransportPort, protocolIdentifier, flowStartSeconds)
    on the first message and every template_refresh records thereafter,
    then one data set per event; transport is a pluggable send(bytes)
    (UDP socket in production, capture list in tests)."""

    TEMPLATE_ID = 256
    # (information element id, length)
    FIELDS = [(8, 4), (225, 4), (7, 2), (227, 2), (4, 1), (150, 4)]

    def __init__(self, send, observation_domain: int = 1,
                 template_refresh: int = 100):
        import struct as st
        self._st = st
        self.send = send
        self.domain = observation_domain
        self.template_refresh = template_refresh
        self.seq = 0
        self._since_template = None     # None => template never sent

    def _msg(self, sets: bytes, now: int) -> bytes:
        st = self._st
        return st.pack(">HHIII", 10, 16 + len(sets), now, self.seq,
                       self.domain) + sets

    def _template_set(self) -> bytes:
        st = self._st
        body = st.pack(">HH", self.TEMPLATE_ID, len(self.FIELDS))
        for ie, ln in self.FIELDS:
            body += st.pack(">HH", ie, ln)
        return st.pack(">HH", 2, 4 + len(body)) + body

    def export(self, ev: "Event") -> bool:
        """NAT-category events only (ref Export :210-214)."""
        if ev.category != CAT_NAT:
            return False
        st = self._st
        d = ev.details
        rec = st.pack(
            ">IIHHBI",
            int(d.get("private_ip", 0) or 0),
            int(d.get("public_ip", 0) or 0),
            int(d.get("private_port", 0) or 0),
            int(d.get("public_port", 0) or 0),
            int(d.get("protocol", 0) or 0),
            int(ev.timestamp))
        sets = b""
        if self._since_template is None or \
                self._since_template >= self.template_refresh:
            sets += self._template_set()
            self._since_template = 0
        sets += st.pack(">HH", self.TEMPLATE_ID, 4 + len(rec)) + rec
        self.send(self._msg(sets, int(ev.timestamp)))
        self.seq += 1
        self._since_template += 1
        return True


class KafkaExporter:
    """Topic-routing audit exporter with a pluggable producer (the
    reference ships this as a stub over kafka-go, export.go:436-533;
    ours takes produce(topic, key, value) so any client slots in)."""

    def __init__(self, produce=None, topic: str = "bng-audit",
                 topic_by_category: bool = False,
                 topic_prefix: str = "bng-audit-",
                 key_field: str = "subscriber"):
        self.produce = produce or (lambda t, k, v: None)
        self.topic = topic
        self.topic_by_category = topic_by_category
        self.topic_prefix = topic_prefix
        self.key_field = key_field
        self.exported = 0

    def export(self, ev: "Event"):
        import json as _json
        topic = (self.topic_prefix + ev.category
                 if self.topic_by_category else self.topic)
        key = getattr(ev, self.key_field, "") or ev.id
        value = _json.dumps({
            "id": ev.id, "category": ev.category, "action": ev.action,
            "timestamp": ev.timestamp, "subscriber": ev.subscriber,
            "ip": ev.ip, "outcome": ev.outcome, "details": ev.details})
        self.produce(topic, key, value)
        self.exported += 1
