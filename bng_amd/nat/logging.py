"""NAT compliance logging pipeline (ref pkg/nat/logging.go:63-702):
formats (JSON / CSV / syslog / NEL), file rotation + gzip compression,
retention, and RFC 6908 bulk (port-block) mode which logs block
assignments instead of per-session records."""
from __future__ import annotations

import gzip
import json
import os
import threading
import time
from typing import Dict, List, Optional

from ..dataplane import abi
from ..dataplane.packets import u32_to_ip

EVENT_NAMES = {
    abi.LOG_SESSION_CREATE: "session_create",
    abi.LOG_SESSION_DELETE: "session_delete",
    abi.LOG_PB_ASSIGN: "port_block_assign",
    abi.LOG_PB_RELEASE: "port_block_release",
    abi.LOG_PORT_EXHAUSTION: "port_exhaustion",
    abi.LOG_HAIRPIN: "hairpin",
    abi.LOG_ALG_TRIGGER: "alg_trigger",
}


def format_event(e: dict, fmt: str) -> str:
    name = EVENT_NAMES.get(e.get("event_type", 0), "unknown")
    fields = {
        "ts": e.get("timestamp", 0),
        "event": name,
        "subscriber_id": e.get("subscriber_id", 0),
        "private_ip": u32_to_ip(e.get("private_ip", 0)),
        "private_port": e.get("private_port", 0),
        "public_ip": u32_to_ip(e.get("public_ip", 0)),
        "public_port": e.get("public_port", 0),
        "dest_ip": u32_to_ip(e.get("dest_ip", 0)),
        "dest_port": e.get("dest_port", 0),
        "protocol": e.get("protocol", 0),
    }
    if fmt == "json":
        return json.dumps(fields)
    if fmt == "csv":
        return ",".join(str(fields[k]) for k in (
            "ts", "event", "subscriber_id", "private_ip", "private_port",
            "public_ip", "public_port", "dest_ip", "dest_port", "protocol"))
    if fmt == "syslog":
        return (f"<134>1 - bng nat - - - {name} sub={fields['subscriber_id']}"
                f" priv={fields['private_ip']}:{fields['private_port']}"
                f" pub={fields['public_ip']}:{fields['public_port']}"
                f" dst={fields['dest_ip']}:{fields['dest_port']}")
    if fmt == "nel":
        # NAT Event Logging (IPFIX-style key=value)
        return " ".join(f"{k}={v}" for k, v in fields.items())
    raise ValueError(f"unknown format {fmt}")


class ComplianceLogger:
    def __init__(self, path: Optional[str] = None, fmt: str = "json",
                 rotate_bytes: int = 10 << 20, compress: bool = True,
                 retention: int = 10, bulk_mode: bool = False):
        """bulk_mode (RFC 6908): only port-block assign/release events are
        logged — per-session create/delete records are suppressed, cutting
        volume by orders of magnitude (ref logging.go bulk mode)."""
        self.path = path
        self.fmt = fmt
        self.rotate_bytes = rotate_bytes
        self.compress = compress
        self.retention = retention
        self.bulk_mode = bulk_mode
        self.records: List[str] = []       # in-memory tail (tests/metrics)
        self.dropped = 0
        self._fh = open(path, "a") if path else None
        self._written = os.path.getsize(path) if path and \
            os.path.exists(path) else 0
        self._lock = threading.Lock()
        self.counters: Dict[str, int] = {}

    def log_event(self, e: dict) -> bool:
        et = e.get("event_type", 0)
        if self.bulk_mode and et in (abi.LOG_SESSION_CREATE,
                                     abi.LOG_SESSION_DELETE):
            return False
        line = format_event(e, self.fmt)
        name = EVENT_NAMES.get(et, "unknown")
        with self._lock:
            self.counters[name] = self.counters.get(name, 0) + 1
            self.records.append(line)
            if len(self.records) > 10000:
                self.records = self.records[-5000:]
            if self._fh is not None:
                self._fh.write(line + "\n")
                self._written += len(line) + 1
                if self._written >= self.rotate_bytes:
                    self._rotate_locked()
        return True

    # ---------------------------------------------------------- rotation
    def _rotate_locked(self):
        self._fh.close()
        stamp = time.strftime("%Y%m%d-%H%M%S")
        rotated = f"{self.path}.{stamp}"
        os.rename(self.path, rotated)
        if self.compress:
            with open(rotated, "rb") as src, \
                    gzip.open(rotated + ".gz", "wb") as dst:
                dst.write(src.read())
            os.unlink(rotated)
        self._fh = open(self.path, "a")
        self._written = 0
        self._apply_retention()

    def rotate(self):
        with self._lock:
            if self._fh is not None:
                self._rotate_locked()

    def _apply_retention(self):
        d = os.path.dirname(self.path) or "."
        base = os.path.basename(self.path)
        rotated = sorted(f for f in os.listdir(d)
                         if f.startswith(base + "."))
        while len(rotated) > self.retention:
            os.unlink(os.path.join(d, rotated.pop(0)))

    def close(self):
        with self._lock:
            if self._fh is not None:
                self._fh.close()
                self._fh = None

    # ----------------------------------------- law-enforcement queries
    def _iter_json_events(self):
        """Every JSON event: rotated files (incl. gzipped) oldest
        first, then the live file, then the in-memory tail for unsynced
        lines.  Only meaningful for fmt=json — compliance queries need
        the structured format (ref logging.go notes the same)."""
        if self.fmt != "json":
            raise ValueError("compliance queries require fmt=json")
        seen = set()
        paths = []
        if self.path:
            d = os.path.dirname(self.path) or "."
            base = os.path.basename(self.path)
            paths = [os.path.join(d, f) for f in sorted(
                f for f in os.listdir(d) if f.startswith(base + "."))]
            paths.append(self.path)
        for p in paths:
            try:
                opener = gzip.open if p.endswith(".gz") else open
                with opener(p, "rt") as f:
                    for line in f:
                        line = line.strip()
                        if line:
                            seen.add(line)
                            yield json.loads(line)
            except (OSError, ValueError):
                continue
        with self._lock:
            tail = list(self.records)
        for line in tail:
            if line not in seen:
                try:
                    yield json.loads(line)
                except ValueError:
                    continue

    def query_by_public_endpoint(self, public_ip: str, public_port: int,
                                 at_time: float = 0.0) -> List[dict]:
        """Who was behind public ip:port (optionally at a moment in
        time)?  The CGNAT legal question (ref QueryByPublicEndpoint):
        returns the matching mapping events — with at_time, only
        mappings whose assign..release interval covers it (a port-block
        event matches any port inside its block)."""
        sessions: Dict[tuple, dict] = {}
        out = []
        for e in self._iter_json_events():
            if e.get("public_ip") != public_ip:
                continue
            ev = e.get("event")
            if ev in ("session_create", "session_delete"):
                if e.get("public_port") != public_port:
                    continue
                key = ("s", e.get("subscriber_id"),
                       e.get("private_ip"), e.get("private_port"),
                       e.get("public_port"), e.get("protocol"))
            elif ev in ("port_block_assign", "port_block_release"):
                # block events encode [start, end] in the private/
                # public port fields (manager.py PB_ASSIGN shape)
                lo = e.get("private_port", 0)
                hi = e.get("public_port", 0) or lo
                if not (lo <= public_port <= (hi if hi >= lo else lo)):
                    continue
                key = ("b", e.get("subscriber_id"), e.get("private_ip"),
                       lo)
            else:
                continue
            if ev.endswith("create") or ev.endswith("assign"):
                sessions[key] = {"start": e, "end": None}
            else:
                if key in sessions:
                    sessions[key]["end"] = e
                else:
                    sessions[key] = {"start": None, "end": e}
        for rec in sessions.values():
            if rec["start"] is None:
                continue
            t0 = rec["start"].get("ts", 0)
            t1 = rec["end"].get("ts", float("inf")) if rec["end"] \
                else float("inf")
            if at_time and not (t0 <= at_time <= t1):
                continue
            out.append({**rec["start"],
                        "released_ts": None if rec["end"] is None
                        else rec["end"].get("ts")})
        return out

    def export_for_compliance(self, start: float = 0.0,
                              end: float = 0.0) -> List[dict]:
        """All events in [start, end] across rotated + live + buffered
        storage (ref ExportForCompliance)."""
        end = end or float("inf")
        return [e for e in self._iter_json_events()
                if start <= e.get("ts", 0) <= end]
