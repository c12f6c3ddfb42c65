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
