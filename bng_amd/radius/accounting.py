"""Accounting manager: session records, interim updates, pending-record
retry queue, disk persistence + orphan recovery on restart
(ref pkg/radius/accounting.go:189-877)."""
from __future__ import annotations

import json
import os
import threading
import time
import uuid
from dataclasses import asdict, dataclass, field
from typing import Dict, List, Optional

from . import packet as rp
from .client import Client, RadiusTimeout


@dataclass
class SessionRecord:
    session_id: str
    username: str
    mac: str = ""
    framed_ip: str = ""
    start_time: float = field(default_factory=time.time)
    input_octets: int = 0
    output_octets: int = 0
    last_interim: float = 0.0
    stopped: bool = False
    terminate_cause: int = 0


class AccountingManager:
    def __init__(self, client: Client, interim_interval: float = 300.0,
                 persist_path: Optional[str] = None,
                 retry_interval: float = 30.0, max_pending: int = 10000):
        self.client = client
        self.interim_interval = interim_interval
        self.persist_path = persist_path
        self.retry_interval = retry_interval
        self.max_pending = max_pending
        self.sessions: Dict[str, SessionRecord] = {}
        self.pending: List[dict] = []   # failed sends awaiting retry
        self._lock = threading.RLock()
        self._stop = threading.Event()
        self._threads: List[threading.Thread] = []
        self._counter_fetcher = None
        if persist_path:
            self._recover_orphans()

    def set_counter_fetcher(self, fn):
        """Live octet-counter pull before each interim/stop record (ref
        accounting.go SetCounterFetcher) — the dataplane integration
        point: fn(session_record) -> (input_octets, output_octets) or
        None to keep the pushed values."""
        self._counter_fetcher = fn

    def _refresh_counters(self, rec: "SessionRecord"):
        if self._counter_fetcher is None:
            return
        try:
            got = self._counter_fetcher(rec)
        except Exception:
            return
        if got:
            rec.input_octets, rec.output_octets = got

    # ---------------------------------------------------------- lifecycle
    def start(self):
        for fn in (self._interim_loop, self._retry_loop):
            t = threading.Thread(target=fn, daemon=True)
            t.start()
            self._threads.append(t)
        return self

    def stop(self):
        self._stop.set()
        self._persist()

    # ------------------------------------------------------------ session
    def start_session(self, username: str, mac: str = "",
                      framed_ip: str = "",
                      session_id: Optional[str] = None) -> str:
        sid = session_id or uuid.uuid4().hex[:16]
        rec = SessionRecord(sid, username, mac, framed_ip)
        with self._lock:
            self.sessions[sid] = rec
        self._persist()
        if not self._try_send(rp.ACCT_START, rec):
            self._queue(rp.ACCT_START, rec)
        return sid

    def update_counters(self, session_id: str, input_octets: int,
                        output_octets: int):
        with self._lock:
            rec = self.sessions.get(session_id)
            if rec:
                rec.input_octets = input_octets
                rec.output_octets = output_octets

    def stop_session(self, session_id: str, terminate_cause: int = 1):
        with self._lock:
            rec = self.sessions.get(session_id)
        if rec is not None:
            self._refresh_counters(rec)
        with self._lock:
            rec = self.sessions.pop(session_id, None)
        if rec is None:
            return
        rec.stopped = True
        rec.terminate_cause = terminate_cause
        self._persist()
        if not self._try_send(rp.ACCT_STOP, rec):
            self._queue(rp.ACCT_STOP, rec)

    # ------------------------------------------------------------- sends
    def _try_send(self, status: int, rec: SessionRecord) -> bool:
        try:
            return self.client.send_accounting(
                status, rec.session_id, rec.username, rec.framed_ip,
                rec.input_octets, rec.output_octets,
                int(time.time() - rec.start_time), rec.terminate_cause,
                rec.mac)
        except Exception:
            return False

    def _queue(self, status: int, rec: SessionRecord):
        """Pending-record queue with bound (ref accounting.go retry queue)."""
        with self._lock:
            if len(self.pending) < self.max_pending:
                self.pending.append({"status": status, "rec": asdict(rec),
                                     "queued_at": time.time()})
        self._persist()

    def _interim_loop(self):
        while not self._stop.wait(min(self.interim_interval, 1.0)):
            now = time.time()
            due = []
            with self._lock:
                for rec in self.sessions.values():
                    anchor = rec.last_interim or rec.start_time
                    if now - anchor >= self.interim_interval:
                        rec.last_interim = now
                        due.append(rec)
            for rec in due:
                self._refresh_counters(rec)
                if not self._try_send(rp.ACCT_INTERIM, rec):
                    self._queue(rp.ACCT_INTERIM, rec)

    def _retry_loop(self):
        while not self._stop.wait(min(self.retry_interval, 0.5)):
            self.flush_pending()

    def flush_pending(self) -> int:
        """Retry queued records; returns number delivered."""
        with self._lock:
            todo, self.pending = self.pending, []
        delivered = 0
        for item in todo:
            rec = SessionRecord(**item["rec"])
            if self._try_send(item["status"], rec):
                delivered += 1
            else:
                with self._lock:
                    self.pending.append(item)
        if delivered:
            self._persist()
        return delivered

    # -------------------------------------------------------- persistence
    def _persist(self):
        if not self.persist_path:
            return
        with self._lock:
            state = {
                "sessions": {k: asdict(v) for k, v in self.sessions.items()},
                "pending": self.pending,
            }
        tmp = self.persist_path + ".tmp"
        with open(tmp, "w") as f:
            json.dump(state, f)
        os.replace(tmp, self.persist_path)

    def _recover_orphans(self):
        """On restart, sessions persisted but never stopped are orphans:
        emit their Stop records (ref accounting.go:729-877)."""
        if not os.path.exists(self.persist_path):
            return
        try:
            with open(self.persist_path) as f:
                state = json.load(f)
        except Exception:
            return
        with self._lock:
            self.pending = list(state.get("pending", []))
            for sid, d in state.get("sessions", {}).items():
                rec = SessionRecord(**d)
                rec.stopped = True
                rec.terminate_cause = 9   # NAS-Error: lost on restart
                self.pending.append({"status": rp.ACCT_STOP,
                                     "rec": asdict(rec),
                                     "queued_at": time.time(),
                                     "orphan": True})
