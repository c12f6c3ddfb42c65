"""RADIUS degradation modes during partitions
(ref pkg/resilience/radius_handler.go:52-509): reject / cached / allow —
cached subscriber profiles answer auth while RADIUS is unreachable, and
accounting records are buffered for replay on recovery."""
from __future__ import annotations

import threading
import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional

from ..radius.client import AuthResult, RadiusTimeout

MODE_REJECT = "reject"
MODE_CACHED = "cached"
MODE_ALLOW = "allow"


@dataclass
class CachedProfile:
    username: str
    result: AuthResult
    cached_at: float = field(default_factory=time.time)


class ResilientRadius:
    """Wraps radius.Client with partition behavior."""

    def __init__(self, client, mode: str = MODE_CACHED,
                 cache_ttl: float = 86400.0, max_buffer: int = 100000):
        self.client = client
        self.mode = mode
        self.cache_ttl = cache_ttl
        self.max_buffer = max_buffer
        self.cache: Dict[str, CachedProfile] = {}
        self.buffered_acct: List[tuple] = []
        self._lock = threading.RLock()
        self.stats = {"cache_hits": 0, "cache_answers": 0,
                      "allow_answers": 0, "rejects": 0,
                      "acct_buffered": 0, "acct_replayed": 0,
                      "reauths_ok": 0, "reauths_failed": 0}
        # usernames admitted on degraded answers -> owed a real
        # authentication once the partition heals (ref
        # radius_handler.go GetDegradedSessions/ProcessReauths)
        self._degraded: Dict[str, dict] = {}
        self._reauth_queue: List[str] = []

    def authenticate(self, username: str, password: str, **kw) -> AuthResult:
        try:
            res = self.client.authenticate(username, password, **kw)
            if res.success:
                with self._lock:
                    self.cache[username] = CachedProfile(username, res)
            return res
        except RadiusTimeout:
            return self._degraded_auth(username)
        except Exception:
            return self._degraded_auth(username)

    def _degraded_auth(self, username: str) -> AuthResult:
        if self.mode == MODE_REJECT:
            self.stats["rejects"] += 1
            return AuthResult(False, reply_message="radius unreachable")
        if self.mode == MODE_CACHED:
            with self._lock:
                prof = self.cache.get(username)
            if prof is not None and \
                    time.time() - prof.cached_at <= self.cache_ttl:
                self.stats["cache_answers"] += 1
                with self._lock:
                    self._degraded[username] = {"mode": "cached",
                                                "at": time.time()}
                return prof.result
            self.stats["rejects"] += 1
            return AuthResult(False, reply_message="no cached profile")
        # MODE_ALLOW: admit with defaults (degraded service)
        self.stats["allow_answers"] += 1
        with self._lock:
            self._degraded[username] = {"mode": "allow",
                                        "at": time.time()}
        return AuthResult(True, policy_name="")

    # ------------------------------------------------------- accounting
    def send_accounting(self, *args, **kw) -> bool:
        try:
            if self.client.send_accounting(*args, **kw):
                return True
        except Exception:
            pass
        with self._lock:
            if len(self.buffered_acct) < self.max_buffer:
                self.buffered_acct.append((args, kw))
                self.stats["acct_buffered"] += 1
        return False

    def replay_buffered(self) -> int:
        """On partition recovery, flush buffered accounting
        (ref AccountingBuffering scenario)."""
        with self._lock:
            todo, self.buffered_acct = self.buffered_acct, []
        done = 0
        for args, kw in todo:
            try:
                if self.client.send_accounting(*args, **kw):
                    done += 1
                    continue
            except Exception:
                pass
            with self._lock:
                self.buffered_acct.append((args, kw))
        self.stats["acct_replayed"] += done
        return done


    # -------------------------------------------- recovery re-auth
    def degraded_sessions(self) -> List[str]:
        """Usernames admitted on cached/allow answers during the
        partition (ref GetDegradedSessions)."""
        with self._lock:
            return list(self._degraded)

    def queue_reauth(self, username: str):
        with self._lock:
            if username not in self._reauth_queue:
                self._reauth_queue.append(username)

    def process_reauths(self, rate_limit: int = 0,
                        credentials=None) -> tuple:
        """Re-authenticate queued degraded sessions against the real
        RADIUS, at most rate_limit per call (0 = all).  credentials:
        optional username -> (password, kwargs) provider; without it
        the cached profile is revalidated by a bare authenticate.
        Returns (completed, failed); failures stay degraded (ref
        ProcessReauths rate-limited loop)."""
        with self._lock:
            batch = self._reauth_queue[:rate_limit or None]
            self._reauth_queue = self._reauth_queue[len(batch):]
        done = failed = 0
        for user in batch:
            try:
                if credentials is not None:
                    pw, kw = credentials(user)
                    res = self.client.authenticate(user, pw, **kw)
                else:
                    res = self.client.authenticate(user, "")
                ok = bool(res.success)
            except Exception:
                ok = False
            if ok:
                done += 1
                with self._lock:
                    self._degraded.pop(user, None)
            else:
                failed += 1
        self.stats["reauths_ok"] += done
        self.stats["reauths_failed"] += failed
        return done, failed
