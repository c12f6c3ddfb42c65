"""CLSet — CRDT-backed Store with peer sync and persistence.

Python re-design of the reference's CLSetStore (pkg/nexus/clset.go:45-427
+ crdt_backend.go:1-320): an eventually-consistent replicated K/V store
that keeps serving reads/writes through partitions and converges on
merge.

CRDT: last-writer-wins register map with tombstones — each entry carries
(lamport, node_id); merge order is (lamport, node_id) lexicographic, so
concurrent writes converge identically on every replica.  The reference
uses a libp2p gossip CLSet over a badger datastore; here:

  * persistence = snapshot file + append-only WAL in `data_dir`
    (the badger analog: every local or merged write is durable before
    it is acknowledged; the WAL compacts into the snapshot);
  * sync = pull-based anti-entropy over HTTP with per-peer reconnect
    backoff (the gossip analog at control-plane rates);
  * discovery = snapshots advertise known peer URLs, so reachable
    replicas learn the full mesh transitively (gossipsub analog).
"""
from __future__ import annotations

import json
import os
import threading
import time
from typing import Dict, List, Optional, Tuple

from .store import Store, WatchEvent


class CLSetStore(Store):
    WAL_COMPACT_EVERY = 1024     # records before auto-compaction

    def __init__(self, node_id: str, sync_interval: float = 1.0,
                 data_dir: Optional[str] = None,
                 advertise_url: str = "",
                 backoff_base: float = 1.0, backoff_max: float = 30.0):
        self.node_id = node_id
        self.sync_interval = sync_interval
        self.advertise_url = advertise_url.rstrip("/")
        self.backoff_base = backoff_base
        self.backoff_max = backoff_max
        # key -> [value|None, lamport, node_id, deleted]
        self._entries: Dict[str, Tuple[Optional[bytes], int, str, bool]] = {}
        self._lamport = 0
        self._lock = threading.RLock()
        self._watchers: List[tuple] = []
        self._peers: List["CLSetStore"] = []
        self._peer_urls: List[str] = []
        # url -> {"fails": n, "next_try": ts, "last_ok": ts}
        self._peer_state: Dict[str, dict] = {}
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self.stats = {"syncs_ok": 0, "syncs_failed": 0, "adopted": 0,
                      "peers_discovered": 0, "compactions": 0}
        self._membership_cbs: List = []
        self.peer_ttl = 300.0
        self._data_dir = data_dir
        self._wal = None
        self._wal_records = 0
        if data_dir:
            os.makedirs(data_dir, exist_ok=True)
            self._load()
            self._wal = open(self._wal_path, "a", encoding="utf-8")

    # ------------------------------------------------------ persistence
    @property
    def _snap_path(self):
        return os.path.join(self._data_dir, "clset_snapshot.json")

    @property
    def _wal_path(self):
        return os.path.join(self._data_dir, "clset_wal.jsonl")

    def _load(self):
        import base64
        if os.path.exists(self._snap_path):
            with open(self._snap_path, encoding="utf-8") as f:
                snap = json.load(f)
            self._lamport = int(snap.get("lamport", 0))
            for k, (v64, lam, nid, dead) in snap.get("entries",
                                                     {}).items():
                v = None if v64 is None else base64.b64decode(v64)
                self._entries[k] = (v, lam, nid, dead)
        if os.path.exists(self._wal_path):
            with open(self._wal_path, encoding="utf-8") as f:
                for line in f:
                    line = line.strip()
                    if not line:
                        continue
                    try:
                        rec = json.loads(line)
                    except ValueError:
                        break        # torn tail record: stop replay
                    k, v64, lam, nid, dead = rec
                    v = None if v64 is None else base64.b64decode(v64)
                    cur = self._entries.get(k)
                    if cur is None or (cur[1], cur[2]) < (lam, nid):
                        self._entries[k] = (v, lam, nid, dead)
                    self._lamport = max(self._lamport, lam)
                    self._wal_records += 1

    def _wal_append(self, key, v, lam, nid, dead):
        if self._wal is None:
            return
        import base64
        v64 = None if v is None else base64.b64encode(v).decode()
        self._wal.write(json.dumps([key, v64, lam, nid, dead]) + "\n")
        self._wal.flush()
        self._wal_records += 1
        if self._wal_records >= self.WAL_COMPACT_EVERY:
            self.compact()

    def compact(self):
        """Fold the WAL into the snapshot (badger-compaction analog)."""
        if self._data_dir is None:
            return
        tmp = self._snap_path + ".tmp"
        with open(tmp, "w", encoding="utf-8") as f:
            json.dump(self.snapshot(), f)
        os.replace(tmp, self._snap_path)
        if self._wal is not None:
            self._wal.close()
        self._wal = open(self._wal_path, "w", encoding="utf-8")
        self._wal_records = 0
        self.stats["compactions"] += 1

    # ------------------------------------------------------------- Store
    def get(self, key):
        with self._lock:
            e = self._entries.get(key)
            return None if e is None or e[3] else e[0]

    def put(self, key, value):
        if isinstance(value, str):
            value = value.encode()
        with self._lock:
            self._lamport += 1
            self._entries[key] = (bytes(value), self._lamport, self.node_id,
                                  False)
            self._wal_append(key, bytes(value), self._lamport,
                             self.node_id, False)
            watchers = [w for w in self._watchers if key.startswith(w[0])]
        for _, cb in watchers:
            cb(WatchEvent("put", key, bytes(value)))

    def delete(self, key):
        with self._lock:
            self._lamport += 1
            self._entries[key] = (None, self._lamport, self.node_id, True)
            self._wal_append(key, None, self._lamport, self.node_id, True)
            watchers = [w for w in self._watchers if key.startswith(w[0])]
        for _, cb in watchers:
            cb(WatchEvent("delete", key))

    def list(self, prefix):
        with self._lock:
            return {k: e[0] for k, e in self._entries.items()
                    if k.startswith(prefix) and not e[3]}

    def watch(self, prefix, callback):
        ent = (prefix, callback)
        with self._lock:
            self._watchers.append(ent)

        def cancel():
            with self._lock:
                if ent in self._watchers:
                    self._watchers.remove(ent)
        return cancel

    # ------------------------------------------------------------- CRDT
    def snapshot(self) -> dict:
        """Serializable replica state for anti-entropy."""
        import base64
        with self._lock:
            known = [u for u in ([self.advertise_url] + self._peer_urls)
                     if u]
            return {
                "node_id": self.node_id,
                "lamport": self._lamport,
                "peers": known,        # gossip-style peer discovery
                "entries": {
                    k: [None if v is None
                        else base64.b64encode(v).decode(), lam, nid, dead]
                    for k, (v, lam, nid, dead) in self._entries.items()
                },
            }

    def merge(self, snap: dict) -> int:
        """LWW merge; returns number of entries adopted."""
        import base64
        adopted = 0
        events = []
        with self._lock:
            self._lamport = max(self._lamport, int(snap.get("lamport", 0)))
            for k, (v64, lam, nid, dead) in snap.get("entries", {}).items():
                cur = self._entries.get(k)
                if cur is not None and (cur[1], cur[2]) >= (lam, nid):
                    continue
                v = None if v64 is None else base64.b64decode(v64)
                self._entries[k] = (v, lam, nid, dead)
                self._wal_append(k, v, lam, nid, dead)   # merged = durable
                adopted += 1
                for w in self._watchers:
                    if k.startswith(w[0]):
                        events.append((w[1], WatchEvent(
                            "delete" if dead else "put", k, v)))
            # transitive peer discovery: adopt unknown advertised URLs
            for u in snap.get("peers", []):
                u = u.rstrip("/")
                if u and u != self.advertise_url and \
                        u not in self._peer_urls:
                    self._peer_urls.append(u)
                    self.stats["peers_discovered"] += 1
        for cb, ev in events:
            cb(ev)
        self.stats["adopted"] += adopted
        return adopted

    # -------------------------------------------------------------- sync
    def add_peer(self, peer: "CLSetStore"):
        self._peers.append(peer)

    def add_peer_url(self, url: str):
        self._peer_urls.append(url.rstrip("/"))
        self._fire_membership()

    def on_membership(self, cb):
        """cb(members: dict url -> {last_ok, fails, alive}) whenever
        the peer set or a peer's reachability changes (ref clset.go
        WithMembershipHook :57-60 + peer TTL :62-65)."""
        self._membership_cbs.append(cb)
        self._fire_membership()

    def members(self) -> dict:
        now = time.monotonic()
        out = {}
        for u in self._peer_urls:
            st = self._peer_state.get(u, {})
            last_ok = st.get("last_ok", 0.0)
            out[u] = {"last_ok": last_ok, "fails": st.get("fails", 0),
                      "alive": st.get("fails", 0) == 0 or
                      (last_ok and now - last_ok < self.peer_ttl)}
        return out

    def _fire_membership(self):
        for cb in getattr(self, "_membership_cbs", []):
            try:
                cb(self.members())
            except Exception:
                pass

    def expire_peers(self, now: Optional[float] = None) -> int:
        """Drop peers silent past peer_ttl (ref WithPeerTTL); they
        re-enter via gossip if they come back."""
        now = now if now is not None else time.monotonic()
        dead = []
        for u in list(self._peer_urls):
            st = self._peer_state.get(u)
            if st is None:
                continue
            last = st.get("last_ok", 0.0)
            ref = last or st.get("first_seen", 0.0)
            if st.get("fails", 0) > 0 and ref and \
                    now - ref > self.peer_ttl:
                self._peer_urls.remove(u)
                self._peer_state.pop(u, None)
                dead.append(u)
        if dead:
            self._fire_membership()
        return len(dead)

    def sync_once(self, now: Optional[float] = None) -> int:
        """One anti-entropy round against every reachable peer, with
        per-peer exponential reconnect backoff (the gossip transport's
        reconnect semantics, crdt_backend.go peer management)."""
        now = now if now is not None else time.monotonic()
        adopted = 0
        for p in self._peers:
            adopted += self.merge(p.snapshot())
            p.merge(self.snapshot())
        if self._peer_urls:
            import requests
            for url in list(self._peer_urls):
                st = self._peer_state.setdefault(
                    url, {"fails": 0, "next_try": 0.0, "last_ok": 0.0,
                          "first_seen": now})
                if now < st["next_try"]:
                    continue          # still backing off
                try:
                    r = requests.post(f"{url}/clset/sync",
                                      json=self.snapshot(), timeout=5)
                    r.raise_for_status()
                    adopted += self.merge(r.json())
                    st["fails"] = 0
                    st["next_try"] = 0.0
                    st["last_ok"] = now
                    self.stats["syncs_ok"] += 1
                    if st.get("was_down"):
                        st["was_down"] = False
                        self._fire_membership()
                except Exception:
                    st["fails"] += 1
                    if st["fails"] == 1:
                        st["was_down"] = True
                        self._fire_membership()
                    st["next_try"] = now + min(
                        self.backoff_base * (2 ** (st["fails"] - 1)),
                        self.backoff_max)
                    self.stats["syncs_failed"] += 1
        return adopted

    def peer_status(self) -> Dict[str, dict]:
        return {u: dict(s) for u, s in self._peer_state.items()}

    def start(self):
        self._thread = threading.Thread(target=self._loop, daemon=True)
        self._thread.start()
        return self

    def close(self):
        self._stop.set()
        if self._thread:
            self._thread.join(timeout=2)
        if self._wal is not None:
            self.compact()            # durable, minimal restart replay
            self._wal.close()
            self._wal = None

    def _loop(self):
        while not self._stop.wait(self.sync_interval):
            try:
                self.sync_once()
            except Exception:
                pass


class CLSetHTTPServer:
    """HTTP sync endpoint for a CLSetStore: POST /clset/sync with a
    snapshot merges it and returns ours (bidirectional anti-entropy)."""

    def __init__(self, store: CLSetStore, host="127.0.0.1", port=0):
        from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
        st = store

        class Handler(BaseHTTPRequestHandler):
            def log_message(self, *a):
                pass

            def do_POST(self):
                if self.path != "/clset/sync":
                    self.send_response(404)
                    self.end_headers()
                    return
                n = int(self.headers.get("Content-Length", 0))
                snap = json.loads(self.rfile.read(n) or b"{}")
                st.merge(snap)
                body = json.dumps(st.snapshot()).encode()
                self.send_response(200)
                self.send_header("Content-Type", "application/json")
                self.send_header("Content-Length", str(len(body)))
                self.end_headers()
                self.wfile.write(body)

        self.httpd = ThreadingHTTPServer((host, port), Handler)
        self.port = self.httpd.server_address[1]
        self._thread = threading.Thread(target=self.httpd.serve_forever,
                                        daemon=True)

    @property
    def url(self):
        return f"http://127.0.0.1:{self.port}"

    def start(self):
        self._thread.start()
        return self

    def stop(self):
        self.httpd.shutdown()
        self.httpd.server_close()
