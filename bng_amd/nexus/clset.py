"""CLSet — CRDT-backed Store with peer sync.

Python re-design of the reference's CLSetStore (pkg/nexus/clset.go:45-427
+ crdt_backend.go): an eventually-consistent replicated K/V store that
keeps serving reads/writes through partitions and converges on merge.

CRDT: last-writer-wins register map with tombstones — each entry carries
(lamport, node_id); merge order is (lamport, node_id) lexicographic, so
concurrent writes converge identically on every replica.  The reference
uses a libp2p gossip CLSet; here sync is pull-based anti-entropy over
HTTP (or direct peer references in tests), which fits the BNG's
control-plane rates.
"""
from __future__ import annotations

import json
import threading
import time
from typing import Dict, List, Optional, Tuple

from .store import Store, WatchEvent


class CLSetStore(Store):
    def __init__(self, node_id: str, sync_interval: float = 1.0):
        self.node_id = node_id
        self.sync_interval = sync_interval
        # key -> [value_b64|None, lamport, node_id, deleted]
        self._entries: Dict[str, Tuple[Optional[bytes], int, str, bool]] = {}
        self._lamport = 0
        self._lock = threading.RLock()
        self._watchers: List[tuple] = []
        self._peers: List["CLSetStore"] = []
        self._peer_urls: List[str] = []
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None

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
            watchers = [w for w in self._watchers if key.startswith(w[0])]
        for _, cb in watchers:
            cb(WatchEvent("put", key, bytes(value)))

    def delete(self, key):
        with self._lock:
            self._lamport += 1
            self._entries[key] = (None, self._lamport, self.node_id, True)
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
            return {
                "node_id": self.node_id,
                "lamport": self._lamport,
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
                adopted += 1
                for w in self._watchers:
                    if k.startswith(w[0]):
                        events.append((w[1], WatchEvent(
                            "delete" if dead else "put", k, v)))
        for cb, ev in events:
            cb(ev)
        return adopted

    # -------------------------------------------------------------- sync
    def add_peer(self, peer: "CLSetStore"):
        self._peers.append(peer)

    def add_peer_url(self, url: str):
        self._peer_urls.append(url.rstrip("/"))

    def sync_once(self) -> int:
        """One anti-entropy round against every reachable peer."""
        adopted = 0
        for p in self._peers:
            adopted += self.merge(p.snapshot())
            p.merge(self.snapshot())
        if self._peer_urls:
            import requests
            for url in self._peer_urls:
                try:
                    r = requests.post(f"{url}/clset/sync",
                                      json=self.snapshot(), timeout=5)
                    if r.status_code == 200:
                        adopted += self.merge(r.json())
                except Exception:
                    continue
        return adopted

    def start(self):
        self._thread = threading.Thread(target=self._loop, daemon=True)
        self._thread.start()
        return self

    def close(self):
        self._stop.set()
        if self._thread:
            self._thread.join(timeout=2)

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
