"""Peer pool — distributed IP allocation WITHOUT a central Nexus
(ref pkg/pool/peer.go): rendezvous/HRW hashing decides which peer owns a
subscriber (peer.go:721-760); non-owners forward allocate/release over
HTTP (:316-440); a health prober (10s interval / 3 failures) drives
owner fallback to the next-ranked healthy peer (:242-268).
"""
from __future__ import annotations

import json
import threading
import time
from typing import Dict, List, Optional

from ..parallel.hashring import RendezvousRing
from ..allocator.bitmap import BitmapAllocator, PoolExhaustedError


class PeerPoolError(Exception):
    pass


class PeerPool:
    def __init__(self, node_id: str, peers: Dict[str, str], cidr: str,
                 listen_port: int = 0, health_interval: float = 10.0,
                 health_threshold: int = 3, reserve_head: int = 2,
                 reserve_tail: int = 0):
        """peers: node_id -> base_url (this node excluded or included)."""
        self.node_id = node_id
        self.peer_urls = {k: v for k, v in peers.items() if k != node_id}
        self.ring = RendezvousRing(sorted(set(peers) | {node_id}))
        self.local = BitmapAllocator(cidr, 32, reserve_head, reserve_tail)
        self._lock = threading.RLock()
        self._fail_counts: Dict[str, int] = {}
        self.health_interval = health_interval
        self.health_threshold = health_threshold
        self._stop = threading.Event()
        self._prober: Optional[threading.Thread] = None
        self._httpd = None
        self._listen_port = listen_port

    # ------------------------------------------------------------- HTTP
    def start(self):
        from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
        pool = self

        class Handler(BaseHTTPRequestHandler):
            def log_message(self, *a):
                pass

            def _send(self, code, obj):
                body = json.dumps(obj).encode()
                self.send_response(code)
                self.send_header("Content-Type", "application/json")
                self.send_header("Content-Length", str(len(body)))
                self.end_headers()
                self.wfile.write(body)

            def do_GET(self):
                if self.path == "/pool/health":
                    return self._send(200, {"node": pool.node_id,
                                            "status": "ok"})
                self._send(404, {})

            def do_POST(self):
                n = int(self.headers.get("Content-Length", 0))
                d = json.loads(self.rfile.read(n) or b"{}")
                sid = d.get("subscriber_id", "")
                try:
                    if self.path == "/pool/allocate":
                        ip = pool.allocate_local(sid)
                        return self._send(200, {"ip": ip,
                                                "owner": pool.node_id})
                    if self.path == "/pool/release":
                        pool.release_local(sid)
                        return self._send(200, {})
                except PoolExhaustedError as e:
                    return self._send(409, {"error": str(e)})
                self._send(404, {})

        self._httpd = ThreadingHTTPServer(("127.0.0.1", self._listen_port),
                                          Handler)
        self._listen_port = self._httpd.server_address[1]
        threading.Thread(target=self._httpd.serve_forever,
                         daemon=True).start()
        self._prober = threading.Thread(target=self._probe_loop, daemon=True)
        self._prober.start()
        return self

    @property
    def url(self) -> str:
        return f"http://127.0.0.1:{self._listen_port}"

    def stop(self):
        self._stop.set()
        if self._httpd:
            self._httpd.shutdown()
            self._httpd.server_close()

    # ------------------------------------------------------------ health
    def _probe_loop(self):
        while not self._stop.wait(self.health_interval):
            self.probe_once()

    def probe_once(self):
        """Probe every peer's /pool/health (ref peer.go health prober)."""
        import requests
        for node, base in self.peer_urls.items():
            ok = False
            try:
                r = requests.get(f"{base}/pool/health", timeout=2)
                ok = r.status_code == 200
            except Exception:
                ok = False
            with self._lock:
                if ok:
                    self._fail_counts[node] = 0
                    self.ring.set_healthy(node, True)
                else:
                    self._fail_counts[node] = \
                        self._fail_counts.get(node, 0) + 1
                    if self._fail_counts[node] >= self.health_threshold:
                        self.ring.set_healthy(node, False)

    # --------------------------------------------------------------- API
    def owner_of(self, subscriber_id: str) -> str:
        return self.ring.owner(subscriber_id)

    def allocate(self, subscriber_id: str) -> str:
        """Owner-routed allocation with health fallback
        (ref peer.go:230-330)."""
        owner = self.owner_of(subscriber_id)
        if owner == self.node_id:
            return self.allocate_local(subscriber_id)
        base = self.peer_urls.get(owner)
        if base is None:
            return self.allocate_local(subscriber_id)
        import requests
        try:
            r = requests.post(f"{base}/pool/allocate",
                              json={"subscriber_id": subscriber_id},
                              timeout=5)
            if r.status_code == 200:
                return r.json()["ip"]
            raise PeerPoolError(f"owner {owner} returned {r.status_code}")
        except PeerPoolError:
            raise
        except Exception:
            # owner unreachable: mark and fall back locally (ref :242-268)
            with self._lock:
                self._fail_counts[owner] = \
                    self._fail_counts.get(owner, 0) + 1
                if self._fail_counts[owner] >= self.health_threshold:
                    self.ring.set_healthy(owner, False)
            return self.allocate_local(subscriber_id)

    def release(self, subscriber_id: str) -> None:
        owner = self.owner_of(subscriber_id)
        if owner == self.node_id or owner not in self.peer_urls:
            return self.release_local(subscriber_id)
        import requests
        try:
            requests.post(f"{self.peer_urls[owner]}/pool/release",
                          json={"subscriber_id": subscriber_id}, timeout=5)
        except Exception:
            self.release_local(subscriber_id)

    def allocate_local(self, subscriber_id: str) -> str:
        with self._lock:
            return self.local.allocate(subscriber_id).split("/")[0]

    def release_local(self, subscriber_id: str) -> None:
        with self._lock:
            self.local.release(subscriber_id)

    def lookup(self, subscriber_id: str) -> Optional[str]:
        with self._lock:
            p = self.local.lookup(subscriber_id)
        return p.split("/")[0] if p else None
