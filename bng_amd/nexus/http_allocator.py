"""HTTP allocator — REST client to a central Nexus allocation service
(ref pkg/nexus/http_allocator.go:80-541).

API surface kept compatible:
  POST /api/v1/allocations                -> allocate IPv4/IPv6
  GET  /api/v1/allocations?subscriber_id= -> lookup (404 => NoAllocation,
       the walled-garden signal the DHCP slow path keys on)
  DELETE /api/v1/allocations              -> release
  GET/POST /api/v1/pools[/{id}]           -> pool info / create
  GET  /health                            -> health check
"""
from __future__ import annotations

import json
import threading
import time
from typing import Dict, List, Optional, Tuple


class NoAllocationError(Exception):
    """Subscriber has no allocation in Nexus — the DHCP slow path treats
    this as 'quarantine to walled garden' (ref http_allocator.go:225-226)."""


class HTTPAllocatorError(Exception):
    pass


class HTTPAllocator:
    def __init__(self, base_url: str, timeout: float = 5.0,
                 auth_headers: Optional[Dict[str, str]] = None,
                 session=None, client_cert: str = "",
                 client_key: str = "", ca_cert: str = "",
                 insecure: bool = False):
        import requests
        self.base_url = base_url.rstrip("/")
        self.timeout = timeout
        self.session = session or requests.Session()
        if auth_headers:
            self.session.headers.update(auth_headers)
        # device->Nexus mTLS (ref deviceauth mtls.go + transport.go):
        # client cert/key presented on every request; server verified
        # against ca_cert unless explicitly insecure
        if client_cert and client_key:
            self.session.cert = (client_cert, client_key)
        if insecure:
            self.session.verify = False
        elif ca_cert:
            self.session.verify = ca_cert

    def _url(self, path: str) -> str:
        return self.base_url + path

    # -------------------------------------------------------- allocations
    def allocate_ipv4(self, pool_id: str, subscriber_id: str) -> str:
        """ref http_allocator.go:95-178 AllocateIPv4."""
        r = self.session.post(self._url("/api/v1/allocations"),
                              json={"pool_id": pool_id,
                                    "subscriber_id": subscriber_id},
                              timeout=self.timeout)
        if r.status_code >= 400:
            raise HTTPAllocatorError(
                f"allocate failed: {r.status_code} {r.text[:200]}")
        return r.json()["ip"]

    def lookup_ipv4(self, subscriber_id: str) -> Tuple[str, str]:
        """Pure-read lookup for DHCP time (ref :181-225 LookupIPv4).
        Returns (ip, pool_id); raises NoAllocationError on 404."""
        r = self.session.get(self._url("/api/v1/allocations"),
                             params={"subscriber_id": subscriber_id},
                             timeout=self.timeout)
        if r.status_code == 404:
            raise NoAllocationError(subscriber_id)
        if r.status_code >= 400:
            raise HTTPAllocatorError(f"lookup failed: {r.status_code}")
        d = r.json()
        return d["ip"], d.get("pool_id", "")

    def allocate_ipv6(self, pool_id: str, subscriber_id: str) -> Tuple[str, int]:
        r = self.session.post(self._url("/api/v1/allocations"),
                              json={"pool_id": pool_id,
                                    "subscriber_id": subscriber_id,
                                    "family": 6},
                              timeout=self.timeout)
        if r.status_code >= 400:
            raise HTTPAllocatorError(f"allocate6 failed: {r.status_code}")
        d = r.json()
        return d["ip"], d.get("prefix", 64)

    def release(self, pool_id: str, subscriber_id: str) -> None:
        r = self.session.delete(self._url("/api/v1/allocations"),
                                json={"pool_id": pool_id,
                                      "subscriber_id": subscriber_id},
                                timeout=self.timeout)
        if r.status_code >= 400:
            raise HTTPAllocatorError(f"release failed: {r.status_code}")

    # -------------------------------------------------------------- pools
    def get_pool_info(self, pool_id: str) -> dict:
        r = self.session.get(self._url(f"/api/v1/pools/{pool_id}"),
                             timeout=self.timeout)
        if r.status_code >= 400:
            raise HTTPAllocatorError(f"pool info failed: {r.status_code}")
        return r.json()

    def create_pool(self, pool_id: str, cidr: str, gateway: str = "",
                    dns: Optional[List[str]] = None) -> None:
        r = self.session.post(self._url("/api/v1/pools"),
                              json={"id": pool_id, "cidr": cidr,
                                    "gateway": gateway, "dns": dns or []},
                              timeout=self.timeout)
        if r.status_code >= 400:
            raise HTTPAllocatorError(f"create pool failed: {r.status_code}")

    def health_check(self) -> bool:
        try:
            r = self.session.get(self._url("/health"), timeout=self.timeout)
            return r.status_code == 200
        except Exception:
            return False


class NexusAllocatorServer:
    """Minimal in-process Nexus allocation REST service implementing the
    endpoints HTTPAllocator speaks — used by tests (the reference mocks
    this with httpmock) and as a standalone central allocator for small
    deployments.  Allocation is the deterministic hashring (client.py)."""

    def __init__(self, host: str = "127.0.0.1", port: int = 0):
        from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
        from .client import Client as _C

        pools: Dict[str, dict] = {}
        allocations: Dict[str, dict] = {}
        lock = threading.Lock()
        self.pools, self.allocations, self._lock = pools, allocations, lock

        outer = self

        class Handler(BaseHTTPRequestHandler):
            def log_message(self, *a):
                pass

            def _send(self, code, obj=None):
                body = json.dumps(obj or {}).encode()
                self.send_response(code)
                self.send_header("Content-Type", "application/json")
                self.send_header("Content-Length", str(len(body)))
                self.end_headers()
                self.wfile.write(body)

            def _body(self):
                n = int(self.headers.get("Content-Length", 0))
                return json.loads(self.rfile.read(n) or b"{}")

            def do_GET(self):
                from urllib.parse import parse_qs, urlparse
                u = urlparse(self.path)
                if u.path == "/health":
                    return self._send(200, {"status": "ok"})
                if u.path == "/api/v1/allocations":
                    sid = parse_qs(u.query).get("subscriber_id", [""])[0]
                    with lock:
                        a = allocations.get(sid)
                    if a is None:
                        return self._send(404, {"error": "no allocation"})
                    return self._send(200, a)
                if u.path.startswith("/api/v1/pools/"):
                    pid = u.path.rsplit("/", 1)[1]
                    with lock:
                        p = pools.get(pid)
                    if p is None:
                        return self._send(404, {"error": "no pool"})
                    return self._send(200, p)
                self._send(404, {"error": "not found"})

            def do_POST(self):
                d = self._body()
                if self.path == "/api/v1/pools":
                    with lock:
                        pools[d["id"]] = {"id": d["id"], "cidr": d["cidr"],
                                          "prefix": int(d["cidr"].split("/")[1]),
                                          "gateway": d.get("gateway", ""),
                                          "dns": d.get("dns", [])}
                    return self._send(201, pools[d["id"]])
                if self.path == "/api/v1/allocations":
                    pid, sid = d["pool_id"], d["subscriber_id"]
                    with lock:
                        if sid in allocations:
                            return self._send(200, allocations[sid])
                        p = pools.get(pid)
                    if p is None:
                        return self._send(404, {"error": "no pool"})
                    ip = _C.allocate_from_pool(p["cidr"], sid)
                    a = {"pool_id": pid, "subscriber_id": sid, "ip": ip,
                         "timestamp": time.time()}
                    with lock:
                        allocations[sid] = a
                    return self._send(200, a)
                self._send(404, {"error": "not found"})

            def do_DELETE(self):
                d = self._body()
                if self.path == "/api/v1/allocations":
                    with lock:
                        allocations.pop(d.get("subscriber_id", ""), None)
                    return self._send(200, {})
                self._send(404, {"error": "not found"})

        self.httpd = ThreadingHTTPServer((host, port), Handler)
        self.port = self.httpd.server_address[1]
        self._thread = threading.Thread(target=self.httpd.serve_forever,
                                        daemon=True)

    @property
    def url(self) -> str:
        return f"http://127.0.0.1:{self.port}"

    def start(self):
        self._thread.start()
        return self

    def stop(self):
        self.httpd.shutdown()
        self.httpd.server_close()
