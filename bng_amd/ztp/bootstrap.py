"""Zero-touch provisioning (ref pkg/ztp/bootstrap.go:23-173, client.go):
read the hardware serial, discover Nexus (static / DHCP option 224/43),
register, poll until approved, receive DeviceConfig (+ HA partner and
pool assignment)."""
from __future__ import annotations

import json
import os
import threading
import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional


@dataclass
class DeviceConfig:
    device_id: str
    role: str = "active"              # active | standby
    ha_partner: str = ""
    pool_network: str = ""
    pool_gateway: str = ""
    radius_servers: List[str] = field(default_factory=list)
    radius_secret: str = ""
    extra: Dict[str, str] = field(default_factory=dict)

    @classmethod
    def from_dict(cls, d):
        return cls(**{k: v for k, v in d.items()
                      if k in cls.__dataclass_fields__})


def read_dmi_serial(path: str = "/sys/class/dmi/id/product_serial") -> str:
    """ref bootstrap.go DMI serial read; falls back to hostname."""
    try:
        with open(path) as f:
            s = f.read().strip()
            if s:
                return s
    except OSError:
        pass
    import socket
    return f"host-{socket.gethostname()}"


def discover_nexus_from_dhcp_options(options: Dict[int, bytes]) -> Optional[str]:
    """Nexus URL from DHCP option 224 (private) or 43 (vendor-specific)
    (ref ztp/client.go)."""
    v = options.get(224) or options.get(43)
    if not v:
        return None
    try:
        url = v.decode().strip()
        return url if url.startswith("http") else None
    except UnicodeDecodeError:
        return None


class BootstrapClient:
    """ref bootstrap.go:110 NewBootstrapClient."""

    def __init__(self, nexus_url: str, serial: Optional[str] = None,
                 poll_interval: float = 1.0, session=None,
                 auth=None, device_id_hint: str = ""):
        import requests
        self.nexus_url = nexus_url.rstrip("/")
        self.serial = serial or read_dmi_serial()
        self.poll_interval = poll_interval
        self.session = session or requests.Session()
        self.auth = auth
        self.device_id_hint = device_id_hint
        self.state = "init"    # init -> registered -> approved

    def _headers(self):
        if self.auth is not None:
            return self.auth.headers(self.device_id_hint or self.serial)
        return {}

    def register(self) -> dict:
        """POST /api/v1/devices/register with serial + capabilities."""
        r = self.session.post(
            f"{self.nexus_url}/api/v1/devices/register",
            json={"serial": self.serial,
                  "capabilities": ["dhcp", "pppoe", "nat44", "qos",
                                   "gpu-dataplane"]},
            headers=self._headers(), timeout=10)
        r.raise_for_status()
        self.state = "registered"
        return r.json()

    def poll_until_approved(self, timeout: float = 600.0) -> DeviceConfig:
        """Poll /api/v1/devices/{serial}/config until the operator
        approves the device (ref bootstrap.go poll loop)."""
        deadline = time.time() + timeout
        while time.time() < deadline:
            r = self.session.get(
                f"{self.nexus_url}/api/v1/devices/{self.serial}/config",
                headers=self._headers(), timeout=10)
            if r.status_code == 200:
                d = r.json()
                if d.get("approved"):
                    self.state = "approved"
                    return DeviceConfig.from_dict(d.get("config", {}))
            time.sleep(self.poll_interval)
        raise TimeoutError("device never approved")

    def bootstrap(self, timeout: float = 600.0) -> DeviceConfig:
        self.register()
        return self.poll_until_approved(timeout)


class ZTPServer:
    """In-process Nexus ZTP endpoint (tests / lab): registration queue +
    operator approve()."""

    def __init__(self, host="127.0.0.1", port=0):
        from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
        devices: Dict[str, dict] = {}
        self.devices = devices

        class Handler(BaseHTTPRequestHandler):
            def log_message(self, *a):
                pass

            def _send(self, code, obj):
                body = json.dumps(obj).encode()
                self.send_response(code)
                self.send_header("Content-Length", str(len(body)))
                self.end_headers()
                self.wfile.write(body)

            def do_POST(self):
                if self.path == "/api/v1/devices/register":
                    n = int(self.headers.get("Content-Length", 0))
                    d = json.loads(self.rfile.read(n))
                    serial = d["serial"]
                    devices.setdefault(serial, {
                        "serial": serial, "approved": False,
                        "capabilities": d.get("capabilities", []),
                        "config": {}})
                    return self._send(200, {"status": "registered",
                                            "serial": serial})
                self._send(404, {})

            def do_GET(self):
                if self.path.startswith("/api/v1/devices/") and \
                        self.path.endswith("/config"):
                    serial = self.path.split("/")[4]
                    dev = devices.get(serial)
                    if dev is None:
                        return self._send(404, {})
                    return self._send(200, dev)
                self._send(404, {})

        self.httpd = ThreadingHTTPServer((host, port), Handler)
        self.port = self.httpd.server_address[1]
        threading.Thread(target=self.httpd.serve_forever,
                         daemon=True).start()

    @property
    def url(self):
        return f"http://127.0.0.1:{self.port}"

    def approve(self, serial: str, config: dict):
        self.devices[serial]["approved"] = True
        self.devices[serial]["config"] = config

    def stop(self):
        self.httpd.shutdown()
        self.httpd.server_close()
