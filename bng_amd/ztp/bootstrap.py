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
    # full assignment shape (ref bootstrap.go DeviceConfig :92-101)
    node_id: str = ""
    site_id: str = ""
    pools: List[dict] = field(default_factory=list)   # {pool_id, cidr, subnets}
    cluster: Dict[str, object] = field(default_factory=dict)

    @classmethod
    def from_dict(cls, d):
        return cls(**{k: v for k, v in d.items()
                      if k in cls.__dataclass_fields__})


@dataclass
class SystemInfo:
    """Hardware identity sent at registration (ref bootstrap.go
    SystemInfo :173-179 / detectSystemInfo :181-217)."""
    serial: str = ""
    mac: str = ""
    model: str = ""
    firmware: str = ""


def _read_sys(path: str) -> str:
    try:
        with open(path) as f:
            return f.read().strip()
    except OSError:
        return ""


def detect_system_info(interface: str = "") -> SystemInfo:
    """DMI serial/model/firmware + management-interface MAC (ref
    detectSystemInfo, detectSerial :340-367, findPrimaryMAC :383-409,
    detectModel :411-428, detectFirmware :430-447)."""
    info = SystemInfo(serial=read_dmi_serial(),
                      model=_read_sys("/sys/class/dmi/id/product_name")
                      or "generic",
                      firmware=_read_sys("/sys/class/dmi/id/bios_version"))
    if interface:
        info.mac = _read_sys(f"/sys/class/net/{interface}/address")
    if not info.mac:
        # first non-loopback, non-virtual interface with a MAC
        try:
            for name in sorted(os.listdir("/sys/class/net")):
                if name == "lo" or name.startswith(("veth", "docker",
                                                    "br-", "virbr")):
                    continue
                mac = _read_sys(f"/sys/class/net/{name}/address")
                if mac and mac != "00:00:00:00:00:00":
                    info.mac = mac
                    break
        except OSError:
            pass
    return info


def parse_vendor_options(data: bytes) -> str:
    """Nexus URL from option-43 vendor TLVs: sub-option 1 carries the
    URL (ref ztp/client.go parseVendorOptions :122-141)."""
    i = 0
    while i + 2 <= len(data):
        typ, ln = data[i], data[i + 1]
        i += 2
        if i + ln > len(data):
            break
        if typ == 1:
            try:
                return data[i:i + ln].decode()
            except UnicodeDecodeError:
                return ""
        i += ln
    return ""


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
    if 224 not in options:
        # option 43 may carry vendor TLVs (sub-option 1 = URL)
        url = parse_vendor_options(v)
        if url.startswith("http"):
            return url
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

    def healthcheck(self) -> bool:
        """GET /health on the Nexus (ref Healthcheck
        bootstrap.go:466-485)."""
        try:
            r = self.session.get(f"{self.nexus_url}/health",
                                 headers=self._headers(), timeout=5)
            return r.status_code == 200
        except Exception:
            return False

    def register_and_wait(self, system_info: Optional[SystemInfo] = None,
                          max_retries: int = 0,
                          initial_backoff: float = 5.0,
                          max_backoff: float = 300.0,
                          deadline: float = 0.0) -> DeviceConfig:
        """Full bootstrap loop (ref bootstrap.go registerAndWait
        :219-300): re-POST registration until the server answers
        status="configured".  Transport errors retry with exponential
        backoff (doubling to max_backoff, reset after any successful
        round-trip); a "pending" answer waits the server-suggested
        retry_after (falling back to the backoff) and counts toward
        max_retries (0 = unlimited)."""
        info = system_info or detect_system_info()
        if not info.serial:
            info.serial = self.serial
        backoff = initial_backoff
        retries = 0
        end = time.time() + deadline if deadline else None
        while True:
            if end is not None and time.time() >= end:
                raise TimeoutError("bootstrap deadline exceeded")
            try:
                r = self.session.post(
                    f"{self.nexus_url}/api/v1/devices/register",
                    json={"serial": info.serial, "mac": info.mac,
                          "model": info.model, "firmware": info.firmware,
                          "capabilities": ["dhcp", "pppoe", "nat44",
                                           "qos", "gpu-dataplane"]},
                    headers=self._headers(), timeout=10)
                r.raise_for_status()
                resp = r.json()
            except Exception:
                time.sleep(backoff)
                backoff = min(backoff * 2, max_backoff)
                continue
            self.state = "registered"
            if resp.get("status") == "configured" or resp.get("approved"):
                self.state = "approved"
                cfg = resp.get("config", {})
                for k in ("node_id", "site_id", "role", "pools",
                          "cluster"):
                    if k in resp and k not in cfg:
                        cfg[k] = resp[k]
                cfg.setdefault("device_id",
                               cfg.get("node_id", "") or info.serial)
                return DeviceConfig.from_dict(cfg)
            retries += 1
            if max_retries and retries >= max_retries:
                raise TimeoutError(
                    f"max retries ({max_retries}) exceeded while pending")
            wait = float(resp.get("retry_after", 0) or 0) or backoff
            time.sleep(wait)
            backoff = initial_backoff     # reset after a good response


def plan_interface_config(iface: str, ip: str, prefix_len: int,
                          gateway: str = "",
                          dns: Optional[List[str]] = None) -> List[str]:
    """The `ip(8)` commands that would apply a ZTP DHCP result to the
    management interface (ref ztp/client.go Configure :144-158 — the
    reference logs the plan rather than mutating the host; so do we,
    and the caller decides whether to execute)."""
    cmds = [f"ip addr add {ip}/{prefix_len} dev {iface}",
            f"ip link set {iface} up"]
    if gateway:
        cmds.append(f"ip route add default via {gateway} dev {iface}")
    for server in dns or []:
        cmds.append(f"resolvectl dns {iface} {server}")
    return cmds


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
                    dev = devices.setdefault(serial, {
                        "serial": serial, "approved": False,
                        "capabilities": d.get("capabilities", []),
                        "config": {}, "retry_after": 0})
                    for k in ("mac", "model", "firmware"):
                        if d.get(k):
                            dev[k] = d[k]
                    # pending/configured contract (ref BootstrapResponse
                    # bootstrap.go:79-90)
                    if dev["approved"]:
                        return self._send(200, {
                            "status": "configured", "serial": serial,
                            "approved": True, "config": dev["config"]})
                    return self._send(200, {
                        "status": "pending", "serial": serial,
                        "retry_after": dev.get("retry_after", 0)})
                self._send(404, {})

            def do_GET(self):
                if self.path == "/health":
                    return self._send(200, {"status": "ok"})
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
