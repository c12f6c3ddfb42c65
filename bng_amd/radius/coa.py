"""CoA / Disconnect-Request server (RFC 5176) + processor wiring
(ref pkg/radius/coa.go:119-530 and coa_handler.go:46-70).

The CoAProcessor connects incoming Disconnect/CoA requests to session
lookup, session termination, and policy updates — including the GPU QoS
table updater hook (the reference's eBPF QoS updater analog).
"""
from __future__ import annotations

import socket
import threading
from dataclasses import dataclass
from typing import Callable, Dict, List, Optional

from . import packet as rp


@dataclass
class CoARequest:
    code: int
    username: str = ""
    session_id: str = ""
    framed_ip: str = ""
    mac: str = ""
    policy_name: str = ""


class CoAServer:
    """Listens for CoA-Request (43) / Disconnect-Request (40) from the
    RADIUS server side; validates the Request Authenticator; dispatches
    to a handler returning (ack: bool, error_cause: int)."""

    def __init__(self, secret: bytes, host: str = "127.0.0.1",
                 port: int = 0,
                 handler: Optional[Callable[[CoARequest], tuple]] = None):
        self.secret = secret if isinstance(secret, bytes) else secret.encode()
        self.handler = handler or (lambda req: (False, 503))
        self.sock = socket.socket(socket.AF_INET, socket.SOCK_DGRAM)
        self.sock.bind((host, port))
        self.port = self.sock.getsockname()[1]
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self.stats = {"coa_ack": 0, "coa_nak": 0, "disconnect_ack": 0,
                      "disconnect_nak": 0, "bad_auth": 0}

    def start(self):
        self._thread = threading.Thread(target=self._loop, daemon=True)
        self._thread.start()
        return self

    def stop(self):
        self._stop.set()
        self.sock.close()

    def _loop(self):
        self.sock.settimeout(0.2)
        while not self._stop.is_set():
            try:
                data, addr = self.sock.recvfrom(4096)
            except socket.timeout:
                continue
            except OSError:
                break
            resp = self._handle(data)
            if resp:
                try:
                    self.sock.sendto(resp, addr)
                except OSError:
                    break

    def _handle(self, data: bytes) -> Optional[bytes]:
        try:
            req = rp.Packet.decode(data)
        except rp.RadiusError:
            return None
        if req.code not in (rp.COA_REQUEST, rp.DISCONNECT_REQUEST):
            return None
        if not rp.verify_request_authenticator(data, self.secret):
            self.stats["bad_auth"] += 1
            return None
        creq = CoARequest(
            code=req.code,
            username=req.get_str(rp.USER_NAME) or "",
            session_id=req.get_str(rp.ACCT_SESSION_ID) or "",
            framed_ip=socket.inet_ntoa(req.get(rp.FRAMED_IP_ADDRESS))
            if req.get(rp.FRAMED_IP_ADDRESS) else "",
            mac=req.get_str(rp.CALLING_STATION_ID) or "",
            policy_name=req.get_str(rp.FILTER_ID) or "")
        try:
            ack, error_cause = self.handler(creq)
        except Exception:
            ack, error_cause = False, 504
        if req.code == rp.COA_REQUEST:
            code = rp.COA_ACK if ack else rp.COA_NAK
            self.stats["coa_ack" if ack else "coa_nak"] += 1
        else:
            code = rp.DISCONNECT_ACK if ack else rp.DISCONNECT_NAK
            self.stats["disconnect_ack" if ack else "disconnect_nak"] += 1
        resp = rp.Packet(code, req.identifier)
        if not ack and error_cause:
            resp.add(rp.ERROR_CAUSE, error_cause)
        return rp.sign_response(resp, req.authenticator, self.secret)


class CoAProcessor:
    """Wires CoA requests to the session store and dataplane
    (ref coa_handler.go:46-70): Disconnect terminates the session;
    CoA with Filter-Id re-applies the named QoS policy through the
    qos_updater hook (GPU table write)."""

    def __init__(self, session_lookup: Callable[[CoARequest], Optional[object]],
                 terminate: Callable[[object], bool],
                 qos_updater: Optional[Callable[[object, str], bool]] = None):
        self.session_lookup = session_lookup
        self.terminate = terminate
        self.qos_updater = qos_updater

    def __call__(self, req: CoARequest):
        session = self.session_lookup(req)
        if session is None:
            return False, 503       # Session-Context-Not-Found
        if req.code == rp.DISCONNECT_REQUEST:
            return (True, 0) if self.terminate(session) else (False, 504)
        # CoA: policy update
        if req.policy_name and self.qos_updater is not None:
            ok = self.qos_updater(session, req.policy_name)
            return (ok, 0 if ok else 504)
        return False, 404           # unsupported CoA contents


def send_coa(server_addr: str, secret: bytes, code: int,
             session_id: str = "", username: str = "", framed_ip: str = "",
             policy_name: str = "", timeout: float = 2.0) -> Optional[int]:
    """Client side (for tests / external tooling): send a CoA/Disconnect
    request, return the response code."""
    secret = secret if isinstance(secret, bytes) else secret.encode()
    pkt = rp.Packet(code, 1)
    if username:
        pkt.add(rp.USER_NAME, username)
    if session_id:
        pkt.add(rp.ACCT_SESSION_ID, session_id)
    if framed_ip:
        pkt.add(rp.FRAMED_IP_ADDRESS, socket.inet_aton(framed_ip))
    if policy_name:
        pkt.add(rp.FILTER_ID, policy_name)
    req_auth = rp.acct_request_authenticator(pkt, secret)
    pkt.authenticator = req_auth
    host, _, port = server_addr.rpartition(":")
    with socket.socket(socket.AF_INET, socket.SOCK_DGRAM) as s:
        s.settimeout(timeout)
        s.sendto(pkt.encode(), (host, int(port)))
        try:
            data, _ = s.recvfrom(4096)
        except socket.timeout:
            return None
    if not rp.verify_response(data, req_auth, secret):
        return None
    return rp.Packet.decode(data).code
