"""In-process RADIUS server (auth + accounting) — the test substrate the
reference covers with mocks, also usable for small lab deployments.
Verifies Message-Authenticator and accounting Request Authenticators."""
from __future__ import annotations

import socket
import threading
from typing import Dict, List, Optional, Tuple

from . import packet as rp


class RadiusServer:
    def __init__(self, secret: bytes, users: Optional[Dict[str, dict]] = None,
                 host: str = "127.0.0.1", port: int = 0):
        """users: name -> {password, framed_ip, policy, session_timeout}."""
        self.secret = secret if isinstance(secret, bytes) else secret.encode()
        self.users = users or {}
        self.sock = socket.socket(socket.AF_INET, socket.SOCK_DGRAM)
        self.sock.bind((host, port))
        self.port = self.sock.getsockname()[1]
        self.acct_sock = socket.socket(socket.AF_INET, socket.SOCK_DGRAM)
        try:
            self.acct_sock.bind((host, self.port + 1))
        except OSError:
            self.acct_sock.bind((host, 0))
        self.acct_port = self.acct_sock.getsockname()[1]
        self.acct_records: List[rp.Packet] = []
        self.auth_requests: List[rp.Packet] = []
        self.drop_requests = False     # simulate partition
        self._stop = threading.Event()
        self._threads: List[threading.Thread] = []
        self._lock = threading.Lock()

    @property
    def addr(self) -> str:
        return f"127.0.0.1:{self.port}"

    def start(self):
        for sock, fn in ((self.sock, self._handle_auth),
                         (self.acct_sock, self._handle_acct)):
            t = threading.Thread(target=self._loop, args=(sock, fn),
                                 daemon=True)
            t.start()
            self._threads.append(t)
        return self

    def stop(self):
        self._stop.set()
        self.sock.close()
        self.acct_sock.close()

    def _loop(self, sock, handler):
        sock.settimeout(0.2)
        while not self._stop.is_set():
            try:
                data, addr = sock.recvfrom(4096)
            except socket.timeout:
                continue
            except OSError:
                break
            if self.drop_requests:
                continue
            try:
                resp = handler(data)
            except Exception:
                continue
            if resp:
                try:
                    sock.sendto(resp, addr)
                except OSError:
                    break

    def _handle_auth(self, data: bytes) -> Optional[bytes]:
        req = rp.Packet.decode(data)
        if req.code != rp.ACCESS_REQUEST:
            return None
        if not rp.verify_message_authenticator(req, self.secret):
            return None
        with self._lock:
            self.auth_requests.append(req)
        user = req.get_str(rp.USER_NAME) or ""
        enc = req.get(rp.USER_PASSWORD)
        chap = req.get(rp.CHAP_PASSWORD)
        rec = self.users.get(user)
        ok = False
        if rec is not None and enc is not None:
            pw = rp.decrypt_user_password(enc, self.secret,
                                          req.authenticator)
            ok = pw.decode(errors="replace") == rec.get("password", "")
        elif rec is not None and chap is not None and len(chap) == 17:
            # CHAP: value = MD5(ident | secret | challenge) (RFC 2865 §2.2)
            import hashlib
            ident, value = chap[0], chap[1:]
            challenge = req.get(rp.CHAP_CHALLENGE) or req.authenticator
            expect = hashlib.md5(bytes([ident]) +
                                 rec.get("password", "").encode() +
                                 challenge).digest()
            ok = value == expect
        resp = rp.Packet(rp.ACCESS_ACCEPT if ok else rp.ACCESS_REJECT,
                         req.identifier)
        if ok:
            if rec.get("framed_ip"):
                resp.add(rp.FRAMED_IP_ADDRESS,
                         socket.inet_aton(rec["framed_ip"]))
            if rec.get("policy"):
                resp.add(rp.FILTER_ID, rec["policy"])
            if rec.get("session_timeout"):
                resp.add(rp.SESSION_TIMEOUT, int(rec["session_timeout"]))
            if rec.get("class"):
                resp.add(rp.CLASS, rec["class"])
        else:
            resp.add(rp.REPLY_MESSAGE, "denied")
        if req.get(rp.MESSAGE_AUTHENTICATOR) is not None:
            # requests carrying MA get MA'd responses (blast-RADIUS)
            return rp.sign_response_with_ma(resp, req.authenticator,
                                            self.secret)
        return rp.sign_response(resp, req.authenticator, self.secret)

    def _handle_acct(self, data: bytes) -> Optional[bytes]:
        req = rp.Packet.decode(data)
        if req.code != rp.ACCOUNTING_REQUEST:
            return None
        if not rp.verify_request_authenticator(data, self.secret):
            return None
        with self._lock:
            self.acct_records.append(req)
        resp = rp.Packet(rp.ACCOUNTING_RESPONSE, req.identifier)
        if req.get(rp.MESSAGE_AUTHENTICATOR) is not None:
            # requests carrying MA get MA'd responses (blast-RADIUS)
            return rp.sign_response_with_ma(resp, req.authenticator,
                                            self.secret)
        return rp.sign_response(resp, req.authenticator, self.secret)
