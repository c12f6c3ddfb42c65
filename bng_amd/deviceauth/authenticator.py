"""Device -> Nexus authentication (ref pkg/deviceauth): none / PSK(HMAC)
/ mTLS authenticators (types.go:35-194, psk.go, mtls.go) and an
authenticated HTTP transport (transport.go:8-40)."""
from __future__ import annotations

import hashlib
import hmac
import time
from typing import Dict, Optional

MODE_NONE = "none"
MODE_PSK = "psk"
MODE_MTLS = "mtls"


class AuthError(Exception):
    pass


class Authenticator:
    """Produces the auth headers a device presents to Nexus and verifies
    them on the Nexus side."""

    mode = MODE_NONE

    def headers(self, device_id: str) -> Dict[str, str]:
        return {}

    def verify(self, headers: Dict[str, str]) -> Optional[str]:
        """-> device_id if valid, else raises AuthError."""
        return headers.get("X-Device-ID") or None


class PSKAuthenticator(Authenticator):
    """HMAC-SHA256 over (device_id | timestamp) with a pre-shared key;
    replay-protected by a timestamp window (ref psk.go)."""

    mode = MODE_PSK

    def __init__(self, psk: bytes, window: float = 300.0):
        self.psk = psk if isinstance(psk, bytes) else psk.encode()
        self.window = window

    def _sig(self, device_id: str, ts: str) -> str:
        return hmac.new(self.psk, f"{device_id}|{ts}".encode(),
                        hashlib.sha256).hexdigest()

    def headers(self, device_id: str) -> Dict[str, str]:
        ts = str(int(time.time()))
        return {"X-Device-ID": device_id, "X-Auth-Timestamp": ts,
                "X-Auth-Signature": self._sig(device_id, ts)}

    def verify(self, headers: Dict[str, str]) -> str:
        did = headers.get("X-Device-ID", "")
        ts = headers.get("X-Auth-Timestamp", "")
        sig = headers.get("X-Auth-Signature", "")
        if not did or not ts or not sig:
            raise AuthError("missing auth headers")
        try:
            age = abs(time.time() - int(ts))
        except ValueError:
            raise AuthError("bad timestamp")
        if age > self.window:
            raise AuthError("timestamp outside replay window")
        if not hmac.compare_digest(self._sig(did, ts), sig):
            raise AuthError("bad signature")
        return did


class MTLSAuthenticator(Authenticator):
    """mTLS: identity from the client certificate CN; here modeled over
    the cert fingerprint registry (the TLS handshake itself is the HTTP
    stack's job; ref mtls.go validates peer certs the same way)."""

    mode = MODE_MTLS

    def __init__(self, ca_fingerprints: Optional[Dict[str, str]] = None):
        # device_id -> expected cert sha256 fingerprint
        self.registry = ca_fingerprints or {}

    def register(self, device_id: str, cert_pem: bytes):
        self.registry[device_id] = hashlib.sha256(cert_pem).hexdigest()

    def headers(self, device_id: str) -> Dict[str, str]:
        return {"X-Device-ID": device_id}

    def verify_cert(self, device_id: str, cert_pem: bytes) -> str:
        fp = hashlib.sha256(cert_pem).hexdigest()
        want = self.registry.get(device_id)
        if want is None:
            raise AuthError(f"unknown device {device_id}")
        if not hmac.compare_digest(want, fp):
            raise AuthError("certificate fingerprint mismatch")
        return device_id

    def verify(self, headers: Dict[str, str]) -> str:
        did = headers.get("X-Device-ID", "")
        if did not in self.registry:
            raise AuthError("unknown device")
        return did


def new_authenticator(mode: str, **kw) -> Authenticator:
    """Factory (ref authenticator.go:16 NewAuthenticator); TPM mode is
    recognized but unimplemented, exactly like the reference (:33)."""
    if mode == MODE_NONE:
        return Authenticator()
    if mode == MODE_PSK:
        return PSKAuthenticator(kw["psk"], kw.get("window", 300.0))
    if mode == MODE_MTLS:
        return MTLSAuthenticator(kw.get("registry"))
    if mode == "tpm":
        raise NotImplementedError(
            "TPM authentication not yet implemented")
    raise ValueError(f"unknown auth mode {mode}")


class AuthenticatedSession:
    """requests.Session wrapper injecting device auth headers
    (ref transport.go:8-40)."""

    def __init__(self, auth: Authenticator, device_id: str, session=None):
        import requests
        self.auth = auth
        self.device_id = device_id
        self.session = session or requests.Session()

    def request(self, method: str, url: str, **kw):
        headers = dict(kw.pop("headers", {}) or {})
        headers.update(self.auth.headers(self.device_id))
        return self.session.request(method, url, headers=headers, **kw)

    def get(self, url, **kw):
        return self.request("GET", url, **kw)

    def post(self, url, **kw):
        return self.request("POST", url, **kw)
