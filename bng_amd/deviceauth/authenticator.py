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

    def rotate(self, new_psk) -> None:
        """Swap the key in place; production floor of 16 chars is
        enforced here, not at load (ref psk.go RotatePSK :293-311)."""
        raw = new_psk if isinstance(new_psk, bytes) else new_psk.encode()
        if len(raw) < 16:
            raise AuthError("rotated PSK must be at least 16 characters")
        self.psk = raw


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

    def certificate_expires_within(self, cert_path: str,
                                   days: float) -> bool:
        """ref mtls.go CertificateExpiresWithin :408-418."""
        from ..agent.tls import is_certificate_expiring_soon
        expiring, _ = is_certificate_expiring_soon(cert_path, days)
        return expiring

    def renewal_request(self, device_id: str, common_name: str = "",
                        reason: str = "") -> Dict[str, str]:
        """CSR + metadata a device POSTs to Nexus for cert renewal
        (ref types.go CertificateRenewalRequest :149-162); returns the
        request dict with the fresh private key under '_key_pem' for
        the caller to store (never transmitted)."""
        csr, key = generate_csr(common_name or device_id)
        return {"device_id": device_id, "csr": csr,
                "reason": reason, "_key_pem": key}


def sanitize_id(s: str) -> str:
    """Keep [A-Za-z0-9_-] (ref authenticator.go sanitizeID :251-260)."""
    return "".join(c for c in s if c.isalnum() or c in "-_")


def generate_device_id(serial: str = "", mac: str = "") -> str:
    """Stable device id: serial first, MAC second, random last (ref
    authenticator.go generateDeviceID :233-249)."""
    if serial:
        return "bng-" + sanitize_id(serial)
    if mac:
        return "bng-" + mac.replace(":", "")
    import uuid
    return "bng-" + uuid.uuid4().hex[:16]


def read_device_identity(interface: str = "") -> Dict[str, str]:
    """Hardware identity: DMI serial + primary MAC -> device id (ref
    ReadDeviceIdentity authenticator.go:137-159)."""
    from ..ztp.bootstrap import detect_system_info
    info = detect_system_info(interface)
    return {"device_id": generate_device_id(info.serial, info.mac),
            "serial": info.serial, "mac": info.mac,
            "model": info.model, "firmware": info.firmware}


def load_psk(key: str = "", key_file: str = "") -> bytes:
    """PSK from inline key or file; short keys are allowed for dev but
    rotation enforces the 16-char floor (ref psk.go loadPSK :75-113)."""
    if key_file:
        with open(key_file) as f:
            psk = f.read().strip()
    else:
        psk = key
    if not psk:
        raise AuthError("PSK is required (key or key_file)")
    return psk.encode()


def validate_config(mode: str, psk_key: str = "", psk_key_file: str = "",
                    cert_file: str = "", key_file: str = "",
                    ca_file: str = "",
                    insecure_skip_verify: bool = False) -> None:
    """Reject half-configured auth before a device goes to the field
    (ref ValidateConfig authenticator.go:262-308)."""
    if mode == MODE_NONE:
        return
    if mode == MODE_PSK:
        if not psk_key and not psk_key_file:
            raise AuthError("PSK key or key_file is required")
        return
    if mode == MODE_MTLS:
        if not cert_file:
            raise AuthError("mTLS cert_file is required")
        if not key_file:
            raise AuthError("mTLS key_file is required")
        if not ca_file and not insecure_skip_verify:
            raise AuthError(
                "mTLS ca_file is required (or set insecure_skip_verify)")
        return
    if mode == "tpm":
        raise NotImplementedError("TPM mode not yet implemented")
    raise AuthError(f"unknown authentication mode: {mode}")


def generate_csr(common_name: str, key_bits: int = 2048):
    """(csr_pem, key_pem) for certificate renewal (ref mtls.go
    GenerateCSR :362-406; key generated fresh per request).  Shells to
    openssl — the only X.509 writer in this image."""
    import subprocess
    import tempfile

    with tempfile.TemporaryDirectory() as td:
        key_path = f"{td}/key.pem"
        csr_path = f"{td}/req.pem"
        subprocess.run(
            ["openssl", "req", "-new", "-newkey", f"rsa:{key_bits}",
             "-nodes", "-keyout", key_path, "-out", csr_path,
             "-subj", f"/CN={common_name}"],
            check=True, capture_output=True)
        with open(csr_path) as f:
            csr = f.read()
        with open(key_path) as f:
            key = f.read()
    return csr, key


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
