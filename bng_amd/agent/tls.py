"""TLS configuration for agent<->controller links (ref pkg/agent/tls.go).

Builds an ssl.SSLContext (or requests-style kwargs) from a declarative
config: CA material by file or inline PEM, client cert/key for mTLS,
min-version selection, server-name override, certificate pinning by
SHA-256 fingerprint of the DER encoding, and an explicitly-insecure
skip-verify mode for testing (GetCertFingerprint tls.go:202-221,
BuildTLSConfig :63-155, ValidateTLSConfig :157-199)."""
from __future__ import annotations

import hashlib
import os
import ssl
import tempfile
from dataclasses import dataclass, field
from typing import List, Optional


class TLSError(Exception):
    pass


@dataclass
class TLSConfig:
    enabled: bool = True
    ca_cert_file: str = ""
    ca_cert_pem: str = ""
    cert_file: str = ""
    key_file: str = ""
    pinned_certs: List[str] = field(default_factory=list)
    server_name: str = ""
    min_version: str = "1.2"
    insecure_skip_verify: bool = False


def default_tls_config() -> TLSConfig:
    return TLSConfig()


def validate_tls_config(c: TLSConfig) -> None:
    """ref ValidateTLSConfig: reject half-configured mTLS, bad versions,
    malformed pins, unreadable files."""
    if not c.enabled:
        return
    if c.min_version not in ("", "1.2", "1.3"):
        raise TLSError(
            f"invalid TLS min_version: {c.min_version} (use '1.2' or '1.3')")
    if bool(c.cert_file) != bool(c.key_file):
        raise TLSError("cert_file and key_file must both be set for mTLS")
    for path in (c.ca_cert_file, c.cert_file, c.key_file):
        if path and not os.path.exists(path):
            raise TLSError(f"TLS file not found: {path}")
    for fp in c.pinned_certs:
        h = fp.replace(":", "").lower()
        if len(h) != 64 or any(ch not in "0123456789abcdef" for ch in h):
            raise TLSError(f"pinned cert is not a SHA-256 hex digest: {fp}")
    if c.cert_file and c.key_file:
        try:
            ctx = ssl.SSLContext(ssl.PROTOCOL_TLS_CLIENT)
            ctx.load_cert_chain(c.cert_file, c.key_file)
        except ssl.SSLError as e:
            raise TLSError(f"cert/key mismatch: {e}") from e


def build_ssl_context(c: TLSConfig) -> Optional[ssl.SSLContext]:
    """ref BuildTLSConfig; returns None when TLS handling is disabled
    (caller uses library defaults)."""
    if not c.enabled:
        return None
    validate_tls_config(c)
    ctx = ssl.SSLContext(ssl.PROTOCOL_TLS_CLIENT)
    ctx.minimum_version = (ssl.TLSVersion.TLSv1_3 if c.min_version == "1.3"
                           else ssl.TLSVersion.TLSv1_2)
    if c.ca_cert_file:
        ctx.load_verify_locations(cafile=c.ca_cert_file)
    if c.ca_cert_pem:
        ctx.load_verify_locations(cadata=c.ca_cert_pem)
    if not c.ca_cert_file and not c.ca_cert_pem:
        ctx.load_default_certs()
    if c.cert_file:
        ctx.load_cert_chain(c.cert_file, c.key_file)
    if c.insecure_skip_verify:
        ctx.check_hostname = False
        ctx.verify_mode = ssl.CERT_NONE
    return ctx


def get_cert_fingerprint(cert_path: str) -> str:
    """SHA-256 over the DER encoding, hex (ref GetCertFingerprint)."""
    with open(cert_path) as f:
        pem = f.read()
    der = ssl.PEM_cert_to_DER_cert(pem)
    return hashlib.sha256(der).hexdigest()


def verify_pinned(der_bytes: bytes, pinned: List[str]) -> bool:
    """Post-handshake pin check: compare the peer cert's DER SHA-256
    against the allowed list (ref VerifyPeerCertificate closure)."""
    if not pinned:
        return True
    fp = hashlib.sha256(der_bytes).hexdigest()
    return any(fp == p.replace(":", "").lower() for p in pinned)


def extract_cert_info(cert_path: str) -> dict:
    """Subject/issuer/validity for audit events (ref
    audit.ExtractCertInfo); uses the stdlib decoder."""
    import _ssl
    d = _ssl._test_decode_cert(cert_path)
    subject = {k: v for part in d.get("subject", ())
               for (k, v) in part}
    issuer = {k: v for part in d.get("issuer", ()) for (k, v) in part}
    return {"subject": subject.get("commonName", ""),
            "issuer": issuer.get("commonName", ""),
            "serial": d.get("serialNumber", ""),
            "not_before": d.get("notBefore", ""),
            "not_after": d.get("notAfter", "")}


def is_certificate_expiring_soon(cert_path: str,
                                 within_days: float = 30.0):
    """(expiring, days_left) for renewal warnings (ref ztp/tls.go
    IsCertificateExpiringSoon :508-522)."""
    import time as _t
    info = extract_cert_info(cert_path)
    na = info.get("not_after", "")
    if not na:
        raise TLSError(f"no notAfter in {cert_path}")
    expires = ssl.cert_time_to_seconds(na)
    days_left = (expires - _t.time()) / 86400.0
    return days_left <= within_days, days_left


def extract_server_name_from_url(url: str) -> str:
    """Hostname for SNI/verification (ref ztp/tls.go
    ExtractServerNameFromURL :524-537)."""
    from urllib.parse import urlparse
    host = urlparse(url).hostname
    return host or ""


def requests_kwargs(c: TLSConfig) -> dict:
    """Map the config onto requests' verify=/cert= kwargs (the
    AuthenticatedTransport analog for our HTTP clients)."""
    if not c.enabled:
        return {}
    kw = {}
    if c.insecure_skip_verify:
        kw["verify"] = False
    elif c.ca_cert_file:
        kw["verify"] = c.ca_cert_file
    elif c.ca_cert_pem:
        f = tempfile.NamedTemporaryFile("w", suffix=".pem", delete=False)
        f.write(c.ca_cert_pem)
        f.close()
        kw["verify"] = f.name
    if c.cert_file:
        kw["cert"] = (c.cert_file, c.key_file)
    return kw
