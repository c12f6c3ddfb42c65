"""RADIUS wire codec (RFC 2865/2866/5176) — packets, attributes,
authenticators.  The transport-level substrate for the client/accounting/
CoA components (ref pkg/radius, which uses layeh.com/radius; here the
codec is implemented directly)."""
from __future__ import annotations

import hashlib
import hmac
import os
import struct
from typing import List, Optional, Tuple

# codes
ACCESS_REQUEST = 1
ACCESS_ACCEPT = 2
ACCESS_REJECT = 3
ACCOUNTING_REQUEST = 4
ACCOUNTING_RESPONSE = 5
ACCESS_CHALLENGE = 11
DISCONNECT_REQUEST = 40
DISCONNECT_ACK = 41
DISCONNECT_NAK = 42
COA_REQUEST = 43
COA_ACK = 44
COA_NAK = 45

# attribute types
USER_NAME = 1
USER_PASSWORD = 2
CHAP_PASSWORD = 3
NAS_IP_ADDRESS = 4
NAS_PORT = 5
SERVICE_TYPE = 6
FRAMED_IP_ADDRESS = 8
FILTER_ID = 11
FRAMED_MTU = 12
REPLY_MESSAGE = 18
STATE = 24
CLASS = 25
VENDOR_SPECIFIC = 26
SESSION_TIMEOUT = 27
IDLE_TIMEOUT = 28
CALLED_STATION_ID = 30
CALLING_STATION_ID = 31
NAS_IDENTIFIER = 32
ACCT_STATUS_TYPE = 40
ACCT_DELAY_TIME = 41
ACCT_INPUT_OCTETS = 42
ACCT_OUTPUT_OCTETS = 43
ACCT_SESSION_ID = 44
ACCT_SESSION_TIME = 46
ACCT_INPUT_PACKETS = 47
ACCT_OUTPUT_PACKETS = 48
ACCT_TERMINATE_CAUSE = 49
CHAP_CHALLENGE = 60
NAS_PORT_TYPE = 61
ERROR_CAUSE = 101
MESSAGE_AUTHENTICATOR = 80

# Acct-Status-Type values
ACCT_START = 1
ACCT_STOP = 2
ACCT_INTERIM = 3

HDR = struct.Struct(">BBH16s")


class RadiusError(Exception):
    pass


class Packet:
    def __init__(self, code: int, identifier: int = 0,
                 authenticator: bytes = b"\x00" * 16):
        self.code = code
        self.identifier = identifier
        self.authenticator = authenticator
        self.attributes: List[Tuple[int, bytes]] = []

    # ------------------------------------------------------- attributes
    def add(self, typ: int, value) -> "Packet":
        if isinstance(value, str):
            value = value.encode()
        elif isinstance(value, int):
            value = struct.pack(">I", value)
        self.attributes.append((typ, bytes(value)))
        return self

    def get(self, typ: int) -> Optional[bytes]:
        for t, v in self.attributes:
            if t == typ:
                return v
        return None

    def get_all(self, typ: int) -> List[bytes]:
        return [v for t, v in self.attributes if t == typ]

    def get_int(self, typ: int) -> Optional[int]:
        v = self.get(typ)
        return None if v is None else struct.unpack(">I", v)[0]

    def get_str(self, typ: int) -> Optional[str]:
        v = self.get(typ)
        return None if v is None else v.decode(errors="replace")

    # ------------------------------------------------------------ wire
    def _attr_bytes(self) -> bytes:
        out = b""
        for t, v in self.attributes:
            if len(v) > 253:
                raise RadiusError(f"attribute {t} too long")
            out += bytes([t, len(v) + 2]) + v
        return out

    def encode(self) -> bytes:
        attrs = self._attr_bytes()
        return HDR.pack(self.code, self.identifier, 20 + len(attrs),
                        self.authenticator) + attrs

    @classmethod
    def decode(cls, data: bytes) -> "Packet":
        if len(data) < 20:
            raise RadiusError("short packet")
        code, ident, length, auth = HDR.unpack(data[:20])
        if length < 20 or length > len(data):
            raise RadiusError("bad length")
        p = cls(code, ident, auth)
        i = 20
        while i + 2 <= length:
            t, ln = data[i], data[i + 1]
            if ln < 2 or i + ln > length:
                raise RadiusError("bad attribute")
            p.attributes.append((t, data[i + 2:i + ln]))
            i += ln
        return p


# --------------------------------------------------------- authenticators
def random_authenticator() -> bytes:
    return os.urandom(16)


def encrypt_user_password(password: bytes, secret: bytes,
                          req_auth: bytes) -> bytes:
    """RFC 2865 §5.2 User-Password obfuscation."""
    if len(password) > 128:
        raise RadiusError("password too long")
    pad = (-len(password)) % 16 or 0
    p = password + b"\x00" * pad
    if not p:
        p = b"\x00" * 16
    out = b""
    prev = req_auth
    for i in range(0, len(p), 16):
        h = hashlib.md5(secret + prev).digest()
        chunk = bytes(a ^ b for a, b in zip(p[i:i + 16], h))
        out += chunk
        prev = chunk
    return out


def decrypt_user_password(enc: bytes, secret: bytes, req_auth: bytes) -> bytes:
    out = b""
    prev = req_auth
    for i in range(0, len(enc), 16):
        h = hashlib.md5(secret + prev).digest()
        out += bytes(a ^ b for a, b in zip(enc[i:i + 16], h))
        prev = enc[i:i + 16]
    return out.rstrip(b"\x00")


def sign_message_authenticator(pkt: Packet, secret: bytes) -> None:
    """Add Message-Authenticator = HMAC-MD5(packet with MA zeroed)
    (ref client.go:405-427 — always included against blast-RADIUS)."""
    pkt.attributes = [(t, v) for t, v in pkt.attributes
                      if t != MESSAGE_AUTHENTICATOR]
    pkt.attributes.append((MESSAGE_AUTHENTICATOR, b"\x00" * 16))
    raw = pkt.encode()
    mac = hmac.new(secret, raw, hashlib.md5).digest()
    pkt.attributes[-1] = (MESSAGE_AUTHENTICATOR, mac)


def verify_message_authenticator(pkt: Packet, secret: bytes,
                                 req_auth: Optional[bytes] = None) -> bool:
    ma = pkt.get(MESSAGE_AUTHENTICATOR)
    if ma is None:
        return False
    clone = Packet(pkt.code, pkt.identifier,
                   req_auth if req_auth is not None else pkt.authenticator)
    clone.attributes = [(t, v if t != MESSAGE_AUTHENTICATOR else b"\x00" * 16)
                        for t, v in pkt.attributes]
    raw = clone.encode()
    return hmac.compare_digest(hmac.new(secret, raw, hashlib.md5).digest(),
                               ma)


def sign_response_with_ma(pkt: "Packet", req_auth: bytes,
                          secret: bytes) -> bytes:
    """Response with Message-Authenticator (blast-RADIUS / CVE-2024-3596
    mitigation): MA = HMAC-MD5 over the response with the REQUEST
    authenticator in the authenticator field and MA zeroed, computed
    BEFORE the Response Authenticator."""
    pkt.attributes = [(t, v) for t, v in pkt.attributes
                      if t != MESSAGE_AUTHENTICATOR]
    pkt.attributes.append((MESSAGE_AUTHENTICATOR, b"\x00" * 16))
    clone = Packet(pkt.code, pkt.identifier, req_auth)
    clone.attributes = list(pkt.attributes)
    mac = hmac.new(secret, clone.encode(), hashlib.md5).digest()
    pkt.attributes[-1] = (MESSAGE_AUTHENTICATOR, mac)
    return sign_response(pkt, req_auth, secret)


def response_authenticator(code: int, identifier: int, attrs: bytes,
                           req_auth: bytes, secret: bytes) -> bytes:
    """RFC 2865 §3: MD5(Code+ID+Length+RequestAuth+Attributes+Secret)."""
    length = 20 + len(attrs)
    return hashlib.md5(struct.pack(">BBH", code, identifier, length)
                       + req_auth + attrs + secret).digest()


def sign_response(pkt: Packet, req_auth: bytes, secret: bytes) -> bytes:
    attrs = pkt._attr_bytes()
    pkt.authenticator = response_authenticator(pkt.code, pkt.identifier,
                                               attrs, req_auth, secret)
    return pkt.encode()


def verify_response(data: bytes, req_auth: bytes, secret: bytes) -> bool:
    if len(data) < 20:
        return False
    code, ident = data[0], data[1]
    expect = response_authenticator(code, ident, data[20:], req_auth, secret)
    return hmac.compare_digest(expect, data[4:20])


def acct_request_authenticator(pkt: Packet, secret: bytes) -> bytes:
    """RFC 2866 §3: MD5(Code+ID+Length+16 zeros+Attributes+Secret)."""
    attrs = pkt._attr_bytes()
    length = 20 + len(attrs)
    return hashlib.md5(struct.pack(">BBH", pkt.code, pkt.identifier, length)
                       + b"\x00" * 16 + attrs + secret).digest()


def verify_request_authenticator(data: bytes, secret: bytes) -> bool:
    """For Accounting-Request / CoA / Disconnect requests (RFC 2866/5176)."""
    if len(data) < 20:
        return False
    expect = hashlib.md5(data[:4] + b"\x00" * 16 + data[20:]
                         + secret).digest()
    return hmac.compare_digest(expect, data[4:20])
