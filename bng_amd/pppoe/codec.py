"""PPPoE / PPP wire codec (RFC 2516, RFC 1661) — discovery and session
frames, PPP protocol payloads, LCP/IPCP/IPV6CP option TLVs
(ref pkg/pppoe/protocol handling in server.go/lcp.go/ipcp.go)."""
from __future__ import annotations

import struct
from typing import List, Optional, Tuple

ETH_PPPOE_DISC = 0x8863
ETH_PPPOE_SESS = 0x8864

# PPPoE discovery codes (RFC 2516)
PADI = 0x09
PADO = 0x07
PADR = 0x19
PADS = 0x65
PADT = 0xA7

# discovery tags
TAG_EOL = 0x0000
TAG_SERVICE_NAME = 0x0101
TAG_AC_NAME = 0x0102
TAG_HOST_UNIQ = 0x0103
TAG_AC_COOKIE = 0x0104
TAG_RELAY_SESSION_ID = 0x0110
TAG_SERVICE_NAME_ERROR = 0x0201
TAG_AC_SYSTEM_ERROR = 0x0202
TAG_GENERIC_ERROR = 0x0203

# PPP protocols
PROTO_LCP = 0xC021
PROTO_PAP = 0xC023
PROTO_CHAP = 0xC223
PROTO_IPCP = 0x8021
PROTO_IPV6CP = 0x8057
PROTO_IPV4 = 0x0021
PROTO_IPV6 = 0x0057

# CP codes (RFC 1661)
CONF_REQ = 1
CONF_ACK = 2
CONF_NAK = 3
CONF_REJ = 4
TERM_REQ = 5
TERM_ACK = 6
CODE_REJ = 7
PROTO_REJ = 8
ECHO_REQ = 9
ECHO_REP = 10
DISCARD_REQ = 11

# LCP options
LCP_OPT_MRU = 1
LCP_OPT_AUTH = 3
LCP_OPT_QUALITY = 4
LCP_OPT_MAGIC = 5
LCP_OPT_PFC = 7
LCP_OPT_ACFC = 8

# IPCP options (RFC 1332)
IPCP_OPT_IP = 3
IPCP_OPT_DNS1 = 129
IPCP_OPT_DNS2 = 131

# IPV6CP options (RFC 5072)
IPV6CP_OPT_IFID = 1

# auth protocols
AUTH_PAP = 0xC023
AUTH_CHAP_MD5 = (0xC223, 5)

# CHAP codes
CHAP_CHALLENGE = 1
CHAP_RESPONSE = 2
CHAP_SUCCESS = 3
CHAP_FAILURE = 4

# PAP codes
PAP_AUTH_REQ = 1
PAP_AUTH_ACK = 2
PAP_AUTH_NAK = 3


def encode_tags(tags: List[Tuple[int, bytes]]) -> bytes:
    return b"".join(struct.pack(">HH", t, len(v)) + v for t, v in tags)


def decode_tags(data: bytes) -> List[Tuple[int, bytes]]:
    tags = []
    i = 0
    while i + 4 <= len(data):
        t, ln = struct.unpack_from(">HH", data, i)
        if i + 4 + ln > len(data):
            break
        tags.append((t, data[i + 4:i + 4 + ln]))
        i += 4 + ln
    return tags


def get_tag(tags: List[Tuple[int, bytes]], t: int) -> Optional[bytes]:
    for tt, v in tags:
        if tt == t:
            return v
    return None


class DiscoveryPacket:
    def __init__(self, code: int, session_id: int = 0,
                 tags: Optional[List[Tuple[int, bytes]]] = None,
                 src_mac: bytes = b"\x00" * 6, dst_mac: bytes = b"\xff" * 6):
        self.code = code
        self.session_id = session_id
        self.tags = tags or []
        self.src_mac = src_mac
        self.dst_mac = dst_mac

    def encode(self) -> bytes:
        payload = encode_tags(self.tags)
        return (self.dst_mac + self.src_mac +
                struct.pack(">H", ETH_PPPOE_DISC) +
                struct.pack(">BBHH", 0x11, self.code, self.session_id,
                            len(payload)) + payload)

    @classmethod
    def decode(cls, frame: bytes) -> "DiscoveryPacket":
        if len(frame) < 20:
            raise ValueError("short PPPoE discovery frame")
        et = struct.unpack_from(">H", frame, 12)[0]
        if et != ETH_PPPOE_DISC:
            raise ValueError("not PPPoE discovery")
        ver_type, code, sid, ln = struct.unpack_from(">BBHH", frame, 14)
        if ver_type != 0x11:
            raise ValueError("bad PPPoE version/type")
        p = cls(code, sid, decode_tags(frame[20:20 + ln]),
                src_mac=frame[6:12], dst_mac=frame[0:6])
        return p


class SessionPacket:
    """PPPoE session frame carrying one PPP protocol payload."""

    def __init__(self, session_id: int, ppp_proto: int, payload: bytes,
                 src_mac: bytes = b"\x00" * 6, dst_mac: bytes = b"\x00" * 6):
        self.session_id = session_id
        self.ppp_proto = ppp_proto
        self.payload = payload
        self.src_mac = src_mac
        self.dst_mac = dst_mac

    def encode(self) -> bytes:
        inner = struct.pack(">H", self.ppp_proto) + self.payload
        return (self.dst_mac + self.src_mac +
                struct.pack(">H", ETH_PPPOE_SESS) +
                struct.pack(">BBHH", 0x11, 0x00, self.session_id,
                            len(inner)) + inner)

    @classmethod
    def decode(cls, frame: bytes) -> "SessionPacket":
        if len(frame) < 22:
            raise ValueError("short PPPoE session frame")
        if struct.unpack_from(">H", frame, 12)[0] != ETH_PPPOE_SESS:
            raise ValueError("not PPPoE session")
        _vt, _code, sid, ln = struct.unpack_from(">BBHH", frame, 14)
        proto = struct.unpack_from(">H", frame, 20)[0]
        return cls(sid, proto, frame[22:20 + ln], src_mac=frame[6:12],
                   dst_mac=frame[0:6])


class CPPacket:
    """Control-protocol packet (LCP/IPCP/IPV6CP/auth share the shape)."""

    def __init__(self, code: int, identifier: int, data: bytes = b""):
        self.code = code
        self.identifier = identifier
        self.data = data

    def encode(self) -> bytes:
        return struct.pack(">BBH", self.code, self.identifier,
                           4 + len(self.data)) + self.data

    @classmethod
    def decode(cls, payload: bytes) -> "CPPacket":
        if len(payload) < 4:
            raise ValueError("short CP packet")
        code, ident, ln = struct.unpack_from(">BBH", payload, 0)
        if ln < 4 or ln > len(payload):
            raise ValueError("bad CP length")
        return cls(code, ident, payload[4:ln])


def encode_opts(opts: List[Tuple[int, bytes]]) -> bytes:
    return b"".join(bytes([t, len(v) + 2]) + v for t, v in opts)


def decode_opts(data: bytes) -> List[Tuple[int, bytes]]:
    opts = []
    i = 0
    while i + 2 <= len(data):
        t, ln = data[i], data[i + 1]
        if ln < 2 or i + ln > len(data):
            break
        opts.append((t, data[i + 2:i + ln]))
        i += ln
    return opts


def get_opt(opts, t):
    for tt, v in opts:
        if tt == t:
            return v
    return None


def chap_md5_response(ident: int, secret: bytes, challenge: bytes) -> bytes:
    import hashlib
    return hashlib.md5(bytes([ident]) + secret + challenge).digest()
