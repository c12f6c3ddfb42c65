"""DHCP message codec for the slow path (UDP payload level) — the analog
of the reference's insomniacslk/dhcp dependency (pkg/dhcp/server.go),
implemented directly."""
from __future__ import annotations

import struct
from typing import List, Optional, Tuple

DISCOVER, OFFER, REQUEST, DECLINE, ACK, NAK, RELEASE, INFORM = range(1, 9)
MAGIC = 0x63825363

OPT_SUBNET_MASK = 1
OPT_ROUTER = 3
OPT_DNS = 6
OPT_HOSTNAME = 12
OPT_REQUESTED_IP = 50
OPT_LEASE_TIME = 51
OPT_MSG_TYPE = 53
OPT_SERVER_ID = 54
OPT_PARAM_REQ_LIST = 55
OPT_RENEWAL_TIME = 58
OPT_REBIND_TIME = 59
OPT_VENDOR_CLASS = 60
OPT_CLIENT_ID = 61
OPT_RELAY_AGENT = 82
OPT_END = 255


class DHCPMessage:
    def __init__(self):
        self.op = 1
        self.htype = 1
        self.hlen = 6
        self.hops = 0
        self.xid = 0
        self.secs = 0
        self.flags = 0
        self.ciaddr = 0
        self.yiaddr = 0
        self.siaddr = 0
        self.giaddr = 0
        self.chaddr = b"\x00" * 6
        self.sname = b""
        self.file = b""
        self.options: List[Tuple[int, bytes]] = []

    # ------------------------------------------------------------ options
    def get_option(self, code: int) -> Optional[bytes]:
        for c, v in self.options:
            if c == code:
                return v
        return None

    def set_option(self, code: int, value: bytes):
        self.options = [(c, v) for c, v in self.options if c != code]
        self.options.append((code, value))

    @property
    def msg_type(self) -> int:
        v = self.get_option(OPT_MSG_TYPE)
        return v[0] if v else 0

    @property
    def requested_ip(self) -> int:
        v = self.get_option(OPT_REQUESTED_IP)
        return struct.unpack(">I", v)[0] if v and len(v) == 4 else 0

    @property
    def client_mac(self) -> bytes:
        return self.chaddr[:6]

    def circuit_id(self) -> bytes:
        """Option 82 sub-option 1 (ref parseOption82, server.go:199)."""
        v = self.get_option(OPT_RELAY_AGENT)
        if not v:
            return b""
        i = 0
        while i + 2 <= len(v):
            sc, sl = v[i], v[i + 1]
            if i + 2 + sl > len(v):
                break
            if sc == 1:
                return v[i + 2:i + 2 + sl]
            i += 2 + sl
        return b""

    def remote_id(self) -> bytes:
        v = self.get_option(OPT_RELAY_AGENT)
        if not v:
            return b""
        i = 0
        while i + 2 <= len(v):
            sc, sl = v[i], v[i + 1]
            if i + 2 + sl > len(v):
                break
            if sc == 2:
                return v[i + 2:i + 2 + sl]
            i += 2 + sl
        return b""

    @property
    def vendor_class(self) -> str:
        v = self.get_option(OPT_VENDOR_CLASS)
        return v.decode(errors="replace") if v else ""

    # -------------------------------------------------------------- wire
    def encode(self) -> bytes:
        out = struct.pack(">BBBBIHHIIII", self.op, self.htype, self.hlen,
                          self.hops, self.xid, self.secs, self.flags,
                          self.ciaddr, self.yiaddr, self.siaddr, self.giaddr)
        out += (self.chaddr + b"\x00" * 16)[:16]
        out += (self.sname + b"\x00" * 64)[:64]
        out += (self.file + b"\x00" * 128)[:128]
        out += struct.pack(">I", MAGIC)
        for c, v in self.options:
            out += bytes([c, len(v)]) + v
        out += bytes([OPT_END])
        return out

    @classmethod
    def decode(cls, data: bytes) -> "DHCPMessage":
        if len(data) < 240:
            raise ValueError("short DHCP message")
        m = cls()
        (m.op, m.htype, m.hlen, m.hops, m.xid, m.secs, m.flags, m.ciaddr,
         m.yiaddr, m.siaddr, m.giaddr) = struct.unpack(">BBBBIHHIIII",
                                                       data[:28])
        m.chaddr = data[28:34]
        m.sname = data[44:108].rstrip(b"\x00")
        m.file = data[108:236].rstrip(b"\x00")
        if struct.unpack(">I", data[236:240])[0] != MAGIC:
            raise ValueError("bad magic cookie")
        i = 240
        while i < len(data):
            code = data[i]
            if code == 0:
                i += 1
                continue
            if code == OPT_END:
                break
            if i + 1 >= len(data):
                break
            ln = data[i + 1]
            if i + 2 + ln > len(data):
                break
            m.options.append((code, data[i + 2:i + 2 + ln]))
            i += 2 + ln
        return m


def build_request(mac: bytes, msg_type: int, xid: int = 1,
                  requested_ip: int = 0, server_id: int = 0,
                  ciaddr: int = 0, giaddr: int = 0,
                  circuit_id: bytes = b"", vendor_class: str = "",
                  broadcast: bool = False) -> DHCPMessage:
    m = DHCPMessage()
    m.op = 1
    m.xid = xid
    m.flags = 0x8000 if broadcast else 0
    m.chaddr = mac
    m.ciaddr = ciaddr
    m.giaddr = giaddr
    m.set_option(OPT_MSG_TYPE, bytes([msg_type]))
    if requested_ip:
        m.set_option(OPT_REQUESTED_IP, struct.pack(">I", requested_ip))
    if server_id:
        m.set_option(OPT_SERVER_ID, struct.pack(">I", server_id))
    if vendor_class:
        m.set_option(OPT_VENDOR_CLASS, vendor_class.encode())
    if circuit_id:
        m.set_option(OPT_RELAY_AGENT,
                     bytes([1, len(circuit_id)]) + circuit_id)
    return m
