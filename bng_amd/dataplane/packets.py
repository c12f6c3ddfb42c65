"""Synthetic packet construction/parsing for tests, the golden model and bench.

Builds the same wire formats the reference dataplane parses:
Ethernet [+ 802.1Q / QinQ 802.1ad+802.1Q] + IPv4 + UDP + DHCP
(ref bpf/dhcp_fastpath.c:352-428), and plain IPv4 TCP/UDP/ICMP frames for
the NAT44/QoS/antispoof paths (ref bpf/nat44.c:565-660).
"""
from __future__ import annotations

import ipaddress
import struct

ETH_P_IP = 0x0800
ETH_P_IPV6 = 0x86DD
ETH_P_8021Q = 0x8100
ETH_P_8021AD = 0x88A8

DHCP_DISCOVER, DHCP_OFFER, DHCP_REQUEST, DHCP_DECLINE, DHCP_ACK, DHCP_NAK, \
    DHCP_RELEASE, DHCP_INFORM = range(1, 9)

BOOTREQUEST, BOOTREPLY = 1, 2
DHCP_MAGIC = 0x63825363
DHCP_SERVER_PORT, DHCP_CLIENT_PORT = 67, 68
BROADCAST_MAC = b"\xff" * 6


def ip2u32(ip) -> int:
    """dotted quad -> u32 in NETWORK byte order as a host int (big-endian
    bytes interpreted big-endian, i.e. 10.0.0.1 -> 0x0A000001)."""
    return int(ipaddress.IPv4Address(ip))


def u32_to_ip(v: int) -> str:
    return str(ipaddress.IPv4Address(v & 0xFFFFFFFF))


def mac_bytes(mac) -> bytes:
    if isinstance(mac, bytes):
        return mac
    if isinstance(mac, int):
        return mac.to_bytes(6, "big")
    return bytes(int(x, 16) for x in mac.split(":"))


def ipv4_header(src: int, dst: int, payload_len: int, proto: int = 17,
                ttl: int = 64, ident: int = 0) -> bytes:
    """20-byte IPv4 header with correct checksum."""
    total = 20 + payload_len
    hdr = struct.pack(">BBHHHBBH", 0x45, 0, total, ident, 0, ttl, proto, 0)
    hdr += struct.pack(">II", src, dst)
    csum = ipv4_checksum(hdr)
    return hdr[:10] + struct.pack(">H", csum) + hdr[12:]


def ipv4_checksum(hdr20: bytes) -> int:
    s = 0
    for i in range(0, 20, 2):
        s += struct.unpack(">H", hdr20[i:i + 2])[0]
    s = (s & 0xFFFF) + (s >> 16)
    s = (s & 0xFFFF) + (s >> 16)
    return (~s) & 0xFFFF


def l4_checksum(src: int, dst: int, proto: int, l4: bytes) -> int:
    """TCP/UDP checksum over pseudo-header + segment."""
    pseudo = struct.pack(">IIBBH", src, dst, 0, proto, len(l4))
    data = pseudo + l4
    if len(data) & 1:
        data += b"\x00"
    s = 0
    for i in range(0, len(data), 2):
        s += struct.unpack(">H", data[i:i + 2])[0]
    s = (s & 0xFFFF) + (s >> 16)
    s = (s & 0xFFFF) + (s >> 16)
    v = (~s) & 0xFFFF
    return v


def eth_header(dst: bytes, src: bytes, proto: int,
               s_tag: int = 0, c_tag: int = 0) -> bytes:
    """Ethernet header, optionally 802.1Q (c_tag only) or QinQ (both)."""
    hdr = dst + src
    if s_tag and c_tag:
        hdr += struct.pack(">HH", ETH_P_8021AD, s_tag)
        hdr += struct.pack(">HH", ETH_P_8021Q, c_tag)
    elif s_tag or c_tag:
        hdr += struct.pack(">HH", ETH_P_8021Q, s_tag or c_tag)
    hdr += struct.pack(">H", proto)
    return hdr


def build_dhcp_request(client_mac, msg_type: int = DHCP_DISCOVER,
                       xid: int = 0x12345678, *, s_tag: int = 0,
                       c_tag: int = 0, giaddr: int = 0, ciaddr: int = 0,
                       broadcast: bool = False, circuit_id: bytes = b"",
                       src_mac=None, extra_opts: bytes = b"",
                       pad_before_53: int = 0) -> bytes:
    """Full Ethernet frame carrying a DHCP DISCOVER/REQUEST.

    Mirrors the shapes the reference fast path accepts
    (bpf/dhcp_fastpath.c:619-695): optional VLAN/QinQ tags, optional
    option 82 circuit-id, optional relay giaddr.
    """
    cmac = mac_bytes(client_mac)
    smac = mac_bytes(src_mac) if src_mac is not None else cmac
    flags = 0x8000 if broadcast else 0
    dhcp = struct.pack(">BBBBIHHIIII", BOOTREQUEST, 1, 6, 0, xid, 0, flags,
                       ciaddr, 0, 0, giaddr)
    dhcp += cmac + b"\x00" * 10           # chaddr (16)
    dhcp += b"\x00" * 64                  # sname
    dhcp += b"\x00" * 128                 # file
    dhcp += struct.pack(">I", DHCP_MAGIC)
    opts = b"\x00" * pad_before_53
    opts += bytes([53, 1, msg_type])
    if circuit_id:
        sub = bytes([1, len(circuit_id)]) + circuit_id
        opts += bytes([82, len(sub)]) + sub
    opts += extra_opts
    opts += bytes([255])
    dhcp += opts

    udp_len = 8 + len(dhcp)
    src_ip = ciaddr
    dst_ip = 0xFFFFFFFF if giaddr == 0 else 0  # direct requests broadcast
    udp = struct.pack(">HHHH", DHCP_CLIENT_PORT, DHCP_SERVER_PORT, udp_len, 0)
    ip = ipv4_header(src_ip, dst_ip, udp_len, proto=17)
    eth = eth_header(BROADCAST_MAC, smac, ETH_P_IP, s_tag=s_tag, c_tag=c_tag)
    return eth + ip + udp + dhcp


def build_ipv4(src_mac, dst_mac, src_ip: int, dst_ip: int, proto: int = 17,
               sport: int = 40000, dport: int = 80, payload: bytes = b"",
               tcp_flags: int = 0x10, icmp_id: int = 0,
               pad_to: int = 0) -> bytes:
    """Plain IPv4 frame (UDP/TCP/ICMP) for NAT/QoS/antispoof paths."""
    if proto == 17:
        l4 = struct.pack(">HHHH", sport, dport, 8 + len(payload), 0) + payload
        ck = l4_checksum(src_ip, dst_ip, 17, l4)
        l4 = l4[:6] + struct.pack(">H", ck) + l4[8:]
    elif proto == 6:
        l4 = struct.pack(">HHIIBBHHH", sport, dport, 1, 1, 0x50, tcp_flags,
                         8192, 0, 0) + payload
        ck = l4_checksum(src_ip, dst_ip, 6, l4)
        l4 = l4[:16] + struct.pack(">H", ck) + l4[18:]
    elif proto == 1:
        body = struct.pack(">BBHHH", 8, 0, 0, icmp_id, 1) + payload
        s = 0
        d = body if len(body) % 2 == 0 else body + b"\x00"
        for i in range(0, len(d), 2):
            s += struct.unpack(">H", d[i:i + 2])[0]
        s = (s & 0xFFFF) + (s >> 16)
        s = (s & 0xFFFF) + (s >> 16)
        l4 = body[:2] + struct.pack(">H", (~s) & 0xFFFF) + body[4:]
    else:
        l4 = payload
    ip = ipv4_header(src_ip, dst_ip, len(l4), proto=proto)
    frame = eth_header(mac_bytes(dst_mac), mac_bytes(src_mac), ETH_P_IP) + ip + l4
    if pad_to and len(frame) < pad_to:
        frame += b"\x00" * (pad_to - len(frame))
    return frame


# ------------------------------------------------------------- parsing
class ParsedDHCP:
    __slots__ = ("eth_dst", "eth_src", "s_tag", "c_tag", "vlan_offset",
                 "src_ip", "dst_ip", "ip_checksum_ok", "sport", "dport",
                 "op", "xid", "flags", "ciaddr", "yiaddr", "siaddr",
                 "giaddr", "chaddr", "options", "msg_type")


def parse_dhcp_frame(frame: bytes) -> ParsedDHCP:
    """Parse an Ethernet+[VLAN]+IPv4+UDP+DHCP frame (for test assertions)."""
    p = ParsedDHCP()
    p.eth_dst, p.eth_src = frame[0:6], frame[6:12]
    off = 12
    proto = struct.unpack(">H", frame[off:off + 2])[0]
    p.s_tag = p.c_tag = 0
    p.vlan_offset = 0
    if proto in (ETH_P_8021Q, ETH_P_8021AD):
        p.s_tag = struct.unpack(">H", frame[off + 2:off + 4])[0] & 0xFFF
        proto = struct.unpack(">H", frame[off + 4:off + 6])[0]
        off += 4
        p.vlan_offset = 4
        if proto == ETH_P_8021Q:
            p.c_tag = struct.unpack(">H", frame[off + 2:off + 4])[0] & 0xFFF
            proto = struct.unpack(">H", frame[off + 4:off + 6])[0]
            off += 4
            p.vlan_offset = 8
    assert proto == ETH_P_IP, f"not IPv4: {proto:#x}"
    off += 2
    ip = frame[off:off + 20]
    p.ip_checksum_ok = ipv4_checksum(ip[:10] + b"\x00\x00" + ip[12:]) == \
        struct.unpack(">H", ip[10:12])[0]
    p.src_ip, p.dst_ip = struct.unpack(">II", ip[12:20])
    ihl = (ip[0] & 0xF) * 4
    off += ihl
    p.sport, p.dport = struct.unpack(">HH", frame[off:off + 4])
    off += 8
    d = frame[off:]
    p.op = d[0]
    p.xid = struct.unpack(">I", d[4:8])[0]
    p.flags = struct.unpack(">H", d[10:12])[0]
    p.ciaddr, p.yiaddr, p.siaddr, p.giaddr = struct.unpack(">IIII", d[12:28])
    p.chaddr = d[28:34]
    magic = struct.unpack(">I", d[236:240])[0]
    assert magic == DHCP_MAGIC
    p.options = {}
    i = 240
    p.msg_type = 0
    while i < len(d):
        code = d[i]
        if code == 0:
            i += 1
            continue
        if code == 255:
            break
        ln = d[i + 1]
        p.options[code] = d[i + 2:i + 2 + ln]
        i += 2 + ln
    if 53 in p.options:
        p.msg_type = p.options[53][0]
    return p
