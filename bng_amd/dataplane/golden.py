"""CPU golden-model dataplane.

Packet-exact reference implementation of the four dataplane kernels:

  * DHCP fast path   (ref bpf/dhcp_fastpath.c:619-813)
  * NAT44 SNAT/DNAT  (ref bpf/nat44.c:565-948)
  * QoS token bucket (ref bpf/qos_ratelimit.c:126-222)
  * Antispoof uRPF   (ref bpf/antispoof.c:189-293)

Every HIP kernel is differential-tested against this model on random packet
batches (tests/test_kernels_gpu.py), the analog of the reference testing its
BPF programs through the kernel verifier + unit tests.

Reference quirks are preserved deliberately where the control plane depends
on them (e.g. antispoof's binding+LOOSE fallthrough, the EIM port-collision
heuristic); deviations are design upgrades shared by golden model and HIP
kernel alike (full TLV option scan instead of the reference's fixed-offset
verifier workaround, dhcp_fastpath.c:216-250).
"""
from __future__ import annotations

import struct
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Tuple

from . import abi
from .packets import (ETH_P_8021AD, ETH_P_8021Q, ETH_P_IP, ETH_P_IPV6,
                      ipv4_checksum)

# verdicts
PASS, TX, DROP, FWD = abi.PASS, abi.TX, abi.DROP, abi.FWD


# ------------------------------------------------------------- table rows
@dataclass
class SubRecord:
    pool_id: int
    allocated_ip: int          # network-order u32
    lease_expiry: int          # unix seconds
    vlan_id: int = 0
    client_class: int = 0
    flags: int = 0


@dataclass
class PoolRecord:
    network: int
    prefix_len: int
    gateway: int
    dns_primary: int = 0
    dns_secondary: int = 0
    lease_time: int = 3600


@dataclass
class NatSessionRec:
    nat_ip: int
    nat_port: int              # network order
    orig_port: int             # network order
    orig_ip: int
    dest_ip: int
    dest_port: int
    state: int = abi.NAT_NEW
    is_hairpin: int = 0
    last_seen: int = 0
    created: int = 0
    packets_out: int = 0
    packets_in: int = 0
    bytes_out: int = 0
    bytes_in: int = 0
    protocol: int = 0


@dataclass
class EimRec:
    external_ip: int
    external_port: int         # HOST order (ref eim_mapping nat44.c:112-120)
    created: int = 0
    last_used: int = 0
    ref_count: int = 1
    flags: int = 0


@dataclass
class SubnatRec:
    public_ip: int
    port_start: int
    port_end: int
    next_port: int
    subscriber_id: int = 0
    sessions_active: int = 0
    sessions_total: int = 0
    bytes_out: int = 0
    bytes_in: int = 0
    ports_in_use: int = 0
    allocated_at: int = 0


@dataclass
class QosBucketRec:
    rate_bps: int
    burst_bytes: int
    tokens: int = 0
    last_update: int = 0
    priority: int = 0


@dataclass
class BindingRec:
    ipv4_addr: int = 0
    ipv4_valid: int = 0
    ipv6_addr: bytes = b"\x00" * 16
    ipv6_valid: int = 0
    mode: int = abi.AS_DISABLED


DEFAULT_PRIVATE_RANGES = [
    # is_private_ip inline ranges, ref nat44.c:340-363
    (0x0A000000, 0xFF000000),                    # 10.0.0.0/8
    (0xAC100000, 0xFFF00000),                    # 172.16.0.0/12
    (0xC0A80000, 0xFFFF0000),                    # 192.168.0.0/16
    (0x64400000, 0xFFC00000),                    # 100.64.0.0/10 CGNAT
]


class GoldenDataplane:
    """Semantic reference for the GPU dataplane; one instance = one shard."""

    def __init__(self, now_ns: int = 0):
        self.now_ns = now_ns
        # DHCP fast path tables
        self.subscribers: Dict[int, SubRecord] = {}   # tagged u64 key
        self.pools: Dict[int, PoolRecord] = {}
        self.server_mac = b"\x00" * 6
        self.server_ip = 0
        self.dhcp_stats = [0] * abi.DHCP_NSTATS
        # NAT tables
        self.nat_sessions: Dict[Tuple, NatSessionRec] = {}
        self.nat_reverse: Dict[Tuple, Tuple] = {}
        self.eim: Dict[Tuple, EimRec] = {}
        self.subnat: Dict[int, SubnatRec] = {}
        self.hairpin_ips: set = set()
        self.alg_ports: set = set()                   # (port, proto)
        self.nat_flags = abi.NAT_FLAG_EIM
        self.private_ranges: List[Tuple[int, int]] = list(DEFAULT_PRIVATE_RANGES)
        self.nat_stats = [0] * abi.NAT_NSTATS
        self.nat_log: List[dict] = []
        # QoS
        self.qos_egress: Dict[int, QosBucketRec] = {}
        self.qos_ingress: Dict[int, QosBucketRec] = {}
        self.qos_stats = [0] * abi.QOS_NSTATS
        # antispoof
        self.bindings: Dict[int, BindingRec] = {}
        self.as_default_mode = abi.AS_DISABLED
        self.as_log_violations = 0
        self.allowed_ranges: List[Tuple[int, int]] = []
        self.as_stats = [0] * abi.AS_NSTATS
        self.spoof_events: List[dict] = []

    # ------------------------------------------------------------- helpers
    @property
    def now_sec(self) -> int:
        return self.now_ns // 1_000_000_000

    def _is_private(self, ip: int) -> bool:
        for net, mask in self.private_ranges:
            if (ip & mask) == net:
                return True
        return False

    # =================================================== DHCP fast path K1
    def dhcp_fastpath(self, frame: bytearray) -> Tuple[int, int]:
        """Process one frame; returns (verdict, out_len).  On TX the frame
        is rewritten in place into the OFFER/ACK (ref dhcp_fastpath.c:619)."""
        n = len(frame)
        if n < 14:
            return PASS, n
        proto = struct.unpack_from(">H", frame, 12)[0]
        off = 14
        vlan_offset = 0
        s_tag = c_tag = 0
        tagged = False
        if proto in (ETH_P_8021Q, ETH_P_8021AD):
            if n < off + 4:
                return PASS, n
            tagged = True
            s_tag = struct.unpack_from(">H", frame, off)[0] & 0xFFF
            proto = struct.unpack_from(">H", frame, off + 2)[0]
            off += 4
            vlan_offset = 4
            self.dhcp_stats[abi.ST_VLAN_PACKETS] += 1
            if proto == ETH_P_8021Q:
                if n < off + 4:
                    return PASS, n
                c_tag = struct.unpack_from(">H", frame, off)[0] & 0xFFF
                proto = struct.unpack_from(">H", frame, off + 2)[0]
                off += 4
                vlan_offset = 8
        if proto != ETH_P_IP or n < off + 20:
            return PASS, n
        ip_off = off
        if frame[ip_off + 9] != 17:      # UDP
            return PASS, n
        ihl = (frame[ip_off] & 0xF) * 4
        udp_off = ip_off + ihl
        if n < udp_off + 8:
            return PASS, n
        dport = struct.unpack_from(">H", frame, udp_off + 2)[0]
        if dport != 67:
            return PASS, n
        dhcp_off = udp_off + 8
        if n < dhcp_off + 240:
            return PASS, n
        if frame[dhcp_off] != 1:         # BOOTREQUEST
            return PASS, n
        magic = struct.unpack_from(">I", frame, dhcp_off + 236)[0]
        if magic != 0x63825363:
            return PASS, n

        self.dhcp_stats[abi.ST_TOTAL_REQUESTS] += 1

        msg_type, circuit_id = self._scan_options(frame, dhcp_off + 240, n)
        if msg_type not in (1, 3):       # DISCOVER / REQUEST only
            self.dhcp_stats[abi.ST_FASTPATH_MISSES] += 1
            return PASS, n

        # 3-way lookup priority: VLAN -> circuit-ID -> MAC
        # (ref dhcp_fastpath.c:647-687)
        sub: Optional[SubRecord] = None
        if tagged:
            sub = self.subscribers.get(abi.vlan_key(s_tag, c_tag))
        if sub is None and circuit_id:
            sub = self.subscribers.get(abi.circuit_key(circuit_id))
            if sub is not None:
                self.dhcp_stats[abi.ST_OPTION82_PRESENT] += 1
        if sub is None:
            chaddr = bytes(frame[dhcp_off + 28:dhcp_off + 34])
            sub = self.subscribers.get(abi.mac_to_u64(chaddr))
        if sub is None:
            self.dhcp_stats[abi.ST_FASTPATH_MISSES] += 1
            return PASS, n
        if self.now_sec > sub.lease_expiry:
            self.dhcp_stats[abi.ST_CACHE_EXPIRED] += 1
            return PASS, n
        pool = self.pools.get(sub.pool_id)
        if pool is None:
            self.dhcp_stats[abi.ST_ERRORS] += 1
            return PASS, n
        self.dhcp_stats[abi.ST_FASTPATH_HITS] += 1

        reply_type = 2 if msg_type == 1 else 5   # OFFER : ACK
        giaddr = struct.unpack_from(">I", frame, dhcp_off + 24)[0]
        server_ip = self.server_ip if self.server_ip else pool.gateway

        if giaddr != 0:
            # relayed: unicast to relay agent (ref :726-743)
            frame[0:6] = frame[6:12]
            frame[6:12] = self.server_mac
            struct.pack_into(">I", frame, ip_off + 12, server_ip)
            struct.pack_into(">I", frame, ip_off + 16, giaddr)
            daddr_for_udp = giaddr
            struct.pack_into(">HH", frame, udp_off, 67, 67)
            self.dhcp_stats[abi.ST_UNICAST_REPLIES] += 1
        else:
            flags = struct.unpack_from(">H", frame, dhcp_off + 10)[0]
            ciaddr = struct.unpack_from(">I", frame, dhcp_off + 12)[0]
            chaddr = bytes(frame[dhcp_off + 28:dhcp_off + 34])
            use_broadcast = bool(flags & 0x8000) or ciaddr == 0
            if use_broadcast:
                frame[0:6] = b"\xff" * 6
                self.dhcp_stats[abi.ST_BROADCAST_REPLIES] += 1
            else:
                frame[0:6] = chaddr
                self.dhcp_stats[abi.ST_UNICAST_REPLIES] += 1
            frame[6:12] = self.server_mac
            struct.pack_into(">I", frame, ip_off + 12, server_ip)
            struct.pack_into(">I", frame, ip_off + 16, 0xFFFFFFFF)
            struct.pack_into(">HH", frame, udp_off, 67, 68)

        frame[ip_off + 8] = 64           # TTL
        struct.pack_into(">H", frame, udp_off + 6, 0)  # UDP csum 0

        # DHCP reply fixed fields (ref :759-766)
        frame[dhcp_off] = 2              # BOOTREPLY
        frame[dhcp_off + 3] = 0          # hops
        struct.pack_into(">I", frame, dhcp_off + 16, sub.allocated_ip)  # yiaddr
        struct.pack_into(">I", frame, dhcp_off + 20, server_ip)         # siaddr
        frame[dhcp_off + 44:dhcp_off + 236] = b"\x00" * 192  # sname+file

        # grow the buffer for the reply options if the request was shorter
        # (the XDP analog is bpf_xdp_adjust_tail, ref :799-809; the GPU path
        # has fixed-stride slots with headroom instead)
        need = dhcp_off + 240 + 64
        if len(frame) < need:
            frame.extend(b"\x00" * (need - len(frame)))
        opt_len = self._build_reply_options(
            frame, dhcp_off + 240, reply_type, pool, server_ip)

        dhcp_len = 240 + opt_len
        udp_len = 8 + dhcp_len
        ip_len = 20 + udp_len
        total = 14 + vlan_offset + ip_len
        struct.pack_into(">H", frame, ip_off + 2, ip_len)
        struct.pack_into(">H", frame, udp_off + 4, udp_len)
        struct.pack_into(">H", frame, ip_off + 10, 0)
        csum = ipv4_checksum(bytes(frame[ip_off:ip_off + 20]))
        struct.pack_into(">H", frame, ip_off + 10, csum)
        return TX, total

    @staticmethod
    def _scan_options(frame, opt_off: int, end: int) -> Tuple[int, bytes]:
        """Full TLV scan for option 53 and option-82 circuit-id.  Upgrade
        over the reference's fixed-offset scan (dhcp_fastpath.c:216-323);
        bounded like its MAX_DHCP_OPTIONS_ITER/SCAN_LEN (maps.h:19-22)."""
        msg_type = 0
        circuit_id = b""
        i = opt_off
        limit = min(end, opt_off + 312)
        iters = 0
        while i < limit and iters < 64:
            iters += 1
            code = frame[i]
            if code == 0:
                i += 1
                continue
            if code == 255:
                break
            if i + 1 >= limit:
                break
            ln = frame[i + 1]
            if i + 2 + ln > limit:
                break
            if code == 53 and ln == 1:
                msg_type = frame[i + 2]
            elif code == 82:
                j = i + 2
                sub_end = i + 2 + ln
                while j + 2 <= sub_end:
                    sc, sl = frame[j], frame[j + 1]
                    if j + 2 + sl > sub_end:
                        break
                    if sc == 1 and 0 < sl <= 32:
                        circuit_id = bytes(frame[j + 2:j + 2 + sl])
                    j += 2 + sl
            i += 2 + ln
        return msg_type, circuit_id

    @staticmethod
    def _build_reply_options(frame, off: int, reply_type: int,
                             pool: PoolRecord, server_ip: int) -> int:
        """Options 53/54/51/1/3/6/58/59/255 (ref build_dhcp_options :519-602)."""
        o = off
        frame[o:o + 3] = bytes([53, 1, reply_type]); o += 3
        frame[o:o + 2] = bytes([54, 4]); struct.pack_into(">I", frame, o + 2, server_ip); o += 6
        frame[o:o + 2] = bytes([51, 4]); struct.pack_into(">I", frame, o + 2, pool.lease_time); o += 6
        mask = 0 if pool.prefix_len == 0 else (0xFFFFFFFF << (32 - min(pool.prefix_len, 32))) & 0xFFFFFFFF
        frame[o:o + 2] = bytes([1, 4]); struct.pack_into(">I", frame, o + 2, mask); o += 6
        frame[o:o + 2] = bytes([3, 4]); struct.pack_into(">I", frame, o + 2, pool.gateway); o += 6
        if pool.dns_primary:
            dns_len = 8 if pool.dns_secondary else 4
            frame[o:o + 2] = bytes([6, dns_len])
            struct.pack_into(">I", frame, o + 2, pool.dns_primary)
            if pool.dns_secondary:
                struct.pack_into(">I", frame, o + 6, pool.dns_secondary)
            o += 2 + dns_len
        frame[o:o + 2] = bytes([58, 4]); struct.pack_into(">I", frame, o + 2, pool.lease_time // 2); o += 6
        frame[o:o + 2] = bytes([59, 4]); struct.pack_into(">I", frame, o + 2, (pool.lease_time * 7) // 8); o += 6
        frame[o] = 255; o += 1
        return o - off

    # ========================================================= NAT44 K2
    @staticmethod
    def _upd_csum(csum: int, old: int, new: int) -> int:
        """Incremental checksum for a 32-bit field (ref update_csum :384-391)."""
        s = (~csum) & 0xFFFF
        s += (~old & 0xFFFF) + ((~old >> 16) & 0xFFFF)
        s += (new & 0xFFFF) + (new >> 16)
        s = (s & 0xFFFF) + (s >> 16)
        s = (s & 0xFFFF) + (s >> 16)
        return (~s) & 0xFFFF

    @staticmethod
    def _upd_csum16(csum: int, old: int, new: int) -> int:
        s = (~csum) & 0xFFFF
        s += (~old & 0xFFFF) + (new & 0xFFFF)
        s = (s & 0xFFFF) + (s >> 16)
        s = (s & 0xFFFF) + (s >> 16)
        return (~s) & 0xFFFF

    def _alloc_port(self, blk: SubnatRec, preserve_parity: bool,
                    orig_port_host: int, internal_ip: int, proto: int) -> int:
        """RFC 6431 port rotor with EIM-collision check
        (ref allocate_port_from_block nat44.c:408-466)."""
        orig_parity = orig_port_host & 1
        for _ in range(64):
            port = blk.next_port & 0xFFFF
            blk.next_port += 1
            if port > blk.port_end:
                port = blk.port_start
            if blk.next_port > blk.port_end:
                blk.next_port = blk.port_start
            if preserve_parity and (port & 1) != orig_parity:
                continue
            if (internal_ip, port, proto) in self.eim:
                continue
            return port
        return 0

    def _parse_l3l4(self, frame) -> Optional[dict]:
        n = len(frame)
        if n < 34 or struct.unpack_from(">H", frame, 12)[0] != ETH_P_IP:
            return None
        ip_off = 14
        ihl = (frame[ip_off] & 0xF) * 4
        proto = frame[ip_off + 9]
        saddr, daddr = struct.unpack_from(">II", frame, ip_off + 12)
        l4 = ip_off + ihl
        r = dict(ip_off=ip_off, l4=l4, proto=proto, saddr=saddr, daddr=daddr)
        if proto == 6 and n >= l4 + 20:
            r["sport"], r["dport"] = struct.unpack_from(">HH", frame, l4)
            r["tcp_flags"] = frame[l4 + 13]
        elif proto == 17 and n >= l4 + 8:
            r["sport"], r["dport"] = struct.unpack_from(">HH", frame, l4)
        elif proto == 1 and n >= l4 + 8:
            r["icmp_id"] = struct.unpack_from(">H", frame, l4 + 4)[0]
        else:
            return r if proto not in (6, 17, 1) else None
        return r

    def nat44_egress(self, frame: bytearray) -> int:
        """SNAT (ref nat44_egress nat44.c:565-802).  Returns verdict."""
        h = self._parse_l3l4(frame)
        if h is None:
            return FWD
        if not self._is_private(h["saddr"]):
            return FWD
        blk = self.subnat.get(h["saddr"])
        if blk is None:
            self.nat_stats[abi.NS_PASSED] += 1
            return PASS
        proto = h["proto"]
        if proto == 6:
            sport, dport = h["sport"], h["dport"]
            if self.nat_flags & (abi.NAT_FLAG_ALG_FTP | abi.NAT_FLAG_ALG_SIP):
                if (dport, 6) in self.alg_ports:
                    self.nat_stats[abi.NS_ALG_TRIGGERS] += 1
                    self._log(abi.LOG_ALG_TRIGGER, blk.subscriber_id,
                              h["saddr"], 0, sport, 0, h["daddr"], dport, 6)
                    return PASS
        elif proto == 17:
            sport, dport = h["sport"], h["dport"]
            if self.nat_flags & abi.NAT_FLAG_ALG_SIP:
                if (dport, 17) in self.alg_ports:
                    self.nat_stats[abi.NS_ALG_TRIGGERS] += 1
                    self._log(abi.LOG_ALG_TRIGGER, blk.subscriber_id,
                              h["saddr"], 0, sport, 0, h["daddr"], dport, 17)
                    return PASS
        elif proto == 1:
            sport, dport = h["icmp_id"], 0
        else:
            return FWD

        is_hairpin = 0
        if (self.nat_flags & abi.NAT_FLAG_HAIRPIN) and h["daddr"] in self.hairpin_ips:
            is_hairpin = 1
            self.nat_stats[abi.NS_HAIRPIN] += 1

        key = (h["saddr"], h["daddr"], sport, dport, proto)
        sess = self.nat_sessions.get(key)
        now = self.now_ns
        if sess is not None:
            nat_ip, nat_port = sess.nat_ip, sess.nat_port
            sess.last_seen = now
            sess.packets_out += 1
            sess.bytes_out += len(frame)
        else:
            nat_ip = nat_port = None
            if self.nat_flags & abi.NAT_FLAG_EIM:
                ek = (h["saddr"], sport, proto)
                eim = self.eim.get(ek)
                if eim is not None:
                    eim.last_used = now
                    eim.ref_count += 1
                    self.nat_stats[abi.NS_EIM_HITS] += 1
                    nat_ip, nat_port = eim.external_ip, eim.external_port
                else:
                    p = self._alloc_port(blk, bool(self.nat_flags & abi.NAT_FLAG_PARITY),
                                         sport, h["saddr"], proto)
                    if p == 0:
                        self.nat_stats[abi.NS_PORT_EXHAUSTION] += 1
                        self.nat_stats[abi.NS_DROPPED] += 1
                        self._log(abi.LOG_PORT_EXHAUSTION, blk.subscriber_id,
                                  h["saddr"], blk.public_ip, sport, 0,
                                  h["daddr"], dport, proto)
                        return DROP
                    self.eim[ek] = EimRec(blk.public_ip, p, now, now, 1)
                    self.nat_stats[abi.NS_EIM_MISSES] += 1
                    nat_ip, nat_port = blk.public_ip, p
            if nat_ip is None:
                p = self._alloc_port(blk, bool(self.nat_flags & abi.NAT_FLAG_PARITY),
                                     sport, h["saddr"], proto)
                if p == 0:
                    self.nat_stats[abi.NS_PORT_EXHAUSTION] += 1
                    self.nat_stats[abi.NS_DROPPED] += 1
                    self._log(abi.LOG_PORT_EXHAUSTION, blk.subscriber_id,
                              h["saddr"], blk.public_ip, sport, 0,
                              h["daddr"], dport, proto)
                    return DROP
                nat_ip, nat_port = blk.public_ip, p
            sess = NatSessionRec(nat_ip, nat_port, sport, h["saddr"],
                                 h["daddr"], dport, abi.NAT_NEW, is_hairpin,
                                 now, now, 1, 0, len(frame), 0, proto)
            self.nat_sessions[key] = sess
            rev = (h["daddr"], nat_ip, dport, nat_port, proto)
            self.nat_reverse[rev] = key
            blk.sessions_active += 1
            blk.sessions_total += 1
            self.nat_stats[abi.NS_SESS_CREATED] += 1
            self._log(abi.LOG_SESSION_CREATE, blk.subscriber_id, h["saddr"],
                      nat_ip, sport, nat_port, h["daddr"], dport, proto,
                      is_hairpin)

        # rewrite (ref :752-798)
        ip_off, l4 = h["ip_off"], h["l4"]
        old_ip = h["saddr"]
        struct.pack_into(">I", frame, ip_off + 12, nat_ip)
        ipck = struct.unpack_from(">H", frame, ip_off + 10)[0]
        struct.pack_into(">H", frame, ip_off + 10,
                         self._upd_csum(ipck, old_ip, nat_ip))
        if proto == 6:
            old_port = struct.unpack_from(">H", frame, l4)[0]
            struct.pack_into(">H", frame, l4, nat_port)
            ck = struct.unpack_from(">H", frame, l4 + 16)[0]
            ck = self._upd_csum(ck, old_ip, nat_ip)
            ck = self._upd_csum16(ck, old_port, nat_port)
            struct.pack_into(">H", frame, l4 + 16, ck)
        elif proto == 17:
            old_port = struct.unpack_from(">H", frame, l4)[0]
            struct.pack_into(">H", frame, l4, nat_port)
            ck = struct.unpack_from(">H", frame, l4 + 6)[0]
            if ck != 0:
                ck = self._upd_csum(ck, old_ip, nat_ip)
                ck = self._upd_csum16(ck, old_port, nat_port)
                if ck == 0:
                    ck = 0xFFFF
                struct.pack_into(">H", frame, l4 + 6, ck)
        elif proto == 1:
            old_id = struct.unpack_from(">H", frame, l4 + 4)[0]
            struct.pack_into(">H", frame, l4 + 4, nat_port)
            ck = struct.unpack_from(">H", frame, l4 + 2)[0]
            struct.pack_into(">H", frame, l4 + 2,
                             self._upd_csum16(ck, old_id, nat_port))
        self.nat_stats[abi.NS_SNAT] += 1
        return FWD

    def nat44_ingress(self, frame: bytearray) -> int:
        """DNAT (ref nat44_ingress nat44.c:805-948)."""
        h = self._parse_l3l4(frame)
        if h is None:
            return FWD
        proto = h["proto"]
        if proto == 6 or proto == 17:
            sport, dport = h["sport"], h["dport"]
        elif proto == 1:
            sport, dport = 0, h["icmp_id"]
        else:
            return FWD
        rev = (h["saddr"], h["daddr"], sport, dport, proto)
        orig_key = self.nat_reverse.get(rev)
        if orig_key is None:
            self.nat_stats[abi.NS_PASSED] += 1
            return FWD
        sess = self.nat_sessions.get(orig_key)
        if sess is None:
            del self.nat_reverse[rev]
            self.nat_stats[abi.NS_SESS_EXPIRED] += 1
            return FWD
        sess.last_seen = self.now_ns
        sess.packets_in += 1
        sess.bytes_in += len(frame)
        if proto == 6:
            fl = h["tcp_flags"]
            if fl & 0x05:                       # FIN|RST
                sess.state = abi.NAT_CLOSING
            elif sess.state == abi.NAT_NEW and fl & 0x10:  # ACK
                sess.state = abi.NAT_ESTABLISHED

        ip_off, l4 = h["ip_off"], h["l4"]
        old_ip, new_ip = h["daddr"], sess.orig_ip
        struct.pack_into(">I", frame, ip_off + 16, new_ip)
        ipck = struct.unpack_from(">H", frame, ip_off + 10)[0]
        struct.pack_into(">H", frame, ip_off + 10,
                         self._upd_csum(ipck, old_ip, new_ip))
        new_port = sess.orig_port
        if proto == 6:
            old_port = struct.unpack_from(">H", frame, l4 + 2)[0]
            struct.pack_into(">H", frame, l4 + 2, new_port)
            ck = struct.unpack_from(">H", frame, l4 + 16)[0]
            ck = self._upd_csum(ck, old_ip, new_ip)
            ck = self._upd_csum16(ck, old_port, new_port)
            struct.pack_into(">H", frame, l4 + 16, ck)
        elif proto == 17:
            old_port = struct.unpack_from(">H", frame, l4 + 2)[0]
            struct.pack_into(">H", frame, l4 + 2, new_port)
            ck = struct.unpack_from(">H", frame, l4 + 6)[0]
            if ck != 0:
                ck = self._upd_csum(ck, old_ip, new_ip)
                ck = self._upd_csum16(ck, old_port, new_port)
                if ck == 0:
                    ck = 0xFFFF
                struct.pack_into(">H", frame, l4 + 6, ck)
        elif proto == 1:
            old_id = struct.unpack_from(">H", frame, l4 + 4)[0]
            struct.pack_into(">H", frame, l4 + 4, new_port)
            ck = struct.unpack_from(">H", frame, l4 + 2)[0]
            struct.pack_into(">H", frame, l4 + 2,
                             self._upd_csum16(ck, old_id, new_port))
        self.nat_stats[abi.NS_DNAT] += 1
        return FWD

    def _log(self, ev, sub_id, priv_ip, pub_ip, priv_port, pub_port,
             dest_ip, dest_port, proto, flags=0):
        self.nat_log.append(dict(
            timestamp=self.now_ns, event_type=ev, subscriber_id=sub_id,
            private_ip=priv_ip, public_ip=pub_ip, private_port=priv_port,
            public_port=pub_port, dest_ip=dest_ip, dest_port=dest_port,
            protocol=proto, flags=flags))

    # =========================================================== QoS K3
    def _tb_check(self, tb: QosBucketRec, pkt_len: int) -> bool:
        """Token bucket (ref token_bucket_check qos_ratelimit.c:70-104)."""
        if tb.rate_bps == 0:
            return True
        elapsed = self.now_ns - tb.last_update
        tb.tokens += (elapsed * (tb.rate_bps // 8)) // 1_000_000_000
        if tb.tokens > tb.burst_bytes:
            tb.tokens = tb.burst_bytes
        tb.last_update = self.now_ns
        if tb.tokens >= pkt_len:
            tb.tokens -= pkt_len
            return True
        return False

    def qos(self, frame: bytes, direction: str) -> int:
        """direction 'egress' keys dst IP (download), 'ingress' src IP."""
        if len(frame) < 34 or struct.unpack_from(">H", frame, 12)[0] != ETH_P_IP:
            return FWD
        saddr, daddr = struct.unpack_from(">II", frame, 14 + 12)
        table = self.qos_egress if direction == "egress" else self.qos_ingress
        tb = table.get(daddr if direction == "egress" else saddr)
        if tb is None:
            return FWD
        allowed = self._tb_check(tb, len(frame))
        if allowed:
            self.qos_stats[abi.QS_PKT_PASSED] += 1
            self.qos_stats[abi.QS_BYTES_PASSED] += len(frame)
            return FWD
        self.qos_stats[abi.QS_PKT_DROPPED] += 1
        self.qos_stats[abi.QS_BYTES_DROPPED] += len(frame)
        return DROP

    # ====================================================== antispoof K4
    def antispoof(self, frame: bytes) -> int:
        """uRPF source validation (ref antispoof_ingress antispoof.c:189-293).
        The reference's binding+LOOSE fallthrough quirk (a valid binding in
        LOOSE mode is never range-checked and always violates, :227-235) is
        preserved for behavioral parity."""
        if len(frame) < 14:
            return FWD
        mac = abi.mac_to_u64(bytes(frame[6:12]))
        b = self.bindings.get(mac)
        mode = b.mode if b is not None else self.as_default_mode
        if mode == abi.AS_DISABLED:
            self.as_stats[abi.AS_ALLOWED] += 1
            return FWD
        proto = struct.unpack_from(">H", frame, 12)[0]
        if proto == ETH_P_IP:
            if len(frame) < 34:
                return FWD
            src_ip = struct.unpack_from(">I", frame, 14 + 12)[0]
            allowed = False
            if b is not None and b.ipv4_valid:
                if mode in (abi.AS_STRICT, abi.AS_LOG_ONLY):
                    allowed = src_ip == b.ipv4_addr
            elif mode == abi.AS_LOOSE:
                allowed = any((src_ip & m) == net
                              for net, m in self.allowed_ranges)
            if not allowed:
                if self.as_log_violations:
                    self.spoof_events.append(dict(
                        timestamp=self.now_ns, src_mac=bytes(frame[6:12]),
                        protocol=4, spoofed_ip=src_ip,
                        allowed_ip=b.ipv4_addr if b else 0))
                    self.as_stats[abi.AS_LOGGED] += 1
                if mode == abi.AS_LOG_ONLY:
                    self.as_stats[abi.AS_ALLOWED] += 1
                    return FWD
                self.as_stats[abi.AS_DROPPED] += 1
                self.as_stats[abi.AS_V4_VIOLATIONS] += 1
                return DROP
            self.as_stats[abi.AS_ALLOWED] += 1
            return FWD
        if proto == ETH_P_IPV6:
            if len(frame) < 14 + 40:
                return FWD
            src6 = bytes(frame[14 + 8:14 + 24])
            allowed = False
            if b is not None and b.ipv6_valid:
                allowed = src6 == b.ipv6_addr
            elif mode == abi.AS_LOOSE:
                allowed = True
            if not allowed and mode != abi.AS_LOG_ONLY:
                if self.as_log_violations:
                    self.spoof_events.append(dict(
                        timestamp=self.now_ns, src_mac=bytes(frame[6:12]),
                        protocol=6, spoofed_ip=0, allowed_ip=0))
                    self.as_stats[abi.AS_LOGGED] += 1
                self.as_stats[abi.AS_DROPPED] += 1
                self.as_stats[abi.AS_V6_VIOLATIONS] += 1
                return DROP
            self.as_stats[abi.AS_ALLOWED] += 1
            return FWD
        self.as_stats[abi.AS_ALLOWED] += 1
        return FWD
