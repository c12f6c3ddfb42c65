"""Python mirror of the host<->device table ABI (csrc/bng_abi.h).

The analog of the reference's Go mirror structs (pkg/ebpf/loader.go:21-71),
whose layout is verified against bpf/maps.h by test/ebpf/maps_test.go:15-80.
Here tests/test_abi.py asserts these ctypes layouts against the compiled
extension's sizeof/offsetof report.
"""
from __future__ import annotations

import ctypes as C

# ------------------------------------------------------------------ sizing
MAX_SUBSCRIBERS_LOG2 = 21
MAX_POOLS = 16384
MAX_NAT_SESSIONS_LOG2 = 23
MAX_EIM_LOG2 = 22
MAX_SUBNAT_LOG2 = 21
MAX_QOS_LOG2 = 21
MAX_BINDINGS_LOG2 = 21
MAX_PROBE = 128
MAX_PRIVATE_RANGES = 64
MAX_ALLOWED_RANGES = 256
MAX_HAIRPIN_IPS = 1024
MAX_ALG_PORTS = 64
LOG_RING_LOG2 = 15
SPOOF_RING_LOG2 = 14

# key-space tags (bits 63:62)
KEY_MAC = 0
KEY_VLAN = 1 << 62
KEY_CIRCUIT = 2 << 62
KEY_EMPTY = 0
KEY_TOMBSTONE = (1 << 64) - 1

# verdicts
PASS, TX, DROP, FWD = 0, 1, 2, 3

# DHCP stat indices (order matches ref bpf/maps.h:171-184)
ST_TOTAL_REQUESTS = 0
ST_FASTPATH_HITS = 1
ST_FASTPATH_MISSES = 2
ST_ERRORS = 3
ST_CACHE_EXPIRED = 4
ST_OPTION82_PRESENT = 5
ST_OPTION82_ABSENT = 6
ST_BROADCAST_REPLIES = 7
ST_UNICAST_REPLIES = 8
ST_VLAN_PACKETS = 9
DHCP_NSTATS = 10
DHCP_STAT_NAMES = [
    "total_requests", "fastpath_hits", "fastpath_misses", "errors",
    "cache_expired", "option82_present", "option82_absent",
    "broadcast_replies", "unicast_replies", "vlan_packets",
]

# NAT flags (ref nat44.c:56-62)
NAT_FLAG_EIM = 0x01
NAT_FLAG_EIF = 0x02
NAT_FLAG_HAIRPIN = 0x04
NAT_FLAG_ALG_FTP = 0x08
NAT_FLAG_ALG_SIP = 0x10
NAT_FLAG_PARITY = 0x20
NAT_FLAG_CONTIG = 0x40

# NAT states (ref nat44.c:65-71)
NAT_NEW, NAT_ESTABLISHED, NAT_FIN_WAIT, NAT_CLOSING, NAT_TIME_WAIT = range(5)

# NAT stat indices (ref nat44.c:176-190)
NS_SNAT, NS_DNAT, NS_HAIRPIN, NS_DROPPED, NS_PASSED, NS_SESS_CREATED, \
    NS_SESS_EXPIRED, NS_PORT_EXHAUSTION, NS_EIM_HITS, NS_EIM_MISSES, \
    NS_ALG_TRIGGERS, NS_CT_LOOKUPS, NS_CT_HITS = range(13)
NAT_NSTATS = 13
NAT_STAT_NAMES = [
    "packets_snat", "packets_dnat", "packets_hairpin", "packets_dropped",
    "packets_passed", "sessions_created", "sessions_expired",
    "port_exhaustion", "eim_hits", "eim_misses", "alg_triggers",
    "conntrack_lookups", "conntrack_hits",
]

# NAT log events (ref nat44.c:74-82)
LOG_SESSION_CREATE = 1
LOG_SESSION_DELETE = 2
LOG_PB_ASSIGN = 3
LOG_PB_RELEASE = 4
LOG_PORT_EXHAUSTION = 5
LOG_HAIRPIN = 6
LOG_ALG_TRIGGER = 7

# QoS stat indices (ref qos_ratelimit.c:53-58)
QS_PKT_PASSED, QS_PKT_DROPPED, QS_BYTES_PASSED, QS_BYTES_DROPPED = range(4)
QOS_NSTATS = 4
QOS_STAT_NAMES = ["packets_passed", "packets_dropped",
                  "bytes_passed", "bytes_dropped"]

# antispoof modes (ref antispoof.c:30-33)
AS_DISABLED, AS_STRICT, AS_LOOSE, AS_LOG_ONLY = range(4)

# antispoof stat indices (ref antispoof.c:58-65)
AS_ALLOWED, AS_DROPPED, AS_LOGGED, AS_V4_VIOLATIONS, AS_V6_VIOLATIONS, \
    AS_UNKNOWN_MAC = range(6)
AS_NSTATS = 6
AS_STAT_NAMES = ["packets_allowed", "packets_dropped", "packets_logged",
                 "ipv4_violations", "ipv6_violations", "unknown_mac"]


class SubEntry(C.Structure):
    _fields_ = [("key", C.c_uint64), ("pool_id", C.c_uint32),
                ("allocated_ip", C.c_uint32), ("lease_expiry", C.c_uint64),
                ("vlan_id", C.c_uint16), ("client_class", C.c_uint8),
                ("flags", C.c_uint8), ("_pad", C.c_uint32)]


class IpPool(C.Structure):
    _fields_ = [("network", C.c_uint32), ("gateway", C.c_uint32),
                ("dns_primary", C.c_uint32), ("dns_secondary", C.c_uint32),
                ("lease_time", C.c_uint32), ("prefix_len", C.c_uint8),
                ("valid", C.c_uint8), ("_pad", C.c_uint16),
                ("_pad2", C.c_uint32)]


class ServerConfig(C.Structure):
    _fields_ = [("server_mac", C.c_uint8 * 6), ("_pad", C.c_uint16),
                ("server_ip", C.c_uint32), ("if_index", C.c_uint32)]


class NatTuple(C.Structure):
    _fields_ = [("src_ip", C.c_uint32), ("dst_ip", C.c_uint32),
                ("src_port", C.c_uint16), ("dst_port", C.c_uint16),
                ("protocol", C.c_uint8), ("_pad", C.c_uint8 * 3)]


class NatSession(C.Structure):
    _fields_ = [("sig", C.c_uint64), ("key", NatTuple),
                ("nat_ip", C.c_uint32), ("nat_port", C.c_uint16),
                ("orig_port", C.c_uint16), ("orig_ip", C.c_uint32),
                ("state", C.c_uint8), ("is_hairpin", C.c_uint8),
                ("ready", C.c_uint8), ("_pad", C.c_uint8),
                ("last_seen", C.c_uint64),
                ("created", C.c_uint64), ("packets_out", C.c_uint64),
                ("packets_in", C.c_uint64), ("bytes_out", C.c_uint64),
                ("bytes_in", C.c_uint64), ("_pad2", C.c_uint64 * 5)]


class NatReverse(C.Structure):
    _fields_ = [("sig", C.c_uint64), ("key", NatTuple), ("orig", NatTuple),
                ("ready", C.c_uint8), ("_pad", C.c_uint8 * 7)]


class EimEntry(C.Structure):
    _fields_ = [("sig", C.c_uint64), ("internal_ip", C.c_uint32),
                ("internal_port", C.c_uint16), ("protocol", C.c_uint8),
                ("ready", C.c_uint8), ("external_ip", C.c_uint32),
                ("external_port", C.c_uint16), ("_pad", C.c_uint16),
                ("created", C.c_uint64), ("last_used", C.c_uint64),
                ("ref_count", C.c_uint32), ("flags", C.c_uint32)]


class SubCtx(C.Structure):
    """Merged per-subscriber uplink context: RFC6431 port block (ref
    subscriber_nat nat44.c:157-164) + ingress token bucket (ref
    qos_ingress qos_ratelimit.c:44-50) in one 64-B entry."""
    _fields_ = [("key_ip", C.c_uint32), ("public_ip", C.c_uint32),
                ("port_start", C.c_uint16), ("port_end", C.c_uint16),
                ("qos_valid", C.c_uint8), ("nat_valid", C.c_uint8),
                ("priority", C.c_uint8), ("flags", C.c_uint8),
                ("rate_bps", C.c_uint64), ("tokens", C.c_int64),
                ("last_update", C.c_uint64), ("burst_bytes", C.c_uint32),
                ("next_port", C.c_uint32), ("subscriber_id", C.c_uint32),
                ("sessions_active", C.c_uint32),
                ("sessions_total", C.c_uint32), ("_pad", C.c_uint32)]


CTX_SET_NAT, CTX_SET_QOS, CTX_CLR_QOS, CTX_CLR_NAT = 1, 2, 4, 8


class NatConfig(C.Structure):
    _fields_ = [("flags", C.c_uint32), ("port_range_start", C.c_uint16),
                ("port_range_end", C.c_uint16),
                ("default_ports_per_sub", C.c_uint32),
                ("n_private_ranges", C.c_uint32), ("n_alg_ports", C.c_uint32),
                ("_pad", C.c_uint32),
                ("priv_lo", C.c_uint32 * MAX_PRIVATE_RANGES),
                ("priv_hi", C.c_uint32 * MAX_PRIVATE_RANGES),
                ("alg_key", C.c_uint32 * MAX_ALG_PORTS)]


class NatLogEntry(C.Structure):
    _fields_ = [("timestamp", C.c_uint64), ("event_type", C.c_uint32),
                ("subscriber_id", C.c_uint32), ("private_ip", C.c_uint32),
                ("public_ip", C.c_uint32), ("private_port", C.c_uint16),
                ("public_port", C.c_uint16), ("dest_ip", C.c_uint32),
                ("dest_port", C.c_uint16), ("protocol", C.c_uint8),
                ("flags", C.c_uint8)]


class QosBucket(C.Structure):
    _fields_ = [("key_ip", C.c_uint32), ("valid", C.c_uint8),
                ("priority", C.c_uint8), ("_pad", C.c_uint16),
                ("rate_bps", C.c_uint64), ("tokens", C.c_int64),
                ("last_update", C.c_uint64), ("burst_bytes", C.c_uint32),
                ("_pad2", C.c_uint32), ("_pad3", C.c_uint64 * 3)]


class BindingEntry(C.Structure):
    _fields_ = [("key_mac", C.c_uint64), ("ipv4_addr", C.c_uint32),
                ("ipv4_valid", C.c_uint8), ("ipv6_valid", C.c_uint8),
                ("mode", C.c_uint8), ("_pad", C.c_uint8),
                ("ipv6_addr", C.c_uint8 * 16)]


class AntispoofConfig(C.Structure):
    _fields_ = [("default_mode", C.c_uint8), ("log_violations", C.c_uint8),
                ("_pad", C.c_uint16), ("n_allowed_ranges", C.c_uint32),
                ("allowed_lo", C.c_uint32 * MAX_ALLOWED_RANGES),
                ("allowed_hi", C.c_uint32 * MAX_ALLOWED_RANGES)]


class SpoofEvent(C.Structure):
    _fields_ = [("timestamp", C.c_uint64), ("src_mac", C.c_uint8 * 6),
                ("protocol", C.c_uint8), ("_pad", C.c_uint8),
                ("spoofed_ip", C.c_uint32), ("allowed_ip", C.c_uint32),
                ("spoofed_ipv6", C.c_uint8 * 16),
                ("allowed_ipv6", C.c_uint8 * 16)]


class RingHeader(C.Structure):
    _fields_ = [("widx", C.c_uint32), ("dropped", C.c_uint32),
                ("capacity", C.c_uint32), ("_pad", C.c_uint32)]


class SessExport(C.Structure):
    """Compact NAT-session record for HA snapshot/delta sync."""
    _fields_ = [("src_ip", C.c_uint32), ("dst_ip", C.c_uint32),
                ("src_port", C.c_uint16), ("dst_port", C.c_uint16),
                ("protocol", C.c_uint8), ("state", C.c_uint8),
                ("is_hairpin", C.c_uint8), ("flags", C.c_uint8),
                ("nat_ip", C.c_uint32), ("nat_port", C.c_uint16),
                ("eim_port", C.c_uint16), ("created", C.c_uint64),
                ("last_seen", C.c_uint64), ("_pad", C.c_uint64)]


SESS_EXPORT_DTYPE = [("src_ip", "<u4"), ("dst_ip", "<u4"),
                     ("src_port", "<u2"), ("dst_port", "<u2"),
                     ("protocol", "u1"), ("state", "u1"),
                     ("is_hairpin", "u1"), ("flags", "u1"),
                     ("nat_ip", "<u4"), ("nat_port", "<u2"),
                     ("eim_port", "<u2"), ("created", "<u8"),
                     ("last_seen", "<u8"), ("_pad", "<u8")]


class SvcCtrl(C.Structure):
    """Pinned-host doorbell of the persistent DHCP service kernel."""
    _fields_ = [("head", C.c_uint32), ("tail", C.c_uint32),
                ("run", C.c_uint32), ("n_pkts", C.c_uint32),
                ("now_sec", C.c_uint64), ("stride", C.c_uint32),
                ("idle_exit_k", C.c_uint32), ("served", C.c_uint64),
                ("batches", C.c_uint64), ("exited", C.c_uint32),
                ("_pad", C.c_uint8 * 12)]


EXPECTED_SIZES = {
    "bng_sub_entry": (SubEntry, 32),
    "bng_ip_pool": (IpPool, 28),
    "bng_server_config": (ServerConfig, 16),
    "bng_nat_tuple": (NatTuple, 16),
    "bng_nat_session": (NatSession, 128),
    "bng_nat_reverse": (NatReverse, 48),
    "bng_eim_entry": (EimEntry, 48),
    "bng_subctx": (SubCtx, 64),
    "bng_nat_config": (NatConfig, 24 + 64 * 8 + 64 * 4),
    "bng_nat_log_entry": (NatLogEntry, 40),
    "bng_qos_bucket": (QosBucket, 64),
    "bng_binding_entry": (BindingEntry, 32),
    "bng_antispoof_config": (AntispoofConfig, 8 + 256 * 8),
    "bng_spoof_event": (SpoofEvent, 56),
    "bng_ring_header": (RingHeader, 16),
    "bng_svc_ctrl": (SvcCtrl, 64),
    "bng_sess_export": (SessExport, 48),
}


# ---------------------------------------------------------------- hashing
def fnv1a64(data: bytes) -> int:
    """FNV-1a 64-bit, matching the reference's circuit-ID hash
    (pkg/ebpf/loader.go FNV-1a) and the device-side bng_fnv1a64."""
    h = 0xCBF29CE484222325
    for b in data:
        h ^= b
        h = (h * 0x100000001B3) & 0xFFFFFFFFFFFFFFFF
    return h


def mix64(x: int) -> int:
    """splitmix64 finalizer; must match device bng_mix64 bit-for-bit."""
    x &= 0xFFFFFFFFFFFFFFFF
    x = (x + 0x9E3779B97F4A7C15) & 0xFFFFFFFFFFFFFFFF
    z = x
    z = ((z ^ (z >> 30)) * 0xBF58476D1CE4E5B9) & 0xFFFFFFFFFFFFFFFF
    z = ((z ^ (z >> 27)) * 0x94D049BB133111EB) & 0xFFFFFFFFFFFFFFFF
    return z ^ (z >> 31)


def mac_to_u64(mac: bytes) -> int:
    """MAC bytes -> u64, big-endian like ref bpf/dhcp_fastpath.c:175-182."""
    v = 0
    for b in mac[:6]:
        v = (v << 8) | b
    return v


def vlan_key(s_tag: int, c_tag: int) -> int:
    return KEY_VLAN | ((s_tag & 0xFFFF) << 16) | (c_tag & 0xFFFF)


def circuit_key(circuit_id: bytes) -> int:
    """32-byte zero-padded/truncated circuit-id (ref maps.h:216-220) hashed."""
    cid = (circuit_id[:32] + b"\x00" * 32)[:32]
    return KEY_CIRCUIT | (fnv1a64(cid) >> 2)


def tuple_sig(src_ip: int, dst_ip: int, src_port: int, dst_port: int,
              proto: int) -> int:
    """64-bit slot signature of a 5-tuple; must match device bng_tuple_sig.
    Low bit forced to 1 so a signature can never equal EMPTY(0)."""
    a = ((src_ip & 0xFFFFFFFF) << 32) | (dst_ip & 0xFFFFFFFF)
    b = ((src_port & 0xFFFF) << 24) | ((dst_port & 0xFFFF) << 8) | (proto & 0xFF)
    s = mix64(mix64(a) ^ b)
    s |= 1
    if s == KEY_TOMBSTONE:
        s -= 2
    return s


def eim_sig(internal_ip: int, internal_port: int, proto: int) -> int:
    s = mix64(((internal_ip & 0xFFFFFFFF) << 24)
              | ((internal_port & 0xFFFF) << 8) | (proto & 0xFF))
    s |= 1
    if s == KEY_TOMBSTONE:
        s -= 2
    return s


def prefixes_to_intervals(ranges):
    """[(net, mask)] host-order prefixes -> sorted merged [lo, hi]
    intervals (the launcher-side fold that turns the reference's LPM
    membership maps into the GPU's binary-searchable tables)."""
    iv = sorted((net & mask, (net & mask) | (~mask & 0xFFFFFFFF))
                for net, mask in ranges)
    out = []
    for lo, hi in iv:
        if out and lo <= out[-1][1] + 1:
            out[-1][1] = max(out[-1][1], hi)
        else:
            out.append([lo, hi])
    return out
