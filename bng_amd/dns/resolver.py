"""Caching DNS resolver with TTL clamps, per-client rate limiting and
intercept rules — the walled-garden DNS (ref pkg/dns/resolver.go:16-51,
cache.go, types.go:223).

Wire-level DNS codec for A/AAAA queries; upstream is a pluggable
callable (UDP forwarder in production, fakes in tests)."""
from __future__ import annotations

import struct
import threading
import time
from dataclasses import dataclass, field
from typing import Callable, Dict, List, Optional, Tuple

QTYPE_A = 1
QTYPE_CNAME = 5
QTYPE_SRV = 33
QTYPE_AAAA = 28


def encode_qname(name: str) -> bytes:
    out = b""
    for label in name.strip(".").split("."):
        out += bytes([len(label)]) + label.encode()
    return out + b"\x00"


def decode_qname(data: bytes, off: int) -> Tuple[str, int]:
    labels = []
    while off < len(data):
        ln = data[off]
        if ln == 0:
            off += 1
            break
        if ln & 0xC0:        # compression pointer
            ptr = struct.unpack_from(">H", data, off)[0] & 0x3FFF
            sub, _ = decode_qname(data, ptr)
            labels.append(sub)
            off += 2
            return ".".join(labels), off
        labels.append(data[off + 1:off + 1 + ln].decode(errors="replace"))
        off += 1 + ln
    return ".".join(labels), off


def build_query(name: str, qtype: int = QTYPE_A, txid: int = 0x1234) -> bytes:
    return struct.pack(">HHHHHH", txid, 0x0100, 1, 0, 0, 0) + \
        encode_qname(name) + struct.pack(">HH", qtype, 1)


def build_response(query: bytes, addrs: List[str], ttl: int = 300,
                   rcode: int = 0) -> bytes:
    txid = struct.unpack_from(">H", query, 0)[0]
    qname, off = decode_qname(query, 12)
    qtype, qclass = struct.unpack_from(">HH", query, off)
    hdr = struct.pack(">HHHHHH", txid, 0x8180 | rcode, 1, len(addrs), 0, 0)
    out = hdr + query[12:off + 4]
    import ipaddress
    for a in addrs:
        ip = ipaddress.ip_address(a)
        rd = ip.packed
        rtype = QTYPE_A if ip.version == 4 else QTYPE_AAAA
        out += b"\xc0\x0c" + struct.pack(">HHIH", rtype, 1, ttl, len(rd)) + rd
    return out


def build_cname_response(query: bytes, target: str,
                         ttl: int = 300) -> bytes:
    """CNAME answer (ref createCNAMEResponse resolver.go:515-530)."""
    txid = struct.unpack_from(">H", query, 0)[0]
    _qname, off = decode_qname(query, 12)
    rd = encode_qname(target)
    hdr = struct.pack(">HHHHHH", txid, 0x8180, 1, 1, 0, 0)
    return hdr + query[12:off + 4] + b"\xc0\x0c" + \
        struct.pack(">HHIH", QTYPE_CNAME, 1, ttl, len(rd)) + rd


def parse_response(data: bytes) -> Tuple[str, List[str], int]:
    """-> (qname, addresses, min_ttl)"""
    import ipaddress
    _txid, _flags, qd, an, _ns, _ar = struct.unpack_from(">HHHHHH", data, 0)
    off = 12
    qname = ""
    for _ in range(qd):
        qname, off = decode_qname(data, off)
        off += 4
    addrs, min_ttl = [], 2**31
    for _ in range(an):
        _, off = decode_qname(data, off)
        rtype, _rc, ttl, rdlen = struct.unpack_from(">HHIH", data, off)
        off += 10
        rd = data[off:off + rdlen]
        off += rdlen
        if rtype in (QTYPE_A, QTYPE_AAAA):
            addrs.append(str(ipaddress.ip_address(rd)))
            min_ttl = min(min_ttl, ttl)
    return qname, addrs, 0 if min_ttl == 2**31 else min_ttl


@dataclass
class CacheEntry:
    addrs: List[str]
    expires: float


class Resolver:
    def __init__(self, upstream: Callable[[bytes], Optional[bytes]],
                 min_ttl: int = 30, max_ttl: int = 3600,
                 rate_limit: float = 0.0, rate_burst: int = 50,
                 max_entries: int = 100_000, negative_ttl: int = 30):
        from collections import OrderedDict
        self.upstream = upstream
        self.min_ttl = min_ttl
        self.max_ttl = max_ttl
        self.max_entries = max_entries
        self.negative_ttl = negative_ttl
        self.cache: "OrderedDict[Tuple[str, int], CacheEntry]" = \
            OrderedDict()
        self.intercepts: Dict[str, List[str]] = {}   # name -> portal IPs
        self.intercept_all_to: Optional[List[str]] = None
        self.rules: List[dict] = []          # typed interception rules
        self.walled_clients: set = set()     # client IPs in quarantine
        self._lock = threading.RLock()
        self.rate_limit = rate_limit
        self.rate_burst = rate_burst
        self._buckets: Dict[str, List[float]] = {}
        self.stats = {"queries": 0, "cache_hits": 0, "intercepted": 0,
                      "rate_limited": 0, "upstream_fail": 0,
                      "negative_hits": 0, "evicted": 0}

    # -------------------------------------------------------- intercepts
    def add_intercept(self, name: str, addrs: List[str]):
        """Walled-garden DNS rule: this name resolves to the portal."""
        with self._lock:
            self.intercepts[name.rstrip(".").lower()] = addrs

    def set_intercept_all(self, addrs: Optional[List[str]]):
        """Quarantined clients: EVERY name resolves to the portal."""
        self.intercept_all_to = addrs

    def remove_intercept(self, name: str):
        with self._lock:
            self.intercepts.pop(name.rstrip(".").lower(), None)

    # typed interception rules (ref checkInterceptionRules
    # resolver.go:444-530; actions allow/block/redirect/cname, match
    # modes exact / suffix / wildcard-subdomain :468-490)
    def add_rule(self, domain: str = "", action: str = "block",
                 redirect: Optional[List[str]] = None, cname: str = "",
                 exact: bool = False, suffix: str = ""):
        if action not in ("allow", "block", "redirect", "cname"):
            raise ValueError(f"unknown intercept action {action}")
        with self._lock:
            self.rules.append({"domain": domain.rstrip(".").lower(),
                               "suffix": suffix.lower(),
                               "action": action,
                               "redirect": redirect or [],
                               "cname": cname, "exact": exact})

    def remove_rule(self, domain: str) -> bool:
        d = domain.rstrip(".").lower()
        with self._lock:
            before = len(self.rules)
            self.rules = [r for r in self.rules if r["domain"] != d]
            return len(self.rules) != before

    @staticmethod
    def _rule_matches(rule: dict, qname: str) -> bool:
        if rule["exact"]:
            return qname == rule["domain"]
        if rule["suffix"]:
            return qname.endswith(rule["suffix"])
        d = rule["domain"]
        return bool(d) and (qname == d or qname.endswith("." + d))

    def _check_rules(self, query: bytes, qname: str):
        """-> response bytes or None (allow)."""
        with self._lock:
            rules = list(self.rules)
        for r in rules:
            if not self._rule_matches(r, qname):
                continue
            if r["action"] == "allow":
                return None
            self.stats["intercepted"] += 1
            if r["action"] == "block":
                return build_response(query, [], ttl=30, rcode=3)
            if r["action"] == "redirect":
                return build_response(query, r["redirect"], ttl=300)
            return build_cname_response(query, r["cname"])
        return None

    # per-client walled-garden registry (ref AddWalledGardenClient
    # resolver.go:238-270): queries from these IPs resolve everything
    # to the portal
    def add_walled_client(self, ip: str):
        with self._lock:
            self.walled_clients.add(ip)

    def remove_walled_client(self, ip: str) -> bool:
        with self._lock:
            had = ip in self.walled_clients
            self.walled_clients.discard(ip)
            return had

    def is_walled(self, ip: str) -> bool:
        with self._lock:
            return ip in self.walled_clients

    # ------------------------------------------------------------ resolve
    def _allowed(self, client: str) -> bool:
        if self.rate_limit <= 0:
            return True
        now = time.monotonic()
        with self._lock:
            b = self._buckets.setdefault(client, [float(self.rate_burst),
                                                  now])
            b[0] = min(self.rate_burst, b[0] + (now - b[1]) * self.rate_limit)
            b[1] = now
            if b[0] >= 1:
                b[0] -= 1
                return True
            return False

    def handle_query(self, query: bytes, client: str = "",
                     quarantined: bool = False) -> Optional[bytes]:
        self.stats["queries"] += 1
        if client and not self._allowed(client):
            self.stats["rate_limited"] += 1
            return None
        try:
            qname, off = decode_qname(query, 12)
            qtype = struct.unpack_from(">H", query, off)[0]
        except (struct.error, IndexError):
            return None
        key = qname.lower()
        # typed rules first (block/redirect/cname)
        ruled = self._check_rules(query, key)
        if ruled is not None:
            return ruled
        # walled garden: flagged caller or registered client IP
        if client and not quarantined:
            quarantined = self.is_walled(client)
        if quarantined and self.intercept_all_to:
            self.stats["intercepted"] += 1
            return build_response(query, self.intercept_all_to, ttl=30)
        with self._lock:
            hit = self.intercepts.get(key)
        if hit is not None:
            self.stats["intercepted"] += 1
            return build_response(query, hit, ttl=30)
        # cache (LRU: a hit moves the entry to the back)
        with self._lock:
            ce = self.cache.get((key, qtype))
            if ce is not None and ce.expires > time.time():
                self.cache.move_to_end((key, qtype))
                if not ce.addrs:
                    # negative cache: answer NXDOMAIN without upstream
                    self.stats["negative_hits"] += 1
                    return build_response(query, [], ttl=self.negative_ttl,
                                          rcode=3)
                self.stats["cache_hits"] += 1
                ttl = max(1, int(ce.expires - time.time()))
                return build_response(query, ce.addrs, ttl=ttl)
        # upstream
        resp = None
        try:
            resp = self.upstream(query)
        except Exception:
            resp = None
        if resp is None:
            self.stats["upstream_fail"] += 1
            return None
        _qn, addrs, ttl = parse_response(resp)
        ttl = max(self.min_ttl, min(self.max_ttl, ttl))   # TTL clamp
        with self._lock:
            if addrs:
                self.cache[(key, qtype)] = CacheEntry(addrs,
                                                      time.time() + ttl)
            else:
                # empty answer / NXDOMAIN: negative-cache it (RFC 2308)
                self.cache[(key, qtype)] = CacheEntry(
                    [], time.time() + self.negative_ttl)
            self.cache.move_to_end((key, qtype))
            while len(self.cache) > self.max_entries:   # LRU eviction
                self.cache.popitem(last=False)
                self.stats["evicted"] += 1
        return resp

    def cleanup(self, now: Optional[float] = None) -> int:
        """Drop expired cache entries (ref resolver cache cleanup)."""
        now = now if now is not None else time.time()
        with self._lock:
            dead = [k for k, ce in self.cache.items() if ce.expires <= now]
            for k in dead:
                del self.cache[k]
            return len(dead)


# ------------------------------------------------------------- SRV (RFC 2782)
def build_srv_response(query: bytes, records, ttl: int = 300) -> bytes:
    """records: [(priority, weight, port, target_name)] — used by tests
    and by the embedded DNS server for service records."""
    txid = struct.unpack_from(">H", query, 0)[0]
    _, off = decode_qname(query, 12)
    hdr = struct.pack(">HHHHHH", txid, 0x8180, 1, len(records), 0, 0)
    out = hdr + query[12:off + 4]
    for pri, weight, port, target in records:
        rdata = (struct.pack(">HHH", pri, weight, port) +
                 encode_qname(target))
        out += b"\xc0\x0c" + struct.pack(">HHIH", QTYPE_SRV, 1, ttl,
                                          len(rdata)) + rdata
    return out


def parse_srv_response(data: bytes):
    """-> [(priority, weight, port, target)] from a DNS SRV answer."""
    _, flags, qd, an, _, _ = struct.unpack_from(">HHHHHH", data, 0)
    off = 12
    for _ in range(qd):
        _, off = decode_qname(data, off)
        off += 4
    out = []
    for _ in range(an):
        _, off = decode_qname(data, off)
        rtype, rclass, ttl, rdlen = struct.unpack_from(">HHIH", data, off)
        off += 10
        if rtype == QTYPE_SRV and rdlen >= 7:
            pri, weight, port = struct.unpack_from(">HHH", data, off)
            target, _ = decode_qname(data, off + 6)
            out.append((pri, weight, port, target))
        off += rdlen
    return out


def resolve_srv(name: str, server: str, timeout: float = 2.0):
    """One-shot SRV lookup over UDP (the reference's DNS-SRV peer
    discovery, peer-discovery=dns + peer-service flags)."""
    import socket as _s
    host, _, port = server.rpartition(":")
    addr = (host or server, int(port or 53))
    q = build_query(name, QTYPE_SRV, txid=0x5273)
    with _s.socket(_s.AF_INET, _s.SOCK_DGRAM) as sk:
        sk.settimeout(timeout)
        sk.sendto(q, addr)
        data, _ = sk.recvfrom(4096)
    return parse_srv_response(data)
