"""SLAAC router advertisements (ref pkg/slaac/radvd.go:104): RADVD-style
RA construction — prefixes with on-link/autonomous flags, M/O flags, MTU,
RDNSS/DNSSL options, min/max interval scheduling, and solicited replies
(RS -> RA).  Transport-agnostic: build_ra() returns the ICMPv6 payload;
the raw ICMPv6 socket layer plugs in at deployment."""
from __future__ import annotations

import ipaddress
import struct
import threading
import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Tuple

ND_ROUTER_SOLICIT = 133
ND_ROUTER_ADVERT = 134

OPT_SOURCE_LLADDR = 1
OPT_PREFIX_INFO = 3
OPT_MTU = 5
OPT_RDNSS = 25
OPT_DNSSL = 31


@dataclass
class PrefixConfig:
    prefix: str                  # e.g. "2001:db8:1::/64"
    on_link: bool = True
    autonomous: bool = True
    valid_lifetime: int = 86400
    preferred_lifetime: int = 14400


@dataclass
class RAConfig:
    prefixes: List[PrefixConfig] = field(default_factory=list)
    managed: bool = False        # M flag: addresses via DHCPv6
    other_config: bool = False   # O flag: other config via DHCPv6
    router_lifetime: int = 1800
    mtu: int = 0
    rdnss: List[str] = field(default_factory=list)
    dnssl: List[str] = field(default_factory=list)
    dns_lifetime: int = 3600
    hop_limit: int = 64
    min_interval: float = 200.0
    max_interval: float = 600.0
    source_lladdr: bytes = b""


def build_ra(cfg: RAConfig) -> bytes:
    """ICMPv6 Router Advertisement payload (checksum left to the stack)."""
    flags = (0x80 if cfg.managed else 0) | (0x40 if cfg.other_config else 0)
    out = struct.pack(">BBHBBHII", ND_ROUTER_ADVERT, 0, 0, cfg.hop_limit,
                      flags, cfg.router_lifetime, 0, 0)
    if cfg.source_lladdr:
        out += struct.pack(">BB", OPT_SOURCE_LLADDR, 1) + \
            (cfg.source_lladdr + b"\x00" * 6)[:6]
    if cfg.mtu:
        out += struct.pack(">BBHI", OPT_MTU, 1, 0, cfg.mtu)
    for p in cfg.prefixes:
        net = ipaddress.IPv6Network(p.prefix, strict=False)
        pflags = (0x80 if p.on_link else 0) | (0x40 if p.autonomous else 0)
        out += struct.pack(">BBBBIII", OPT_PREFIX_INFO, 4, net.prefixlen,
                           pflags, p.valid_lifetime, p.preferred_lifetime,
                           0)
        out += net.network_address.packed
    if cfg.rdnss:
        body = struct.pack(">HI", 0, cfg.dns_lifetime) + b"".join(
            ipaddress.IPv6Address(a).packed for a in cfg.rdnss)
        out += struct.pack(">BB", OPT_RDNSS, 1 + len(cfg.rdnss) * 2) + body
    if cfg.dnssl:
        enc = b""
        for name in cfg.dnssl:
            for label in name.strip(".").split("."):
                enc += bytes([len(label)]) + label.encode()
            enc += b"\x00"
        pad = (-len(enc)) % 8
        enc += b"\x00" * pad
        out += struct.pack(">BBHI", OPT_DNSSL, 1 + len(enc) // 8, 0,
                           cfg.dns_lifetime) + enc
    return out


def parse_ra(data: bytes) -> dict:
    """Parse an RA payload (for tests)."""
    t, _c, _ck, hop, flags, lifetime, reach, retrans = struct.unpack_from(
        ">BBHBBHII", data, 0)
    assert t == ND_ROUTER_ADVERT
    r = {"hop_limit": hop, "managed": bool(flags & 0x80),
         "other": bool(flags & 0x40), "router_lifetime": lifetime,
         "prefixes": [], "mtu": None, "rdnss": [], "dnssl": []}
    i = 16
    while i + 2 <= len(data):
        ot, oln = data[i], data[i + 1]
        if oln == 0:
            break
        body = data[i + 2:i + oln * 8]
        if ot == OPT_PREFIX_INFO:
            plen, pflags, valid, pref, _ = struct.unpack_from(">BBIII",
                                                              body, 0)
            addr = ipaddress.IPv6Address(body[14:30])
            r["prefixes"].append({
                "prefix": f"{addr}/{plen}",
                "on_link": bool(pflags & 0x80),
                "autonomous": bool(pflags & 0x40),
                "valid": valid, "preferred": pref})
        elif ot == OPT_MTU:
            r["mtu"] = struct.unpack_from(">I", body, 2)[0]
        elif ot == OPT_RDNSS:
            for j in range(6, len(body), 16):
                r["rdnss"].append(str(ipaddress.IPv6Address(
                    body[j:j + 16])))
        elif ot == OPT_DNSSL:
            j = 6
            while j < len(body):
                labels = []
                while j < len(body) and body[j]:
                    ln = body[j]
                    labels.append(body[j + 1:j + 1 + ln].decode())
                    j += 1 + ln
                j += 1
                if labels:
                    r["dnssl"].append(".".join(labels))
        i += oln * 8
    return r


class Server:
    """RA scheduler: periodic unsolicited RAs within [min,max] interval +
    immediate solicited RAs on RS (ref radvd.go send loop)."""

    def __init__(self, cfg: RAConfig, send_fn=None):
        self.cfg = cfg
        self.send_fn = send_fn or (lambda payload, dst: None)
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self.stats = {"ra_sent": 0, "ra_solicited": 0, "rs_received": 0}

    def start(self):
        self._thread = threading.Thread(target=self._loop, daemon=True)
        self._thread.start()
        return self

    def stop(self):
        self._stop.set()

    def _loop(self):
        import random
        while True:
            iv = random.uniform(self.cfg.min_interval, self.cfg.max_interval)
            if self._stop.wait(iv):
                return
            self.advertise()

    def advertise(self, dst: str = "ff02::1"):
        self.send_fn(build_ra(self.cfg), dst)
        self.stats["ra_sent"] += 1

    def add_prefix(self, p: "PrefixConfig"):
        """ref AddPrefix radvd.go:491-496; takes effect next RA."""
        self.cfg.prefixes.append(p)

    def remove_prefix(self, prefix: str):
        """ref RemovePrefix radvd.go:498-509."""
        self.cfg.prefixes = [q for q in self.cfg.prefixes
                             if q.prefix != prefix]

    def send_immediate_ra(self):
        """ref SendImmediateRA radvd.go:519."""
        self.advertise()

    def handle_rs(self, data: bytes, src: str = "") -> Optional[bytes]:
        """Router Solicitation -> immediate unicast RA."""
        if not data or data[0] != ND_ROUTER_SOLICIT:
            return None
        self.stats["rs_received"] += 1
        self.stats["ra_solicited"] += 1
        self.stats["ra_sent"] += 1
        payload = build_ra(self.cfg)
        self.send_fn(payload, src or "ff02::1")
        return payload


# ----------------------------------------------------------------------
# address generation + classification (ref slaac/types.go:124-185)

def generate_slaac_address(prefix: str, mac: str) -> str:
    """Modified EUI-64: ff:fe spliced into the MAC, universal/local bit
    flipped, appended to the /64 prefix (ref GenerateSLAACAddress
    types.go:124-148)."""
    m = bytes(int(b, 16) for b in mac.split(":"))
    if len(m) != 6:
        raise ValueError(f"bad MAC {mac}")
    eui64 = bytes([m[0] ^ 0x02, m[1], m[2], 0xFF, 0xFE, m[3], m[4], m[5]])
    net = ipaddress.IPv6Network(prefix, strict=False)
    return str(ipaddress.IPv6Address(net.network_address.packed[:8] +
                                     eui64))


def generate_stable_privacy_address(prefix: str, interface_id: bytes,
                                    secret_key: bytes,
                                    dad_counter: int = 0) -> str:
    """RFC 7217 stable privacy address: SHA-256 over
    (prefix | interface_id | dad_counter | secret) -> IID (the reference
    ships a simplified XOR and notes production should hash,
    types.go:150-168 — we do the hash)."""
    import hashlib
    net = ipaddress.IPv6Network(prefix, strict=False)
    digest = hashlib.sha256(net.network_address.packed[:8] +
                            interface_id +
                            dad_counter.to_bytes(4, "big") +
                            secret_key).digest()
    iid = bytearray(digest[:8])
    iid[0] &= ~0x02                    # clear universal/local bit
    return str(ipaddress.IPv6Address(net.network_address.packed[:8] +
                                     bytes(iid)))


def is_link_local(ip: str) -> bool:
    a = ipaddress.ip_address(ip)
    return a.is_link_local


def is_global_unicast(ip: str) -> bool:
    a = ipaddress.ip_address(ip)
    return a.is_global and not a.is_private


# ----------------------------------------------------------------------
# neighbor cache (ref types.go NeighborEntry :102-122; RFC 4861 §7.3
# reachability states)

N_INCOMPLETE = "INCOMPLETE"
N_REACHABLE = "REACHABLE"
N_STALE = "STALE"
N_DELAY = "DELAY"
N_PROBE = "PROBE"


class NeighborCache:
    """Minimal RFC 4861 neighbor state machine: confirmations make an
    entry REACHABLE; it decays to STALE after reachable_time; sending to
    a STALE entry moves it DELAY, and an unanswered DELAY probes."""

    def __init__(self, reachable_time: float = 30.0,
                 delay_time: float = 5.0):
        self.reachable_time = reachable_time
        self.delay_time = delay_time
        self._entries: Dict[str, dict] = {}

    def confirm(self, ip: str, mac: str = "",
                is_router: Optional[bool] = None, now: float = 0.0):
        """Reachability confirmed (NA received / upper-layer hint);
        is_router only changes when the confirmation says so (an NA
        without the R flag conveys nothing about router-ness here)."""
        now = now or time.time()
        e = self._entries.setdefault(ip, {"is_router": False, "mac": ""})
        if mac:
            e["mac"] = mac
        if is_router is not None:
            e["is_router"] = is_router
        e.update(state=N_REACHABLE, last_seen=now)

    def incomplete(self, ip: str, now: float = 0.0):
        """NS sent, no answer yet."""
        self._entries[ip] = {"mac": "", "state": N_INCOMPLETE,
                             "last_seen": now or time.time(),
                             "is_router": False}

    def state(self, ip: str, now: float = 0.0) -> str:
        e = self._entries.get(ip)
        if e is None:
            return ""
        now = now or time.time()
        if e["state"] == N_REACHABLE and \
                now - e["last_seen"] > self.reachable_time:
            e["state"] = N_STALE
        elif e["state"] == N_DELAY and \
                now - e.get("delay_at", now) > self.delay_time:
            e["state"] = N_PROBE
        return e["state"]

    def used(self, ip: str, now: float = 0.0):
        """A packet was sent to the neighbor: STALE -> DELAY."""
        now = now or time.time()
        if self.state(ip, now) == N_STALE:
            e = self._entries[ip]
            e["state"] = N_DELAY
            e["delay_at"] = now

    def lookup(self, ip: str) -> Optional[dict]:
        return self._entries.get(ip)

    def routers(self):
        return [ip for ip, e in self._entries.items() if e["is_router"]]

    def purge(self, max_age: float, now: float = 0.0) -> int:
        now = now or time.time()
        dead = [ip for ip, e in self._entries.items()
                if now - e["last_seen"] > max_age]
        for ip in dead:
            del self._entries[ip]
        return len(dead)
