"""DHCPv6 server (ref pkg/dhcpv6: protocol.go:98-160, server.go:111):
SOLICIT/ADVERTISE/REQUEST/REPLY/RENEW/RELEASE with IA_NA address
assignment and IA_PD prefix delegation, DUID-based client identity,
rapid commit."""
from __future__ import annotations

import ipaddress
import struct
import threading
import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Tuple

from ..allocator.bitmap import BitmapAllocator, PoolExhaustedError

# message types (RFC 8415)
SOLICIT, ADVERTISE, REQUEST, CONFIRM, RENEW, REBIND, REPLY, RELEASE, \
    DECLINE, RECONFIGURE, INFORMATION_REQUEST, RELAY_FORW, RELAY_REPL = \
    range(1, 14)

# options
OPT_CLIENTID = 1
OPT_SERVERID = 2
OPT_IA_NA = 3
OPT_IAADDR = 5
OPT_ORO = 6
OPT_PREFERENCE = 7
OPT_STATUS_CODE = 13
OPT_RAPID_COMMIT = 14
OPT_DNS_SERVERS = 23
OPT_DOMAIN_LIST = 24
OPT_IA_PD = 25
OPT_IAPREFIX = 26

STATUS_SUCCESS = 0
STATUS_NOADDRS = 2
STATUS_NOBINDING = 3
STATUS_NOPREFIX = 6


class DHCPv6Message:
    def __init__(self, msg_type: int, txn_id: int = 0):
        self.msg_type = msg_type
        self.txn_id = txn_id
        self.options: List[Tuple[int, bytes]] = []

    def add(self, t: int, v: bytes):
        self.options.append((t, v))
        return self

    def get(self, t: int) -> Optional[bytes]:
        for tt, v in self.options:
            if tt == t:
                return v
        return None

    def get_all(self, t: int) -> List[bytes]:
        return [v for tt, v in self.options if tt == t]

    def encode(self) -> bytes:
        out = struct.pack(">I", (self.msg_type << 24) | self.txn_id)
        for t, v in self.options:
            out += struct.pack(">HH", t, len(v)) + v
        return out

    @classmethod
    def decode(cls, data: bytes) -> "DHCPv6Message":
        if len(data) < 4:
            raise ValueError("short DHCPv6 message")
        w = struct.unpack(">I", data[:4])[0]
        m = cls(w >> 24, w & 0xFFFFFF)
        i = 4
        while i + 4 <= len(data):
            t, ln = struct.unpack_from(">HH", data, i)
            if i + 4 + ln > len(data):
                break
            m.options.append((t, data[i + 4:i + 4 + ln]))
            i += 4 + ln
        return m


def encode_ia_na(iaid: int, t1: int, t2: int,
                 addrs: List[Tuple[str, int, int]]) -> bytes:
    body = struct.pack(">III", iaid, t1, t2)
    for addr, pref, valid in addrs:
        a = ipaddress.IPv6Address(addr).packed
        body += struct.pack(">HH", OPT_IAADDR, 24) + a + \
            struct.pack(">II", pref, valid)
    return body


def encode_ia_pd(iaid: int, t1: int, t2: int,
                 prefixes: List[Tuple[str, int, int]]) -> bytes:
    body = struct.pack(">III", iaid, t1, t2)
    for prefix, pref, valid in prefixes:
        net = ipaddress.IPv6Network(prefix, strict=False)
        body += struct.pack(">HH", OPT_IAPREFIX, 25) + \
            struct.pack(">IIB", pref, valid, net.prefixlen) + \
            net.network_address.packed
    return body


def parse_ia(body: bytes) -> Tuple[int, int, int, List[bytes]]:
    iaid, t1, t2 = struct.unpack_from(">III", body, 0)
    subs = []
    i = 12
    while i + 4 <= len(body):
        t, ln = struct.unpack_from(">HH", body, i)
        subs.append((t, body[i + 4:i + 4 + ln]))
        i += 4 + ln
    return iaid, t1, t2, subs


@dataclass
class Binding:
    duid: bytes
    iaid: int
    value: str              # address or prefix
    expiry: float
    is_pd: bool = False


class DHCPv6Server:
    def __init__(self, duid: bytes = b"\x00\x03\x00\x01\x02\x00\x00\x00\x00\x01",
                 na_pool: str = "2001:db8:1::/64",
                 pd_pool: str = "2001:db8:f000::/40",
                 pd_prefix_len: int = 56,
                 dns: Optional[List[str]] = None,
                 domains: Optional[List[str]] = None,
                 preferred_lifetime: int = 1800, valid_lifetime: int = 3600,
                 rapid_commit: bool = True):
        self.server_duid = duid
        self.na_alloc = BitmapAllocator(na_pool, 128, reserve_head=1)
        self.pd_alloc = BitmapAllocator(pd_pool, pd_prefix_len)
        self.dns = dns or []
        self.domains = domains or []      # RFC 8415 option 24
        self._declined: dict = {}   # addr -> quarantine expiry
        self.preferred = preferred_lifetime
        self.valid = valid_lifetime
        self.rapid_commit = rapid_commit
        self.bindings: Dict[Tuple[bytes, int, bool], Binding] = {}
        self._lock = threading.RLock()
        self.stats = {k: 0 for k in (
            "solicit", "advertise", "request", "renew", "rebind", "release",
            "reply", "rapid_commits", "no_addrs", "info_request")}

    # ------------------------------------------------------------ handle
    def handle(self, data: bytes) -> Optional[bytes]:
        try:
            msg_type = data[0] if data else 0
        except IndexError:
            return None
        if msg_type == RELAY_FORW:
            return self._relay(data)
        try:
            msg = DHCPv6Message.decode(data)
        except ValueError:
            return None
        duid = msg.get(OPT_CLIENTID)
        if duid is None and msg.msg_type != INFORMATION_REQUEST:
            return None
        h = {SOLICIT: self._solicit, REQUEST: self._request,
             RENEW: self._renew, REBIND: self._renew,
             RELEASE: self._release, CONFIRM: self._confirm,
             DECLINE: self._decline,
             INFORMATION_REQUEST: self._inforeq}.get(msg.msg_type)
        if h is None:
            return None
        resp = h(msg, duid)
        return resp.encode() if resp is not None else None

    def _base_reply(self, msg: DHCPv6Message, msg_type: int,
                    duid: Optional[bytes]) -> DHCPv6Message:
        r = DHCPv6Message(msg_type, msg.txn_id)
        r.add(OPT_SERVERID, self.server_duid)
        if duid is not None:
            r.add(OPT_CLIENTID, duid)
        if self.dns:
            r.add(OPT_DNS_SERVERS, b"".join(
                ipaddress.IPv6Address(d).packed for d in self.dns))
        if self.domains:
            enc = b""
            for name in self.domains:
                for label in name.strip(".").split("."):
                    enc += bytes([len(label)]) + label.encode()
                enc += b"\x00"
            r.add(OPT_DOMAIN_LIST, enc)
        return r

    def _fill_ias(self, msg: DHCPv6Message, resp: DHCPv6Message,
                  duid: bytes, commit: bool) -> None:
        sub_key = duid.hex()
        for body in msg.get_all(OPT_IA_NA):
            iaid, _, _, _ = parse_ia(body)
            try:
                with self._lock:
                    self._expire_quarantine()
                    prefix = self.na_alloc.allocate(f"{sub_key}/{iaid}")
                    addr = prefix.split("/")[0]
                if commit:
                    self.bindings[(duid, iaid, False)] = Binding(
                        duid, iaid, addr, time.time() + self.valid)
                resp.add(OPT_IA_NA, encode_ia_na(
                    iaid, self.preferred // 2, self.preferred * 4 // 5,
                    [(addr, self.preferred, self.valid)]))
            except PoolExhaustedError:
                self.stats["no_addrs"] += 1
                resp.add(OPT_IA_NA, struct.pack(">III", iaid, 0, 0) +
                         struct.pack(">HH", OPT_STATUS_CODE, 2 + 9) +
                         struct.pack(">H", STATUS_NOADDRS) + b"no addrs")
        for body in msg.get_all(OPT_IA_PD):
            iaid, _, _, _ = parse_ia(body)
            try:
                with self._lock:
                    prefix = self.pd_alloc.allocate(f"{sub_key}/pd/{iaid}")
                if commit:
                    self.bindings[(duid, iaid, True)] = Binding(
                        duid, iaid, prefix, time.time() + self.valid, True)
                resp.add(OPT_IA_PD, encode_ia_pd(
                    iaid, self.preferred // 2, self.preferred * 4 // 5,
                    [(prefix, self.preferred, self.valid)]))
            except PoolExhaustedError:
                self.stats["no_addrs"] += 1
                resp.add(OPT_IA_PD, struct.pack(">III", iaid, 0, 0) +
                         struct.pack(">HH", OPT_STATUS_CODE, 2 + 10) +
                         struct.pack(">H", STATUS_NOPREFIX) + b"no prefix")

    # ----------------------------------------------------------- handlers
    def _solicit(self, msg, duid):
        self.stats["solicit"] += 1
        if self.rapid_commit and msg.get(OPT_RAPID_COMMIT) is not None:
            self.stats["rapid_commits"] += 1
            self.stats["reply"] += 1
            resp = self._base_reply(msg, REPLY, duid)
            resp.add(OPT_RAPID_COMMIT, b"")
            self._fill_ias(msg, resp, duid, commit=True)
            return resp
        self.stats["advertise"] += 1
        resp = self._base_reply(msg, ADVERTISE, duid)
        resp.add(OPT_PREFERENCE, b"\xff")
        self._fill_ias(msg, resp, duid, commit=False)
        return resp

    def _request(self, msg, duid):
        if msg.get(OPT_SERVERID) != self.server_duid:
            return None                      # not for us
        self.stats["request"] += 1
        self.stats["reply"] += 1
        resp = self._base_reply(msg, REPLY, duid)
        self._fill_ias(msg, resp, duid, commit=True)
        return resp

    def _renew(self, msg, duid):
        self.stats["renew"] += 1
        resp = self._base_reply(msg, REPLY, duid)
        for body in msg.get_all(OPT_IA_NA):
            iaid, _, _, _ = parse_ia(body)
            b = self.bindings.get((duid, iaid, False))
            if b is None:
                resp.add(OPT_IA_NA, struct.pack(">III", iaid, 0, 0) +
                         struct.pack(">HH", OPT_STATUS_CODE, 2 + 10) +
                         struct.pack(">H", STATUS_NOBINDING) + b"nobinding")
                continue
            b.expiry = time.time() + self.valid
            resp.add(OPT_IA_NA, encode_ia_na(
                iaid, self.preferred // 2, self.preferred * 4 // 5,
                [(b.value, self.preferred, self.valid)]))
        for body in msg.get_all(OPT_IA_PD):
            iaid, _, _, _ = parse_ia(body)
            b = self.bindings.get((duid, iaid, True))
            if b is None:
                resp.add(OPT_IA_PD, struct.pack(">III", iaid, 0, 0) +
                         struct.pack(">HH", OPT_STATUS_CODE, 2 + 10) +
                         struct.pack(">H", STATUS_NOBINDING) + b"nobinding")
                continue
            b.expiry = time.time() + self.valid
            resp.add(OPT_IA_PD, encode_ia_pd(
                iaid, self.preferred // 2, self.preferred * 4 // 5,
                [(b.value, self.preferred, self.valid)]))
        self.stats["reply"] += 1
        return resp

    def _release(self, msg, duid):
        self.stats["release"] += 1
        sub_key = duid.hex()
        for body in msg.get_all(OPT_IA_NA):
            iaid, _, _, _ = parse_ia(body)
            with self._lock:
                self.bindings.pop((duid, iaid, False), None)
                self.na_alloc.release(f"{sub_key}/{iaid}")
        for body in msg.get_all(OPT_IA_PD):
            iaid, _, _, _ = parse_ia(body)
            with self._lock:
                self.bindings.pop((duid, iaid, True), None)
                self.pd_alloc.release(f"{sub_key}/pd/{iaid}")
        self.stats["reply"] += 1
        resp = self._base_reply(msg, REPLY, duid)
        resp.add(OPT_STATUS_CODE,
                 struct.pack(">H", STATUS_SUCCESS) + b"released")
        return resp

    def _confirm(self, msg, duid):
        self.stats["reply"] += 1
        resp = self._base_reply(msg, REPLY, duid)
        resp.add(OPT_STATUS_CODE, struct.pack(">H", STATUS_SUCCESS) + b"ok")
        return resp

    DECLINE_QUARANTINE = 3600.0

    def _quarantine(self, addr: str):
        """Hold a conflicted address out of the pool by parking it on a
        sentinel allocation; lazily released when the quarantine
        expires (the bitmap itself enforces the blacklist)."""
        self._declined[addr] = time.time() + self.DECLINE_QUARANTINE
        try:
            self.na_alloc.allocate_specific(f"__declined__/{addr}",
                                            f"{addr}/128")
        except (PoolExhaustedError, ValueError, KeyError):
            pass

    def _is_declined(self, addr: str) -> bool:
        exp = self._declined.get(addr)
        return exp is not None and time.time() < exp

    def _expire_quarantine(self):
        now = time.time()
        for addr in [a for a, e in self._declined.items() if now >= e]:
            del self._declined[addr]
            try:
                self.na_alloc.release(f"__declined__/{addr}")
            except KeyError:
                pass

    def _decline(self, msg, duid):
        """Client detected an address conflict (RFC 8415 §18.3.8):
        release the binding AND quarantine the address so the next
        allocation doesn't hand the conflicted address straight back
        (the reference releases only, dhcpv6/server.go:684-693; the v4
        pool's decline blacklist is the model here)."""
        self.stats["decline"] = self.stats.get("decline", 0) + 1
        sub_key = duid.hex()
        import ipaddress
        for body in msg.get_all(OPT_IA_NA):
            iaid, _, _, subs = parse_ia(body)
            with self._lock:
                b = self.bindings.pop((duid, iaid, False), None)
                self.na_alloc.release(f"{sub_key}/{iaid}")
                if b is not None:
                    self._quarantine(b.value)
                for t, sub in subs:
                    if t == OPT_IAADDR and len(sub) >= 16:
                        a = str(ipaddress.IPv6Address(sub[:16]))
                        if not self._is_declined(a):
                            self._quarantine(a)
        resp = self._base_reply(msg, REPLY, duid)
        resp.add(OPT_STATUS_CODE,
                 struct.pack(">H", STATUS_SUCCESS) + b"declined")
        self.stats["reply"] += 1
        return resp

    OPT_RELAY_MSG = 9
    OPT_INTERFACE_ID = 18

    def _relay(self, data: bytes) -> Optional[bytes]:
        """RELAY-FORW: unwrap the inner message, process it, wrap the
        answer in RELAY-REPL echoing peer/link addresses and
        Interface-Id (RFC 8415 §19)."""
        if len(data) < 34:
            return None
        hop = data[1]
        link_peer = data[2:34]
        inner = None
        iface_id = None
        i = 34
        while i + 4 <= len(data):
            t, ln = struct.unpack_from(">HH", data, i)
            if i + 4 + ln > len(data):
                break
            if t == self.OPT_RELAY_MSG:
                inner = data[i + 4:i + 4 + ln]
            elif t == self.OPT_INTERFACE_ID:
                iface_id = data[i + 4:i + 4 + ln]
            i += 4 + ln
        if inner is None:
            return None
        answer = self.handle(inner)     # recursion covers relay chains
        if answer is None:
            return None
        out = bytes([RELAY_REPL, hop]) + link_peer
        if iface_id is not None:
            out += struct.pack(">HH", self.OPT_INTERFACE_ID,
                               len(iface_id)) + iface_id
        out += struct.pack(">HH", self.OPT_RELAY_MSG, len(answer)) + answer
        return out

    def _inforeq(self, msg, duid):
        self.stats["info_request"] += 1
        self.stats["reply"] += 1
        return self._base_reply(msg, REPLY, duid)

    def sweep_expired(self, now: Optional[float] = None) -> int:
        now = now or time.time()
        dead = []
        with self._lock:
            for key, b in list(self.bindings.items()):
                if b.expiry <= now:
                    dead.append(key)
                    del self.bindings[key]
                    sub_key = b.duid.hex()
                    if b.is_pd:
                        self.pd_alloc.release(f"{sub_key}/pd/{b.iaid}")
                    else:
                        self.na_alloc.release(f"{sub_key}/{b.iaid}")
        return len(dead)
