"""Raw-rtnetlink routing platform (ref pkg/routing/netlink_linux.go
:20-235): route add/delete/dump per table, policy rules (ip rule) for
per-ISP table steering, interface up/down — no iproute2 subprocess, no
libnl; the netlink(7) messages are built directly like the AF_XDP
module builds its link requests.

Implements the RoutingPlatform protocol from routing.manager so the
Manager's per-ISP tables and subscriber rules run on a live kernel
where CAP_NET_ADMIN exists, and on MemoryPlatform elsewhere."""
from __future__ import annotations

import os
import socket
import struct
from typing import List, Optional

from .manager import Route, Rule

# message types (rtnetlink(7))
RTM_NEWROUTE, RTM_DELROUTE, RTM_GETROUTE = 24, 25, 26
RTM_NEWRULE, RTM_DELRULE, RTM_GETRULE = 32, 33, 34
RTM_NEWLINK = 16

NLM_F_REQUEST = 0x01
NLM_F_ACK = 0x04
NLM_F_EXCL = 0x200
NLM_F_CREATE = 0x400
NLM_F_REPLACE = 0x100
NLM_F_DUMP = 0x300
NLMSG_DONE = 3
NLMSG_ERROR = 2

# rtmsg fields
AF_INET = 2
RT_TABLE_MAIN = 254
RTPROT_STATIC = 4
RT_SCOPE_UNIVERSE = 0
RT_SCOPE_LINK = 253
RTN_UNICAST = 1

# route attributes
RTA_DST = 1
RTA_OIF = 4
RTA_GATEWAY = 5
RTA_PRIORITY = 6
RTA_TABLE = 15

# rule attributes (fib_rules.h)
FRA_DST = 1
FRA_SRC = 2
FRA_PRIORITY = 6
FRA_FWMARK = 10
FRA_TABLE = 15
FR_ACT_TO_TBL = 1

IFF_UP = 1


def _nlattr(t: int, payload: bytes) -> bytes:
    ln = 4 + len(payload)
    return struct.pack("<HH", ln, t) + payload + b"\x00" * ((4 - ln % 4) % 4)


def _parse_attrs(data: bytes) -> dict:
    out = {}
    off = 0
    while off + 4 <= len(data):
        ln, t = struct.unpack_from("<HH", data, off)
        if ln < 4:
            break
        out[t] = data[off + 4:off + ln]
        off += (ln + 3) & ~3
    return out


class NetlinkError(OSError):
    pass


class NetlinkPlatform:
    """One netlink socket per operation keeps this state-free and
    fork-safe (the reference holds a handle; ours are ~30us to open)."""

    # ------------------------------------------------------ plumbing
    @staticmethod
    def _talk(msg_type: int, flags: int, payload: bytes) -> List[bytes]:
        sk = socket.socket(socket.AF_NETLINK, socket.SOCK_RAW, 0)
        msgs: List[bytes] = []
        try:
            sk.bind((0, 0))
            hdr = struct.pack("<IHHII", 16 + len(payload), msg_type,
                              flags, 1, 0)
            sk.send(hdr + payload)
            done = False
            while not done:
                resp = sk.recv(1 << 16)
                off = 0
                while off + 16 <= len(resp):
                    ln, ty, _fl, _seq, _pid = struct.unpack_from(
                        "<IHHII", resp, off)
                    body = resp[off + 16:off + ln]
                    if ty == NLMSG_ERROR:
                        err = struct.unpack_from("<i", body, 0)[0]
                        if err:
                            raise NetlinkError(-err, os.strerror(-err))
                        return msgs
                    if ty == NLMSG_DONE:
                        return msgs
                    msgs.append(body)
                    off += (ln + 3) & ~3
                if not (flags & NLM_F_DUMP):
                    done = True
        finally:
            sk.close()
        return msgs

    @staticmethod
    def _rtmsg(dst_len: int, table: int, scope: int = RT_SCOPE_UNIVERSE,
               rtype: int = RTN_UNICAST, proto: int = RTPROT_STATIC) -> bytes:
        # struct rtmsg: family, dst_len, src_len, tos, table, protocol,
        # scope, type, flags(u32)
        tbl_byte = table if table < 256 else 0   # RTA_TABLE carries big ids
        return struct.pack("<BBBBBBBBI", AF_INET, dst_len, 0, 0,
                           tbl_byte, proto, scope, rtype, 0)

    @staticmethod
    def _prefix(cidr: str):
        ip, _, plen = cidr.partition("/")
        return socket.inet_aton(ip), int(plen or 32)

    # ------------------------------------------------------- routes
    def add_route(self, r: Route):
        """ref AddRoute netlink_linux.go:45-62."""
        dst, plen = self._prefix(r.prefix)
        scope = RT_SCOPE_LINK if not r.next_hop else RT_SCOPE_UNIVERSE
        payload = self._rtmsg(plen, r.table, scope)
        payload += _nlattr(RTA_TABLE, struct.pack("<I", r.table))
        if plen:
            payload += _nlattr(RTA_DST, dst)
        if r.next_hop:
            payload += _nlattr(RTA_GATEWAY, socket.inet_aton(r.next_hop))
        if r.device:
            payload += _nlattr(RTA_OIF, struct.pack(
                "<I", socket.if_nametoindex(r.device)))
        if r.metric:
            payload += _nlattr(RTA_PRIORITY, struct.pack("<I", r.metric))
        self._talk(RTM_NEWROUTE, NLM_F_REQUEST | NLM_F_ACK |
                   NLM_F_CREATE | NLM_F_REPLACE, payload)

    def del_route(self, r: Route):
        dst, plen = self._prefix(r.prefix)
        # delete matches scope; proto/type 0 = wildcard (iproute2
        # convention)
        scope = RT_SCOPE_LINK if not r.next_hop else RT_SCOPE_UNIVERSE
        payload = self._rtmsg(plen, r.table, scope, rtype=0, proto=0)
        payload += _nlattr(RTA_TABLE, struct.pack("<I", r.table))
        if plen:
            payload += _nlattr(RTA_DST, dst)
        self._talk(RTM_DELROUTE, NLM_F_REQUEST | NLM_F_ACK, payload)

    def routes(self, table: int) -> List[Route]:
        """ref GetRoutes netlink_linux.go:82-101 (dump + filter)."""
        payload = self._rtmsg(0, 0, proto=0, rtype=0)
        out: List[Route] = []
        for body in self._talk(RTM_GETROUTE,
                               NLM_F_REQUEST | NLM_F_DUMP, payload):
            fam, dst_len = body[0], body[1]
            tbl_byte = body[4]
            if fam != AF_INET:
                continue
            attrs = _parse_attrs(body[12:])
            tbl = struct.unpack("<I", attrs[RTA_TABLE])[0] \
                if RTA_TABLE in attrs else tbl_byte
            if tbl != table:
                continue
            dst = socket.inet_ntoa(attrs[RTA_DST]) \
                if RTA_DST in attrs else "0.0.0.0"
            gw = socket.inet_ntoa(attrs[RTA_GATEWAY]) \
                if RTA_GATEWAY in attrs else ""
            dev = ""
            if RTA_OIF in attrs:
                try:
                    dev = socket.if_indextoname(
                        struct.unpack("<I", attrs[RTA_OIF])[0])
                except OSError:
                    pass
            metric = struct.unpack("<I", attrs[RTA_PRIORITY])[0] \
                if RTA_PRIORITY in attrs else 0
            out.append(Route(prefix=f"{dst}/{dst_len}", next_hop=gw,
                             device=dev, table=tbl, metric=metric))
        return out

    def flush_table(self, table: int) -> int:
        """ref FlushTable netlink_linux.go:103-118."""
        rs = self.routes(table)
        for r in rs:
            try:
                self.del_route(r)
            except NetlinkError:
                pass
        return len(rs)

    # -------------------------------------------------- policy rules
    @staticmethod
    def _frh(src_len: int, dst_len: int, table: int) -> bytes:
        # struct fib_rule_hdr: family, dst_len, src_len, tos, table,
        # res1, res2, action, flags(u32)
        tbl_byte = table if table < 256 else 0
        return struct.pack("<BBBBBBBBI", AF_INET, dst_len, src_len, 0,
                           tbl_byte, 0, 0, FR_ACT_TO_TBL, 0)

    def add_rule(self, r: Rule):
        """ref AddRule netlink_linux.go:120-133 (ip rule from SRC
        lookup TABLE)."""
        src, src_len = self._prefix(r.src) if r.src else (b"", 0)
        payload = self._frh(src_len, 0, r.table)
        payload += _nlattr(FRA_TABLE, struct.pack("<I", r.table))
        if src_len:
            payload += _nlattr(FRA_SRC, src)
        if r.priority:
            payload += _nlattr(FRA_PRIORITY, struct.pack(
                "<I", r.priority))
        if r.fwmark:
            payload += _nlattr(FRA_FWMARK, struct.pack("<I", r.fwmark))
        self._talk(RTM_NEWRULE, NLM_F_REQUEST | NLM_F_ACK |
                   NLM_F_CREATE | NLM_F_EXCL, payload)

    def del_rule(self, r: Rule):
        src, src_len = self._prefix(r.src) if r.src else (b"", 0)
        payload = self._frh(src_len, 0, r.table)
        payload += _nlattr(FRA_TABLE, struct.pack("<I", r.table))
        if src_len:
            payload += _nlattr(FRA_SRC, src)
        if r.priority:
            payload += _nlattr(FRA_PRIORITY, struct.pack(
                "<I", r.priority))
        self._talk(RTM_DELRULE, NLM_F_REQUEST | NLM_F_ACK, payload)

    def rules(self) -> List[Rule]:
        payload = self._frh(0, 0, 0)
        out: List[Rule] = []
        for body in self._talk(RTM_GETRULE,
                               NLM_F_REQUEST | NLM_F_DUMP, payload):
            fam, _dst_len, src_len = body[0], body[1], body[2]
            if fam != AF_INET:
                continue
            attrs = _parse_attrs(body[12:])
            tbl = struct.unpack("<I", attrs[FRA_TABLE])[0] \
                if FRA_TABLE in attrs else body[4]
            src = ""
            if FRA_SRC in attrs:
                src = f"{socket.inet_ntoa(attrs[FRA_SRC])}/{src_len}"
            prio = struct.unpack("<I", attrs[FRA_PRIORITY])[0] \
                if FRA_PRIORITY in attrs else 0
            fwmark = struct.unpack("<I", attrs[FRA_FWMARK])[0] \
                if FRA_FWMARK in attrs else 0
            out.append(Rule(src=src, table=tbl, priority=prio,
                            fwmark=fwmark))
        return out

    # ---------------------------------------------------- interfaces
    def set_interface_up(self, name: str):
        """ref SetInterfaceUp netlink_linux.go:197-209."""
        from ..dataplane.afxdp import link_up
        link_up(name)

    def interface_index(self, name: str) -> int:
        return socket.if_nametoindex(name)
