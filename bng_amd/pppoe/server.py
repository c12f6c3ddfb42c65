"""PPPoE access concentrator (ref pkg/pppoe — the reference's largest
package): discovery PADI->PADO->PADR->PADS->PADT (server.go:335-465),
RFC1661 LCP negotiation state machines (lcp.go), PAP+CHAP authentication
against RADIUS with rate limiting (auth.go:202-580), IPCP and IPV6CP
address negotiation (ipcp.go:92-731, ipv6cp.go:90-674), LCP echo
keepalives (keepalive.go:46-295) and graceful teardown (teardown.go).

Transport-agnostic: `handle_frame(bytes) -> [bytes]` consumes one
Ethernet frame and returns the frames to transmit, so tests drive the
full handshake in-process (the AF_PACKET raw-socket transport of
socket_linux.go plugs in at deployment; the reference stubs it the same
way for non-Linux tests, socket_stub.go).
"""
from __future__ import annotations

import hashlib
import hmac
import os
import struct
import threading
import time
from dataclasses import dataclass, field
from typing import Callable, Dict, List, Optional, Tuple

from ..dataplane.packets import ip2u32, u32_to_ip
from . import codec as C

# session phases
PH_DISCOVERY = "discovery"
PH_LCP = "lcp"
PH_AUTH = "auth"
PH_NETWORK = "network"
PH_OPEN = "open"
PH_TERMINATED = "terminated"

# mini-FSM states for each CP (server side of RFC 1661)
ST_CLOSED = "closed"
ST_REQ_SENT = "req-sent"
ST_ACK_RCVD = "ack-rcvd"
ST_ACK_SENT = "ack-sent"
ST_OPENED = "opened"


@dataclass
class CPState:
    state: str = ST_CLOSED
    our_ident: int = 0
    retransmits: int = 0     # Restart counter sends (RFC1661 Max-Configure)
    failures: int = 0        # Nak/Rej rounds (RFC1661 Max-Failure)
    last_req: bytes = b""

    def reset(self):
        self.state = ST_CLOSED
        self.retransmits = 0
        self.failures = 0


@dataclass
class Session:
    session_id: int
    client_mac: bytes
    server_mac: bytes
    phase: str = PH_LCP
    lcp: CPState = field(default_factory=CPState)
    ipcp: CPState = field(default_factory=CPState)
    ipv6cp: CPState = field(default_factory=CPState)
    our_magic: int = 0
    peer_magic: int = 0
    peer_mru: int = 1492
    our_mru: int = 0               # 0 = announce server default
    magic_loops: int = 0           # looped-link detections (RFC1661 §6.4)
    lcp_suppressed: frozenset = frozenset()   # options the peer Rejected
    auth_proto: int = C.PROTO_CHAP
    chap_challenge: bytes = b""
    chap_ident: int = 0
    username: str = ""
    ip: int = 0
    dns1: int = 0
    dns2: int = 0
    ifid: bytes = b""
    policy_name: str = ""
    created: float = field(default_factory=time.time)
    # keepalive
    echo_ident: int = 0
    echo_pending: int = 0
    last_echo_reply: float = field(default_factory=time.time)
    auth_attempts: int = 0
    acct_session_id: str = ""


class PPPoEServer:
    MAX_AUTH_ATTEMPTS = 3      # ref auth.go rate limiting (per session)
    MAX_CONFIGURE = 10         # RFC1661 Max-Configure restart counter
    MAX_FAILURE = 5            # RFC1661 Max-Failure Nak/Rej rounds
    MAX_MAGIC_LOOPS = 3        # looped-link give-up (RFC1661 §6.4)
    AUTH_FAIL_WINDOW = 60.0    # per-MAC throttle (ref auth.go:202-580:
    AUTH_FAIL_LIMIT = 5        # rate limiting survives session churn)
    AUTH_LOCKOUT = 30.0

    def __init__(self, server_mac: bytes, ac_name: str = "bng-amd",
                 service_name: str = "", auth: str = "chap",
                 mru: int = 1492, cookie_secret: Optional[bytes] = None,
                 echo_interval: float = 30.0, echo_fails: int = 3,
                 session_timeout: float = 0.0):
        self.server_mac = server_mac
        self.ac_name = ac_name
        self.service_name = service_name
        self.auth_kind = auth            # chap | pap | none
        self.mru = mru
        self.cookie_secret = cookie_secret or os.urandom(16)
        self.echo_interval = echo_interval
        self.echo_fails = echo_fails
        # absolute session lifetime, 0 = unlimited (ref
        # pppoe-session-timeout flag / session.go lifetime handling)
        self.session_timeout = session_timeout
        self.sessions: Dict[int, Session] = {}
        self.by_mac: Dict[bytes, int] = {}
        self._next_sid = 1
        self._lock = threading.RLock()
        # collaborators
        self.radius = None               # radius.Client (PAP/CHAP verify)
        self.local_users: Dict[str, str] = {}
        self.allocator: Optional[Callable[[str], str]] = None  # user -> ip
        self.releaser: Optional[Callable[[str], None]] = None
        self.dns = (0, 0)
        self.on_session_open: Optional[Callable[[Session], None]] = None
        self.on_session_close: Optional[Callable[[Session], None]] = None
        self.stats = {k: 0 for k in (
            "padi", "pado", "padr", "pads", "padt_rx", "padt_tx",
            "lcp_opened", "auth_ok", "auth_fail", "ipcp_opened",
            "ipv6cp_opened", "sessions_open", "echo_timeout",
            "term_rx", "loopback_detected", "restart_exhausted",
            "auth_throttled")}
        # per-MAC auth-failure timestamps (survives session teardown, so
        # a client cannot reset the limit by re-discovering)
        self._auth_fails: Dict[bytes, List[float]] = {}

    # ------------------------------------------------------------ entry
    def handle_frame(self, frame: bytes) -> List[bytes]:
        if len(frame) < 14:
            return []
        et = struct.unpack_from(">H", frame, 12)[0]
        try:
            if et == C.ETH_PPPOE_DISC:
                return self._handle_discovery(C.DiscoveryPacket.decode(frame))
            if et == C.ETH_PPPOE_SESS:
                return self._handle_session(C.SessionPacket.decode(frame))
        except ValueError:
            return []
        return []

    # -------------------------------------------------------- discovery
    def _cookie(self, mac: bytes) -> bytes:
        return hmac.new(self.cookie_secret, mac, hashlib.sha256).digest()[:16]

    def _handle_discovery(self, p: C.DiscoveryPacket) -> List[bytes]:
        if p.code == C.PADI:
            self.stats["padi"] += 1
            svc = C.get_tag(p.tags, C.TAG_SERVICE_NAME) or b""
            if self.service_name and svc and \
                    svc.decode(errors="replace") != self.service_name:
                return []     # not our service
            tags = [(C.TAG_AC_NAME, self.ac_name.encode()),
                    (C.TAG_SERVICE_NAME, svc),
                    (C.TAG_AC_COOKIE, self._cookie(p.src_mac))]
            hu = C.get_tag(p.tags, C.TAG_HOST_UNIQ)
            if hu is not None:
                tags.append((C.TAG_HOST_UNIQ, hu))
            self.stats["pado"] += 1
            return [C.DiscoveryPacket(C.PADO, 0, tags,
                                      src_mac=self.server_mac,
                                      dst_mac=p.src_mac).encode()]
        if p.code == C.PADR:
            self.stats["padr"] += 1
            if self._auth_locked(p.src_mac):
                self.stats["auth_throttled"] += 1
                return [C.DiscoveryPacket(
                    C.PADS, 0,
                    [(C.TAG_GENERIC_ERROR, b"too many auth failures")],
                    src_mac=self.server_mac, dst_mac=p.src_mac).encode()]
            cookie = C.get_tag(p.tags, C.TAG_AC_COOKIE)
            if cookie != self._cookie(p.src_mac):
                return [C.DiscoveryPacket(
                    C.PADS, 0,
                    [(C.TAG_GENERIC_ERROR, b"bad cookie")],
                    src_mac=self.server_mac, dst_mac=p.src_mac).encode()]
            with self._lock:
                old_sid = self.by_mac.get(p.src_mac)
                if old_sid is not None:
                    self._destroy(old_sid, notify=False)
                sid = self._next_sid
                self._next_sid = self._next_sid % 0xFFFE + 1
                s = Session(sid, p.src_mac, self.server_mac)
                s.our_magic = struct.unpack(
                    ">I", hashlib.md5(os.urandom(8)).digest()[:4])[0]
                if self.auth_kind == "pap":
                    s.auth_proto = C.PROTO_PAP
                self.sessions[sid] = s
                self.by_mac[p.src_mac] = sid
            tags = [(C.TAG_AC_NAME, self.ac_name.encode()),
                    (C.TAG_SERVICE_NAME,
                     C.get_tag(p.tags, C.TAG_SERVICE_NAME) or b"")]
            hu = C.get_tag(p.tags, C.TAG_HOST_UNIQ)
            if hu is not None:
                tags.append((C.TAG_HOST_UNIQ, hu))
            self.stats["pads"] += 1
            out = [C.DiscoveryPacket(C.PADS, sid, tags,
                                     src_mac=self.server_mac,
                                     dst_mac=p.src_mac).encode()]
            out += self._send_lcp_req(s)
            return out
        if p.code == C.PADT:
            self.stats["padt_rx"] += 1
            with self._lock:
                sid = self.by_mac.get(p.src_mac)
            if sid is not None:
                self._destroy(sid)
            return []
        return []

    # ------------------------------------------------------------- LCP
    def _auth_locked(self, mac: bytes, now: Optional[float] = None) -> bool:
        now = now or time.time()
        fails = self._auth_fails.get(mac)
        if not fails:
            return False
        fails[:] = [t for t in fails if now - t < self.AUTH_FAIL_WINDOW]
        if not fails:
            del self._auth_fails[mac]
            return False
        return (len(fails) >= self.AUTH_FAIL_LIMIT and
                now - fails[-1] < self.AUTH_LOCKOUT)

    def _send_lcp_req(self, s: Session) -> List[bytes]:
        """(Re)send our Configure-Request, honoring options the peer
        Rejected and values it Nak-suggested, bounded by the RFC1661
        Max-Configure restart counter (ref lcp.go option negotiation)."""
        s.lcp.retransmits += 1
        if s.lcp.retransmits > self.MAX_CONFIGURE:
            self.stats["restart_exhausted"] += 1
            return self._teardown(s)
        s.lcp.our_ident = (s.lcp.our_ident + 1) & 0xFF
        opts = []
        if C.LCP_OPT_MRU not in s.lcp_suppressed:
            opts.append((C.LCP_OPT_MRU,
                         struct.pack(">H", s.our_mru or self.mru)))
        if C.LCP_OPT_MAGIC not in s.lcp_suppressed:
            opts.append((C.LCP_OPT_MAGIC, struct.pack(">I", s.our_magic)))
        if C.LCP_OPT_AUTH not in s.lcp_suppressed:
            if self.auth_kind == "chap":
                opts.append((C.LCP_OPT_AUTH,
                             struct.pack(">HB", C.PROTO_CHAP, 5)))
            elif self.auth_kind == "pap":
                opts.append((C.LCP_OPT_AUTH,
                             struct.pack(">H", C.PROTO_PAP)))
        req = C.CPPacket(C.CONF_REQ, s.lcp.our_ident,
                         C.encode_opts(opts)).encode()
        s.lcp.last_req = req
        if s.lcp.state == ST_CLOSED:
            s.lcp.state = ST_REQ_SENT
        return [self._sess_frame(s, C.PROTO_LCP, req)]

    def _sess_frame(self, s: Session, proto: int, payload: bytes) -> bytes:
        return C.SessionPacket(s.session_id, proto, payload,
                               src_mac=self.server_mac,
                               dst_mac=s.client_mac).encode()

    def _handle_session(self, p: C.SessionPacket) -> List[bytes]:
        with self._lock:
            s = self.sessions.get(p.session_id)
        if s is None or s.client_mac != p.src_mac:
            return []
        if p.ppp_proto == C.PROTO_LCP:
            return self._lcp(s, C.CPPacket.decode(p.payload))
        if p.ppp_proto == C.PROTO_CHAP and s.phase == PH_AUTH:
            return self._chap(s, C.CPPacket.decode(p.payload))
        if p.ppp_proto == C.PROTO_PAP and s.phase == PH_AUTH:
            return self._pap(s, C.CPPacket.decode(p.payload))
        if p.ppp_proto == C.PROTO_IPCP and s.phase in (PH_NETWORK, PH_OPEN):
            return self._ncp(s, s.ipcp, C.CPPacket.decode(p.payload),
                             C.PROTO_IPCP)
        if p.ppp_proto == C.PROTO_IPV6CP and s.phase in (PH_NETWORK, PH_OPEN):
            return self._ncp(s, s.ipv6cp, C.CPPacket.decode(p.payload),
                             C.PROTO_IPV6CP)
        return []

    def _lcp(self, s: Session, cp: C.CPPacket) -> List[bytes]:
        out: List[bytes] = []
        if cp.code == C.CONF_REQ:
            opts = C.decode_opts(cp.data)
            naks, rejs = [], []
            for t, v in opts:
                if t == C.LCP_OPT_MRU:
                    mru = struct.unpack(">H", v)[0]
                    # PPPoE caps MRU at 1492 (RFC 2516 §7); too-small
                    # MRUs are also bargained up (ref lcp.go NAK rules)
                    if mru < 576 or mru > 1492:
                        naks.append((t, struct.pack(">H", self.mru)))
                    else:
                        s.peer_mru = mru
                elif t == C.LCP_OPT_MAGIC:
                    peer_magic = struct.unpack(">I", v)[0]
                    if peer_magic == 0:
                        # zero magic defeats loop detection: NAK with
                        # a proper random number (ref lcp_test "NAK
                        # zero magic number")
                        naks.append((t, os.urandom(4)))
                    elif peer_magic == s.our_magic:
                        # our own magic coming back: looped link
                        # (RFC1661 §6.4) — Nak with a fresh number;
                        # give up after MAX_MAGIC_LOOPS
                        s.magic_loops += 1
                        self.stats["loopback_detected"] += 1
                        if s.magic_loops >= self.MAX_MAGIC_LOOPS:
                            return self._teardown(s)
                        naks.append((t, os.urandom(4)))
                    else:
                        s.peer_magic = peer_magic
                elif t in (C.LCP_OPT_PFC, C.LCP_OPT_ACFC):
                    rejs.append((t, v))       # we don't compress
                elif t == C.LCP_OPT_AUTH:
                    rejs.append((t, v))       # client must not auth US
                else:
                    rejs.append((t, v))
            if rejs:
                out.append(self._sess_frame(s, C.PROTO_LCP, C.CPPacket(
                    C.CONF_REJ, cp.identifier,
                    C.encode_opts(rejs)).encode()))
            elif naks:
                out.append(self._sess_frame(s, C.PROTO_LCP, C.CPPacket(
                    C.CONF_NAK, cp.identifier,
                    C.encode_opts(naks)).encode()))
            else:
                out.append(self._sess_frame(s, C.PROTO_LCP, C.CPPacket(
                    C.CONF_ACK, cp.identifier, cp.data).encode()))
                if s.lcp.state == ST_REQ_SENT:
                    s.lcp.state = ST_ACK_SENT
                elif s.lcp.state == ST_ACK_RCVD:
                    s.lcp.state = ST_OPENED
                    out += self._lcp_opened(s)
        elif cp.code == C.CONF_ACK:
            s.lcp.retransmits = 0
            s.lcp.failures = 0
            if s.lcp.state == ST_REQ_SENT:
                s.lcp.state = ST_ACK_RCVD
            elif s.lcp.state == ST_ACK_SENT:
                s.lcp.state = ST_OPENED
                out += self._lcp_opened(s)
        elif cp.code in (C.CONF_NAK, C.CONF_REJ):
            # real option bargaining (RFC1661; ref lcp.go): adopt Nak'd
            # values, drop Rejected options, bounded by Max-Failure
            s.lcp.failures += 1
            if s.lcp.failures > self.MAX_FAILURE:
                self.stats["restart_exhausted"] += 1
                return self._teardown(s)
            peer_opts = C.decode_opts(cp.data)
            if cp.code == C.CONF_NAK:
                for t, v in peer_opts:
                    if t == C.LCP_OPT_MRU and len(v) == 2:
                        sug = struct.unpack(">H", v)[0]
                        if sug >= 576:
                            s.our_mru = min(sug, self.mru)
                    elif t == C.LCP_OPT_MAGIC and len(v) == 4:
                        s.our_magic = struct.unpack(">I", v)[0]
                    # auth-proto Nak: we do not downgrade the
                    # configured authenticator — re-request (the
                    # Max-Failure bound terminates a peer that will
                    # never take it; ref auth is non-negotiable too)
            else:
                dropped = {t for t, _ in peer_opts}
                if C.LCP_OPT_AUTH in dropped and self.auth_kind != "none":
                    # peer refuses to authenticate: no service
                    return self._teardown(s)
                s.lcp_suppressed = frozenset(s.lcp_suppressed | dropped)
            out += self._send_lcp_req(s)
        elif cp.code == C.ECHO_REQ:
            if (len(cp.data) >= 4 and
                    struct.unpack(">I", cp.data[:4])[0] == s.our_magic):
                # echo carrying OUR magic: looped link (RFC1661 §5.8)
                s.magic_loops += 1
                self.stats["loopback_detected"] += 1
                if s.magic_loops >= self.MAX_MAGIC_LOOPS:
                    return self._teardown(s)
            out.append(self._sess_frame(s, C.PROTO_LCP, C.CPPacket(
                C.ECHO_REP, cp.identifier,
                struct.pack(">I", s.our_magic)).encode()))
        elif cp.code == C.ECHO_REP:
            s.echo_pending = 0
            s.last_echo_reply = time.time()
        elif cp.code == C.TERM_REQ:
            self.stats["term_rx"] += 1
            out.append(self._sess_frame(s, C.PROTO_LCP, C.CPPacket(
                C.TERM_ACK, cp.identifier).encode()))
            self._destroy(s.session_id)
        return out

    def _lcp_opened(self, s: Session) -> List[bytes]:
        self.stats["lcp_opened"] += 1
        if self.auth_kind == "none":
            return self._start_network(s)
        s.phase = PH_AUTH
        if self.auth_kind == "chap":
            s.chap_ident = (s.chap_ident + 1) & 0xFF
            s.chap_challenge = os.urandom(16)
            data = bytes([len(s.chap_challenge)]) + s.chap_challenge + \
                self.ac_name.encode()
            return [self._sess_frame(s, C.PROTO_CHAP, C.CPPacket(
                C.CHAP_CHALLENGE, s.chap_ident, data).encode())]
        return []    # PAP: wait for client Auth-Req

    # ------------------------------------------------------------- auth
    def _auth_fail(self, s: Session, proto: int, ident: int,
                   msg: bytes) -> List[bytes]:
        self.stats["auth_fail"] += 1
        s.auth_attempts += 1
        self._auth_fails.setdefault(s.client_mac, []).append(time.time())
        code = C.CHAP_FAILURE if proto == C.PROTO_CHAP else C.PAP_AUTH_NAK
        body = msg if proto == C.PROTO_CHAP else bytes([len(msg)]) + msg
        out = [self._sess_frame(s, proto,
                                C.CPPacket(code, ident, body).encode())]
        if s.auth_attempts >= self.MAX_AUTH_ATTEMPTS:
            out += self._teardown(s)          # rate limit (ref auth.go)
        return out

    def _chap(self, s: Session, cp: C.CPPacket) -> List[bytes]:
        if cp.code != C.CHAP_RESPONSE or not cp.data:
            return []
        vlen = cp.data[0]
        value = cp.data[1:1 + vlen]
        username = cp.data[1 + vlen:].decode(errors="replace")
        s.username = username
        ok, policy = self._verify_chap(username, cp.identifier, value,
                                       s.chap_challenge)
        if not ok:
            return self._auth_fail(s, C.PROTO_CHAP, cp.identifier,
                                   b"authentication failed")
        s.policy_name = policy
        self.stats["auth_ok"] += 1
        out = [self._sess_frame(s, C.PROTO_CHAP, C.CPPacket(
            C.CHAP_SUCCESS, cp.identifier, b"welcome").encode())]
        out += self._start_network(s)
        return out

    def _pap(self, s: Session, cp: C.CPPacket) -> List[bytes]:
        if cp.code != C.PAP_AUTH_REQ or not cp.data:
            return []
        ulen = cp.data[0]
        username = cp.data[1:1 + ulen].decode(errors="replace")
        plen = cp.data[1 + ulen]
        password = cp.data[2 + ulen:2 + ulen + plen].decode(errors="replace")
        s.username = username
        ok, policy = self._verify_pap(username, password)
        # password zeroization discipline (ref auth.go): drop the ref now
        password = ""
        if not ok:
            return self._auth_fail(s, C.PROTO_PAP, cp.identifier, b"denied")
        s.policy_name = policy
        self.stats["auth_ok"] += 1
        out = [self._sess_frame(s, C.PROTO_PAP, C.CPPacket(
            C.PAP_AUTH_ACK, cp.identifier, b"\x07welcome").encode())]
        out += self._start_network(s)
        return out

    def _verify_pap(self, username: str, password: str) -> Tuple[bool, str]:
        if self.radius is not None:
            try:
                res = self.radius.authenticate(username, password)
                return res.success, res.policy_name
            except Exception:
                return False, ""
        pw = self.local_users.get(username)
        return (pw is not None and pw == password), ""

    def _verify_chap(self, username: str, ident: int, value: bytes,
                     challenge: bytes) -> Tuple[bool, str]:
        if self.radius is not None:
            try:
                res = self.radius.authenticate(
                    username, "", chap=(challenge, bytes([ident]) + value))
                return res.success, res.policy_name
            except Exception:
                return False, ""
        pw = self.local_users.get(username)
        if pw is None:
            return False, ""
        expect = C.chap_md5_response(ident, pw.encode(), challenge)
        return hmac.compare_digest(expect, value), ""

    # ---------------------------------------------------------- network
    def _start_network(self, s: Session) -> List[bytes]:
        s.phase = PH_NETWORK
        if self.allocator is not None:
            try:
                s.ip = ip2u32(self.allocator(s.username or
                                             s.client_mac.hex()))
            except Exception:
                return self._teardown(s)
        s.dns1, s.dns2 = self.dns
        # our IPCP Conf-Req announces the BNG-side address
        s.ipcp.our_ident += 1
        req = C.CPPacket(C.CONF_REQ, s.ipcp.our_ident, C.encode_opts(
            [(C.IPCP_OPT_IP, struct.pack(">I", ip2u32("10.255.255.1")))]))
        s.ipcp.state = ST_REQ_SENT
        out = [self._sess_frame(s, C.PROTO_IPCP, req.encode())]
        # IPV6CP in parallel: our interface id
        s.ifid = hashlib.md5(self.server_mac).digest()[:8]
        s.ipv6cp.our_ident += 1
        req6 = C.CPPacket(C.CONF_REQ, s.ipv6cp.our_ident, C.encode_opts(
            [(C.IPV6CP_OPT_IFID, s.ifid)]))
        s.ipv6cp.state = ST_REQ_SENT
        out.append(self._sess_frame(s, C.PROTO_IPV6CP, req6.encode()))
        return out

    def _ncp(self, s: Session, st: CPState, cp: C.CPPacket,
             proto: int) -> List[bytes]:
        out: List[bytes] = []
        if cp.code == C.CONF_REQ:
            opts = C.decode_opts(cp.data)
            naks = []
            rejects = []
            if proto == C.PROTO_IPCP:
                # unsupported options (VJ IP-compression etc.) are
                # Configure-Rejected, not silently accepted (RFC 1332;
                # ref ipcp.go compression rejection)
                known = {C.IPCP_OPT_IP, C.IPCP_OPT_DNS1, C.IPCP_OPT_DNS2}
                rejects = [(t, v) for (t, v) in opts if t not in known]
                want = C.get_opt(opts, C.IPCP_OPT_IP)
                if want is None or struct.unpack(">I", want)[0] != s.ip:
                    naks.append((C.IPCP_OPT_IP, struct.pack(">I", s.ip)))
                d1 = C.get_opt(opts, C.IPCP_OPT_DNS1)
                if d1 is not None and s.dns1 and \
                        struct.unpack(">I", d1)[0] != s.dns1:
                    naks.append((C.IPCP_OPT_DNS1,
                                 struct.pack(">I", s.dns1)))
                d2 = C.get_opt(opts, C.IPCP_OPT_DNS2)
                if d2 is not None and s.dns2 and \
                        struct.unpack(">I", d2)[0] != s.dns2:
                    naks.append((C.IPCP_OPT_DNS2,
                                 struct.pack(">I", s.dns2)))
            else:
                ifid = C.get_opt(opts, C.IPV6CP_OPT_IFID)
                if ifid is None or ifid == b"\x00" * 8 or ifid == s.ifid:
                    # zero or colliding interface-id: suggest one
                    # derived from the client MAC (RFC 5072 NAK rule)
                    naks.append((C.IPV6CP_OPT_IFID,
                                 hashlib.md5(s.client_mac).digest()[:8]))
            if rejects:
                out.append(self._sess_frame(s, proto, C.CPPacket(
                    C.CONF_REJ, cp.identifier,
                    C.encode_opts(rejects)).encode()))
            elif naks:
                out.append(self._sess_frame(s, proto, C.CPPacket(
                    C.CONF_NAK, cp.identifier,
                    C.encode_opts(naks)).encode()))
            else:
                out.append(self._sess_frame(s, proto, C.CPPacket(
                    C.CONF_ACK, cp.identifier, cp.data).encode()))
                if st.state == ST_REQ_SENT:
                    st.state = ST_ACK_SENT
                elif st.state == ST_ACK_RCVD:
                    st.state = ST_OPENED
                    out += self._ncp_opened(s, proto)
        elif cp.code == C.CONF_ACK:
            if st.state == ST_REQ_SENT:
                st.state = ST_ACK_RCVD
            elif st.state == ST_ACK_SENT:
                st.state = ST_OPENED
                out += self._ncp_opened(s, proto)
        elif cp.code in (C.CONF_NAK, C.CONF_REJ):
            pass   # keep our request; client retries
        elif cp.code == C.TERM_REQ:
            out.append(self._sess_frame(s, proto, C.CPPacket(
                C.TERM_ACK, cp.identifier).encode()))
            st.reset()
        return out

    def _ncp_opened(self, s: Session, proto: int) -> List[bytes]:
        if proto == C.PROTO_IPCP:
            self.stats["ipcp_opened"] += 1
        else:
            self.stats["ipv6cp_opened"] += 1
        if s.phase != PH_OPEN and s.ipcp.state == ST_OPENED:
            s.phase = PH_OPEN
            self.stats["sessions_open"] += 1
            if self.on_session_open:
                try:
                    self.on_session_open(s)
                except Exception:
                    pass
        return []

    # --------------------------------------------------------- teardown
    def _teardown(self, s: Session) -> List[bytes]:
        """Graceful teardown (ref teardown.go): TERM-REQ then PADT."""
        out = [self._sess_frame(s, C.PROTO_LCP,
                                C.CPPacket(C.TERM_REQ,
                                           (s.lcp.our_ident + 1) & 0xFF,
                                           b"teardown").encode()),
               C.DiscoveryPacket(C.PADT, s.session_id,
                                 [(C.TAG_GENERIC_ERROR, b"closed")],
                                 src_mac=self.server_mac,
                                 dst_mac=s.client_mac).encode()]
        self.stats["padt_tx"] += 1
        self._destroy(s.session_id)
        return out

    def terminate_session(self, session_id: int) -> List[bytes]:
        with self._lock:
            s = self.sessions.get(session_id)
        return self._teardown(s) if s else []

    def _destroy(self, sid: int, notify: bool = True):
        with self._lock:
            s = self.sessions.pop(sid, None)
            if s is None:
                return
            if self.by_mac.get(s.client_mac) == sid:
                del self.by_mac[s.client_mac]
        s.phase = PH_TERMINATED
        if self.releaser is not None and s.username:
            try:
                self.releaser(s.username)
            except Exception:
                pass
        if notify and self.on_session_close:
            try:
                self.on_session_close(s)
            except Exception:
                pass

    # -------------------------------------------------------- keepalive
    def tick(self, now: Optional[float] = None) -> List[bytes]:
        """Periodic driver: send LCP echo requests on open sessions and
        tear down sessions past the miss threshold (ref keepalive.go)."""
        now = now or time.time()
        out: List[bytes] = []
        with self._lock:
            open_sessions = [s for s in self.sessions.values()
                             if s.phase == PH_OPEN]
            pending_lcp = [s for s in self.sessions.values()
                           if s.phase == PH_LCP and
                           s.lcp.state in (ST_REQ_SENT, ST_ACK_RCVD,
                                           ST_ACK_SENT)]
        # restart timer: retransmit our un-Acked Configure-Request,
        # bounded by Max-Configure (RFC1661 restart counter)
        for s in pending_lcp:
            out += self._send_lcp_req(s)
        for s in open_sessions:
            if self.session_timeout and \
                    now - s.created >= self.session_timeout:
                self.stats["session_timeout"] = \
                    self.stats.get("session_timeout", 0) + 1
                out += self._teardown(s)
                continue
            if s.echo_pending >= self.echo_fails:
                self.stats["echo_timeout"] += 1
                out += self._teardown(s)
                continue
            s.echo_ident = (s.echo_ident + 1) & 0xFF
            s.echo_pending += 1
            out.append(self._sess_frame(s, C.PROTO_LCP, C.CPPacket(
                C.ECHO_REQ, s.echo_ident,
                struct.pack(">I", s.our_magic)).encode()))
        return out

    def session_count(self) -> int:
        with self._lock:
            return len(self.sessions)
