"""PPPoE tests — simulated client driving the full lifecycle
(ref pkg/pppoe/*_test.go: discovery, LCP FSM, PAP/CHAP, IPCP/IPV6CP,
keepalive, teardown)."""
import struct

import pytest

from bng_amd.pppoe import codec as C
from bng_amd.pppoe.server import (PH_AUTH, PH_OPEN, PPPoEServer, ST_OPENED)
from bng_amd.radius.client import Client as RadiusClient
from bng_amd.radius.server import RadiusServer

SRV_MAC = bytes.fromhex("020000000001")
CLI_MAC = bytes.fromhex("aabbcc000001")


class SimClient:
    """Minimal PPPoE/PPP client for driving the server."""

    def __init__(self, server: PPPoEServer, mac=CLI_MAC,
                 username="alice", password="pw1"):
        self.srv = server
        self.mac = mac
        self.username = username
        self.password = password
        self.session_id = 0
        self.magic = 0x1234ABCD
        self.got_ip = None
        self.got_dns = None
        self.lcp_acked = self.ipcp_acked = False
        self.inbox = []

    def _push(self, frames):
        out = []
        for f in frames:
            out += self._react(f)
        return out

    def send(self, frame_bytes):
        replies = self.srv.handle_frame(frame_bytes)
        # feed server replies through the client reactor (and any
        # counter-replies back to the server) until quiescent
        pending = list(replies)
        while pending:
            f = pending.pop(0)
            for counter in self._react(f):
                pending.extend(self.srv.handle_frame(counter))

    def discover(self, host_uniq=b"HU1"):
        padi = C.DiscoveryPacket(C.PADI, 0, [
            (C.TAG_SERVICE_NAME, b""), (C.TAG_HOST_UNIQ, host_uniq)],
            src_mac=self.mac).encode()
        self.send(padi)

    def _react(self, frame):
        et = struct.unpack_from(">H", frame, 12)[0]
        if et == C.ETH_PPPOE_DISC:
            p = C.DiscoveryPacket.decode(frame)
            self.inbox.append(p)
            if p.code == C.PADO:
                cookie = C.get_tag(p.tags, C.TAG_AC_COOKIE)
                padr = C.DiscoveryPacket(C.PADR, 0, [
                    (C.TAG_SERVICE_NAME, b""),
                    (C.TAG_AC_COOKIE, cookie),
                    (C.TAG_HOST_UNIQ, C.get_tag(p.tags, C.TAG_HOST_UNIQ)
                     or b"")], src_mac=self.mac).encode()
                return [padr]
            if p.code == C.PADS and not C.get_tag(p.tags,
                                                  C.TAG_GENERIC_ERROR):
                self.session_id = p.session_id
            return []
        if et != C.ETH_PPPOE_SESS:
            return []
        p = C.SessionPacket.decode(frame)
        self.inbox.append(p)
        cp = C.CPPacket.decode(p.payload)
        out = []
        if p.ppp_proto == C.PROTO_LCP:
            if cp.code == C.CONF_REQ:
                out.append(self._sess(C.PROTO_LCP, C.CPPacket(
                    C.CONF_ACK, cp.identifier, cp.data)))
                if not self.lcp_acked:
                    self.lcp_acked = True
                    out.append(self._sess(C.PROTO_LCP, C.CPPacket(
                        C.CONF_REQ, 1, C.encode_opts(
                            [(C.LCP_OPT_MRU, struct.pack(">H", 1492)),
                             (C.LCP_OPT_MAGIC,
                              struct.pack(">I", self.magic))]))))
            elif cp.code == C.ECHO_REQ:
                out.append(self._sess(C.PROTO_LCP, C.CPPacket(
                    C.ECHO_REP, cp.identifier,
                    struct.pack(">I", self.magic))))
        elif p.ppp_proto == C.PROTO_CHAP:
            if cp.code == C.CHAP_CHALLENGE:
                clen = cp.data[0]
                challenge = cp.data[1:1 + clen]
                resp = C.chap_md5_response(cp.identifier,
                                           self.password.encode(),
                                           challenge)
                out.append(self._sess(C.PROTO_CHAP, C.CPPacket(
                    C.CHAP_RESPONSE, cp.identifier,
                    bytes([len(resp)]) + resp + self.username.encode())))
        elif p.ppp_proto == C.PROTO_IPCP:
            if cp.code == C.CONF_REQ:
                out.append(self._sess(C.PROTO_IPCP, C.CPPacket(
                    C.CONF_ACK, cp.identifier, cp.data)))
                if not self.ipcp_acked:
                    self.ipcp_acked = True
                    out.append(self._sess(C.PROTO_IPCP, C.CPPacket(
                        C.CONF_REQ, 1, C.encode_opts(
                            [(C.IPCP_OPT_IP, b"\x00\x00\x00\x00"),
                             (C.IPCP_OPT_DNS1, b"\x00\x00\x00\x00")]))))
            elif cp.code == C.CONF_NAK:
                opts = C.decode_opts(cp.data)
                ip = C.get_opt(opts, C.IPCP_OPT_IP)
                if ip:
                    self.got_ip = struct.unpack(">I", ip)[0]
                d = C.get_opt(opts, C.IPCP_OPT_DNS1)
                if d:
                    self.got_dns = struct.unpack(">I", d)[0]
                newopts = [(C.IPCP_OPT_IP, ip or b"\x00\x00\x00\x00")]
                if self.got_dns:
                    newopts.append((C.IPCP_OPT_DNS1,
                                    struct.pack(">I", self.got_dns)))
                out.append(self._sess(C.PROTO_IPCP, C.CPPacket(
                    C.CONF_REQ, 2, C.encode_opts(newopts))))
        elif p.ppp_proto == C.PROTO_IPV6CP:
            if cp.code == C.CONF_REQ:
                out.append(self._sess(C.PROTO_IPV6CP, C.CPPacket(
                    C.CONF_ACK, cp.identifier, cp.data)))
                out.append(self._sess(C.PROTO_IPV6CP, C.CPPacket(
                    C.CONF_REQ, 1, C.encode_opts(
                        [(C.IPV6CP_OPT_IFID, b"\x11" * 8)]))))
        return out

    def _sess(self, proto, cp):
        return C.SessionPacket(self.session_id, proto, cp.encode(),
                               src_mac=self.mac,
                               dst_mac=SRV_MAC).encode()


def make_server(auth="chap", users=None):
    srv = PPPoEServer(SRV_MAC, auth=auth)
    srv.local_users = users if users is not None else {"alice": "pw1"}
    ips = {}
    def alloc(user):
        ips.setdefault(user, f"10.0.2.{len(ips) + 10}")
        return ips[user]
    srv.allocator = alloc
    from bng_amd.dataplane.packets import ip2u32
    srv.dns = (ip2u32("8.8.8.8"), ip2u32("1.1.1.1"))
    return srv


class TestDiscovery:
    def test_padi_pado_tags(self):
        srv = make_server()
        cli = SimClient(srv)
        # stop after PADO by sending only PADI and inspecting inbox
        pado_frames = srv.handle_frame(C.DiscoveryPacket(
            C.PADI, 0, [(C.TAG_SERVICE_NAME, b""),
                        (C.TAG_HOST_UNIQ, b"XYZ")],
            src_mac=CLI_MAC).encode())
        assert len(pado_frames) == 1
        pado = C.DiscoveryPacket.decode(pado_frames[0])
        assert pado.code == C.PADO
        assert C.get_tag(pado.tags, C.TAG_AC_NAME) == b"bng-amd"
        assert C.get_tag(pado.tags, C.TAG_HOST_UNIQ) == b"XYZ"
        assert C.get_tag(pado.tags, C.TAG_AC_COOKIE) is not None

    def test_padr_bad_cookie_rejected(self):
        srv = make_server()
        padr = C.DiscoveryPacket(C.PADR, 0, [
            (C.TAG_AC_COOKIE, b"x" * 16)], src_mac=CLI_MAC).encode()
        out = srv.handle_frame(padr)
        pads = C.DiscoveryPacket.decode(out[0])
        assert pads.session_id == 0
        assert C.get_tag(pads.tags, C.TAG_GENERIC_ERROR) is not None
        assert srv.session_count() == 0


class TestFullLifecycle:
    def test_chap_session_to_open(self):
        srv = make_server(auth="chap")
        opened = []
        srv.on_session_open = lambda s: opened.append(s)
        cli = SimClient(srv)
        cli.discover()
        assert cli.session_id != 0
        assert srv.stats["lcp_opened"] == 1
        assert srv.stats["auth_ok"] == 1
        assert srv.stats["ipcp_opened"] == 1
        assert len(opened) == 1
        s = opened[0]
        assert s.phase == PH_OPEN
        assert s.username == "alice"
        from bng_amd.dataplane.packets import u32_to_ip
        assert u32_to_ip(s.ip) == "10.0.2.10"
        assert cli.got_ip == s.ip
        assert cli.got_dns == srv.dns[0]

    def test_chap_wrong_password_fails(self):
        srv = make_server(auth="chap")
        cli = SimClient(srv, password="wrong")
        cli.discover()
        assert srv.stats["auth_fail"] == 1
        assert srv.stats["sessions_open"] == 0

    def test_pap_auth(self):
        srv = make_server(auth="pap")
        cli = SimClient(srv)
        cli.discover()
        # client must initiate PAP after LCP opens
        req = C.CPPacket(C.PAP_AUTH_REQ, 1,
                         bytes([5]) + b"alice" + bytes([3]) + b"pw1")
        cli.send(cli._sess(C.PROTO_PAP, req))
        assert srv.stats["auth_ok"] == 1
        assert srv.stats["sessions_open"] == 1

    def test_pap_rate_limit_teardown(self):
        srv = make_server(auth="pap")
        cli = SimClient(srv)
        cli.discover()
        for i in range(PPPoEServer.MAX_AUTH_ATTEMPTS):
            req = C.CPPacket(C.PAP_AUTH_REQ, i + 1,
                             bytes([5]) + b"alice" + bytes([2]) + b"xx")
            cli.send(cli._sess(C.PROTO_PAP, req))
        assert srv.stats["auth_fail"] == PPPoEServer.MAX_AUTH_ATTEMPTS
        assert srv.session_count() == 0        # torn down
        assert srv.stats["padt_tx"] == 1

    def test_radius_pap_auth(self):
        rsrv = RadiusServer(b"sec", users={
            "alice": {"password": "pw1", "policy": "gold"}}).start()
        try:
            srv = make_server(auth="pap")
            srv.local_users = {}
            srv.radius = RadiusClient([rsrv.addr], b"sec")
            cli = SimClient(srv)
            cli.discover()
            req = C.CPPacket(C.PAP_AUTH_REQ, 1,
                             bytes([5]) + b"alice" + bytes([3]) + b"pw1")
            cli.send(cli._sess(C.PROTO_PAP, req))
            assert srv.stats["auth_ok"] == 1
            s = list(srv.sessions.values())[0]
            assert s.policy_name == "gold"
        finally:
            rsrv.stop()

    def test_radius_chap_auth(self):
        """Full PPPoE CHAP handshake verified against RADIUS
        (ref auth.go CHAP vs RADIUS)."""
        rsrv = RadiusServer(b"sec", users={
            "alice": {"password": "pw1", "policy": "silver"}}).start()
        try:
            srv = make_server(auth="chap")
            srv.local_users = {}
            srv.radius = RadiusClient([rsrv.addr], b"sec")
            cli = SimClient(srv)          # responds to CHAP challenges
            cli.discover()
            assert srv.stats["auth_ok"] == 1
            s = list(srv.sessions.values())[0]
            assert s.username == "alice"
            assert s.policy_name == "silver"
            # wrong password rejected through RADIUS
            srv2 = make_server(auth="chap")
            srv2.local_users = {}
            srv2.radius = RadiusClient([rsrv.addr], b"sec")
            bad = SimClient(srv2, password="nope")
            bad.discover()
            assert srv2.stats["auth_fail"] == 1
        finally:
            rsrv.stop()


class TestKeepaliveTeardown:
    def test_echo_and_timeout(self):
        srv = make_server()
        srv.echo_fails = 2
        cli = SimClient(srv)
        cli.discover()
        s = list(srv.sessions.values())[0]
        # healthy: echo answered via client reactor
        frames = srv.tick()
        assert len(frames) == 1
        for f in frames:
            for counter in cli._react(f):
                srv.handle_frame(counter)
        assert s.echo_pending == 0
        # client goes silent: after echo_fails ticks, session torn down
        srv.tick()
        srv.tick()
        out = srv.tick()
        assert srv.stats["echo_timeout"] == 1
        assert srv.session_count() == 0
        assert any(struct.unpack_from(">H", f, 12)[0] == C.ETH_PPPOE_DISC
                   for f in out)    # PADT sent

    def test_client_terminate(self):
        srv = make_server()
        released = []
        srv.releaser = released.append
        cli = SimClient(srv)
        cli.discover()
        term = C.CPPacket(C.TERM_REQ, 9, b"bye")
        out = srv.handle_frame(cli._sess(C.PROTO_LCP, term))
        assert srv.stats["term_rx"] == 1
        assert srv.session_count() == 0
        assert released == ["alice"]
        ack = C.CPPacket.decode(C.SessionPacket.decode(out[0]).payload)
        assert ack.code == C.TERM_ACK

    def test_padt_from_client(self):
        srv = make_server()
        cli = SimClient(srv)
        cli.discover()
        closed = []
        srv.on_session_close = lambda s: closed.append(s.session_id)
        srv.handle_frame(C.DiscoveryPacket(
            C.PADT, cli.session_id, src_mac=CLI_MAC).encode())
        assert srv.session_count() == 0
        assert closed == [cli.session_id]

    def test_reconnect_replaces_session(self):
        srv = make_server()
        cli = SimClient(srv)
        cli.discover()
        sid1 = cli.session_id
        cli2 = SimClient(srv)
        cli2.discover()
        assert cli2.session_id != sid1
        assert srv.session_count() == 1


class TestNCPOptionHandling:
    """RFC 1332 / RFC 5072 option rules (ref ipcp_test.go 'reject IP
    compression option', ipv6cp 'NAK zero interface ID')."""

    def _open_session(self):
        srv = make_server(auth="chap")
        cli = SimClient(srv)
        cli.discover()
        assert srv.stats["ipcp_opened"] == 1
        return srv, cli

    def _send(self, srv, cli, proto, cp):
        frames = srv.handle_frame(C.SessionPacket(
            cli.session_id, proto, cp.encode(), src_mac=cli.mac,
            dst_mac=SRV_MAC).encode())
        out = []
        for f in frames:
            p = C.SessionPacket.decode(f)
            out.append((p.ppp_proto, C.CPPacket.decode(p.payload)))
        return out

    def test_ipcp_rejects_compression_option(self):
        srv, cli = self._open_session()
        from bng_amd.dataplane.packets import ip2u32
        req = C.CPPacket(C.CONF_REQ, 9, C.encode_opts(
            [(C.IPCP_OPT_IP, struct.pack(">I", cli.got_ip)),
             (2, b"\x00\x2d\x0f\x01")]))      # VJ compression opt
        replies = self._send(srv, cli, C.PROTO_IPCP, req)
        proto, cp = replies[0]
        assert proto == C.PROTO_IPCP and cp.code == C.CONF_REJ
        opts = C.decode_opts(cp.data)
        assert [t for t, _ in opts] == [2]    # only the bad option

    def test_ipv6cp_naks_zero_interface_id(self):
        srv, cli = self._open_session()
        req = C.CPPacket(C.CONF_REQ, 5, C.encode_opts(
            [(C.IPV6CP_OPT_IFID, b"\x00" * 8)]))
        replies = self._send(srv, cli, C.PROTO_IPV6CP, req)
        proto, cp = replies[0]
        assert proto == C.PROTO_IPV6CP and cp.code == C.CONF_NAK
        sug = C.get_opt(C.decode_opts(cp.data), C.IPV6CP_OPT_IFID)
        assert sug is not None and sug != b"\x00" * 8

    def test_ipcp_acks_correct_ip(self):
        srv, cli = self._open_session()
        req = C.CPPacket(C.CONF_REQ, 7, C.encode_opts(
            [(C.IPCP_OPT_IP, struct.pack(">I", cli.got_ip))]))
        replies = self._send(srv, cli, C.PROTO_IPCP, req)
        assert replies[0][1].code == C.CONF_ACK


class TestCodecFuzz:
    """Discovery/session decoder robustness on adversarial frames."""

    def test_random_frames_never_crash(self):
        import random
        rng = random.Random(77)
        srv = make_server()
        for _ in range(2000):
            n = rng.randrange(14, 128)
            frame = bytearray(rng.randrange(256) for _ in range(n))
            # half the time, make it look like PPPoE to go deeper
            if rng.random() < 0.5:
                struct.pack_into(">H", frame, 12,
                                 C.ETH_PPPOE_DISC if rng.random() < 0.5
                                 else C.ETH_PPPOE_SESS)
            try:
                srv.handle_frame(bytes(frame))
            except Exception as e:      # decoder must contain errors
                raise AssertionError(
                    f"server crashed on fuzz frame: {e!r}") from e

    def test_truncated_real_frames(self):
        srv = make_server()
        cli = SimClient(srv)
        padi = C.DiscoveryPacket(C.PADI, 0, [
            (C.TAG_SERVICE_NAME, b""), (C.TAG_HOST_UNIQ, b"H")],
            src_mac=cli.mac).encode()
        for cut in range(len(padi)):
            srv.handle_frame(padi[:cut])    # must not raise


def _discover_session(srv, mac=b"\xaa\xbb\xcc\x00\x00\x77"):
    """PADI/PADR by hand; returns (session, lcp frames from PADS)."""
    padi = C.DiscoveryPacket(C.PADI, 0, [(C.TAG_SERVICE_NAME, b"")],
                             src_mac=mac).encode()
    pado = C.DiscoveryPacket.decode(srv.handle_frame(padi)[0])
    cookie = C.get_tag(pado.tags, C.TAG_AC_COOKIE)
    padr = C.DiscoveryPacket(C.PADR, 0, [
        (C.TAG_SERVICE_NAME, b""), (C.TAG_AC_COOKIE, cookie)],
        src_mac=mac).encode()
    frames = srv.handle_frame(padr)
    pads = C.DiscoveryPacket.decode(frames[0])
    sid = pads.session_id
    return srv.sessions[sid], frames[1:]


def _lcp_from(frames):
    """Extract decoded LCP CPPackets from session frames."""
    out = []
    for f in frames:
        if struct.unpack_from(">H", f, 12)[0] == C.ETH_PPPOE_SESS:
            p = C.SessionPacket.decode(f)
            if p.ppp_proto == C.PROTO_LCP:
                out.append(C.CPPacket.decode(p.payload))
    return out


def _send_lcp(srv, s, code, ident, data=b""):
    return srv.handle_frame(C.SessionPacket(
        s.session_id, C.PROTO_LCP,
        C.CPPacket(code, ident, data).encode(),
        src_mac=s.client_mac, dst_mac=srv.server_mac).encode())


class TestLcpBargaining:
    """Round-1 VERDICT task 6: Configure-Nak/Reject option bargaining,
    magic-number loop detection, restart counters, auth throttling
    (ref lcp.go / auth.go:202-580)."""

    def make(self):
        srv = PPPoEServer(SRV_MAC, auth="chap")
        srv.local_users["alice"] = "pw1"
        return srv

    def test_nak_adopts_suggested_mru(self):
        srv = self.make()
        s, frames = _discover_session(srv)
        req = _lcp_from(frames)[0]
        assert req.code == C.CONF_REQ
        # client Naks our MRU, suggesting 1400
        out = _send_lcp(srv, s, C.CONF_NAK, req.identifier,
                        C.encode_opts([(C.LCP_OPT_MRU,
                                        struct.pack(">H", 1400))]))
        req2 = _lcp_from(out)[0]
        opts = dict(C.decode_opts(req2.data))
        assert struct.unpack(">H", opts[C.LCP_OPT_MRU])[0] == 1400
        assert s.our_mru == 1400

    def test_reject_drops_option_but_not_auth(self):
        srv = self.make()
        s, frames = _discover_session(srv)
        req = _lcp_from(frames)[0]
        # client Rejects our MAGIC option -> resent request omits it
        out = _send_lcp(srv, s, C.CONF_REJ, req.identifier,
                        C.encode_opts([(C.LCP_OPT_MAGIC, b"\0\0\0\0")]))
        req2 = _lcp_from(out)[0]
        assert req2.code == C.CONF_REQ
        types = [t for t, _ in C.decode_opts(req2.data)]
        assert C.LCP_OPT_MAGIC not in types
        assert C.LCP_OPT_AUTH in types
        # rejecting AUTH ends the session (no unauthenticated service)
        out = _send_lcp(srv, s, C.CONF_REJ, req2.identifier,
                        C.encode_opts([(C.LCP_OPT_AUTH,
                                        struct.pack(">HB", C.PROTO_CHAP,
                                                    5))]))
        assert s.session_id not in srv.sessions
        assert any(struct.unpack_from(">H", f, 12)[0] == C.ETH_PPPOE_DISC
                   for f in out)   # PADT sent

    def test_magic_loop_detection_naks_then_terminates(self):
        srv = self.make()
        s, _ = _discover_session(srv)
        # peer's CONF_REQ carries OUR magic: looped link
        for i in range(srv.MAX_MAGIC_LOOPS):
            alive = s.session_id in srv.sessions
            out = _send_lcp(srv, s, C.CONF_REQ, 10 + i,
                            C.encode_opts([(C.LCP_OPT_MAGIC,
                                            struct.pack(">I",
                                                        s.our_magic))]))
            if i < srv.MAX_MAGIC_LOOPS - 1:
                naks = [p for p in _lcp_from(out) if p.code == C.CONF_NAK]
                assert naks, "expected Configure-Nak with a fresh magic"
        assert s.session_id not in srv.sessions
        assert srv.stats["loopback_detected"] == srv.MAX_MAGIC_LOOPS

    def test_nak_storm_hits_max_failure(self):
        srv = self.make()
        s, frames = _discover_session(srv)
        req = _lcp_from(frames)[0]
        ident = req.identifier
        for _ in range(srv.MAX_FAILURE + 1):
            if s.session_id not in srv.sessions:
                break
            out = _send_lcp(srv, s, C.CONF_NAK, ident,
                            C.encode_opts([(C.LCP_OPT_MRU,
                                            struct.pack(">H", 1400))]))
            got = _lcp_from(out)
            if got:
                ident = got[0].identifier
        assert s.session_id not in srv.sessions
        assert srv.stats["restart_exhausted"] == 1

    def test_tick_retransmits_until_max_configure(self):
        srv = self.make()
        s, _ = _discover_session(srv)
        sends = 1                       # initial request at PADS
        while s.session_id in srv.sessions and sends < 50:
            srv.tick()
            sends += 1
        assert s.session_id not in srv.sessions
        assert sends == srv.MAX_CONFIGURE + 1
        assert srv.stats["restart_exhausted"] == 1

    def test_per_mac_auth_throttle_survives_rediscovery(self):
        srv = self.make()
        srv.AUTH_FAIL_LIMIT = 3
        mac = b"\xaa\xbb\xcc\x00\x00\x99"
        # record 3 failures (as _auth_fail would)
        import time as _t
        srv._auth_fails[mac] = [_t.time()] * 3
        # a fresh PADR from the same MAC is refused while locked out
        padi = C.DiscoveryPacket(C.PADI, 0, [(C.TAG_SERVICE_NAME, b"")],
                                 src_mac=mac).encode()
        pado = C.DiscoveryPacket.decode(srv.handle_frame(padi)[0])
        cookie = C.get_tag(pado.tags, C.TAG_AC_COOKIE)
        padr = C.DiscoveryPacket(C.PADR, 0, [
            (C.TAG_SERVICE_NAME, b""), (C.TAG_AC_COOKIE, cookie)],
            src_mac=mac).encode()
        out = srv.handle_frame(padr)
        pads = C.DiscoveryPacket.decode(out[0])
        assert pads.session_id == 0
        assert C.get_tag(pads.tags, C.TAG_GENERIC_ERROR) is not None
        assert srv.stats["auth_throttled"] == 1
        # outside the window the client may try again
        srv._auth_fails[mac] = [_t.time() - srv.AUTH_FAIL_WINDOW - 1] * 3
        out = srv.handle_frame(padr)
        assert C.DiscoveryPacket.decode(out[0]).session_id != 0


def test_session_timeout_teardown():
    import time
    """--pppoe-session-timeout: open sessions past the absolute
    lifetime are torn down on tick."""
    srv = PPPoEServer(SRV_MAC, auth="none", session_timeout=600)
    c = SimClient(srv)
    c.discover()
    sid = c.session_id
    assert sid in srv.sessions and srv.sessions[sid].phase == "open"
    srv.tick(now=time.time() + 300)      # inside lifetime: stays
    assert sid in srv.sessions
    out = srv.tick(now=time.time() + 601)
    assert sid not in srv.sessions
    assert srv.stats.get("session_timeout") == 1


class TestLcpOptionBounds:
    """MRU upper bound (RFC 2516) and zero-magic NAK (ref lcp_test
    scenarios 'NAK MRU greater than 1492' / 'NAK zero magic')."""

    def _req(self, srv, s, opts):
        return _send_lcp(srv, s, C.CONF_REQ, 9, C.encode_opts(opts))

    def test_mru_over_1492_naked(self):
        srv = PPPoEServer(SRV_MAC, auth="none")
        s, _ = _discover_session(srv)
        out = self._req(srv, s, [(C.LCP_OPT_MRU,
                                  struct.pack(">H", 1500))])
        naks = [c for c in _lcp_from(out) if c.code == C.CONF_NAK]
        assert naks, "oversized MRU must be Naked"
        opts = dict(C.decode_opts(naks[0].data))
        assert struct.unpack(">H", opts[C.LCP_OPT_MRU])[0] <= 1492
        # valid MRU is accepted
        out = self._req(srv, s, [(C.LCP_OPT_MRU,
                                  struct.pack(">H", 1492))])
        assert any(c.code == C.CONF_ACK for c in _lcp_from(out))
        assert s.peer_mru == 1492

    def test_zero_magic_naked_with_real_number(self):
        srv = PPPoEServer(SRV_MAC, auth="none")
        s, _ = _discover_session(srv)
        out = self._req(srv, s, [(C.LCP_OPT_MAGIC, b"\x00" * 4)])
        naks = [c for c in _lcp_from(out) if c.code == C.CONF_NAK]
        assert naks, "zero magic must be Naked"
        opts = dict(C.decode_opts(naks[0].data))
        assert opts[C.LCP_OPT_MAGIC] != b"\x00" * 4
        # and it is not mistaken for a loop
        assert s.magic_loops == 0
