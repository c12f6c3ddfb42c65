"""DHCPv6 + SLAAC tests (ref pkg/dhcpv6, pkg/slaac test strategies)."""
import ipaddress
import struct
import time

import pytest

from bng_amd.dhcpv6.server import (ADVERTISE, DHCPv6Message, DHCPv6Server,
                                   OPT_CLIENTID, OPT_IA_NA, OPT_IA_PD,
                                   OPT_IAADDR, OPT_IAPREFIX,
                                   OPT_RAPID_COMMIT, OPT_SERVERID,
                                   OPT_STATUS_CODE, OPT_DNS_SERVERS,
                                   RELEASE, RENEW, REPLY, REQUEST, SOLICIT,
                                   STATUS_NOBINDING, parse_ia)
from bng_amd.slaac.radvd import (PrefixConfig, RAConfig, Server as RadvServer,
                                 build_ra, parse_ra)

DUID = b"\x00\x03\x00\x01\xaa\xbb\xcc\x00\x00\x01"


def solicit(duid=DUID, iaid=1, pd=False, rapid=False, txn=0x1234):
    m = DHCPv6Message(SOLICIT, txn)
    m.add(OPT_CLIENTID, duid)
    m.add(OPT_IA_NA, struct.pack(">III", iaid, 0, 0))
    if pd:
        m.add(OPT_IA_PD, struct.pack(">III", iaid, 0, 0))
    if rapid:
        m.add(OPT_RAPID_COMMIT, b"")
    return m


def extract_addr(resp, opt=OPT_IA_NA, sub=OPT_IAADDR):
    body = resp.get(opt)
    _iaid, _t1, _t2, subs = parse_ia(body)
    for t, v in subs:
        if t == sub:
            if sub == OPT_IAADDR:
                return str(ipaddress.IPv6Address(v[:16]))
            plen = v[8]
            return f"{ipaddress.IPv6Address(v[9:25])}/{plen}"
    return None


class TestDHCPv6:
    def test_four_message_exchange(self):
        srv = DHCPv6Server(rapid_commit=False, dns=["2001:4860:4860::8888"])
        adv_raw = srv.handle(solicit().encode())
        adv = DHCPv6Message.decode(adv_raw)
        assert adv.msg_type == ADVERTISE
        addr = extract_addr(adv)
        assert addr.startswith("2001:db8:1:")
        # REQUEST echoes server id
        req = solicit(txn=0x1235)
        req.msg_type = REQUEST
        req.add(OPT_SERVERID, srv.server_duid)
        rep = DHCPv6Message.decode(srv.handle(req.encode()))
        assert rep.msg_type == REPLY
        assert extract_addr(rep) == addr       # sticky per DUID/IAID
        assert (DUID, 1, False) in srv.bindings
        assert rep.get(OPT_DNS_SERVERS) is not None

    def test_rapid_commit(self):
        srv = DHCPv6Server(rapid_commit=True)
        rep = DHCPv6Message.decode(srv.handle(
            solicit(rapid=True).encode()))
        assert rep.msg_type == REPLY
        assert rep.get(OPT_RAPID_COMMIT) is not None
        assert srv.stats["rapid_commits"] == 1
        assert (DUID, 1, False) in srv.bindings

    def test_prefix_delegation(self):
        srv = DHCPv6Server(rapid_commit=True, pd_prefix_len=56)
        rep = DHCPv6Message.decode(srv.handle(
            solicit(pd=True, rapid=True).encode()))
        pd = extract_addr(rep, OPT_IA_PD, OPT_IAPREFIX)
        assert pd.endswith("/56")
        # a second client gets a different prefix
        rep2 = DHCPv6Message.decode(srv.handle(
            solicit(duid=DUID[:-1] + b"\x02", pd=True,
                    rapid=True).encode()))
        assert extract_addr(rep2, OPT_IA_PD, OPT_IAPREFIX) != pd

    def test_renew_extends_and_unknown_nobinding(self):
        srv = DHCPv6Server(rapid_commit=True)
        srv.handle(solicit(rapid=True).encode())
        b = srv.bindings[(DUID, 1, False)]
        old_exp = b.expiry
        time.sleep(0.01)
        ren = solicit(txn=0x2222)
        ren.msg_type = RENEW
        ren.add(OPT_SERVERID, srv.server_duid)
        rep = DHCPv6Message.decode(srv.handle(ren.encode()))
        assert rep.msg_type == REPLY
        assert b.expiry > old_exp
        # unknown binding -> NoBinding status
        other = solicit(duid=b"\x00\x03\x00\x01xxxxxx", txn=1)
        other.msg_type = RENEW
        rep2 = DHCPv6Message.decode(srv.handle(other.encode()))
        _, _, _, subs = parse_ia(rep2.get(OPT_IA_NA))
        st = [v for t, v in subs if t == OPT_STATUS_CODE][0]
        assert struct.unpack(">H", st[:2])[0] == STATUS_NOBINDING

    def test_release_frees_address(self):
        srv = DHCPv6Server(rapid_commit=True)
        rep = DHCPv6Message.decode(srv.handle(
            solicit(rapid=True).encode()))
        addr = extract_addr(rep)
        rel = solicit(txn=3)
        rel.msg_type = RELEASE
        srv.handle(rel.encode())
        assert (DUID, 1, False) not in srv.bindings
        # address is reusable by another DUID
        rep2 = DHCPv6Message.decode(srv.handle(
            solicit(duid=DUID[:-1] + b"\x09", rapid=True).encode()))
        assert extract_addr(rep2) == addr

    def test_sweep_expired(self):
        srv = DHCPv6Server(rapid_commit=True, valid_lifetime=1)
        srv.handle(solicit(rapid=True).encode())
        assert srv.sweep_expired(now=time.time() + 10) == 1
        assert not srv.bindings


class TestSLAAC:
    def cfg(self):
        return RAConfig(
            prefixes=[PrefixConfig("2001:db8:2::/64")],
            managed=False, other_config=True, mtu=1492,
            rdnss=["2001:4860:4860::8888", "2001:4860:4860::8844"],
            dnssl=["isp.example.com"], source_lladdr=b"\x02\x00\x00\x00\x00\x01")

    def test_ra_roundtrip(self):
        ra = parse_ra(build_ra(self.cfg()))
        assert ra["other"] and not ra["managed"]
        assert ra["mtu"] == 1492
        assert ra["prefixes"][0]["prefix"] == "2001:db8:2::/64"
        assert ra["prefixes"][0]["autonomous"]
        assert ra["rdnss"] == ["2001:4860:4860::8888",
                               "2001:4860:4860::8844"]
        assert ra["dnssl"] == ["isp.example.com"]

    def test_managed_flag_for_dhcpv6_deployments(self):
        c = self.cfg()
        c.managed = True
        c.prefixes[0].autonomous = False
        ra = parse_ra(build_ra(c))
        assert ra["managed"] and not ra["prefixes"][0]["autonomous"]

    def test_rs_triggers_solicited_ra(self):
        sent = []
        srv = RadvServer(self.cfg(), send_fn=lambda p, d: sent.append((p, d)))
        rs = bytes([133, 0, 0, 0, 0, 0, 0, 0])
        payload = srv.handle_rs(rs, src="fe80::1")
        assert payload is not None
        assert srv.stats["rs_received"] == 1
        assert sent[0][1] == "fe80::1"
        assert parse_ra(sent[0][0])["prefixes"]
        assert srv.handle_rs(b"\x00" * 8) is None


class TestDHCPv6Relay:
    def test_relay_forw_repl_roundtrip(self):
        import struct as st
        from bng_amd.dhcpv6.server import RELAY_FORW, RELAY_REPL
        srv = DHCPv6Server(rapid_commit=True)
        inner = solicit(rapid=True).encode()
        link = b"\x20\x01" + b"\x00" * 14
        peer = b"\xfe\x80" + b"\x00" * 14
        relay = bytes([RELAY_FORW, 0]) + link + peer + \
            st.pack(">HH", 18, 5) + b"pon01" + \
            st.pack(">HH", 9, len(inner)) + inner
        out = srv.handle(relay)
        assert out is not None and out[0] == RELAY_REPL
        assert out[2:18] == link and out[18:34] == peer
        # interface-id echoed, inner REPLY carried
        i, inner_reply, iface = 34, None, None
        while i + 4 <= len(out):
            t, ln = st.unpack_from(">HH", out, i)
            if t == 18:
                iface = out[i + 4:i + 4 + ln]
            if t == 9:
                inner_reply = out[i + 4:i + 4 + ln]
            i += 4 + ln
        assert iface == b"pon01"
        rep = DHCPv6Message.decode(inner_reply)
        assert rep.msg_type == REPLY
        assert extract_addr(rep) is not None
        assert (DUID, 1, False) in srv.bindings


class TestDecline:
    """RFC 8415 §18.3.8 Decline: conflicted addresses are quarantined
    and the next allocation avoids them (the reference only releases,
    dhcpv6/server.go:684-693; the v4 decline blacklist is the model)."""

    def _solicit_request(self, srv, duid, iaid=1):
        from bng_amd.dhcpv6 import server as srv_mod
        from bng_amd.dhcpv6.server import (DHCPv6Message, OPT_CLIENTID,
                                           OPT_IA_NA, REQUEST, SOLICIT,
                                           parse_ia, OPT_IAADDR)
        import struct as st
        m = DHCPv6Message(REQUEST, 0x111)
        m.add(OPT_CLIENTID, duid)
        m.add(srv_mod.OPT_SERVERID, srv.server_duid)
        m.add(OPT_IA_NA, st.pack(">III", iaid, 0, 0))
        resp = DHCPv6Message.decode(srv.handle(m.encode()))
        body = resp.get(OPT_IA_NA)
        _, _, _, subs = parse_ia(body)
        for t, sub in subs:
            if t == OPT_IAADDR:
                import ipaddress
                return str(ipaddress.IPv6Address(sub[:16]))
        return None

    def test_declined_address_not_reoffered(self):
        from bng_amd.dhcpv6.server import (DECLINE, DHCPv6Message,
                                           DHCPv6Server, OPT_CLIENTID,
                                           OPT_IA_NA, encode_ia_na)
        import struct as st
        srv = DHCPv6Server(rapid_commit=False)
        duid = b"\x00\x01duid-x"
        addr1 = self._solicit_request(srv, duid)
        assert addr1 is not None
        # client detects a conflict and declines
        d = DHCPv6Message(DECLINE, 0x222)
        d.add(OPT_CLIENTID, duid)
        d.add(OPT_IA_NA, encode_ia_na(1, 0, 0, [(addr1, 0, 0)]))
        resp = srv.handle(d.encode())
        assert resp is not None
        assert srv.stats["decline"] == 1
        assert srv._is_declined(addr1)
        # a new request must get a DIFFERENT address
        addr2 = self._solicit_request(srv, duid)
        assert addr2 is not None and addr2 != addr1
        # quarantine expires
        srv._declined[addr1] = 0.0
        assert not srv._is_declined(addr1)


def test_nested_relay_chain():
    """Two-level RELAY-FORW chain (RFC 8415 §19 multi-hop): the reply
    is a matching nested RELAY-REPL with Interface-Id echoed at each
    level and the inner REPLY intact."""
    import struct as st
    from bng_amd.dhcpv6.server import (DHCPv6Message, DHCPv6Server,
                                       OPT_CLIENTID, OPT_IA_NA, RELAY_FORW,
                                       RELAY_REPL, REPLY, SOLICIT)
    srv = DHCPv6Server(rapid_commit=True)
    duid = b"\x00\x01duid-r"
    m = DHCPv6Message(SOLICIT, 0x333)
    m.add(OPT_CLIENTID, duid)
    m.add(14, b"")            # rapid commit
    m.add(OPT_IA_NA, st.pack(">III", 9, 0, 0))
    inner = m.encode()

    def wrap(payload, hop, ifid):
        out = bytes([RELAY_FORW, hop]) + b"\x00" * 32
        out += st.pack(">HH", 18, len(ifid)) + ifid          # iface-id
        out += st.pack(">HH", 9, len(payload)) + payload     # relay-msg
        return out

    lvl1 = wrap(inner, 0, b"eth-cust")
    lvl2 = wrap(lvl1, 1, b"eth-aggr")
    resp = srv.handle(lvl2)
    assert resp is not None and resp[0] == RELAY_REPL and resp[1] == 1

    def unwrap(data, want_ifid):
        i = 34
        msg = None
        ifid = None
        while i + 4 <= len(data):
            t, ln = st.unpack_from(">HH", data, i)
            if t == 9:
                msg = data[i + 4:i + 4 + ln]
            elif t == 18:
                ifid = data[i + 4:i + 4 + ln]
            i += 4 + ln
        assert ifid == want_ifid
        return msg

    lvl1_rep = unwrap(resp, b"eth-aggr")
    assert lvl1_rep[0] == RELAY_REPL and lvl1_rep[1] == 0
    final = unwrap(lvl1_rep, b"eth-cust")
    dec = DHCPv6Message.decode(final)
    assert dec.msg_type == REPLY and dec.txn_id == 0x333
    assert dec.get(OPT_IA_NA) is not None


class TestSLAACAddressing:
    """EUI-64 / RFC 7217 address generation, classification, neighbor
    cache (ref pkg/slaac/types.go:102-185)."""

    def test_eui64_address(self):
        from bng_amd.slaac.radvd import generate_slaac_address
        # known vector: 00:1b:44:11:3a:b7 -> ::21b:44ff:fe11:3ab7
        got = generate_slaac_address("2001:db8:1::/64",
                                     "00:1b:44:11:3a:b7")
        assert got == "2001:db8:1:0:21b:44ff:fe11:3ab7"
        with pytest.raises(ValueError):
            generate_slaac_address("2001:db8::/64", "00:1b:44")

    def test_stable_privacy_address(self):
        from bng_amd.slaac.radvd import generate_stable_privacy_address
        a1 = generate_stable_privacy_address("2001:db8:1::/64", b"eth0",
                                             b"secret")
        a2 = generate_stable_privacy_address("2001:db8:1::/64", b"eth0",
                                             b"secret")
        assert a1 == a2                              # stable
        a3 = generate_stable_privacy_address("2001:db8:2::/64", b"eth0",
                                             b"secret")
        assert a1 != a3                              # per-prefix
        a4 = generate_stable_privacy_address("2001:db8:1::/64", b"eth0",
                                             b"secret", dad_counter=1)
        assert a1 != a4                              # DAD retry moves
        assert a1.startswith("2001:db8:1:")
        # universal/local bit cleared in the IID
        import ipaddress
        assert not (ipaddress.IPv6Address(a1).packed[8] & 0x02)

    def test_classification(self):
        from bng_amd.slaac.radvd import is_global_unicast, is_link_local
        assert is_link_local("fe80::1")
        assert is_link_local("169.254.1.1")
        assert not is_link_local("2001:db8::1")
        assert is_global_unicast("2600::1")
        assert not is_global_unicast("fe80::1")
        assert not is_global_unicast("10.0.0.1")

    def test_neighbor_state_machine(self):
        from bng_amd.slaac.radvd import (N_DELAY, N_PROBE, N_REACHABLE,
                                         N_STALE, NeighborCache)
        nc = NeighborCache(reachable_time=30.0, delay_time=5.0)
        nc.confirm("fe80::1", "aa:bb:cc:00:00:01", is_router=True,
                   now=1000.0)
        assert nc.state("fe80::1", 1010.0) == N_REACHABLE
        assert nc.state("fe80::1", 1031.0) == N_STALE    # decayed
        nc.used("fe80::1", 1032.0)                       # tx to STALE
        assert nc.state("fe80::1", 1033.0) == N_DELAY
        assert nc.state("fe80::1", 1038.0) == N_PROBE    # unanswered
        nc.confirm("fe80::1", now=1039.0)                # NA arrives
        assert nc.state("fe80::1", 1040.0) == N_REACHABLE
        assert nc.routers() == ["fe80::1"]
        assert nc.state("fe80::9") == ""
        nc.incomplete("fe80::9", now=1000.0)
        assert nc.state("fe80::9") == "INCOMPLETE"
        assert nc.purge(max_age=60.0, now=1090.0) == 1   # fe80::9 gone
        assert nc.lookup("fe80::1") is not None

    def test_server_prefix_management(self):
        from bng_amd.slaac.radvd import parse_ra
        sent = []
        cfg = RAConfig(prefixes=[PrefixConfig("2001:db8:1::/64")])
        srv = RadvServer(cfg, send_fn=lambda p, d: sent.append(p))
        srv.add_prefix(PrefixConfig("2001:db8:2::/64"))
        srv.send_immediate_ra()
        got = parse_ra(sent[-1])
        assert [p["prefix"] for p in got["prefixes"]] == \
            ["2001:db8:1::/64", "2001:db8:2::/64"]
        srv.remove_prefix("2001:db8:1::/64")
        srv.send_immediate_ra()
        got = parse_ra(sent[-1])
        assert [p["prefix"] for p in got["prefixes"]] == \
            ["2001:db8:2::/64"]
