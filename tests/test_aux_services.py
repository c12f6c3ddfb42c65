"""Tests for dns / deviceauth / ztp / agent / pon / direct / wifi / qinq."""
import time

import pytest

from bng_amd.agent.agent import (S_CONNECTED, S_PARTITIONED, Agent)
from bng_amd.deviceauth.authenticator import (AuthError, MTLSAuthenticator,
                                              PSKAuthenticator,
                                              new_authenticator)
from bng_amd.direct.authenticator import (Authenticator as DirectAuth,
                                          StubBSS)
from bng_amd.dns.resolver import (Resolver, build_query, build_response,
                                  parse_response)
from bng_amd.nexus.client import Client as NexusClient
from bng_amd.nexus.model import Subscriber
from bng_amd.nexus.store import MemoryStore
from bng_amd.pon.manager import Manager as PONManager, QoSProfile
from bng_amd.qinq.mapper import Mapper as QinQMapper, QinQError
from bng_amd.wifi.gateway import Gateway
from bng_amd.ztp.bootstrap import (BootstrapClient, ZTPServer,
                                   discover_nexus_from_dhcp_options)


class TestDNS:
    def upstream(self, answers):
        def fn(q):
            return build_response(q, answers, ttl=500)
        return fn

    def test_resolve_and_cache(self):
        calls = []
        up = self.upstream(["93.184.216.34"])
        def counting(q):
            calls.append(1)
            return up(q)
        r = Resolver(counting)
        resp = r.handle_query(build_query("example.com"))
        _, addrs, _ = parse_response(resp)
        assert addrs == ["93.184.216.34"]
        r.handle_query(build_query("example.com"))
        assert len(calls) == 1            # second from cache
        assert r.stats["cache_hits"] == 1

    def test_ttl_clamp(self):
        r = Resolver(self.upstream(["1.2.3.4"]), min_ttl=60, max_ttl=120)
        r.handle_query(build_query("x.com"))
        ce = r.cache[("x.com", 1)]
        assert 110 <= ce.expires - time.time() <= 121

    def test_intercept_walled_garden(self):
        r = Resolver(self.upstream(["1.2.3.4"]))
        r.add_intercept("portal.isp.com", ["10.0.0.10"])
        resp = r.handle_query(build_query("portal.isp.com"))
        _, addrs, _ = parse_response(resp)
        assert addrs == ["10.0.0.10"]
        # quarantined client: EVERYTHING goes to the portal
        r.set_intercept_all(["10.0.0.10"])
        resp = r.handle_query(build_query("facebook.com"),
                              quarantined=True)
        _, addrs, _ = parse_response(resp)
        assert addrs == ["10.0.0.10"]
        assert r.stats["intercepted"] == 2

    def test_rate_limit(self):
        r = Resolver(self.upstream(["1.2.3.4"]), rate_limit=1,
                     rate_burst=2)
        q = build_query("a.com")
        assert r.handle_query(q, client="c1") is not None
        assert r.handle_query(build_query("b.com"), client="c1") is not None
        assert r.handle_query(build_query("c.com"), client="c1") is None
        assert r.stats["rate_limited"] == 1


class TestDeviceAuth:
    def test_psk_roundtrip_and_replay_window(self):
        a = PSKAuthenticator(b"secret", window=300)
        h = a.headers("olt-1")
        assert a.verify(h) == "olt-1"
        h2 = dict(h)
        h2["X-Auth-Timestamp"] = str(int(time.time()) - 10_000)
        with pytest.raises(AuthError):
            a.verify(h2)
        h3 = dict(h)
        h3["X-Auth-Signature"] = "00" * 32
        with pytest.raises(AuthError):
            a.verify(h3)

    def test_mtls_fingerprints(self):
        a = MTLSAuthenticator()
        a.register("olt-1", b"CERT-PEM-BYTES")
        assert a.verify_cert("olt-1", b"CERT-PEM-BYTES") == "olt-1"
        with pytest.raises(AuthError):
            a.verify_cert("olt-1", b"EVIL")
        with pytest.raises(AuthError):
            a.verify_cert("olt-2", b"CERT-PEM-BYTES")

    def test_factory(self):
        assert new_authenticator("none").mode == "none"
        assert new_authenticator("psk", psk=b"x").mode == "psk"
        with pytest.raises(ValueError):
            new_authenticator("wat")


class TestZTP:
    def test_bootstrap_flow(self):
        srv = ZTPServer()
        try:
            c = BootstrapClient(srv.url, serial="SN123",
                                poll_interval=0.05)
            c.register()
            assert c.state == "registered"
            assert "SN123" in srv.devices
            # operator approves with a config incl. HA partner + pool
            srv.approve("SN123", {
                "device_id": "bng-7", "role": "standby",
                "ha_partner": "http://10.0.0.8:8443",
                "pool_network": "10.7.0.0/16"})
            cfg = c.poll_until_approved(timeout=5)
            assert c.state == "approved"
            assert cfg.device_id == "bng-7"
            assert cfg.role == "standby"
            assert cfg.ha_partner.endswith(":8443")
            assert cfg.pool_network == "10.7.0.0/16"
        finally:
            srv.stop()

    def test_dhcp_option_discovery(self):
        assert discover_nexus_from_dhcp_options(
            {224: b"https://nexus.isp:8443"}) == "https://nexus.isp:8443"
        assert discover_nexus_from_dhcp_options(
            {43: b"http://n"}) == "http://n"
        assert discover_nexus_from_dhcp_options({43: b"\xff\x01x"}) is None
        assert discover_nexus_from_dhcp_options({}) is None


class TestAgent:
    def test_config_watch_and_states(self):
        store = MemoryStore()
        a = Agent(store, "node-1", partition_after=0.0)
        states, cfgs = [], []
        a.on_state_change(lambda o, n: states.append(n))
        a.on_config_change(cfgs.append)
        a.start()
        try:
            assert a.state == S_CONNECTED
            store.put("nexus/device_configs/node-1",
                      b'{"pool": "10.9.0.0/24"}')
            assert cfgs and cfgs[-1]["pool"] == "10.9.0.0/24"
            # partition: store writes fail
            orig = store.put
            store.put = lambda *a_, **k: (_ for _ in ()).throw(OSError())
            assert not a.heartbeat_once()
            assert a.state == S_PARTITIONED
            store.put = orig
            a.heartbeat_once()
            a.heartbeat_once()
            assert a.state == S_CONNECTED
            assert S_PARTITIONED in states
        finally:
            a.stop()


class TestPON:
    def test_discovery_and_provisioning(self):
        store = MemoryStore()
        qm = QinQMapper()
        qm.add_range(100, 2, 100)
        pon = PONManager(store, vlan_mapper=qm)
        pon.add_profile(QoSProfile("residential", 1000, 200))
        events = []
        pon.on_event(lambda ev, n: events.append((ev, n.id)))
        nte = pon.ont_discovered("ONT123", "pon0/1")
        assert nte.state == "discovered"
        p = pon.provision(nte.id, profile="residential")
        assert p.provisioned and p.state == "active"
        assert (p.s_tag, p.c_tag) == (100, 2)
        pon.ont_offline(nte.id)
        assert [e[0] for e in events] == ["discovered", "provisioned",
                                         "offline"]
        with pytest.raises(KeyError):
            pon.provision(nte.id, profile="nope")


class TestDirect:
    def make(self):
        store = MemoryStore()
        c = NexusClient(store)
        c.save_subscriber(Subscriber("sub-1", s_tag=100, c_tag=5,
                                     mac="aa:bb:cc:00:00:01",
                                     isp_id="isp-a"))
        return c

    def test_vlan_and_mac_identity(self):
        c = self.make()
        d = DirectAuth(c)
        r = d.authenticate_by_vlan(100, 5)
        assert r.success and r.subscriber_id == "sub-1"
        r2 = d.authenticate_by_mac("aa:bb:cc:00:00:01")
        assert r2.success and r2.isp_id == "isp-a"
        assert not d.authenticate_by_vlan(1, 1).success

    def test_bss_suspension(self):
        c = self.make()
        d = DirectAuth(c, bss=StubBSS({"sub-1": "suspended"}))
        r = d.authenticate_by_mac("aa:bb:cc:00:00:01")
        assert not r.success and "suspended" in r.reason


class TestWiFi:
    def test_guest_lifecycle(self):
        gw = Gateway(network="192.168.100.0/28")
        s = gw.join("AA:BB:CC:00:00:01")
        assert gw.is_quarantined("aa:bb:cc:00:00:01")
        assert gw.accept_terms("aa:bb:cc:00:00:01")
        assert not gw.is_quarantined("aa:bb:cc:00:00:01")
        # idle guest expires after grace epochs; active one renewed
        gw.join("aa:bb:cc:00:00:02")
        gw.advance_epoch()
        gw.touch("aa:bb:cc:00:00:01")     # renews s1 only
        gw.advance_epoch()
        assert gw.session_count() == 1
        assert gw.stats["expired"] == 1

    def test_exhaustion(self):
        gw = Gateway(network="192.168.100.0/30")   # 2 usable (net,bcast out)
        gw.join("aa:bb:cc:00:00:01")
        gw.join("aa:bb:cc:00:00:02")
        from bng_amd.allocator.epoch_bitmap import PoolExhaustedError
        with pytest.raises(PoolExhaustedError):
            gw.join("aa:bb:cc:00:00:09")
        assert gw.stats["exhausted"] == 1


class TestQinQ:
    def test_register_conflicts_and_auto_assign(self):
        m = QinQMapper()
        m.add_range(200, 2, 4)
        assert m.register("sub-1", 200, 2) == (200, 2)
        with pytest.raises(QinQError):
            m.register("sub-2", 200, 2)
        assert m.auto_assign("sub-2") == (200, 3)
        assert m.lookup(200, 3) == "sub-2"
        m.unregister("sub-2")
        assert m.auto_assign("sub-3") == (200, 3)


class TestAgentTLS:
    """TLS config builder (ref pkg/agent/tls_test.go)."""

    @pytest.fixture(scope="class")
    def certs(self, tmp_path_factory):
        import subprocess
        d = tmp_path_factory.mktemp("tls")
        cert, key = str(d / "cert.pem"), str(d / "key.pem")
        subprocess.run(
            ["openssl", "req", "-x509", "-newkey", "rsa:2048",
             "-keyout", key, "-out", cert, "-days", "1", "-nodes",
             "-subj", "/CN=bng-test/O=bng"],
            check=True, capture_output=True)
        return cert, key

    def test_disabled_returns_none(self):
        from bng_amd.agent.tls import TLSConfig, build_ssl_context
        assert build_ssl_context(TLSConfig(enabled=False)) is None

    def test_default_and_min_versions(self):
        import ssl
        from bng_amd.agent.tls import (TLSConfig, TLSError,
                                       build_ssl_context)
        ctx = build_ssl_context(TLSConfig())
        assert ctx.minimum_version == ssl.TLSVersion.TLSv1_2
        ctx13 = build_ssl_context(TLSConfig(min_version="1.3"))
        assert ctx13.minimum_version == ssl.TLSVersion.TLSv1_3
        with pytest.raises(TLSError):
            build_ssl_context(TLSConfig(min_version="1.1"))

    def test_insecure_skip_verify(self):
        import ssl
        from bng_amd.agent.tls import TLSConfig, build_ssl_context
        ctx = build_ssl_context(TLSConfig(insecure_skip_verify=True))
        assert ctx.verify_mode == ssl.CERT_NONE

    def test_ca_by_file_and_pem_and_mtls(self, certs):
        from bng_amd.agent.tls import (TLSConfig, TLSError,
                                       build_ssl_context,
                                       requests_kwargs)
        cert, key = certs
        ctx = build_ssl_context(TLSConfig(ca_cert_file=cert,
                                          cert_file=cert, key_file=key))
        assert ctx.get_ca_certs()
        pem = open(cert).read()
        ctx2 = build_ssl_context(TLSConfig(ca_cert_pem=pem))
        assert ctx2.get_ca_certs()
        # half-configured mTLS rejected
        with pytest.raises(TLSError):
            build_ssl_context(TLSConfig(cert_file=cert))
        with pytest.raises(TLSError):
            build_ssl_context(TLSConfig(ca_cert_file="/nope.pem"))
        kw = requests_kwargs(TLSConfig(ca_cert_file=cert,
                                       cert_file=cert, key_file=key))
        assert kw["verify"] == cert and kw["cert"] == (cert, key)

    def test_fingerprint_and_pinning(self, certs):
        import ssl as ssl_mod
        from bng_amd.agent.tls import (TLSConfig, TLSError,
                                       get_cert_fingerprint,
                                       validate_tls_config, verify_pinned)
        cert, _ = certs
        fp = get_cert_fingerprint(cert)
        assert len(fp) == 64
        validate_tls_config(TLSConfig(pinned_certs=[fp]))
        with pytest.raises(TLSError):
            validate_tls_config(TLSConfig(pinned_certs=["zz"]))
        der = ssl_mod.PEM_cert_to_DER_cert(open(cert).read())
        assert verify_pinned(der, [fp])
        assert verify_pinned(der, [fp.upper().replace("", "")])
        assert not verify_pinned(der, ["0" * 64])
        assert verify_pinned(der, [])        # no pins -> pass

    def test_extract_cert_info(self, certs):
        from bng_amd.agent.tls import extract_cert_info
        cert, _ = certs
        info = extract_cert_info(cert)
        assert info["subject"] == "bng-test"
        assert info["issuer"] == "bng-test"     # self-signed
        assert info["not_after"]


class TestQinQFull:
    """Mapper behaviors mirrored from ref pkg/qinq/qinq_test.go."""

    def test_invalid_tags_rejected(self):
        m = QinQMapper()
        with pytest.raises(QinQError):
            m.register("s", 0, 5)
        with pytest.raises(QinQError):
            m.register("s", 100, 4095)
        with pytest.raises(QinQError):
            m.add_range(5000, 2, 4)
        with pytest.raises(QinQError):
            m.add_range(100, 10, 5)       # start > end

    def test_update_moves_subscriber_mapping(self):
        m = QinQMapper()
        m.register("sub-1", 100, 2)
        m.register("sub-1", 100, 9)       # re-register: frees old pair
        assert m.lookup(100, 2) is None
        assert m.lookup(100, 9) == "sub-1"
        assert m.lookup_subscriber("sub-1") == (100, 9)

    def test_unregister_by_vlan_and_stats(self):
        m = QinQMapper()
        m.add_range(300, 2, 11)
        m.register("sub-1", 300, 2)
        m.register("sub-2", 300, 3)
        assert m.unregister_by_vlan(300, 2) == "sub-1"
        assert m.unregister_by_vlan(300, 2) is None
        assert m.lookup_subscriber("sub-1") is None
        st = m.stats()
        assert st == {"mappings": 1, "ranges": 1, "capacity": 10,
                      "free": 9}


class TestDNSCacheBehaviors:
    """LRU eviction / negative cache / cleanup (ref
    pkg/dns/resolver_test.go)."""

    def _mk(self, answers=None, **kw):
        from bng_amd.dns.resolver import (Resolver, build_query,
                                          build_response)
        calls = []

        def upstream(q):
            calls.append(q)
            from bng_amd.dns.resolver import decode_qname
            name, _ = decode_qname(q, 12)
            addrs = (answers or {}).get(name, [])
            return build_response(q, addrs, ttl=300,
                                  rcode=0 if addrs else 3)
        return Resolver(upstream, **kw), calls

    def test_lru_eviction(self):
        from bng_amd.dns.resolver import build_query
        r, calls = self._mk({f"h{i}.x": ["10.0.0.1"] for i in range(5)},
                            max_entries=3)
        for i in range(5):
            r.handle_query(build_query(f"h{i}.x"))
        assert r.stats["evicted"] == 2
        # h0/h1 evicted; h4 cached
        r.handle_query(build_query("h4.x"))
        assert r.stats["cache_hits"] == 1
        n = len(calls)
        r.handle_query(build_query("h0.x"))
        assert len(calls) == n + 1            # refetched

    def test_negative_cache(self):
        from bng_amd.dns.resolver import build_query
        r, calls = self._mk({})               # every name NXDOMAIN
        q = build_query("missing.example")
        r.handle_query(q)
        resp2 = r.handle_query(q)
        assert len(calls) == 1                # second hit served locally
        assert r.stats["negative_hits"] == 1
        import struct as st
        assert st.unpack_from(">H", resp2, 2)[0] & 0xF == 3   # NXDOMAIN

    def test_cleanup_expired(self):
        from bng_amd.dns.resolver import build_query
        r, _ = self._mk({"a.x": ["10.0.0.1"]}, min_ttl=1, max_ttl=1)
        r.handle_query(build_query("a.x"))
        assert r.cleanup(now=time.time() + 5) == 1
        assert r.cleanup() == 0


class TestDirectCaching:
    """Cache / binding events / BSS sync (ref
    pkg/direct/authenticator_test.go)."""

    def make(self):
        store = MemoryStore()
        c = NexusClient(store)
        c.save_subscriber(Subscriber("sub-1", s_tag=100, c_tag=5,
                                     mac="aa:bb:cc:00:00:01",
                                     isp_id="isp-a"))
        return c

    def test_positive_results_cached(self):
        d = DirectAuth(self.make())
        assert d.authenticate_by_mac("aa:bb:cc:00:00:01").success
        assert d.authenticate_by_mac("aa:bb:cc:00:00:01").success
        assert d.stats["cache_hits"] == 1
        # negative results are NOT cached
        d.authenticate_by_mac("ff:ff:ff:ff:ff:ff")
        d.authenticate_by_mac("ff:ff:ff:ff:ff:ff")
        assert d.stats["cache_hits"] == 1
        d.invalidate_cache()
        d.authenticate_by_mac("aa:bb:cc:00:00:01")
        assert d.stats["cache_hits"] == 1          # refilled, not hit

    def test_binding_events_reach_bss(self):
        events = []

        class BSS:
            def subscriber_status(self, sid):
                return "active"

            def report_binding(self, ev):
                events.append((ev.subscriber_id, ev.mac, ev.event))
        d = DirectAuth(self.make(), bss=BSS())
        assert d.report_binding_event("sub-1", "aa:bb:cc:00:00:01",
                                      "10.0.1.5")
        assert events == [("sub-1", "aa:bb:cc:00:00:01", "bind")]
        # BSS without the capability: graceful False
        d2 = DirectAuth(self.make())
        assert not d2.report_binding_event("sub-1", "m", "i")

    def test_sync_from_bss_prefills_cache(self):
        class BSS:
            def subscriber_status(self, sid):
                return "active"

            def sync_mappings(self):
                return [{"subscriber_id": "sub-9", "isp_id": "isp-b",
                         "mac": "aa:00:00:00:00:09",
                         "vlan": "100.9"}]
        d = DirectAuth(self.make(), bss=BSS())
        assert d.sync_from_bss() == 2              # mac + vlan keys
        r = d.authenticate_by_mac("aa:00:00:00:00:09")
        assert r.success and r.subscriber_id == "sub-9"
        assert d.stats["cache_hits"] == 1


class TestPONCoverage:
    """Duplicate discovery / callbacks / offline-rediscovery (ref
    pkg/pon/manager_test.go)."""

    def test_duplicate_discovery_is_one_nte(self):
        from bng_amd.pon.manager import Manager as PON
        store = MemoryStore()
        pon = PON(store)
        events = []
        pon.on_event(lambda ev, n: events.append((ev, n.id)))
        a = pon.ont_discovered("SER001", "pon0/1")
        b = pon.ont_discovered("SER001", "pon0/1")     # re-announce
        assert a.id == b.id
        assert pon.stats["discovered"] == 1            # counted once
        assert len(pon.list_ntes()) == 1
        assert events == [("discovered", a.id)] * 2    # both announced

    def test_offline_then_rediscovered_keeps_provisioning(self):
        from bng_amd.pon.manager import Manager as PON, QoSProfile
        store = MemoryStore()
        pon = PON(store)
        pon.add_profile(QoSProfile("res-100", 100_000_000, 20_000_000))
        nte = pon.ont_discovered("SER002", "pon0/2")
        pon.provision(nte.id, profile="res-100", s_tag=100, c_tag=7)
        pon.ont_offline(nte.id)
        assert [n.state for n in pon.list_ntes()] == ["offline"]
        back = pon.ont_discovered("SER002", "pon0/2")
        assert back.provisioned and (back.s_tag, back.c_tag) == (100, 7)
        assert pon.stats["offline"] == 1


class TestAuthenticatedTransport:
    """Header-injecting HTTP transport (ref
    pkg/deviceauth/transport_test.go)."""

    def test_headers_injected_and_verified(self):
        import threading
        from http.server import BaseHTTPRequestHandler, HTTPServer
        from bng_amd.deviceauth.authenticator import (AuthenticatedSession,
                                                      PSKAuthenticator)
        auth = PSKAuthenticator(b"shared-key")
        seen = {}

        class H(BaseHTTPRequestHandler):
            def do_GET(self):
                seen.update({k: v for k, v in self.headers.items()})
                try:
                    seen["verified"] = auth.verify(dict(self.headers))
                except Exception as e:
                    seen["verified"] = f"error:{e}"
                self.send_response(200)
                self.end_headers()
                self.wfile.write(b"ok")

            def log_message(self, *a):
                pass
        srv = HTTPServer(("127.0.0.1", 0), H)
        t = threading.Thread(target=srv.serve_forever, daemon=True)
        t.start()
        try:
            s = AuthenticatedSession(auth, "nte-42")
            r = s.get(f"http://127.0.0.1:{srv.server_port}/cfg")
            assert r.status_code == 200
            assert seen["verified"] == "nte-42"    # server-side verify
            assert "X-Device-Id" in seen or "X-Device-ID" in seen or \
                any(k.lower() == "x-device-id" for k in seen)
            # user headers preserved alongside auth headers
            s.get(f"http://127.0.0.1:{srv.server_port}/cfg",
                  headers={"X-Custom": "1"})
            assert seen.get("X-Custom") == "1"
        finally:
            srv.shutdown()


def test_tpm_mode_recognized_but_unimplemented():
    """Factory parity: TPM is a named mode that raises, exactly like
    the reference (authenticator.go:33)."""
    from bng_amd.deviceauth.authenticator import new_authenticator
    with pytest.raises(NotImplementedError):
        new_authenticator("tpm")
    with pytest.raises(ValueError):
        new_authenticator("retina-scan")


def test_wifi_traffic_stats():
    """ref wifi/gateway_test.go UpdateTrafficStats + Stats."""
    from bng_amd.wifi.gateway import Gateway
    g = Gateway(network="192.168.200.0/28")
    g.join("aa:01")
    g.join("aa:02")
    g.accept_terms("aa:01")
    assert g.update_traffic("aa:01", 1000, 200)
    assert g.update_traffic("aa:01", 500, 100)
    assert not g.update_traffic("none", 1, 1)
    st = g.get_stats()
    assert st["sessions"] == 2 and st["accepted"] == 1
    assert st["bytes_in"] == 1500 and st["bytes_out"] == 300


class TestZTPFullFlow:
    """register_and_wait pending/configured contract (ref
    bootstrap.go:219-300) + vendor TLVs + system-info detection."""

    def test_pending_then_configured(self):
        from bng_amd.ztp.bootstrap import SystemInfo
        srv = ZTPServer()
        try:
            c = BootstrapClient(srv.url, serial="SN-P1",
                                poll_interval=0.05)
            import threading

            def approve_later():
                import time as _t
                while "SN-P1" not in srv.devices:
                    _t.sleep(0.01)
                srv.approve("SN-P1", {
                    "node_id": "bng-42", "site_id": "pop-7",
                    "role": "active",
                    "pools": [{"pool_id": "p1",
                               "cidr": "10.1.0.0/16"}],
                    "cluster": {"peers": ["http://n1", "http://n2"]}})
            threading.Thread(target=approve_later, daemon=True).start()
            cfg = c.register_and_wait(
                SystemInfo(serial="SN-P1", mac="02:00:00:00:00:aa",
                           model="lab", firmware="1.0"),
                initial_backoff=0.05, max_backoff=0.2, deadline=10)
            assert c.state == "approved"
            assert cfg.node_id == "bng-42" and cfg.site_id == "pop-7"
            assert cfg.pools[0]["cidr"] == "10.1.0.0/16"
            assert cfg.cluster["peers"] == ["http://n1", "http://n2"]
            assert cfg.device_id == "bng-42"       # node_id fallback
            # register recorded the hardware identity
            assert srv.devices["SN-P1"]["mac"] == "02:00:00:00:00:aa"
        finally:
            srv.stop()

    def test_pending_max_retries(self):
        srv = ZTPServer()
        try:
            c = BootstrapClient(srv.url, serial="SN-P2")
            with pytest.raises(TimeoutError):
                c.register_and_wait(max_retries=3,
                                    initial_backoff=0.01,
                                    max_backoff=0.02, deadline=10)
        finally:
            srv.stop()

    def test_vendor_tlv_discovery(self):
        from bng_amd.ztp.bootstrap import parse_vendor_options
        tlv = bytes([7, 2, 0, 0, 1, 8]) + b"http://n"
        assert parse_vendor_options(tlv) == "http://n"
        assert parse_vendor_options(b"\x01\xff") == ""   # truncated
        assert parse_vendor_options(b"") == ""
        # option 43 TLV beats plain decode; plain stays supported
        assert discover_nexus_from_dhcp_options(
            {43: tlv}) == "http://n"
        assert discover_nexus_from_dhcp_options(
            {43: b"http://plain"}) == "http://plain"
        # option 224 always wins
        assert discover_nexus_from_dhcp_options(
            {224: b"http://x", 43: tlv}) == "http://x"

    def test_detect_system_info(self):
        from bng_amd.ztp.bootstrap import detect_system_info
        info = detect_system_info()
        assert info.serial            # DMI or hostname fallback
        assert info.model

    def test_plan_interface_config(self):
        from bng_amd.ztp.bootstrap import plan_interface_config
        cmds = plan_interface_config("eth1", "10.0.0.5", 24,
                                     gateway="10.0.0.1",
                                     dns=["1.1.1.1"])
        assert cmds[0] == "ip addr add 10.0.0.5/24 dev eth1"
        assert any("default via 10.0.0.1" in c for c in cmds)
        assert any("1.1.1.1" in c for c in cmds)


class TestTLSCertHelpers:
    def test_expiring_soon_and_server_name(self, tmp_path):
        import subprocess
        from bng_amd.agent.tls import (extract_server_name_from_url,
                                       is_certificate_expiring_soon)
        cert = tmp_path / "c.pem"
        key = tmp_path / "k.pem"
        subprocess.run(
            ["openssl", "req", "-x509", "-newkey", "rsa:2048",
             "-keyout", str(key), "-out", str(cert), "-days", "10",
             "-nodes", "-subj", "/CN=ztp-test"],
            check=True, capture_output=True)
        expiring, days = is_certificate_expiring_soon(str(cert), 30)
        assert expiring and 9 < days <= 10
        expiring2, _ = is_certificate_expiring_soon(str(cert), 5)
        assert not expiring2
        assert extract_server_name_from_url(
            "https://nexus.isp:8443/api") == "nexus.isp"
        assert extract_server_name_from_url("http://10.0.0.1") == \
            "10.0.0.1"


class TestDeviceAuthDepth:
    """Identity derivation, config validation, PSK lifecycle, CSR
    renewal (ref pkg/deviceauth authenticator.go:233-308, psk.go,
    mtls.go:362-418)."""

    def test_device_id_derivation(self):
        from bng_amd.deviceauth.authenticator import (generate_device_id,
                                                      sanitize_id)
        assert sanitize_id("AB-12_c!@#$") == "AB-12_c"
        assert generate_device_id("SN 99/3") == "bng-SN993"
        assert generate_device_id("", "02:aa:bb:cc:dd:ee") == \
            "bng-02aabbccddee"
        a, b = generate_device_id(), generate_device_id()
        assert a.startswith("bng-") and a != b     # random fallback

    def test_read_device_identity(self):
        from bng_amd.deviceauth.authenticator import read_device_identity
        ident = read_device_identity()
        assert ident["device_id"].startswith("bng-")
        assert ident["serial"]

    def test_validate_config(self):
        from bng_amd.deviceauth.authenticator import (AuthError,
                                                      validate_config)
        validate_config("none")
        validate_config("psk", psk_key="k")
        with pytest.raises(AuthError):
            validate_config("psk")
        validate_config("mtls", cert_file="c", key_file="k",
                        ca_file="ca")
        validate_config("mtls", cert_file="c", key_file="k",
                        insecure_skip_verify=True)
        with pytest.raises(AuthError):
            validate_config("mtls", cert_file="c", key_file="k")
        with pytest.raises(AuthError):
            validate_config("mtls", cert_file="c")
        with pytest.raises(NotImplementedError):
            validate_config("tpm")
        with pytest.raises(AuthError):
            validate_config("wat")

    def test_psk_load_and_rotate(self, tmp_path):
        from bng_amd.deviceauth.authenticator import (AuthError,
                                                      PSKAuthenticator,
                                                      load_psk)
        kf = tmp_path / "psk.key"
        kf.write_text("  file-secret-0123456789  \n")
        assert load_psk(key_file=str(kf)) == b"file-secret-0123456789"
        assert load_psk(key="inline") == b"inline"
        with pytest.raises(AuthError):
            load_psk()
        a = PSKAuthenticator(load_psk(key_file=str(kf)))
        h = a.headers("dev-1")
        assert a.verify(h) == "dev-1"
        # rotation enforces the 16-char production floor
        with pytest.raises(AuthError):
            a.rotate("short")
        a.rotate("long-enough-key-123")
        with pytest.raises(AuthError):
            a.verify(h)                    # old-key signature now bad
        assert a.verify(a.headers("dev-1")) == "dev-1"

    def test_csr_renewal_request(self):
        from bng_amd.deviceauth.authenticator import MTLSAuthenticator
        m = MTLSAuthenticator()
        req = m.renewal_request("bng-SN7", reason="expiring")
        assert req["csr"].startswith("-----BEGIN CERTIFICATE REQUEST")
        assert "PRIVATE KEY" in req["_key_pem"]
        assert req["device_id"] == "bng-SN7"

    def test_cert_expires_within(self, tmp_path):
        import subprocess
        from bng_amd.deviceauth.authenticator import MTLSAuthenticator
        cert = tmp_path / "c.pem"
        subprocess.run(
            ["openssl", "req", "-x509", "-newkey", "rsa:2048",
             "-keyout", str(tmp_path / "k.pem"), "-out", str(cert),
             "-days", "10", "-nodes", "-subj", "/CN=d"],
            check=True, capture_output=True)
        m = MTLSAuthenticator()
        assert m.certificate_expires_within(str(cert), 30)
        assert not m.certificate_expires_within(str(cert), 5)


class TestAgentRegistries:
    """Wholesale subscriber/NTE registries + ISP churn (ref
    pkg/agent/agent.go:315-467, types.go:156-240)."""

    def _agent(self):
        store = MemoryStore()
        return Agent(store, "node-1", partition_after=0.0)

    def test_subscriber_indexes(self):
        from bng_amd.agent.agent import Subscriber
        a = self._agent()
        a.set_subscriber(Subscriber("sub-1", nte_id="NTE-7",
                                    mac="AA:BB:CC:00:00:01",
                                    isp_id="isp-a"))
        assert a.get_subscriber("sub-1").isp_id == "isp-a"
        assert a.get_subscriber_by_mac("aa:bb:cc:00:00:01") \
            .subscriber_id == "sub-1"
        assert a.get_subscriber_by_nte("NTE-7").subscriber_id == "sub-1"
        assert a.subscriber_count() == 1
        a.remove_subscriber("sub-1")
        assert a.subscriber_count() == 0
        assert a.get_subscriber_by_mac("aa:bb:cc:00:00:01") is None
        assert a.get_subscriber_by_nte("NTE-7") is None

    def test_isp_churn_event(self):
        from bng_amd.agent.agent import Subscriber
        a = self._agent()
        events = []
        a.on_isp_churn(events.append)
        a.set_subscriber(Subscriber("sub-1", isp_id="isp-a"))
        assert events == []                      # first set: no churn
        a.set_subscriber(Subscriber("sub-1", isp_id="isp-a"))
        assert events == []                      # same ISP: no churn
        a.set_subscriber(Subscriber("sub-1", isp_id="isp-b"))
        assert len(events) == 1
        assert events[0]["old_isp_id"] == "isp-a"
        assert events[0]["new_isp_id"] == "isp-b"
        assert a.subscriber_count_by_isp() == {"isp-b": 1}

    def test_nte_registry_and_health(self):
        from bng_amd.agent.agent import NTE, Subscriber
        a = self._agent()
        a.start()
        try:
            a.set_nte(NTE("SER-1", port=3, status="provisioned",
                          vendor="acme"))
            a.set_subscriber(Subscriber("sub-1"))
            assert a.get_nte("SER-1").vendor == "acme"
            assert a.nte_count() == 1
            h = a.health()
            assert h["status"] == "connected" and h["online"]
            assert h["subscribers"] == 1 and h["ntes"] == 1
            assert h["uptime_seconds"] >= 0
            a.remove_nte("SER-1")
            assert a.nte_count() == 0
        finally:
            a.stop()

    def test_isp_config_lookup(self):
        a = self._agent()
        a.config = {"isps": [{"isp_id": "isp-a", "radius_realm": "a.net"},
                             {"isp_id": "isp-b"}]}
        assert a.get_isp_config("isp-a")["radius_realm"] == "a.net"
        assert a.get_isp_config("nope") is None


class TestWifiManager:
    """Captive-portal session manager with grace periods + dual
    operating modes (ref pkg/wifi/gateway.go:27-553)."""

    def test_wifi_mode_grace_period_flow(self):
        from bng_amd.wifi.gateway import (Manager as WifiMgr, S_ACTIVE,
                                          S_AUTHENTICATED, S_GRACE,
                                          WifiConfig)
        events = {"created": [], "authed": [], "expired": []}
        m = WifiMgr(WifiConfig(lease_duration=100.0, grace_period=10.0))
        m.on_create = lambda s: events["created"].append(s.mac)
        m.on_auth = lambda s: events["authed"].append(s.mac)
        m.on_expire = lambda s: events["expired"].append(s.mac)
        s = m.create_session("AA:BB:CC:00:00:01", hostname="phone",
                             ip="192.168.100.5")
        assert s.state == S_GRACE
        assert m.is_in_grace_period("aa:bb:cc:00:00:01")
        assert m.needs_authentication("aa:bb:cc:00:00:01")
        assert m.needs_authentication("ff:ff:ff:00:00:00")  # unknown
        assert m.get_session_by_ip("192.168.100.5") is s
        assert m.authenticate_session("aa:bb:cc:00:00:01",
                                      user_identity="user@isp")
        assert s.state == S_AUTHENTICATED and s.authenticated
        assert not m.needs_authentication("aa:bb:cc:00:00:01")
        m.update_traffic("aa:bb:cc:00:00:01", bytes_in=100,
                         packets_in=1)
        assert s.state == S_ACTIVE and s.bytes_in == 100
        assert events == {"created": ["aa:bb:cc:00:00:01"],
                          "authed": ["aa:bb:cc:00:00:01"],
                          "expired": []}

    def test_grace_lapse_kills_unauthenticated_only(self):
        import time as _t
        from bng_amd.wifi.gateway import Manager as WifiMgr, WifiConfig
        m = WifiMgr(WifiConfig(lease_duration=1000.0, grace_period=10.0))
        m.create_session("aa:00:00:00:00:01")
        authed = m.create_session("aa:00:00:00:00:02")
        m.authenticate_session("aa:00:00:00:00:02")
        now = _t.time()
        assert m.cleanup_expired(now + 20) == 1       # grace lapsed
        assert m.get_session("aa:00:00:00:00:01") is None
        assert m.get_session("aa:00:00:00:00:02") is authed
        # lease expiry takes the authenticated one too
        assert m.cleanup_expired(now + 2000) == 1
        assert m.manager_stats()["active_sessions"] == 0

    def test_renewal_extends_lease(self):
        import time as _t
        from bng_amd.wifi.gateway import Manager as WifiMgr, WifiConfig
        m = WifiMgr(WifiConfig(lease_duration=100.0,
                               captive_portal_enabled=False))
        s = m.create_session("aa:00:00:00:00:03")
        assert s.authenticated                       # portal off
        assert not m.needs_authentication("aa:00:00:00:00:03")
        first_expiry = s.lease_expiry
        _t.sleep(0.01)
        assert m.renew_session("aa:00:00:00:00:03")
        assert s.lease_expiry > first_expiry
        assert not m.renew_session("none")

    def test_olt_bng_preset(self):
        from bng_amd.wifi.gateway import MODE_OLT_BNG, WifiConfig
        c = WifiConfig.olt_bng()
        assert c.mode == MODE_OLT_BNG
        assert c.allocation_trigger == "radius_auth"
        assert c.deallocation_trigger == "session_termination"
        assert not c.captive_portal_enabled


class TestPONAutoProvisioner:
    """Discovery -> provision pipeline with retries and state machine
    (ref pkg/pon/manager.go:216-330, 398-486)."""

    def _mk(self, **kw):
        from bng_amd.pon.manager import AutoProvisioner, Manager as PON
        from bng_amd.qinq.mapper import Mapper
        store = MemoryStore()
        vm = Mapper()
        vm.add_range(100)
        pon = PON(store, vlan_mapper=vm)
        return pon, AutoProvisioner(pon, retry_delay=0.0, **kw)

    def test_discovery_provisions_with_vlans(self):
        pon, ap = self._mk()
        discovered, provisioned = [], []
        ap.on_discovered = discovered.append
        ap.on_provisioned = provisioned.append
        r = ap.handle_discovery("SER100", "pon0/1")
        assert r["success"] and r["s_tag"] and r["c_tag"]
        assert ap.nte_state("SER100") == "CONNECTED"
        assert ap.list_connected() == ["SER100"]
        assert ap.list_pending() == []
        assert discovered[0]["serial"] == "SER100"
        assert provisioned[0]["nte_id"] == "nte-SER100"

    def test_reconnect_keeps_existing_tags(self):
        pon, ap = self._mk()
        r1 = ap.handle_discovery("SER101", "pon0/1")
        ap.handle_disconnect("SER101")
        assert ap.nte_state("SER101") == "DISCONNECTED"
        assert ap.stats["disconnected"] == 1
        r2 = ap.handle_discovery("SER101", "pon0/1")
        assert r2["success"]
        assert (r2["s_tag"], r2["c_tag"]) == (r1["s_tag"], r1["c_tag"])
        assert ap.stats["reconnected"] == 1

    def test_provisioning_retries_then_fails(self):
        pon, ap = self._mk(retries=2, default_profile="missing-profile")
        results = []
        ap.on_provisioned = results.append
        r = ap.handle_discovery("SER102", "pon0/1")
        assert not r["success"]
        assert "missing-profile" in r["error"]
        assert ap.nte_state("SER102") == "UNCONFIGURED"
        assert len(ap.list_pending()) == 1
        assert ap.stats["failed"] == 1
        # operator adds the profile; rediscovery succeeds
        from bng_amd.pon.manager import QoSProfile
        pon.add_profile(QoSProfile("missing-profile", 10, 5))
        r2 = ap.handle_discovery("SER102", "pon0/1")
        assert r2["success"]
        assert ap.list_pending() == []


class TestONTAuthenticator:
    """ONT-mapping-driven direct auth feeding the subscriber manager's
    rich protocol (ref pkg/direct/authenticator.go:93-470)."""

    def _auth(self):
        from bng_amd.direct.authenticator import (MappingBSS, ONTMapping,
                                                  ONTAuthenticator)
        bss = MappingBSS()
        bss.add_mapping(ONTMapping(
            ont_serial="SER-1", subscriber_id="sub-1",
            circuit_id="olt1/pon0/1", isp_id="isp-a",
            qos_policy="gold", download_bps=500_000_000,
            upload_bps=100_000_000))
        bss.add_mapping(ONTMapping(
            ont_serial="SER-2", subscriber_id="sub-2",
            status="suspended"))
        bss.add_mapping(ONTMapping(
            ont_serial="SER-3", subscriber_id="sub-3",
            status="disconnected"))
        return bss, ONTAuthenticator(bss, default_isp="isp-def",
                                     session_timeout=3600)

    def test_full_flow_through_subscriber_manager(self):
        from bng_amd.subscriber.manager import Manager, S_AUTHENTICATED
        bss, auth = self._auth()
        m = Manager(authenticator=auth)
        s = m.open_session("pending", mac="aa:00:00:00:00:01")
        s.circuit_id = "olt1/pon0/1"
        r = m.authenticate_full(s.id)
        assert r["success"]
        assert s.state == S_AUTHENTICATED
        assert s.subscriber_id == "sub-1" and s.isp_id == "isp-a"
        assert s.qos_policy_id == "gold"
        assert s.download_rate_bps == 500_000_000
        assert s.session_timeout == 3600

    def test_suspended_goes_walled(self):
        from bng_amd.subscriber.manager import Manager, S_WALLED
        bss, auth = self._auth()
        m = Manager(authenticator=auth)
        s = m.open_session("x")
        s.nte_id = "SER-2"
        assert m.authenticate_full(s.id)["success"]
        assert s.state == S_WALLED
        assert "suspended" in s.walled_reason.lower()

    def test_disconnected_and_unknown_fail(self):
        bss, auth = self._auth()

        class S:
            circuit_id = ""
            nte_id = "SER-3"
        r = auth.authenticate_session(S(), {})
        assert not r["success"] and "disconnected" in r["error"]
        S.nte_id = "SER-404"
        assert not auth.authenticate_session(S(), {})["success"]
        assert auth.stats["not_found"] == 1

    def test_cache_and_invalidate(self):
        bss, auth = self._auth()

        class S:
            circuit_id = ""
            nte_id = "SER-1"
        auth.authenticate_session(S(), {})
        auth.authenticate_session(S(), {})
        assert auth.stats["cache_hits"] == 1
        # BSS change invisible until invalidation
        bss.by_serial["SER-1"].status = "suspended"
        assert "walled_garden" not in auth.authenticate_session(S(), {})
        auth.invalidate_cache(ont_serial="SER-1")
        assert auth.authenticate_session(S(), {}).get("walled_garden")
        assert auth.sync_from_bss() == 3

    def test_binding_events(self):
        import pytest as _pt
        bss, auth = self._auth()
        auth.report_binding("assign", "SER-1", "sub-1",
                            mac="aa:00:00:00:00:01", ipv4="10.0.1.5")
        auth.report_binding("release", "SER-1", "sub-1")
        assert [b["event_type"] for b in bss.bindings] == \
            ["assign", "release"]
        with _pt.raises(ValueError):
            auth.report_binding("wat", "SER-1", "sub-1")


class TestWalledGardenClassify:
    """Keyed allowed destinations + flow classification with portal
    redirect (ref pkg/walledgarden/manager.go:56-242)."""

    def _mgr(self):
        from bng_amd.walledgarden.manager import Manager as WG
        return WG(portal_ip="10.255.255.1", portal_port=8080,
                  dns_servers=["8.8.8.8"])

    def test_default_allowed_dests(self):
        from bng_amd.walledgarden.manager import (REASON_DNS,
                                                  REASON_PORTAL)
        m = self._mgr()
        assert m.is_destination_allowed("8.8.8.8", 53, 17)
        assert m.is_destination_allowed("8.8.8.8", 53, 6)
        assert m.is_destination_allowed("10.255.255.1", 8080, 6)
        assert not m.is_destination_allowed("10.255.255.1", 443, 6)
        assert m.allowed_dests[("8.8.8.8", 53, 17)] == REASON_DNS
        assert m.allowed_dests[("10.255.255.1", 8080, 6)] == \
            REASON_PORTAL
        # legacy bare-IP call still works
        assert m.is_destination_allowed("8.8.8.8")

    def test_classification_table(self):
        from bng_amd.walledgarden.manager import (V_DROP, V_FORWARD,
                                                  V_REDIRECT)
        m = self._mgr()
        redirects = []
        m.on_redirect(lambda mac, ip: redirects.append((mac, ip)))
        m.add("AA:00:00:00:00:01", "10.0.1.5")
        # quarantined: DNS ok, HTTP redirects, HTTPS drops
        assert m.classify("aa:00:00:00:00:01", "8.8.8.8", 53, 17) == \
            V_FORWARD
        assert m.classify("aa:00:00:00:00:01", "93.184.216.34", 80,
                          6) == V_REDIRECT
        assert redirects == [("aa:00:00:00:00:01", "93.184.216.34")]
        assert m.classify("aa:00:00:00:00:01", "93.184.216.34", 443,
                          6) == V_DROP
        # custom allowed destination opens up
        m.allow_destination("93.184.216.34", 443, 6)
        assert m.classify("aa:00:00:00:00:01", "93.184.216.34", 443,
                          6) == V_FORWARD
        # activation ends quarantine entirely
        m.activate("aa:00:00:00:00:01")
        assert m.classify("aa:00:00:00:00:01", "1.2.3.4", 443, 6) == \
            V_FORWARD
        # blocked drops everything, allowed or not
        m.block("aa:00:00:00:00:01")
        assert m.classify("aa:00:00:00:00:01", "8.8.8.8", 53, 17) == \
            V_DROP
        # unknown MAC is not the garden's business
        assert m.classify("ff:ff:ff:00:00:00", "1.2.3.4", 443, 6) == \
            V_FORWARD
        assert m.stats["redirects"] == 1


class TestQinQConfiguredMapper:
    """Range-validated VLAN mapping + lookup priority (ref
    pkg/qinq/qinq.go:18-212)."""

    def test_pair_semantics(self):
        from bng_amd.qinq.mapper import VLANPair
        assert str(VLANPair(100, 7)) == "s100.c7"
        assert str(VLANPair(0, 7)) == "c7"
        assert VLANPair(100, 7).is_double_tagged
        assert VLANPair(0, 7).is_single_tagged
        assert VLANPair().is_untagged

    def test_range_validation_and_moves(self):
        from bng_amd.qinq.mapper import (ConfiguredMapper, QinQConfig,
                                         QinQError, VLANPair)
        m = ConfiguredMapper(QinQConfig(
            s_tag_ranges=[(100, 199, "isp-a"), (300, 399, "isp-b")],
            c_tag_range=(10, 20)))
        m.register(VLANPair(100, 10), "sub-1")
        m.register(VLANPair(300, 20), "sub-2")
        with pytest.raises(QinQError):
            m.register(VLANPair(200, 10), "sub-3")   # S out of range
        with pytest.raises(QinQError):
            m.register(VLANPair(100, 50), "sub-3")   # C out of range
        with pytest.raises(QinQError):
            m.register(VLANPair(100, 10), "sub-9")   # owned
        # re-registering moves the subscriber, freeing the old pair
        m.register(VLANPair(100, 11), "sub-1")
        assert m.get_subscriber(VLANPair(100, 10)) is None
        assert m.get_vlan("sub-1") == VLANPair(100, 11)
        m.unregister_subscriber("sub-2")
        assert m.get_subscriber(VLANPair(300, 20)) is None
        assert m.stats()["total_mappings"] == 1

    def test_lookup_priority(self):
        from bng_amd.qinq.mapper import (ConfiguredMapper, QinQConfig,
                                         VLANPair)
        mac_table = {"aa:bb": "sub-mac"}
        pair = VLANPair(100, 110)

        def mk(prio):
            m = ConfiguredMapper(QinQConfig(lookup_priority=prio))
            m.register(pair, "sub-vlan")
            return m

        look = lambda mac: mac_table.get(mac)
        assert mk("vlan_first").lookup(pair, look, "aa:bb") == "sub-vlan"
        assert mk("mac_first").lookup(pair, look, "aa:bb") == "sub-mac"
        assert mk("vlan_only").lookup(pair, look, "aa:bb") == "sub-vlan"
        # miss on the preferred path falls through
        m = mk("vlan_first")
        assert m.lookup(VLANPair(100, 99), look, "aa:bb") == "sub-mac"
        assert mk("vlan_only").lookup(VLANPair(100, 99), look,
                                      "aa:bb") is None
        with pytest.raises(Exception):
            QinQConfig(lookup_priority="wat")


class TestDNSInterceptRules:
    """Typed interception rules + per-client walled garden (ref
    pkg/dns/resolver.go:212-530)."""

    def _resolver(self):
        upstream_log = []

        def upstream(q):
            upstream_log.append(q)
            return build_response(q, ["93.184.216.34"], ttl=300)
        r = Resolver(upstream)
        return r, upstream_log

    def test_block_redirect_cname_actions(self):
        import struct as _st
        from bng_amd.dns.resolver import (build_cname_response,
                                          decode_qname)
        r, log = self._resolver()
        r.add_rule("ads.example", action="block")
        r.add_rule("old.example", action="redirect",
                   redirect=["10.9.9.9"])
        r.add_rule("www.legacy", action="cname",
                   cname="portal.isp.net")
        # block -> NXDOMAIN, no upstream
        resp = r.handle_query(build_query("ads.example"))
        assert _st.unpack_from(">H", resp, 2)[0] & 0xF == 3
        assert log == []
        # subdomain wildcard matches too
        resp = r.handle_query(build_query("x.ads.example"))
        assert _st.unpack_from(">H", resp, 2)[0] & 0xF == 3
        # redirect -> the configured IP
        _, addrs, _ = parse_response(
            r.handle_query(build_query("old.example")))
        assert addrs == ["10.9.9.9"]
        # cname -> CNAME record with the target
        resp = r.handle_query(build_query("www.legacy"))
        an = _st.unpack_from(">H", resp, 6)[0]
        assert an == 1
        _qn, off = decode_qname(resp, 12)
        off += 4 + 2          # question + name pointer
        rtype = _st.unpack_from(">H", resp, off)[0]
        assert rtype == 5     # CNAME
        target, _ = decode_qname(resp, off + 10)
        assert target == "portal.isp.net"
        # non-matching name goes upstream
        r.handle_query(build_query("fine.example"))
        assert len(log) == 1

    def test_match_modes(self):
        r, log = self._resolver()
        r.add_rule("exact.example", action="block", exact=True)
        r.add_rule(action="block", suffix=".tracker.net")
        # exact does not match subdomains
        import struct as _st
        assert _st.unpack_from(
            ">H", r.handle_query(build_query("exact.example")), 2
        )[0] & 0xF == 3
        r.handle_query(build_query("sub.exact.example"))
        assert len(log) == 1                  # went upstream
        assert _st.unpack_from(
            ">H", r.handle_query(build_query("x.tracker.net")), 2
        )[0] & 0xF == 3
        assert r.remove_rule("exact.example")
        r.handle_query(build_query("exact.example"))
        assert len(log) == 2

    def test_walled_client_registry(self):
        r, log = self._resolver()
        r.set_intercept_all(["10.255.255.1"])
        r.add_walled_client("10.0.1.5")
        assert r.is_walled("10.0.1.5")
        _, addrs, _ = parse_response(
            r.handle_query(build_query("anything.example"),
                           client="10.0.1.5"))
        assert addrs == ["10.255.255.1"]      # portal for walled client
        r.handle_query(build_query("anything.example"),
                       client="10.0.2.2")
        assert len(log) == 1                  # other clients normal
        assert r.remove_walled_client("10.0.1.5")
        assert not r.remove_walled_client("10.0.1.5")
        r.handle_query(build_query("other.example"),
                       client="10.0.1.5")
        assert len(log) == 2


class TestWalledGardenDnsGlue:
    def test_quarantine_drives_dns_interception(self):
        """Garden state changes flow into the resolver's walled-client
        registry: quarantined IPs resolve everything to the portal,
        activation restores normal resolution."""
        from bng_amd.walledgarden.manager import Manager as WG, attach_dns
        upstream_hits = []

        def upstream(q):
            upstream_hits.append(q)
            return build_response(q, ["93.184.216.34"])
        r = Resolver(upstream)
        wg = WG(portal_ip="10.255.255.1", dns_servers=["8.8.8.8"])
        attach_dns(wg, r, ["10.255.255.1"])
        wg.add("aa:00:00:00:00:01", "10.0.1.5")
        assert r.is_walled("10.0.1.5")
        _, addrs, _ = parse_response(
            r.handle_query(build_query("x.example"), client="10.0.1.5"))
        assert addrs == ["10.255.255.1"]
        assert upstream_hits == []
        wg.activate("aa:00:00:00:00:01")
        assert not r.is_walled("10.0.1.5")
        r.handle_query(build_query("x.example"), client="10.0.1.5")
        assert len(upstream_hits) == 1
        # expiry also releases
        import time as _t
        wg.add("aa:00:00:00:00:02", "10.0.1.6", ttl=5.0)
        assert r.is_walled("10.0.1.6")
        wg.expire_stale(now=_t.time() + 10)
        assert not r.is_walled("10.0.1.6")
        # pre-existing quarantine adopted at attach time
        wg2 = WG(portal_ip="10.255.255.1")
        wg2.add("aa:00:00:00:00:03", "10.0.1.7")
        r2 = Resolver(upstream)
        attach_dns(wg2, r2, ["10.255.255.1"])
        assert r2.is_walled("10.0.1.7")


def test_wifi_recreate_renews_lease():
    """CreateSession on a known MAC returns the same session and
    extends its lease (ref gateway_test.go SessionRenewalOnCreate)."""
    import time as _t
    from bng_amd.wifi.gateway import Manager as WifiMgr, WifiConfig
    m = WifiMgr(WifiConfig(lease_duration=100.0))
    s1 = m.create_session("aa:00:00:00:00:09")
    first_expiry = s1.lease_expiry
    _t.sleep(0.01)
    s2 = m.create_session("aa:00:00:00:00:09")
    assert s2 is s1
    assert s2.lease_expiry > first_expiry


def test_ztp_healthcheck():
    """/health probe against the Nexus (ref bootstrap.go Healthcheck)."""
    srv = ZTPServer()
    try:
        c = BootstrapClient(srv.url, serial="SN-H")
        assert c.healthcheck() is True
        # a dead endpoint is unhealthy, not an exception
        c2 = BootstrapClient("http://127.0.0.1:1", serial="SN-H")
        assert c2.healthcheck() is False
    finally:
        srv.stop()
