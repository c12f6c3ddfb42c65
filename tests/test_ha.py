"""HA tests — real active+standby HASyncer pairs on localhost HTTP
(ref pkg/ha/sync_test.go, failover_test.go:94-193)."""
import time

import pytest

from bng_amd.ha.failover import (STATE_FAILED_OVER, STATE_NORMAL,
                                 FailoverController)
from bng_amd.ha.health_monitor import HealthMonitor
from bng_amd.ha.protocol import (ROLE_ACTIVE, ROLE_STANDBY, SessionState)
from bng_amd.ha.sync import HASyncer


def wait_for(cond, timeout=5.0, interval=0.02):
    t0 = time.time()
    while time.time() - t0 < timeout:
        if cond():
            return True
        time.sleep(interval)
    return False


@pytest.fixture
def pair():
    active = HASyncer("node-a", ROLE_ACTIVE, heartbeat_interval=0.2).start()
    standby = HASyncer("node-b", ROLE_STANDBY, partner_url=active.url,
                       full_sync_interval=3600,
                       reconnect_backoff=0.1).start()
    yield active, standby
    standby.stop()
    active.stop()


class TestSync:
    def test_delta_stream_replication(self, pair):
        active, standby = pair
        assert wait_for(lambda: standby.connected)
        active.publish_add(SessionState("s1", "sub-1", ip="10.0.1.50"))
        active.publish_add(SessionState("s2", "sub-2", ip="10.0.1.51"))
        assert wait_for(lambda: standby.store.count() == 2)
        assert standby.store.get("s1").ip == "10.0.1.50"
        active.publish_update(SessionState("s1", "sub-1", ip="10.0.1.99"))
        assert wait_for(
            lambda: standby.store.get("s1").ip == "10.0.1.99")
        active.publish_delete("s2")
        assert wait_for(lambda: standby.store.count() == 1)

    def test_full_sync_on_connect(self):
        active = HASyncer("a", ROLE_ACTIVE).start()
        try:
            for i in range(50):
                active.store.put(SessionState(f"s{i}", f"sub-{i}"))
            standby = HASyncer("b", ROLE_STANDBY, partner_url=active.url,
                               reconnect_backoff=0.1).start()
            try:
                assert wait_for(lambda: standby.store.count() == 50)
                assert standby.stats["full_syncs"] >= 1
            finally:
                standby.stop()
        finally:
            active.stop()

    def test_standby_promotion_keeps_state(self, pair):
        """At failover the shadow store becomes authoritative
        (ref SURVEY §3.5)."""
        active, standby = pair
        assert wait_for(lambda: standby.connected)
        active.publish_add(SessionState("s1", "sub-1", ip="10.0.1.50"))
        assert wait_for(lambda: standby.store.count() == 1)
        active.stop()
        standby.promote()
        assert standby.role == ROLE_ACTIVE
        assert standby.store.get("s1").ip == "10.0.1.50"
        # promoted node can now publish
        standby.publish_add(SessionState("s3", "sub-3"))
        assert standby.store.count() == 2


class TestHealthMonitor:
    def test_thresholds(self):
        active = HASyncer("a", ROLE_ACTIVE).start()
        try:
            m = HealthMonitor(active.url, failure_threshold=2,
                              recovery_threshold=2)
            events = []
            m.on_event(lambda e: events.append(e.type))
            assert m.check_once()
            assert m.partner_healthy
            active.stop()
            m.check_once()
            assert m.partner_healthy          # 1 < threshold
            m.check_once()
            assert not m.partner_healthy      # threshold reached
            assert events == ["partner_down"]
        finally:
            active.stop()

    def test_recovery_event(self):
        m = HealthMonitor("http://127.0.0.1:1", failure_threshold=1,
                          recovery_threshold=1, timeout=0.2)
        events = []
        m.on_event(lambda e: events.append(e.type))
        m.check_once()
        assert not m.partner_healthy
        active = HASyncer("a", ROLE_ACTIVE).start()
        try:
            m.partner_url = active.url
            m.check_once()
            assert m.partner_healthy
            assert events == ["partner_down", "partner_up"]
        finally:
            active.stop()


class TestFailover:
    def test_standby_takes_over_on_partner_death(self):
        """End-to-end: active dies -> monitor detects -> controller
        promotes -> role callback fires (ref SURVEY §3.5 call stack)."""
        active = HASyncer("a", ROLE_ACTIVE, heartbeat_interval=0.2).start()
        standby = HASyncer("b", ROLE_STANDBY, partner_url=active.url,
                           reconnect_backoff=0.1).start()
        roles = []
        mon = HealthMonitor(active.url, interval=0.1, timeout=0.3,
                            failure_threshold=2)
        ctl = FailoverController(
            "b", ROLE_STANDBY, monitor=mon,
            role_change_callback=lambda r: (roles.append(r),
                                            standby.promote()))
        try:
            active.publish_add(SessionState("s1", "sub-1"))
            assert wait_for(lambda: standby.store.count() == 1)
            active.stop()
            mon.start()
            assert wait_for(lambda: ctl.role == ROLE_ACTIVE, timeout=10)
            assert ctl.state == STATE_FAILED_OVER
            assert roles == [ROLE_ACTIVE]
            assert standby.role == ROLE_ACTIVE
            assert standby.store.count() == 1
        finally:
            mon.stop()
            standby.stop()

    def test_failback_when_partner_recovers(self):
        ctl = FailoverController("b", ROLE_STANDBY, auto_failback=True)
        assert ctl.initiate_failover(reason="test")
        assert ctl.role == ROLE_ACTIVE and ctl.state == STATE_FAILED_OVER
        from bng_amd.ha.health_monitor import HealthEvent
        ctl.handle_health_event(HealthEvent("partner_up", "x", 0, 1))
        assert ctl.role == ROLE_STANDBY and ctl.state == STATE_NORMAL
        assert [h["event"] for h in ctl.history] == ["failover", "failback"]

    def test_forced_operations(self):
        ctl = FailoverController("b", ROLE_STANDBY, failover_delay=30)
        assert ctl.force_failover()       # skips the delay
        assert ctl.role == ROLE_ACTIVE
        assert ctl.force_failback()
        assert ctl.role == ROLE_STANDBY

    def test_active_ignores_partner_down(self):
        ctl = FailoverController("a", ROLE_ACTIVE)
        from bng_amd.ha.health_monitor import HealthEvent
        ctl.handle_health_event(HealthEvent("partner_down", "x", 0, 3))
        assert ctl.role == ROLE_ACTIVE and ctl.state == STATE_NORMAL


class TestGPUTableSync:
    def test_lease_replication_and_promotion(self):
        """Active's DHCP leases replicate to the standby; at promotion the
        standby rebuilds leases + fast-path table + QoS from the shadow
        store."""
        from bng_amd.cli.main import BNG, build_parser
        from bng_amd.dhcp import message as dm
        from bng_amd.dataplane.packets import mac_bytes
        from bng_amd.ha import session_glue

        def mk(role, partner=""):
            argv = ["run", "--pool-network", "10.0.1.0/24", "--gpu", "off",
                    "--ha-role", role, "--qos-policy", "gold:100:20",
                    "--qos-default-policy", "gold"]
            if partner:
                argv += ["--ha-partner-url", partner]
            return BNG(build_parser().parse_args(argv)).start()

        active = mk("active")
        standby = mk("standby", partner=active.ha.url)
        try:
            assert wait_for(lambda: standby.ha.connected)
            mac = mac_bytes("aa:bb:cc:00:00:07")
            offer = active.dhcp_server.handle(
                dm.build_request(mac, dm.DISCOVER))
            active.dhcp_server.handle(dm.build_request(
                mac, dm.REQUEST, requested_ip=offer.yiaddr))
            assert wait_for(lambda: standby.ha.store.count() == 1)
            # promotion rebuilds live state from the shadow store
            n = session_glue.promote(standby.dhcp_server, standby.ha,
                                     standby.qos)
            assert n == 1
            assert mac in standby.dhcp_server.leases
            lease = standby.dhcp_server.leases[mac]
            assert lease.ip == offer.yiaddr
            # fast-path table rebuilt on the standby
            assert len(standby.launcher.dp.subscribers) == 1
            # release on (old) active propagates deletes
            active.dhcp_server.handle(dm.build_request(mac, dm.RELEASE))
            assert active.ha.store.count() == 0
        finally:
            standby.stop()
            active.stop()

    def test_launcher_snapshot_roundtrip(self):
        from bng_amd.dataplane.launcher import GoldenLauncher
        a, b = GoldenLauncher(), GoldenLauncher()
        a.add_subscriber(b"\xaa\xbb\xcc\x00\x00\x01", 1, 0x0A000105,
                         2_000_000_000)
        a.add_vlan_subscriber(100, 200, 1, 0x0A000106, 2_000_000_000)
        snap = a.export_subscribers()
        assert len(snap) == 2
        assert b.import_subscribers(snap) == 2
        assert b.export_subscribers() and \
            {e["ip"] for e in b.export_subscribers()} == \
            {0x0A000105, 0x0A000106}


class TestFailoverControllerCoverage:
    """Disabled mode, multi-handler, stats, status (ref
    pkg/ha/failover_test.go)."""

    def test_disabled_ignores_health_events(self):
        from bng_amd.ha.failover import FailoverController, STATE_NORMAL
        from bng_amd.ha.health_monitor import (EVENT_PARTNER_DOWN,
                                               HealthEvent)
        fc = FailoverController("n1", ROLE_STANDBY, enabled=False)
        fc.handle_health_event(HealthEvent(EVENT_PARTNER_DOWN, "n2",
                                           time.time(), 3))
        assert fc.role == ROLE_STANDBY and fc.state == STATE_NORMAL
        # forced still works even when automatic failover is off
        assert fc.force_failover()
        assert fc.role == ROLE_ACTIVE and fc.stats["forced"] == 1

    def test_multiple_handlers_and_stats(self):
        from bng_amd.ha.failover import FailoverController
        got1, got2 = [], []
        fc = FailoverController("n1", ROLE_STANDBY,
                                role_change_callback=got1.append)
        fc.on_role_change(got2.append)
        fc.force_failover()
        fc.force_failback()
        assert got1 == [ROLE_ACTIVE, ROLE_STANDBY]
        assert got2 == [ROLE_ACTIVE, ROLE_STANDBY]
        st = fc.status()
        assert st["failovers"] == 1 and st["failbacks"] == 1
        assert st["at_original_role"]

    def test_failover_when_already_active_refused(self):
        from bng_amd.ha.failover import FailoverController
        fc = FailoverController("n1", ROLE_ACTIVE)
        assert not fc.initiate_failover()
        # failback at original (standby) role refused
        fc2 = FailoverController("n2", ROLE_STANDBY)
        assert not fc2.initiate_failback()


def test_health_monitor_tracks_response_time():
    """ref TestHealthMonitor_ResponseTime: each probe records how long
    the partner took to answer."""
    import http.server
    import threading as th
    from bng_amd.ha.health_monitor import HealthMonitor

    class H(http.server.BaseHTTPRequestHandler):
        def do_GET(self):
            time.sleep(0.02)
            self.send_response(200)
            self.end_headers()

        def log_message(self, *a):
            pass
    srv = http.server.HTTPServer(("127.0.0.1", 0), H)
    t = th.Thread(target=srv.serve_forever, daemon=True)
    t.start()
    try:
        m = HealthMonitor(f"http://127.0.0.1:{srv.server_port}")
        assert m.check_once()
        assert m.last_response_time >= 0.02
        assert m.last_check > 0
    finally:
        srv.shutdown()


class TestNatFlowFailover:
    """Round-1 VERDICT task 3: established NAT flows must survive
    promotion.  Active and standby run full golden dataplanes; the flow
    is created by a real egress packet on the active, replicated via
    the device-log-ring glue, and after promotion the RETURN packet is
    DNAT-translated by the standby's own tables."""

    def _mk_dp(self):
        from bng_amd.dataplane.launcher import GoldenLauncher
        from bng_amd.dataplane.packets import ip2u32
        l = GoldenLauncher()
        l.set_nat_config()
        l.add_subscriber_nat(ip2u32("10.0.0.5"), ip2u32("203.0.113.7"),
                             2048, 3071, subscriber_id=42)
        return l

    def test_established_flow_survives_promotion(self):
        import numpy as np
        from bng_amd.dataplane import abi
        from bng_amd.dataplane.packets import build_ipv4, ip2u32
        from bng_amd.ha.nat_glue import NatHaGlue, promote_nat
        from bng_amd.ha.protocol import ROLE_ACTIVE, ROLE_STANDBY

        active_dp = self._mk_dp()
        standby_dp = self._mk_dp()
        active = HASyncer("a", ROLE_ACTIVE, heartbeat_interval=0.2).start()
        standby = HASyncer("b", ROLE_STANDBY, partner_url=active.url,
                           heartbeat_interval=0.2,
                           full_sync_interval=0.3).start()
        glue = NatHaGlue(active_dp, active, interval=0.05)
        try:
            deadline = time.time() + 5
            while not standby.connected and time.time() < deadline:
                time.sleep(0.05)
            # establish a flow on the active: egress UDP packet
            pkt = bytearray(build_ipv4(
                "aa:00:00:00:00:05", "02:00:00:00:00:01",
                ip2u32("10.0.0.5"), ip2u32("93.184.216.34"),
                proto=17, sport=5555, dport=53, payload=b"x" * 22))
            v = active_dp.dp.nat44_egress(pkt)
            assert v == abi.FWD
            nat_port = int.from_bytes(pkt[34:36], "big")  # rewritten sport
            assert glue.pump_once() == 1                  # delta published
            deadline = time.time() + 5
            while not standby.nat_store and time.time() < deadline:
                time.sleep(0.05)
            assert len(standby.nat_store) == 1
            # failover
            standby.promote()
            n = promote_nat(standby_dp, standby)
            assert n == 1
            # the RETURN packet hits the promoted standby and must DNAT
            # back to the subscriber
            ret = bytearray(build_ipv4(
                "02:00:00:00:00:01", "aa:00:00:00:00:05",
                ip2u32("93.184.216.34"), ip2u32("203.0.113.7"),
                proto=17, sport=53, dport=nat_port, payload=b"y" * 22))
            v2 = standby_dp.dp.nat44_ingress(ret)
            assert v2 == abi.FWD
            assert int.from_bytes(ret[30:34], "big") == ip2u32("10.0.0.5")
            assert int.from_bytes(ret[36:38], "big") == 5555
            # and the subscriber's NEXT egress packet keeps the same
            # translation (EIM restored)
            pkt2 = bytearray(build_ipv4(
                "aa:00:00:00:00:05", "02:00:00:00:00:01",
                ip2u32("10.0.0.5"), ip2u32("198.51.100.9"),
                proto=17, sport=5555, dport=443, payload=b"z" * 22))
            v3 = standby_dp.dp.nat44_egress(pkt2)
            assert v3 == abi.FWD
            assert int.from_bytes(pkt2[34:36], "big") == nat_port
        finally:
            glue.stop()
            active.stop()
            standby.stop()

    def test_full_refresh_reconciles_expiry(self):
        """Sessions the sweep expired on the active disappear from the
        replicated set after a full refresh (deletes are not logged;
        the refresh IS the reconciliation)."""
        from bng_amd.dataplane import abi
        from bng_amd.dataplane.packets import build_ipv4, ip2u32
        from bng_amd.ha.nat_glue import NatHaGlue
        from bng_amd.ha.protocol import ROLE_ACTIVE

        dp = self._mk_dp()
        syncer = HASyncer("a", ROLE_ACTIVE).start()
        glue = NatHaGlue(dp, syncer, interval=99, full_refresh=99)
        try:
            pkt = bytearray(build_ipv4(
                "aa:00:00:00:00:05", "02:00:00:00:00:01",
                ip2u32("10.0.0.5"), ip2u32("93.184.216.34"),
                proto=17, sport=7777, dport=53, payload=b"x" * 22))
            assert dp.dp.nat44_egress(pkt) == abi.FWD
            glue.pump_once()
            assert len(syncer.nat_store) == 1
            # expire it (UDP timeout is 120s)
            dp.sweep_nat(now_ns=dp.dp.now_ns + 10**12)
            assert glue.full_refresh_once() == 0
            assert len(syncer.nat_store) == 0
        finally:
            glue.stop()
            syncer.stop()


class TestHAListenHostAuth:
    """Round-1 advisor (medium): non-loopback HA binds need auth."""

    def test_nonloopback_requires_token(self):
        from bng_amd.ha.protocol import ROLE_ACTIVE
        with pytest.raises(ValueError):
            HASyncer("a", ROLE_ACTIVE, listen_host="0.0.0.0")
        # explicit opt-out still allowed
        s = HASyncer("a", ROLE_ACTIVE, listen_host="0.0.0.0",
                     allow_insecure=True)
        assert s.listen_host == "0.0.0.0"

    def test_token_enforced_on_sync_endpoints(self):
        import requests
        from bng_amd.ha.protocol import ROLE_ACTIVE
        active = HASyncer("a", ROLE_ACTIVE, auth_token="s3cret").start()
        try:
            r = requests.get(f"{active.url}/sync/full", timeout=2)
            assert r.status_code == 401
            assert active.stats["auth_rejects"] == 1
            r = requests.get(f"{active.url}/sync/full", timeout=2,
                             headers={"X-BNG-HA-Token": "s3cret"})
            assert r.status_code == 200
            # /health stays open for monitors
            assert requests.get(f"{active.url}/health",
                                timeout=2).status_code == 200
        finally:
            active.stop()

    def test_authed_pair_replicates(self):
        from bng_amd.ha.protocol import ROLE_ACTIVE, ROLE_STANDBY, \
            SessionState
        active = HASyncer("a", ROLE_ACTIVE, auth_token="tok",
                          listen_host="0.0.0.0",
                          heartbeat_interval=0.2).start()
        standby = HASyncer(
            "b", ROLE_STANDBY,
            partner_url=f"http://127.0.0.1:{active._listen_port}",
            auth_token="tok", heartbeat_interval=0.2).start()
        try:
            deadline = time.time() + 5
            while not standby.connected and time.time() < deadline:
                time.sleep(0.05)
            active.publish_add(SessionState(session_id="s1", mac="aa"))
            deadline = time.time() + 5
            while standby.store.count() == 0 and time.time() < deadline:
                time.sleep(0.05)
            assert standby.store.get("s1") is not None
        finally:
            active.stop()
            standby.stop()


def test_failover_canceled_when_partner_recovers_in_grace():
    """The grace delay re-checks partner health; a recovery during it
    cancels the failover (ref failover_test.go CanceledFailover)."""
    from bng_amd.ha.failover import (FailoverController, ROLE_STANDBY,
                                     STATE_NORMAL)

    class Mon:
        partner_healthy = False

        def on_event(self, cb):
            pass

    mon = Mon()
    c = FailoverController("node-b", ROLE_STANDBY, monitor=mon,
                           failover_delay=0.05)
    # partner comes back while we wait out the grace period
    import threading
    threading.Timer(0.01, lambda: setattr(mon, "partner_healthy",
                                          True)).start()
    assert c.initiate_failover("partner down") is False
    assert c.role == ROLE_STANDBY
    assert c.state == STATE_NORMAL
    assert c.stats["canceled"] == 1
    # forced failover skips the grace re-check entirely
    mon.partner_healthy = True
    assert c.initiate_failover("operator", forced=True) is True
    assert c.role == "active"
