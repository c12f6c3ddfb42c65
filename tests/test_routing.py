"""Routing tests: BGP/BFD over a fake FRR executor, policy routing over
the memory platform, subscriber-route retry/reconcile, health hysteresis
(ref pkg/routing bgp_test.go / bfd_test.go patterns)."""
import os

import pytest

from bng_amd.routing.bgp import BFDManager, BGPController
from bng_amd.routing.frr import FakeExecutor, FRRError
from bng_amd.routing.manager import (HealthChecker, Manager, MemoryPlatform,
                                     Route, SubscriberRouteManager)


class TestBGP:
    def test_start_and_neighbors(self):
        exe = FakeExecutor()
        bgp = BGPController(exe, 65001, router_id="10.0.0.1",
                            ecmp_paths=4).start()
        bgp.add_neighbor("192.0.2.1", 65002, description="upstream",
                         bfd=True, route_map_out="EXPORT")
        cmds = exe.all_commands()
        assert "router bgp 65001" in cmds
        assert "bgp router-id 10.0.0.1" in cmds
        assert "maximum-paths 4" in cmds
        assert "neighbor 192.0.2.1 remote-as 65002" in cmds
        assert "neighbor 192.0.2.1 bfd" in cmds
        assert "neighbor 192.0.2.1 route-map EXPORT out" in cmds
        bgp.remove_neighbor("192.0.2.1")
        assert "no neighbor 192.0.2.1" in exe.all_commands()
        assert bgp.neighbors == {}

    def test_announce_withdraw(self):
        exe = FakeExecutor()
        bgp = BGPController(exe, 65001)
        bgp.announce_prefix("203.0.113.0/24")
        assert bgp.announced_prefixes() == ["203.0.113.0/24"]
        bgp.withdraw_prefix("203.0.113.0/24")
        assert bgp.announced_prefixes() == []
        cmds = exe.all_commands()
        assert "network 203.0.113.0/24" in cmds
        assert "no network 203.0.113.0/24" in cmds

    def test_bfd_peers_and_events(self):
        exe = FakeExecutor()
        bfd = BFDManager(exe)
        events = []
        bfd.on_state_change(lambda a, up: events.append((a, up)))
        bfd.add_peer("192.0.2.1", interval_ms=50, multiplier=3)
        cmds = exe.all_commands()
        assert "peer 192.0.2.1" in cmds
        assert "receive-interval 50" in cmds
        bfd.handle_state_change("192.0.2.1", True)
        bfd.handle_state_change("192.0.2.1", True)   # dedup
        bfd.handle_state_change("192.0.2.1", False)
        assert events == [("192.0.2.1", True), ("192.0.2.1", False)]


class TestPolicyRouting:
    def test_isp_tables_and_rules(self):
        plat = MemoryPlatform()
        m = Manager(plat)
        t1 = m.create_isp_table("isp-a", "192.0.2.1")
        t2 = m.create_isp_table("isp-b", "192.0.2.9")
        assert t1 != t2
        assert m.create_isp_table("isp-a", "x") == t1    # idempotent
        assert plat.routes(t1)[0].next_hop == "192.0.2.1"
        m.add_subscriber_rule("10.0.1.50", "isp-a")
        m.add_subscriber_rule("10.0.1.51", "isp-b")
        rules = {(r.src, r.table) for r in plat.rules()}
        assert ("10.0.1.50", t1) in rules and ("10.0.1.51", t2) in rules
        m.remove_subscriber_rule("10.0.1.50", "isp-a")
        assert ("10.0.1.50", t1) not in {(r.src, r.table)
                                         for r in plat.rules()}
        with pytest.raises(KeyError):
            m.add_subscriber_rule("10.0.1.52", "isp-zzz")


class TestSubscriberRoutes:
    def test_install_and_withdraw(self):
        bgp = BGPController(FakeExecutor(), 65001)
        srm = SubscriberRouteManager(bgp)
        srm.add_subscriber_route("10.0.1.50")
        assert "10.0.1.50/32" in bgp.announced_prefixes()
        srm.remove_subscriber_route("10.0.1.50")
        assert bgp.announced_prefixes() == []

    def test_retry_queue_on_frr_failure(self):
        exe = FakeExecutor(fail=True)
        bgp = BGPController(exe, 65001)
        srm = SubscriberRouteManager(bgp)
        srm.add_subscriber_route("10.0.1.50")
        assert srm.retry_queue == {"10.0.1.50/32": 1}
        exe.fail = False                      # FRR recovers
        assert srm.retry_pending() == 1
        assert "10.0.1.50/32" in srm.installed
        assert srm.retry_queue == {}

    def test_reconcile_repairs_drift(self):
        bgp = BGPController(FakeExecutor(), 65001)
        srm = SubscriberRouteManager(bgp)
        srm.add_subscriber_route("10.0.1.50")
        # simulated drift: FRR lost the route
        bgp.announced.clear()
        srm.installed.clear()
        assert srm.reconcile() == 1
        assert "10.0.1.50/32" in bgp.announced_prefixes()
        # stale: installed but no longer desired
        srm.desired.clear()
        assert srm.reconcile() == 1
        assert bgp.announced_prefixes() == []


class TestHealthChecker:
    def test_hysteresis(self):
        state = {"ok": True}
        events = []
        hc = HealthChecker("x", probe=lambda: state["ok"],
                           up_threshold=2, down_threshold=3)
        hc.on_change(events.append)
        for _ in range(3):
            hc.check_once()
        assert hc.healthy
        state["ok"] = False
        hc.check_once(); hc.check_once()
        assert hc.healthy                 # 2 < down_threshold
        hc.check_once()
        assert not hc.healthy
        state["ok"] = True
        hc.check_once()
        assert not hc.healthy             # 1 < up_threshold
        hc.check_once()
        assert hc.healthy
        assert events == [False, True]


class TestSessionRouteIntegration:
    """Session lifecycle -> route injection bridge (ref
    session_integration_test.go)."""

    def make(self, **cfg):
        from bng_amd.routing.manager import (SessionRouteConfig,
                                             SessionRouteIntegration)
        bgp = BGPController(FakeExecutor(), 65001)
        rm = SubscriberRouteManager(bgp)
        integ = SessionRouteIntegration(
            rm, SessionRouteConfig(**cfg) if cfg else None)
        return bgp, rm, integ

    def test_activate_injects_and_terminate_withdraws(self):
        _, rm, integ = self.make()
        assert integ.on_session_activate("s1", "sub-1", "100.64.0.5")
        assert "100.64.0.5/32" in rm.installed
        ts = integ.tracked_sessions()
        assert len(ts) == 1 and ts[0].route_injected
        # duplicate activation is a no-op
        assert not integ.on_session_activate("s1", "sub-1", "100.64.0.5")
        assert integ.on_session_terminate("s1", reason="radius-disconnect")
        assert "100.64.0.5/32" not in rm.installed
        assert integ.tracked_sessions() == []
        # unknown session: no-op
        assert not integ.on_session_terminate("zzz")

    def test_disabled_injection_and_withdrawal(self):
        _, rm, integ = self.make(enable_injection=False)
        assert not integ.on_session_activate("s1", "sub-1", "100.64.0.5")
        assert rm.installed == set()
        _, rm2, integ2 = self.make(enable_withdrawal=False)
        integ2.on_session_activate("s2", "sub-2", "100.64.0.6")
        assert not integ2.on_session_terminate("s2")
        assert "100.64.0.6/32" in rm2.installed

    def test_state_change_routing(self):
        _, rm, integ = self.make()
        assert integ.on_session_state_change(
            "s1", "sub-1", "auth", "active", ipv4="100.64.0.7")
        # intermediate state: no change
        assert not integ.on_session_state_change(
            "s1", "sub-1", "active", "rekey", ipv4="100.64.0.7")
        assert "100.64.0.7/32" in rm.installed
        assert integ.on_session_state_change(
            "s1", "sub-1", "active", "timeout", reason="idle")
        assert "100.64.0.7/32" not in rm.installed

    def test_recover_routes_after_frr_restart(self):
        bgp, rm, integ = self.make()
        for k in range(3):
            integ.on_session_activate(f"s{k}", f"sub-{k}", f"100.64.1.{k}")
        # FRR restart wipes announcements; recover re-injects all
        rm.installed.clear()
        assert integ.recover_routes() == 3
        assert len(rm.installed) == 3


class TestRoutingMetrics:
    """bng_routing_* instrument surface (ref routing/metrics_test.go)."""

    def test_instruments_and_collect(self):
        from prometheus_client import generate_latest
        from bng_amd.routing.bgp import BFDManager
        from bng_amd.routing.metrics import RoutingMetrics
        m = RoutingMetrics()
        bgp = BGPController(FakeExecutor(), 65001)
        bgp.start()
        bgp.add_neighbor("192.0.2.1", 65002)
        bgp.neighbors["192.0.2.1"].established = True
        bgp.add_neighbor("192.0.2.2", 65002)
        bgp.announce_prefix("100.64.0.0/10")
        rm = SubscriberRouteManager(bgp)
        rm.add_subscriber_route("100.64.0.9")
        bfd = BFDManager(bgp.exe)
        bfd.add_peer("192.0.2.1")
        bfd.handle_state_change("192.0.2.1", True)
        m.record_route_injection(0.002)
        m.record_route_injection(ok=False)
        m.record_route_withdrawal(0.001)
        m.bgp_state_changes.labels("192.0.2.1", "Established").inc()
        m.bfd_state_changes.labels("192.0.2.1", "up").inc()
        m.collect(bgp=bgp, route_manager=rm, bfd=bfd)
        text = generate_latest(m.registry).decode()
        assert "bng_routing_subscriber_routes_active 1.0" in text
        assert "bng_routing_bgp_neighbors_total 2.0" in text
        assert "bng_routing_bgp_neighbors_established 1.0" in text
        assert "bng_routing_bgp_prefixes_announced 2.0" in text  # /10 + /32
        assert "bng_routing_bfd_peers_up 1.0" in text
        assert "bng_routing_subscriber_routes_injected_total 1.0" in text
        assert "bng_routing_route_injection_errors_total 1.0" in text


@pytest.mark.skipif(os.geteuid() != 0, reason="netlink needs root")
class TestNetlinkPlatform:
    """Raw-rtnetlink platform against the live kernel (ref
    netlink_linux.go:20-235); runs where CAP_NET_ADMIN exists (build
    container), skips on the GPU pool."""

    IFACE0, IFACE1 = "bngr0", "bngr1"
    TABLE = 177

    @pytest.fixture()
    def dev(self):
        from bng_amd.dataplane import afxdp
        try:
            try:
                afxdp.link_del(self.IFACE0)
            except OSError:
                pass
            afxdp.veth_create(self.IFACE0, self.IFACE1)
            afxdp.link_up(self.IFACE0)
            afxdp.link_up(self.IFACE1)
        except OSError as e:
            pytest.skip(f"no CAP_NET_ADMIN: {e}")
        yield self.IFACE0
        from bng_amd.routing.netlink import NetlinkPlatform
        try:
            NetlinkPlatform().flush_table(self.TABLE)
        except OSError:
            pass
        try:
            afxdp.link_del(self.IFACE0)
        except OSError:
            pass

    def test_route_add_dump_delete(self, dev):
        from bng_amd.routing.manager import Route
        from bng_amd.routing.netlink import NetlinkPlatform
        p = NetlinkPlatform()
        r = Route(prefix="203.0.113.0/24", table=self.TABLE,
                  device=dev, metric=50)
        p.add_route(r)
        got = p.routes(self.TABLE)
        assert any(x.prefix == "203.0.113.0/24" and x.device == dev and
                   x.metric == 50 for x in got), got
        # main table not polluted
        assert not any(x.prefix == "203.0.113.0/24"
                       for x in p.routes(254))
        p.del_route(r)
        assert not any(x.prefix == "203.0.113.0/24"
                       for x in p.routes(self.TABLE))

    def test_flush_table(self, dev):
        from bng_amd.routing.manager import Route
        from bng_amd.routing.netlink import NetlinkPlatform
        p = NetlinkPlatform()
        for i in range(3):
            p.add_route(Route(prefix=f"198.51.{100 + i}.0/24",
                              table=self.TABLE, device=dev))
        assert p.flush_table(self.TABLE) == 3
        assert p.routes(self.TABLE) == []

    def test_policy_rules(self, dev):
        from bng_amd.routing.manager import Rule
        from bng_amd.routing.netlink import NetlinkPlatform
        p = NetlinkPlatform()
        r = Rule(src="10.77.0.0/16", table=self.TABLE, priority=1177)
        p.add_rule(r)
        try:
            got = p.rules()
            assert any(x.src == "10.77.0.0/16" and
                       x.table == self.TABLE and x.priority == 1177
                       for x in got), got
        finally:
            p.del_rule(r)
        assert not any(x.priority == 1177 for x in p.rules())

    def test_manager_on_live_kernel(self, dev):
        """The per-ISP table manager drives real kernel state."""
        from bng_amd.routing.manager import Manager, Route
        from bng_amd.routing.netlink import NetlinkPlatform
        p = NetlinkPlatform()
        m = Manager(platform=p)
        table = m.create_isp_table("isp-x", default_next_hop="",
                                   device=dev)
        try:
            p.add_route(Route(prefix="192.0.2.0/24", table=table,
                              device=dev))
            m.add_subscriber_rule("10.88.0.5", "isp-x")
            assert any(x.table == table for x in p.rules())
            m.remove_subscriber_rule("10.88.0.5", "isp-x")
            assert not any(x.src.startswith("10.88.0.5")
                           for x in p.rules())
        finally:
            p.flush_table(table)
            m.remove_isp_table("isp-x")


class TestBGPAnnouncementOptions:
    """Path attributes, ECMP, neighbor BFD (ref bgp.go:328-470)."""

    def test_announce_with_options_emits_route_map(self):
        exe = FakeExecutor()
        b = BGPController(exe, 65001).start()
        ann = b.announce_prefix_with_options(
            "203.0.113.0/24", community="65001:100",
            local_pref=200, med=50)
        joined = "\n".join(c for b_ in exe.batches for c in b_)
        assert "set community 65001:100" in joined
        assert "set local-preference 200" in joined
        assert "set metric 50" in joined
        assert "network 203.0.113.0/24 route-map" in joined
        assert b.announced["203.0.113.0/24"]["local_pref"] == 200
        # plain announcement: no route-map emitted
        exe2 = FakeExecutor()
        b2 = BGPController(exe2, 65001).start()
        b2.announce_prefix_with_options("198.51.100.0/24")
        j2 = "\n".join(c for b_ in exe2.batches for c in b_)
        assert "route-map" not in j2.split("network")[-1]

    def test_max_paths_and_neighbor_bfd(self):
        exe = FakeExecutor()
        b = BGPController(exe, 65001).start()
        b.enable_max_paths(8)
        assert any("maximum-paths 8" in c
                   for b_ in exe.batches for c in b_)
        with pytest.raises(ValueError):
            b.enable_max_paths(0)
        b.add_neighbor("192.0.2.1", 65002)
        b.configure_bfd_for_neighbor("192.0.2.1")
        assert any("neighbor 192.0.2.1 bfd" in c
                   for b_ in exe.batches for c in b_)
