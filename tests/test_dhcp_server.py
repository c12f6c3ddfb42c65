"""DHCP slow-path server tests — handlers exercised directly with
in-memory pools and launcher=None / GoldenLauncher, like the reference's
server_coverage_test.go:69-458."""
import struct
import time

import pytest

from bng_amd.dataplane.launcher import GoldenLauncher
from bng_amd.dataplane.packets import ip2u32, mac_bytes, u32_to_ip
from bng_amd.dhcp import message as dm
from bng_amd.dhcp.pool import Pool, PoolConfig, PoolManager
from bng_amd.dhcp.server import DHCPServer
from bng_amd.nat.manager import Manager as NATManager
from bng_amd.qos.manager import Manager as QoSManager
from bng_amd.radius.policy import Policy, PolicyManager
from bng_amd.radius.server import RadiusServer
from bng_amd.radius.client import Client as RadiusClient
from bng_amd.walledgarden.manager import Manager as WGManager

MAC = mac_bytes("aa:bb:cc:00:00:01")


def make_server(launcher=None):
    pm = PoolManager(launcher)
    pm.add_pool(PoolConfig(1, "10.0.1.0/24", gateway="10.0.1.1",
                           dns=["8.8.8.8"], lease_time=600))
    srv = DHCPServer(pm, "10.0.0.1")
    if launcher is not None:
        srv.set_launcher(launcher)
    return srv


class TestDORA:
    def test_discover_offer_request_ack(self):
        srv = make_server()
        offer = srv.handle(dm.build_request(MAC, dm.DISCOVER, xid=0xAA))
        assert offer is not None and offer.msg_type == dm.OFFER
        assert offer.xid == 0xAA
        ip = offer.yiaddr
        assert u32_to_ip(ip).startswith("10.0.1.")
        assert struct.unpack(">I", offer.get_option(dm.OPT_LEASE_TIME))[0] == 600
        ack = srv.handle(dm.build_request(MAC, dm.REQUEST, xid=0xAB,
                                          requested_ip=ip))
        assert ack.msg_type == dm.ACK and ack.yiaddr == ip
        # renewal keeps the same IP
        ack2 = srv.handle(dm.build_request(MAC, dm.REQUEST, requested_ip=ip))
        assert ack2.msg_type == dm.ACK and ack2.yiaddr == ip

    def test_request_wrong_ip_naks(self):
        srv = make_server()
        offer = srv.handle(dm.build_request(MAC, dm.DISCOVER))
        bad = srv.handle(dm.build_request(MAC, dm.REQUEST,
                                          requested_ip=offer.yiaddr + 7))
        assert bad.msg_type == dm.NAK

    def test_release_returns_ip_to_pool(self):
        srv = make_server()
        offer = srv.handle(dm.build_request(MAC, dm.DISCOVER))
        srv.handle(dm.build_request(MAC, dm.REQUEST,
                                    requested_ip=offer.yiaddr))
        srv.handle(dm.build_request(MAC, dm.RELEASE, ciaddr=offer.yiaddr))
        assert MAC not in srv.leases
        # another client can get the same IP back
        mac2 = mac_bytes("aa:bb:cc:00:00:02")
        offer2 = srv.handle(dm.build_request(mac2, dm.DISCOVER))
        assert offer2 is not None

    def test_decline_blacklists_ip(self):
        srv = make_server()
        offer = srv.handle(dm.build_request(MAC, dm.DISCOVER))
        ip = offer.yiaddr
        srv.handle(dm.build_request(MAC, dm.DECLINE, requested_ip=ip))
        pool = srv.pools.get_pool(1)
        assert ip in pool.unavailable
        offer2 = srv.handle(dm.build_request(MAC, dm.DISCOVER))
        assert offer2.yiaddr != ip

    def test_inform_returns_config_only(self):
        srv = make_server()
        resp = srv.handle(dm.build_request(MAC, dm.INFORM,
                                           ciaddr=ip2u32("10.0.1.55")))
        assert resp.msg_type == dm.ACK and resp.yiaddr == 0
        assert resp.get_option(dm.OPT_ROUTER) is not None

    def test_lease_expiry_sweeper(self):
        srv = make_server()
        srv.lease_time = 1
        offer = srv.handle(dm.build_request(MAC, dm.DISCOVER))
        srv.handle(dm.build_request(MAC, dm.REQUEST,
                                    requested_ip=offer.yiaddr))
        assert srv.sweep_expired(now=time.time() + 5) == 1
        assert MAC not in srv.leases

    def test_circuit_id_secondary_index(self):
        srv = make_server()
        cid = b"olt1/1/1"
        offer = srv.handle(dm.build_request(MAC, dm.DISCOVER,
                                            circuit_id=cid))
        assert offer is not None
        assert srv.leases_by_circuit[cid].ip == offer.yiaddr


class TestFastPathIntegration:
    def test_ack_populates_fastpath_and_release_clears(self):
        """ref updateFastPathCache server.go:1057: after ACK the GPU path
        answers the next DISCOVER without the slow path."""
        launcher = GoldenLauncher()
        srv = make_server(launcher)
        srv.start()
        srv.stop()
        offer = srv.handle(dm.build_request(MAC, dm.DISCOVER))
        srv.handle(dm.build_request(MAC, dm.REQUEST,
                                    requested_ip=offer.yiaddr))
        # fast path now answers directly
        from bng_amd.dataplane.packets import build_dhcp_request
        frame = bytearray(build_dhcp_request(MAC, 1))
        launcher.dp.now_ns = time.time_ns()
        v, _ = launcher.dp.dhcp_fastpath(frame)
        assert v == 1   # TX
        srv.handle(dm.build_request(MAC, dm.RELEASE))
        frame = bytearray(build_dhcp_request(MAC, 1))
        v, _ = launcher.dp.dhcp_fastpath(frame)
        assert v == 0   # PASS again

    def test_full_provisioning_chain(self):
        """RADIUS auth -> lease -> fast path -> QoS policy -> NAT block
        (ref server.go:595-834)."""
        rsrv = RadiusServer(b"sec", users={
            "aa:bb:cc:00:00:01": {"password": "aa:bb:cc:00:00:01",
                                  "policy": "gold"}}).start()
        try:
            launcher = GoldenLauncher()
            srv = make_server(launcher)
            srv.set_radius(RadiusClient([rsrv.addr], b"sec"),
                           auth_mode="mac")
            pm = PolicyManager()
            pm.add_policy(Policy("gold", 10**8, 10**7))
            qos = QoSManager(launcher, pm)
            srv.set_qos_manager(qos)
            nat = NATManager(launcher)
            nat.add_public_ip("203.0.113.1")
            srv.set_nat_manager(nat)

            offer = srv.handle(dm.build_request(MAC, dm.DISCOVER))
            assert offer is not None
            ack = srv.handle(dm.build_request(MAC, dm.REQUEST,
                                              requested_ip=offer.yiaddr))
            assert ack.msg_type == dm.ACK
            ip = ack.yiaddr
            # QoS buckets installed both directions
            assert ip in launcher.dp.qos_egress
            assert ip in launcher.dp.qos_ingress
            assert launcher.dp.qos_egress[ip].rate_bps == 10**8
            # NAT port block installed
            assert ip in launcher.dp.subnat
            blk = launcher.dp.subnat[ip]
            assert blk.port_end - blk.port_start + 1 == 1024
            # lease carries the RADIUS policy
            assert srv.leases[MAC].policy_name == "gold"
        finally:
            rsrv.stop()

    def test_radius_reject_no_lease(self):
        rsrv = RadiusServer(b"sec", users={}).start()
        try:
            srv = make_server()
            srv.set_radius(RadiusClient([rsrv.addr], b"sec"),
                           auth_mode="mac")
            offer = srv.handle(dm.build_request(MAC, dm.DISCOVER))
            assert offer is None
            assert srv.stats["auth_reject"] == 1
            nak = srv.handle(dm.build_request(MAC, dm.REQUEST,
                                              requested_ip=ip2u32("10.0.1.5")))
            assert nak.msg_type == dm.NAK
        finally:
            rsrv.stop()

    def test_nexus_no_allocation_walled_garden(self):
        """Unknown-to-Nexus subscriber lands in the walled garden
        (ref ErrNoAllocation signal, nexus/http_allocator.go:225)."""
        from bng_amd.nexus.http_allocator import (HTTPAllocator,
                                                  NexusAllocatorServer)
        nx = NexusAllocatorServer().start()
        try:
            srv = make_server()
            wg = WGManager(portal_ip="10.0.0.10")
            srv.set_nexus(allocator=HTTPAllocator(nx.url))
            srv.set_walled_garden(wg)
            offer = srv.handle(dm.build_request(MAC, dm.DISCOVER))
            assert offer is not None      # still gets an IP, quarantined
            assert wg.is_quarantined("aa:bb:cc:00:00:01")
            assert srv.leases[MAC].walled_garden
        finally:
            nx.stop()

    def test_nexus_allocation_used(self):
        from bng_amd.nexus.http_allocator import (HTTPAllocator,
                                                  NexusAllocatorServer)
        nx = NexusAllocatorServer().start()
        try:
            al = HTTPAllocator(nx.url)
            al.create_pool("p1", "10.0.1.0/24")
            want = al.allocate_ipv4("p1", "aa:bb:cc:00:00:01")
            srv = make_server()
            srv.set_nexus(allocator=al)
            offer = srv.handle(dm.build_request(MAC, dm.DISCOVER))
            assert u32_to_ip(offer.yiaddr) == want
        finally:
            nx.stop()


class TestFuzz:
    def test_malformed_messages_no_crash(self):
        """Analog of FuzzDHCPPacketParsing (pkg/dhcp/fuzz_test.go)."""
        import random
        rng = random.Random(5)
        srv = make_server()
        base = dm.build_request(MAC, dm.DISCOVER).encode()
        for _ in range(200):
            data = bytearray(base)
            if rng.random() < 0.5 and len(data) > 4:
                data = data[:rng.randrange(4, len(data))]
            for _ in range(rng.randrange(5)):
                if data:
                    data[rng.randrange(len(data))] = rng.randrange(256)
            try:
                msg = dm.DHCPMessage.decode(bytes(data))
            except (ValueError, struct.error):
                continue
            srv.handle(msg)   # must not raise


class TestServerCoverage:
    """Edge behaviors mirrored from the reference's coverage list
    (ref pkg/dhcp/server_coverage_test.go)."""

    def test_discover_reuses_existing_lease(self):
        srv = make_server()
        offer1 = srv.handle(dm.build_request(MAC, dm.DISCOVER))
        srv.handle(dm.build_request(MAC, dm.REQUEST,
                                    requested_ip=offer1.yiaddr))
        offer2 = srv.handle(dm.build_request(MAC, dm.DISCOVER))
        assert offer2.yiaddr == offer1.yiaddr

    def test_request_uses_ciaddr_when_no_option50(self):
        srv = make_server()
        offer = srv.handle(dm.build_request(MAC, dm.DISCOVER))
        req = dm.build_request(MAC, dm.REQUEST)
        req.ciaddr = offer.yiaddr          # RENEWING state: ciaddr only
        ack = srv.handle(req)
        assert ack.msg_type == dm.ACK and ack.yiaddr == offer.yiaddr
        # mismatched ciaddr -> NAK
        req2 = dm.build_request(MAC, dm.REQUEST)
        req2.ciaddr = offer.yiaddr + 1
        assert srv.handle(req2).msg_type == dm.NAK

    def test_release_nonexistent_is_noop(self):
        srv = make_server()
        assert srv.handle(dm.build_request(MAC, dm.RELEASE)) is None
        assert srv.stats["release"] == 1

    def test_decline_without_requested_ip_is_noop(self):
        srv = make_server()
        assert srv.handle(dm.build_request(MAC, dm.DECLINE)) is None

    def test_unknown_message_type_ignored(self):
        srv = make_server()
        req = dm.build_request(MAC, dm.DISCOVER)
        req.set_option(dm.OPT_MSG_TYPE, bytes([99]))
        assert srv.handle(req) is None
        # BOOTREPLY op ignored entirely
        req2 = dm.build_request(MAC, dm.DISCOVER)
        req2.op = 2
        assert srv.handle(req2) is None

    def test_concurrent_discover_request(self):
        """Thread-hammered DORA: every client ends with a unique IP
        (ref TestConcurrentDiscoverRequests)."""
        import threading
        srv = make_server()
        ips, errs = {}, []

        def worker(k):
            try:
                mac = mac_bytes(f"aa:bb:cc:00:01:{k:02x}")
                offer = srv.handle(dm.build_request(mac, dm.DISCOVER))
                ack = srv.handle(dm.build_request(
                    mac, dm.REQUEST, requested_ip=offer.yiaddr))
                assert ack.msg_type == dm.ACK
                ips[k] = ack.yiaddr
            except Exception as e:        # pragma: no cover
                errs.append(e)
        ts = [threading.Thread(target=worker, args=(k,))
              for k in range(32)]
        [t.start() for t in ts]
        [t.join() for t in ts]
        assert not errs
        assert len(set(ips.values())) == 32


class TestWalledGardenManager:
    """MAC listing + stats (ref pkg/walledgarden/manager_test.go)."""

    def test_list_macs_and_stats(self):
        from bng_amd.walledgarden.manager import Manager as WG
        m = WG(portal_ip="10.0.0.1")
        m.add("aa:01", "10.0.1.5")
        m.add("aa:02", "10.0.1.6")
        m.activate("aa:02")
        m.add("aa:03", "10.0.1.7")
        m.block("aa:03", reason="fraud")
        assert m.list_macs() == ["aa:01", "aa:02", "aa:03"]
        assert m.list_macs("walledgarden") == ["aa:01"]
        st = m.get_stats()
        assert st["tracked"] == 3 and st["state_walledgarden"] == 1
        assert st["added"] == 3 and st["blocked"] == 1
        m.remove("aa:03")
        assert m.get_stats()["tracked"] == 2


class TestReplyAddressing:
    """RFC 2131 §4.1 reply-addressing rules (round-1 advisor: the serve
    loop broadcast every non-relayed reply and ignored ciaddr and the
    BROADCAST flag)."""

    def _req(self, ciaddr=0, flags=0, giaddr=0):
        m = dm.DHCPMessage()
        m.op = 1
        m.ciaddr = ciaddr
        m.flags = flags
        m.giaddr = giaddr
        return m

    def _resp(self, yiaddr):
        m = dm.DHCPMessage()
        m.yiaddr = yiaddr
        return m

    def test_relay_wins(self):
        from bng_amd.dhcp.server import DHCPServer
        d = DHCPServer.reply_dest(
            self._req(ciaddr=ip2u32("10.0.1.5"), giaddr=ip2u32("10.9.9.9")),
            self._resp(ip2u32("10.0.1.5")))
        assert d == ("10.9.9.9", 67)

    def test_renewing_client_unicast_to_ciaddr(self):
        from bng_amd.dhcp.server import DHCPServer
        d = DHCPServer.reply_dest(self._req(ciaddr=ip2u32("10.0.1.5")),
                                  self._resp(ip2u32("10.0.1.5")))
        assert d == ("10.0.1.5", 68)

    def test_broadcast_flag_honored(self):
        from bng_amd.dhcp.server import DHCPServer
        d = DHCPServer.reply_dest(self._req(flags=0x8000),
                                  self._resp(ip2u32("10.0.1.6")))
        assert d == ("255.255.255.255", 68)

    def test_default_unicast_to_yiaddr(self):
        from bng_amd.dhcp.server import DHCPServer
        d = DHCPServer.reply_dest(self._req(),
                                  self._resp(ip2u32("10.0.1.7")))
        assert d == ("10.0.1.7", 68)

    def test_nak_broadcasts(self):
        from bng_amd.dhcp.server import DHCPServer
        d = DHCPServer.reply_dest(self._req(), self._resp(0))
        assert d == ("255.255.255.255", 68)


class TestRequestEdgeBehaviors:
    """Reference coverage-test scenarios (server_coverage_test.go):
    ciaddr-only renewals and option-82 preservation."""

    def test_request_without_requested_ip_uses_ciaddr(self):
        srv = make_server()
        mac = mac_bytes("aa:bb:cc:00:00:14")
        offer = srv.handle(dm.build_request(mac, dm.DISCOVER, xid=1))
        ip = offer.yiaddr
        # RENEWING state: no option 50, ciaddr carries the address
        ack = srv.handle(dm.build_request(mac, dm.REQUEST, xid=2,
                                          ciaddr=ip))
        assert ack is not None and ack.msg_type == dm.ACK
        assert ack.yiaddr == ip
        # a ciaddr that disagrees with the lease still NAKs
        nak = srv.handle(dm.build_request(mac, dm.REQUEST, xid=3,
                                          ciaddr=ip + 1))
        assert nak.msg_type == dm.NAK

    def test_renewal_without_option82_preserves_it(self):
        srv = make_server()
        mac = mac_bytes("aa:bb:cc:00:00:16")
        srv.handle(dm.build_request(mac, dm.DISCOVER, xid=1,
                                    circuit_id=b"olt1/pon0/3"))
        ack = srv.handle(dm.build_request(
            mac, dm.REQUEST, xid=2, circuit_id=b"olt1/pon0/3"))
        assert ack.msg_type == dm.ACK
        lease = srv.leases[mac]
        assert lease.circuit_id == b"olt1/pon0/3"
        # renewal WITHOUT option 82 must not clobber the recorded id
        ack2 = srv.handle(dm.build_request(mac, dm.REQUEST, xid=3,
                                           ciaddr=lease.ip))
        assert ack2.msg_type == dm.ACK
        assert srv.leases[mac].circuit_id == b"olt1/pon0/3"
        assert srv.leases_by_circuit[b"olt1/pon0/3"] is srv.leases[mac]
