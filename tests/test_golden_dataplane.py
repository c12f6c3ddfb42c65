"""Unit tests for the CPU golden-model dataplane (the oracle the HIP kernels
are differential-tested against).  Mirrors the behaviors asserted by the
reference's BPF unit/integration tests."""
import struct

import pytest

from bng_amd.dataplane import abi
from bng_amd.dataplane.golden import (GoldenDataplane, PoolRecord, SubRecord,
                                      SubnatRec, QosBucketRec, BindingRec,
                                      PASS, TX, DROP, FWD)
from bng_amd.dataplane.packets import (build_dhcp_request, build_ipv4,
                                       parse_dhcp_frame, ip2u32, mac_bytes,
                                       DHCP_DISCOVER, DHCP_REQUEST,
                                       DHCP_OFFER, DHCP_ACK)


def make_dp(now_sec=1000):
    dp = GoldenDataplane(now_ns=now_sec * 10**9)
    dp.server_mac = mac_bytes("02:00:00:00:00:01")
    dp.server_ip = ip2u32("10.0.0.1")
    dp.pools[1] = PoolRecord(network=ip2u32("10.0.1.0"), prefix_len=24,
                             gateway=ip2u32("10.0.1.1"),
                             dns_primary=ip2u32("8.8.8.8"),
                             dns_secondary=ip2u32("1.1.1.1"), lease_time=3600)
    return dp


MAC = "aa:bb:cc:00:00:01"


def add_sub(dp, mac=MAC, ip="10.0.1.50", expiry=2000):
    dp.subscribers[abi.mac_to_u64(mac_bytes(mac))] = SubRecord(
        pool_id=1, allocated_ip=ip2u32(ip), lease_expiry=expiry)


class TestDHCPFastpath:
    def test_discover_hit_builds_offer(self):
        dp = make_dp()
        add_sub(dp)
        frame = bytearray(build_dhcp_request(MAC, DHCP_DISCOVER, xid=0xDEAD))
        verdict, out_len = dp.dhcp_fastpath(frame)
        assert verdict == TX
        r = parse_dhcp_frame(bytes(frame[:out_len]))
        assert r.op == 2                       # BOOTREPLY
        assert r.msg_type == DHCP_OFFER
        assert r.xid == 0xDEAD
        assert r.yiaddr == ip2u32("10.0.1.50")
        assert r.siaddr == ip2u32("10.0.0.1")
        assert r.src_ip == ip2u32("10.0.0.1")
        assert r.dst_ip == 0xFFFFFFFF          # broadcast
        assert r.eth_dst == b"\xff" * 6
        assert r.eth_src == dp.server_mac
        assert r.sport == 67 and r.dport == 68
        assert r.ip_checksum_ok
        assert struct.unpack(">I", r.options[54])[0] == ip2u32("10.0.0.1")
        assert struct.unpack(">I", r.options[51])[0] == 3600
        assert struct.unpack(">I", r.options[1])[0] == 0xFFFFFF00
        assert struct.unpack(">I", r.options[3])[0] == ip2u32("10.0.1.1")
        assert r.options[6] == struct.pack(">II", ip2u32("8.8.8.8"),
                                           ip2u32("1.1.1.1"))
        assert struct.unpack(">I", r.options[58])[0] == 1800
        assert struct.unpack(">I", r.options[59])[0] == 3150
        assert dp.dhcp_stats[abi.ST_FASTPATH_HITS] == 1
        assert dp.dhcp_stats[abi.ST_BROADCAST_REPLIES] == 1

    def test_request_hit_builds_ack(self):
        dp = make_dp()
        add_sub(dp)
        frame = bytearray(build_dhcp_request(MAC, DHCP_REQUEST))
        verdict, out_len = dp.dhcp_fastpath(frame)
        assert verdict == TX
        r = parse_dhcp_frame(bytes(frame[:out_len]))
        assert r.msg_type == DHCP_ACK

    def test_unknown_mac_misses(self):
        dp = make_dp()
        frame = bytearray(build_dhcp_request(MAC, DHCP_DISCOVER))
        verdict, _ = dp.dhcp_fastpath(frame)
        assert verdict == PASS
        assert dp.dhcp_stats[abi.ST_FASTPATH_MISSES] == 1
        assert dp.dhcp_stats[abi.ST_TOTAL_REQUESTS] == 1

    def test_expired_lease_passes(self):
        dp = make_dp(now_sec=5000)
        add_sub(dp, expiry=2000)
        frame = bytearray(build_dhcp_request(MAC, DHCP_DISCOVER))
        verdict, _ = dp.dhcp_fastpath(frame)
        assert verdict == PASS
        assert dp.dhcp_stats[abi.ST_CACHE_EXPIRED] == 1

    def test_non_dhcp_and_malformed_pass_untouched(self):
        dp = make_dp()
        add_sub(dp)
        # plain UDP:80 packet is not DHCP
        f = bytearray(build_ipv4(MAC, "ff:ff:ff:ff:ff:ff",
                                 ip2u32("10.0.1.50"), ip2u32("1.2.3.4")))
        orig = bytes(f)
        verdict, _ = dp.dhcp_fastpath(f)
        assert verdict == PASS and bytes(f) == orig
        assert dp.dhcp_stats[abi.ST_TOTAL_REQUESTS] == 0
        # wrong op (BOOTREPLY)
        f2 = bytearray(build_dhcp_request(MAC))
        # locate dhcp op byte: 14 eth + 20 ip + 8 udp
        f2[42] = 2
        verdict, _ = dp.dhcp_fastpath(f2)
        assert verdict == PASS

    def test_other_msg_types_pass(self):
        dp = make_dp()
        add_sub(dp)
        for mt in (4, 7, 8):  # DECLINE, RELEASE, INFORM
            frame = bytearray(build_dhcp_request(MAC, mt))
            verdict, _ = dp.dhcp_fastpath(frame)
            assert verdict == PASS

    def test_vlan_qinq_lookup_priority(self):
        dp = make_dp()
        dp.subscribers[abi.vlan_key(100, 200)] = SubRecord(
            pool_id=1, allocated_ip=ip2u32("10.0.1.60"), lease_expiry=2000)
        frame = bytearray(build_dhcp_request(MAC, DHCP_DISCOVER,
                                             s_tag=100, c_tag=200))
        verdict, out_len = dp.dhcp_fastpath(frame)
        assert verdict == TX
        r = parse_dhcp_frame(bytes(frame[:out_len]))
        assert r.yiaddr == ip2u32("10.0.1.60")
        assert r.s_tag == 100 and r.c_tag == 200   # tags preserved
        assert dp.dhcp_stats[abi.ST_VLAN_PACKETS] == 1

    def test_single_vlan(self):
        dp = make_dp()
        dp.subscribers[abi.vlan_key(300, 0)] = SubRecord(
            pool_id=1, allocated_ip=ip2u32("10.0.1.61"), lease_expiry=2000)
        frame = bytearray(build_dhcp_request(MAC, DHCP_DISCOVER, c_tag=300))
        verdict, out_len = dp.dhcp_fastpath(frame)
        assert verdict == TX
        assert parse_dhcp_frame(bytes(frame[:out_len])).yiaddr == ip2u32("10.0.1.61")

    def test_circuit_id_lookup(self):
        dp = make_dp()
        cid = b"olt1/slot2/port3"
        dp.subscribers[abi.circuit_key(cid)] = SubRecord(
            pool_id=1, allocated_ip=ip2u32("10.0.1.70"), lease_expiry=2000)
        frame = bytearray(build_dhcp_request("de:ad:be:ef:00:01",
                                             DHCP_DISCOVER, circuit_id=cid))
        verdict, out_len = dp.dhcp_fastpath(frame)
        assert verdict == TX
        assert parse_dhcp_frame(bytes(frame[:out_len])).yiaddr == ip2u32("10.0.1.70")
        assert dp.dhcp_stats[abi.ST_OPTION82_PRESENT] == 1

    def test_relay_giaddr_unicast(self):
        dp = make_dp()
        add_sub(dp)
        relay_ip = ip2u32("10.9.9.9")
        frame = bytearray(build_dhcp_request(MAC, DHCP_REQUEST,
                                             giaddr=relay_ip,
                                             src_mac="02:11:22:33:44:55"))
        verdict, out_len = dp.dhcp_fastpath(frame)
        assert verdict == TX
        r = parse_dhcp_frame(bytes(frame[:out_len]))
        assert r.dst_ip == relay_ip
        assert r.eth_dst == mac_bytes("02:11:22:33:44:55")  # back to relay
        assert r.sport == 67 and r.dport == 67
        assert r.giaddr == relay_ip
        assert dp.dhcp_stats[abi.ST_UNICAST_REPLIES] == 1

    def test_unicast_when_ciaddr_set(self):
        dp = make_dp()
        add_sub(dp)
        frame = bytearray(build_dhcp_request(
            MAC, DHCP_REQUEST, ciaddr=ip2u32("10.0.1.50")))
        verdict, out_len = dp.dhcp_fastpath(frame)
        assert verdict == TX
        r = parse_dhcp_frame(bytes(frame[:out_len]))
        assert r.eth_dst == mac_bytes(MAC)
        assert dp.dhcp_stats[abi.ST_UNICAST_REPLIES] == 1

    def test_option53_deep_in_options(self):
        dp = make_dp()
        add_sub(dp)
        # pad before option 53 — the full TLV scan must still find it
        frame = bytearray(build_dhcp_request(MAC, DHCP_DISCOVER,
                                             pad_before_53=7))
        verdict, _ = dp.dhcp_fastpath(frame)
        assert verdict == TX


def nat_dp():
    dp = GoldenDataplane(now_ns=10**9)
    dp.subnat[ip2u32("10.0.1.50")] = SubnatRec(
        public_ip=ip2u32("203.0.113.1"), port_start=1024, port_end=2047,
        next_port=1024, subscriber_id=42)
    return dp


PRIV, PUB, DST = "10.0.1.50", "203.0.113.1", "93.184.216.34"


class TestNAT44:
    def test_snat_udp_creates_session(self):
        dp = nat_dp()
        f = bytearray(build_ipv4(MAC, "02:00:00:00:00:01", ip2u32(PRIV),
                                 ip2u32(DST), proto=17, sport=5555, dport=53))
        verdict = dp.nat44_egress(f)
        assert verdict == FWD
        saddr = struct.unpack_from(">I", f, 26)[0]
        sport = struct.unpack_from(">H", f, 34)[0]
        assert saddr == ip2u32(PUB)
        assert sport == 1024
        # IP checksum still valid after incremental update
        from bng_amd.dataplane.packets import ipv4_checksum
        hdr = bytes(f[14:34])
        assert ipv4_checksum(hdr[:10] + b"\x00\x00" + hdr[12:]) == \
            struct.unpack(">H", hdr[10:12])[0]
        assert dp.nat_stats[abi.NS_SNAT] == 1
        assert dp.nat_stats[abi.NS_SESS_CREATED] == 1
        assert len(dp.nat_log) == 1
        assert dp.nat_log[0]["event_type"] == abi.LOG_SESSION_CREATE

    def test_snat_second_packet_reuses_session(self):
        dp = nat_dp()
        for i in range(2):
            f = bytearray(build_ipv4(MAC, "02:00:00:00:00:01", ip2u32(PRIV),
                                     ip2u32(DST), proto=17, sport=5555,
                                     dport=53))
            dp.nat44_egress(f)
        assert dp.nat_stats[abi.NS_SESS_CREATED] == 1
        sess = list(dp.nat_sessions.values())[0]
        assert sess.packets_out == 2

    def test_eim_reuses_mapping_across_destinations(self):
        dp = nat_dp()
        f1 = bytearray(build_ipv4(MAC, "x", ip2u32(PRIV), ip2u32(DST),
                                  proto=17, sport=7777, dport=53,
                                  dst_mac="02:00:00:00:00:01", src_mac=MAC)) \
            if False else bytearray(build_ipv4(MAC, "02:00:00:00:00:01",
                                               ip2u32(PRIV), ip2u32(DST),
                                               proto=17, sport=7777, dport=53))
        dp.nat44_egress(f1)
        p1 = struct.unpack_from(">H", f1, 34)[0]
        f2 = bytearray(build_ipv4(MAC, "02:00:00:00:00:01", ip2u32(PRIV),
                                  ip2u32("8.8.4.4"), proto=17, sport=7777,
                                  dport=123))
        dp.nat44_egress(f2)
        p2 = struct.unpack_from(">H", f2, 34)[0]
        assert p1 == p2                        # endpoint-independent
        assert dp.nat_stats[abi.NS_EIM_HITS] == 1

    def test_dnat_return_path(self):
        dp = nat_dp()
        out = bytearray(build_ipv4(MAC, "02:00:00:00:00:01", ip2u32(PRIV),
                                   ip2u32(DST), proto=17, sport=5555,
                                   dport=53))
        dp.nat44_egress(out)
        nat_port = struct.unpack_from(">H", out, 34)[0]
        back = bytearray(build_ipv4("02:00:00:00:00:02", "02:00:00:00:00:01",
                                    ip2u32(DST), ip2u32(PUB), proto=17,
                                    sport=53, dport=nat_port))
        verdict = dp.nat44_ingress(back)
        assert verdict == FWD
        daddr = struct.unpack_from(">I", back, 30)[0]
        dport = struct.unpack_from(">H", back, 36)[0]
        assert daddr == ip2u32(PRIV) and dport == 5555
        assert dp.nat_stats[abi.NS_DNAT] == 1

    def test_tcp_state_machine(self):
        dp = nat_dp()
        out = bytearray(build_ipv4(MAC, "02:00:00:00:00:01", ip2u32(PRIV),
                                   ip2u32(DST), proto=6, sport=5555, dport=80,
                                   tcp_flags=0x02))
        dp.nat44_egress(out)
        nat_port = struct.unpack_from(">H", out, 34)[0]
        sess = list(dp.nat_sessions.values())[0]
        assert sess.state == abi.NAT_NEW
        ack = bytearray(build_ipv4("02:00:00:00:00:02", "02:00:00:00:00:01",
                                   ip2u32(DST), ip2u32(PUB), proto=6,
                                   sport=80, dport=nat_port, tcp_flags=0x12))
        dp.nat44_ingress(ack)
        assert sess.state == abi.NAT_ESTABLISHED
        fin = bytearray(build_ipv4("02:00:00:00:00:02", "02:00:00:00:00:01",
                                   ip2u32(DST), ip2u32(PUB), proto=6,
                                   sport=80, dport=nat_port, tcp_flags=0x11))
        dp.nat44_ingress(fin)
        assert sess.state == abi.NAT_CLOSING

    def test_non_private_source_forwards_untouched(self):
        dp = nat_dp()
        f = bytearray(build_ipv4(MAC, "02:00:00:00:00:01", ip2u32("8.8.8.8"),
                                 ip2u32(DST)))
        orig = bytes(f)
        assert dp.nat44_egress(f) == FWD
        assert bytes(f) == orig

    def test_no_allocation_passes(self):
        dp = nat_dp()
        f = bytearray(build_ipv4(MAC, "02:00:00:00:00:01",
                                 ip2u32("10.0.1.99"), ip2u32(DST)))
        assert dp.nat44_egress(f) == PASS
        assert dp.nat_stats[abi.NS_PASSED] == 1

    def test_port_exhaustion_drops(self):
        # The reference's in-block EIM-collision check is keyed by INTERNAL
        # port (nat44.c:450-459), so plain reuse is a tolerated benign race;
        # guaranteed exhaustion happens via the parity filter: a block with
        # only even ports + parity preservation + odd source port never
        # yields a port (64 tries, nat44.c:423-465).
        dp = nat_dp()
        dp.nat_flags |= abi.NAT_FLAG_PARITY
        blk = dp.subnat[ip2u32(PRIV)]
        blk.port_start, blk.port_end, blk.next_port = 1024, 1024, 1024
        f = bytearray(build_ipv4(MAC, "02:00:00:00:00:01", ip2u32(PRIV),
                                 ip2u32(DST), proto=17, sport=13, dport=53))
        assert dp.nat44_egress(f) == DROP
        assert dp.nat_stats[abi.NS_PORT_EXHAUSTION] == 1
        assert dp.nat_log[-1]["event_type"] == abi.LOG_PORT_EXHAUSTION

    def test_alg_trigger_punts(self):
        dp = nat_dp()
        dp.nat_flags |= abi.NAT_FLAG_ALG_FTP
        dp.alg_ports.add((21, 6))
        f = bytearray(build_ipv4(MAC, "02:00:00:00:00:01", ip2u32(PRIV),
                                 ip2u32(DST), proto=6, sport=5555, dport=21))
        assert dp.nat44_egress(f) == PASS
        assert dp.nat_stats[abi.NS_ALG_TRIGGERS] == 1

    def test_icmp_id_translation(self):
        dp = nat_dp()
        f = bytearray(build_ipv4(MAC, "02:00:00:00:00:01", ip2u32(PRIV),
                                 ip2u32(DST), proto=1, icmp_id=777))
        assert dp.nat44_egress(f) == FWD
        new_id = struct.unpack_from(">H", f, 38)[0]
        assert 1024 <= new_id <= 2047

    def test_hairpin_detection(self):
        dp = nat_dp()
        dp.nat_flags |= abi.NAT_FLAG_HAIRPIN
        dp.hairpin_ips.add(ip2u32(PUB))
        f = bytearray(build_ipv4(MAC, "02:00:00:00:00:01", ip2u32(PRIV),
                                 ip2u32(PUB), proto=17, sport=5555, dport=53))
        dp.nat44_egress(f)
        assert dp.nat_stats[abi.NS_HAIRPIN] == 1
        assert list(dp.nat_sessions.values())[0].is_hairpin == 1


class TestQoS:
    def test_no_policy_forwards(self):
        dp = GoldenDataplane(now_ns=10**9)
        f = build_ipv4(MAC, "x" * 0 or "02:00:00:00:00:01",
                       ip2u32("1.1.1.1"), ip2u32("10.0.1.50"))
        assert dp.qos(f, "egress") == FWD

    def test_rate_zero_unlimited(self):
        dp = GoldenDataplane(now_ns=10**9)
        dp.qos_egress[ip2u32("10.0.1.50")] = QosBucketRec(0, 0)
        f = build_ipv4("02:00:00:00:00:01", MAC, ip2u32("1.1.1.1"),
                       ip2u32("10.0.1.50"))
        for _ in range(10):
            assert dp.qos(f, "egress") == FWD

    def test_bucket_drains_and_refills(self):
        dp = GoldenDataplane(now_ns=10**9)
        ip = ip2u32("10.0.1.50")
        # 8000 bps = 1000 bytes/s; burst 150 bytes
        dp.qos_egress[ip] = QosBucketRec(rate_bps=8000, burst_bytes=150,
                                         tokens=150, last_update=10**9)
        f = build_ipv4("02:00:00:00:00:01", MAC, ip2u32("1.1.1.1"), ip,
                       payload=b"x" * 58)  # frame = 100 bytes
        assert len(f) == 100
        assert dp.qos(f, "egress") == FWD       # 150 -> 50
        assert dp.qos(f, "egress") == DROP      # 50 < 100
        dp.now_ns += 100_000_000                # +0.1s -> +100 bytes
        assert dp.qos(f, "egress") == FWD       # 150 -> 50
        assert dp.qos_stats[abi.QS_PKT_DROPPED] == 1
        assert dp.qos_stats[abi.QS_PKT_PASSED] == 2

    def test_ingress_keys_source_ip(self):
        dp = GoldenDataplane(now_ns=10**9)
        ip = ip2u32("10.0.1.50")
        dp.qos_ingress[ip] = QosBucketRec(rate_bps=8, burst_bytes=10,
                                          tokens=0, last_update=10**9)
        f = build_ipv4(MAC, "02:00:00:00:00:01", ip, ip2u32("1.1.1.1"))
        assert dp.qos(f, "ingress") == DROP


class TestAntispoof:
    def test_disabled_allows(self):
        dp = GoldenDataplane()
        f = build_ipv4(MAC, "02:00:00:00:00:01", ip2u32("6.6.6.6"),
                       ip2u32("1.1.1.1"))
        assert dp.antispoof(f) == FWD

    def test_strict_match_and_violation(self):
        dp = GoldenDataplane()
        dp.bindings[abi.mac_to_u64(mac_bytes(MAC))] = BindingRec(
            ipv4_addr=ip2u32("10.0.1.50"), ipv4_valid=1, mode=abi.AS_STRICT)
        ok = build_ipv4(MAC, "02:00:00:00:00:01", ip2u32("10.0.1.50"),
                        ip2u32("1.1.1.1"))
        assert dp.antispoof(ok) == FWD
        bad = build_ipv4(MAC, "02:00:00:00:00:01", ip2u32("6.6.6.6"),
                         ip2u32("1.1.1.1"))
        assert dp.antispoof(bad) == DROP
        assert dp.as_stats[abi.AS_V4_VIOLATIONS] == 1

    def test_log_only_mode(self):
        dp = GoldenDataplane()
        dp.as_log_violations = 1
        dp.bindings[abi.mac_to_u64(mac_bytes(MAC))] = BindingRec(
            ipv4_addr=ip2u32("10.0.1.50"), ipv4_valid=1, mode=abi.AS_LOG_ONLY)
        bad = build_ipv4(MAC, "02:00:00:00:00:01", ip2u32("6.6.6.6"),
                         ip2u32("1.1.1.1"))
        assert dp.antispoof(bad) == FWD
        assert len(dp.spoof_events) == 1
        assert dp.as_stats[abi.AS_LOGGED] == 1

    def test_loose_mode_unknown_mac_range_check(self):
        dp = GoldenDataplane()
        dp.as_default_mode = abi.AS_LOOSE
        dp.allowed_ranges = [(ip2u32("10.0.0.0"), 0xFF000000)]
        ok = build_ipv4(MAC, "02:00:00:00:00:01", ip2u32("10.5.5.5"),
                        ip2u32("1.1.1.1"))
        assert dp.antispoof(ok) == FWD
        bad = build_ipv4(MAC, "02:00:00:00:00:01", ip2u32("6.6.6.6"),
                         ip2u32("1.1.1.1"))
        assert dp.antispoof(bad) == DROP

    def test_ipv6_strict(self):
        dp = GoldenDataplane()
        v6 = bytes(range(16))
        dp.bindings[abi.mac_to_u64(mac_bytes(MAC))] = BindingRec(
            ipv6_addr=v6, ipv6_valid=1, mode=abi.AS_STRICT)
        hdr = mac_bytes("02:00:00:00:00:01") + mac_bytes(MAC) + \
            struct.pack(">H", 0x86DD) + b"\x60\x00\x00\x00\x00\x00\x3b\x40" \
            + v6 + b"\x00" * 16
        assert dp.antispoof(hdr) == FWD
        bad6 = bytes(reversed(range(16)))
        hdr2 = mac_bytes("02:00:00:00:00:01") + mac_bytes(MAC) + \
            struct.pack(">H", 0x86DD) + b"\x60\x00\x00\x00\x00\x00\x3b\x40" \
            + bad6 + b"\x00" * 16
        assert dp.antispoof(hdr2) == DROP
        assert dp.as_stats[abi.AS_V6_VIOLATIONS] == 1

    def test_non_ip_allowed(self):
        dp = GoldenDataplane()
        dp.as_default_mode = abi.AS_STRICT
        arp = mac_bytes("ff:ff:ff:ff:ff:ff") + mac_bytes(MAC) + \
            struct.pack(">H", 0x0806) + b"\x00" * 28
        assert dp.antispoof(arp) == FWD


class TestQoSRefillClamp:
    def test_long_idle_refill_capped_at_burst(self):
        """Elapsed > 100 s credits exactly one burst, not rate*elapsed
        (the clamp that stops counter-wrap abuse; kernel qos_tb_check)."""
        dp = GoldenDataplane()
        ip = ip2u32("10.0.1.50")
        dp.qos_egress[ip] = QosBucketRec(rate_bps=8 * 10**9,  # 1 GB/s
                                         burst_bytes=150, tokens=0,
                                         last_update=0)
        dp.now_ns = 200 * 10**9        # 200 s idle: naive credit = 200 GB
        big = build_ipv4(MAC, "02:00:00:00:00:01", ip2u32(DST), ip2u32(PRIV),
                         proto=17, sport=53, dport=5555,
                         payload=b"x" * 100)
        small = build_ipv4(MAC, "02:00:00:00:00:01", ip2u32(DST),
                           ip2u32(PRIV), proto=17, sport=53, dport=5556)
        # 142-B frame passes on the capped 150-B burst; the 8 B left
        # cannot cover the next 42-B frame — proof the 200-GB naive
        # credit never materialized
        assert dp.qos(bytes(big), "egress") == FWD
        assert dp.qos(bytes(small), "egress") == DROP
