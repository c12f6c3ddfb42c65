"""Struct-ABI contract tests: Python ctypes mirrors vs the compiled
extension's sizeof/offsetof report — the analog of the reference's
test/ebpf/maps_test.go:15-80 (Go mirror structs vs bpf/maps.h)."""
import ctypes

import pytest

from bng_amd.dataplane import abi


def test_ctypes_sizes_match_expected():
    for name, (cls, size) in abi.EXPECTED_SIZES.items():
        assert ctypes.sizeof(cls) == size, f"{name}: {ctypes.sizeof(cls)} != {size}"


def test_key_helpers():
    assert abi.mac_to_u64(bytes([0xAA, 0xBB, 0xCC, 0, 0, 1])) == 0xAABBCC000001
    assert abi.vlan_key(100, 200) == (1 << 62) | (100 << 16) | 200
    k = abi.circuit_key(b"port1")
    assert k >> 62 == 2
    # FNV-1a known vector: fnv1a64("") = offset basis
    assert abi.fnv1a64(b"") == 0xCBF29CE484222325
    assert abi.fnv1a64(b"a") == 0xAF63DC4C8601EC8C


def test_tuple_sig_nonzero_and_odd():
    s = abi.tuple_sig(0, 0, 0, 0, 0)
    assert s & 1 and s != abi.KEY_TOMBSTONE
    assert abi.tuple_sig(1, 2, 3, 4, 17) != abi.tuple_sig(1, 2, 3, 4, 6)


def test_extension_layout_matches_ctypes():
    """Compare against the compiled C++ report when the extension exists
    (it is built in-tree; this runs on CPU — layout_report is host-only)."""
    from bng_amd.dataplane.build import get_ext
    ext = get_ext()
    if ext is None:
        pytest.skip("extension not built")
    rep = ext.layout_report()
    for name, (cls, _) in abi.EXPECTED_SIZES.items():
        assert rep[name] == ctypes.sizeof(cls), name
    off = rep["offsets"]
    assert off["sub_entry.lease_expiry"] == abi.SubEntry.lease_expiry.offset
    assert off["nat_session.last_seen"] == abi.NatSession.last_seen.offset
    assert off["nat_session.ready"] == abi.NatSession.ready.offset
    assert off["eim_entry.created"] == abi.EimEntry.created.offset
    assert off["subctx.rate_bps"] == abi.SubCtx.rate_bps.offset
    assert off["subctx.next_port"] == abi.SubCtx.next_port.offset
    assert off["subctx.sessions_active"] == \
        abi.SubCtx.sessions_active.offset
    assert off["qos_bucket.tokens"] == abi.QosBucket.tokens.offset
    assert off["qos_bucket.last_update"] == abi.QosBucket.last_update.offset
    assert off["binding_entry.ipv6_addr"] == abi.BindingEntry.ipv6_addr.offset
    assert off["nat_config.priv_lo"] == abi.NatConfig.priv_lo.offset
    assert off["nat_config.alg_key"] == abi.NatConfig.alg_key.offset
    assert off["antispoof_config.allowed_lo"] == \
        abi.AntispoofConfig.allowed_lo.offset
    assert off["spoof_event.spoofed_ip"] == abi.SpoofEvent.spoofed_ip.offset


class TestConversions:
    """IP/MAC/key conversion consistency (ref test/ebpf/maps_test.go +
    pkg/ebpf/loader_test.go conversion tests)."""

    def test_ip_round_trip(self):
        from bng_amd.dataplane.packets import ip2u32, u32_to_ip
        for ip in ("0.0.0.0", "10.0.1.5", "203.0.113.255",
                   "255.255.255.255"):
            assert u32_to_ip(ip2u32(ip)) == ip
        assert ip2u32("1.2.3.4") == 0x01020304   # BE wire order as host int

    def test_mac_round_trip(self):
        from bng_amd.dataplane.abi import mac_to_u64
        from bng_amd.dataplane.packets import mac_bytes
        m = mac_bytes("aa:bb:cc:dd:ee:ff")
        v = mac_to_u64(m)
        assert v == 0xAABBCCDDEEFF
        assert bytes((v >> (8 * (5 - i))) & 0xFF for i in range(6)) == m

    def test_key_spaces_disjoint(self):
        """MAC / VLAN / circuit keys share one u64 table but must never
        collide across types (tag bits 62-63, ref maps.h key scheme)."""
        from bng_amd.dataplane.abi import (circuit_key, mac_to_u64,
                                           vlan_key)
        from bng_amd.dataplane.packets import mac_bytes
        mk = mac_to_u64(mac_bytes("aa:bb:cc:00:00:01"))
        vk = vlan_key(100, 200)
        ck = circuit_key(b"pon0/1/2:100")
        assert len({mk >> 62, vk >> 62, ck >> 62}) == 3
        assert vlan_key(100, 200) != vlan_key(200, 100)

    def test_circuit_hash_truncation_and_collisions(self):
        """Circuit IDs hash over a 32-byte zero-padded buffer, matching
        the reference's fixed-size map key (maps.h:216-220, ref
        CircuitIDKeyTruncation): ids differing only beyond byte 32
        collide BY DESIGN; within the window they stay distinct."""
        from bng_amd.dataplane.abi import circuit_key, fnv1a64
        assert circuit_key(b"x" * 40) == circuit_key(b"x" * 50)
        assert circuit_key(b"a" + b"x" * 31) != circuit_key(
            b"b" + b"x" * 31)
        assert circuit_key(b"pad") == circuit_key(b"pad\x00\x00")
        # 1k distinct ids -> 1k distinct keys (collision resistance)
        keys = {circuit_key(f"olt{i}/pon{i % 7}".encode())
                for i in range(1000)}
        assert len(keys) == 1000
        assert fnv1a64(b"") == 0xcbf29ce484222325   # FNV offset basis
