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
    assert off["nat_config.private_net"] == abi.NatConfig.private_net.offset
    assert off["nat_config.alg_key"] == abi.NatConfig.alg_key.offset
    assert off["antispoof_config.allowed_net"] == \
        abi.AntispoofConfig.allowed_net.offset
    assert off["spoof_event.spoofed_ip"] == abi.SpoofEvent.spoofed_ip.offset
