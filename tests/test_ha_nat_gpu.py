"""GPU NAT-session HA sync: export/import kernel round trip, flow
survival through a simulated promotion between two HipLaunchers on one
device, and promotion timing at 1M sessions (round-1 VERDICT task 3)."""
import time

import numpy as np
import pytest

from bng_amd.dataplane import abi
from bng_amd.dataplane.packets import build_ipv4, ip2u32, mac_bytes

torch = pytest.importorskip("torch")
pytestmark = [
    pytest.mark.gpu,
    pytest.mark.skipif(not torch.cuda.is_available(), reason="needs GPU"),
]

NOW = 1_700_000_000
NOW_NS = NOW * 10**9


def _mk_launcher(sub_log2=16):
    from bng_amd.dataplane.launcher import HipLauncher
    l = HipLauncher("cuda:0", sub_log2=sub_log2, sess_log2=sub_log2 + 1,
                    eim_log2=sub_log2, subnat_log2=sub_log2,
                    qos_log2=sub_log2, binding_log2=sub_log2)
    l.set_nat_config()
    l.add_subscriber_nat(ip2u32("10.0.0.5"), ip2u32("203.0.113.7"),
                         2048, 3071, subscriber_id=42)
    return l


def test_flow_survives_gpu_promotion():
    active = _mk_launcher()
    standby = _mk_launcher()
    pkt = build_ipv4("aa:00:00:00:00:05", "02:00:00:00:00:01",
                     ip2u32("10.0.0.5"), ip2u32("93.184.216.34"),
                     proto=17, sport=5555, dport=53, payload=b"x" * 22)
    d, l = active.make_batch([pkt])
    v = active.nat44(d, l, egress=True, now_ns=NOW_NS)
    assert int(v[0].item()) == abi.FWD
    out = d.cpu().numpy()[0]
    nat_port = int.from_bytes(bytes(out[34:36]), "big")

    # replicate: export from active, import on standby (the glue's
    # promote path without the HTTP hop)
    recs = active.export_nat_sessions()
    assert len(recs) == 1
    assert int(recs[0]["nat_port"]) == nat_port
    assert standby.import_nat_sessions(recs) == 1

    # return packet DNATs on the standby
    ret = build_ipv4("02:00:00:00:00:01", "aa:00:00:00:00:05",
                     ip2u32("93.184.216.34"), ip2u32("203.0.113.7"),
                     proto=17, sport=53, dport=nat_port, payload=b"y" * 22)
    d2, l2 = standby.make_batch([ret])
    v2 = standby.nat44(d2, l2, egress=False, now_ns=NOW_NS + 10**6)
    assert int(v2[0].item()) == abi.FWD
    out2 = d2.cpu().numpy()[0]
    assert int.from_bytes(bytes(out2[30:34]), "big") == ip2u32("10.0.0.5")
    assert int.from_bytes(bytes(out2[36:38]), "big") == 5555

    # EIM restored: next egress flow from the same internal endpoint
    # keeps the same external port
    pkt2 = build_ipv4("aa:00:00:00:00:05", "02:00:00:00:00:01",
                      ip2u32("10.0.0.5"), ip2u32("198.51.100.9"),
                      proto=17, sport=5555, dport=443, payload=b"z" * 22)
    d3, l3 = standby.make_batch([pkt2])
    v3 = standby.nat44(d3, l3, egress=True, now_ns=NOW_NS + 2 * 10**6)
    assert int(v3[0].item()) == abi.FWD
    out3 = d3.cpu().numpy()[0]
    assert int.from_bytes(bytes(out3[34:36]), "big") == nat_port


def test_promotion_time_1m_sessions():
    """Bulk import at scale: 1M replicated sessions restored into a
    fresh standby's tables; promotion must complete in seconds (the
    number VERDICT asked to measure)."""
    from bng_amd.dataplane.launcher import HipLauncher
    n = 1_000_000
    rng = np.random.default_rng(3)
    recs = np.zeros(n, dtype=abi.SESS_EXPORT_DTYPE)
    recs["src_ip"] = rng.integers(0x0A000002, 0x0A0F4244, n,
                                  dtype=np.uint32)
    recs["dst_ip"] = rng.integers(1, 0xDF000000, n, dtype=np.uint32)
    recs["src_port"] = rng.integers(1024, 65535, n, dtype=np.uint16)
    recs["dst_port"] = 53
    recs["protocol"] = 17
    recs["nat_ip"] = 0xCB007107
    recs["nat_port"] = rng.integers(1024, 65535, n, dtype=np.uint16)
    recs["flags"] = 1
    recs["eim_port"] = recs["nat_port"]
    recs["created"] = NOW_NS
    recs["last_seen"] = NOW_NS
    standby = HipLauncher("cuda:0", sub_log2=18, sess_log2=21,
                          eim_log2=21, subnat_log2=18, qos_log2=18,
                          binding_log2=18)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    ok = standby.import_nat_sessions(recs)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    print(f"[ha] 1M-session promotion import: {dt*1e3:.0f} ms, "
          f"{ok} restored")
    # distinct random tuples can collide in sig space; ~all must land
    assert ok > n * 0.999
    assert dt < 5.0, f"promotion import took {dt:.1f}s"
    # spot-check: re-export sees the imported population
    back = standby.export_nat_sessions()
    assert len(back) > n * 0.99


def test_export_delta_since():
    """since_ns filters to sessions seen after the cutoff (the delta
    export path)."""
    l = _mk_launcher()
    for i, t_off in enumerate((0, 10**9)):
        pkt = build_ipv4("aa:00:00:00:00:05", "02:00:00:00:00:01",
                         ip2u32("10.0.0.5"), ip2u32("93.184.216.34") + i,
                         proto=17, sport=6000 + i, dport=53,
                         payload=b"x" * 22)
        d, ln = l.make_batch([pkt])
        l.nat44(d, ln, egress=True, now_ns=NOW_NS + t_off)
    assert len(l.export_nat_sessions()) == 2
    assert len(l.export_nat_sessions(since_ns=NOW_NS + 1)) == 1
