"""Host-boundary (PCIe-crossing) path: the packed NIC-ring cell layout
used by bench.py --host-io and the vectorized Pump edge.  CPU tests —
the torch index ops are device-agnostic, so a round-trip proven here is
the same code the GPU runs (the driver's bench run measures the rate)."""
import numpy as np
import pytest
import torch

import bench


def _cells_of(L, stride=512):
    if L > 64:
        return min((int(L) + 64 + 63) // 64, stride // 64)
    return (int(L) + 63) // 64


def _round_trip(lens_np, stride=512, seed=0):
    n = len(lens_np)
    rng = np.random.default_rng(seed)
    data = rng.integers(0, 255, size=(n, stride), dtype=np.uint8)
    # bytes beyond the frame's cell span are not wire bytes; zero them
    # so equality below tests exactly what the packed ring must carry
    for i in range(n):
        data[i, _cells_of(int(lens_np[i]), stride) * 64:] = 0
    slots = torch.from_numpy(data)
    C, slot_idx_np = bench.build_cell_map(lens_np, stride)
    slot_idx = torch.from_numpy(slot_idx_np)
    packed = torch.index_select(slots.view(-1, 64), 0, slot_idx)
    out = torch.zeros_like(slots)
    out.view(-1, 64).index_copy_(0, slot_idx, packed)
    return data, out.numpy(), C


def test_cell_map_round_trip_mixed():
    lens = np.array([64, 342, 64, 400, 128, 64, 65, 1], dtype=np.uint16)
    data, out, C = _round_trip(lens)
    cells_expected = sum(_cells_of(int(L)) for L in lens)
    assert C == cells_expected
    for i, L in enumerate(lens):
        span = _cells_of(int(L)) * 64
        assert (out[i, :span] == data[i, :span]).all()


def test_cell_map_matches_gen_batch_traffic():
    data_np, lens_np = bench.gen_batch(2048, 10_000, 0.1, 512, seed=9)
    slots = torch.from_numpy(data_np)
    C, slot_idx_np = bench.build_cell_map(lens_np, 512)
    slot_idx = torch.from_numpy(slot_idx_np)
    packed = torch.index_select(slots.view(-1, 64), 0, slot_idx)
    out = torch.zeros_like(slots)
    out.view(-1, 64).index_copy_(0, slot_idx, packed)
    # every wire byte survives the ring round trip
    for i, L in enumerate(lens_np):
        assert (out.numpy()[i, :L] == data_np[i, :L]).all()
    # packed size is the host-boundary byte count: 64B data packets are
    # one cell, DHCP frames their size + a 64B growth budget
    n_dhcp = int((lens_np > 64).sum())
    n_data = len(lens_np) - n_dhcp
    per_dhcp = {(int(L) + 64 + 63) // 64 for L in lens_np if L > 64}
    assert C == n_data + sum((int(L) + 64 + 63) // 64
                             for L in lens_np if L > 64)
    assert C * 64 < 2048 * 512 * 0.35  # far below shipping full slots


def test_cell_map_dhcp_growth_headroom():
    """A DHCP OFFER built in place may be longer than the DISCOVER;
    the cell map gives DHCP frames a 64-byte growth budget (the fixed
    option set never grows a reply more than that), capped at the
    slot."""
    lens = np.array([70, 342, 500], dtype=np.uint16)  # all DHCP-sized
    C, idx = bench.build_cell_map(lens, 512)
    assert C == 3 + 7 + 8     # ceil((len+64)/64), capped at 8


def test_bench_downlink_crafting_matches_golden():
    """The bench downlink phase builds return packets from exported
    session records; on the golden dataplane every crafted packet must
    DNAT back to its subscriber (protects the driver-visible
    downlink_mpps from crafting regressions)."""
    from bng_amd.dataplane import abi
    from bng_amd.dataplane.launcher import GoldenLauncher
    from bng_amd.dataplane.packets import build_ipv4, ip2u32

    g = GoldenLauncher()
    g.set_nat_config()
    for i in range(1, 40):
        g.add_subscriber_nat(ip2u32("10.0.0.0") + i,
                             ip2u32("203.0.113.1"),
                             1024 + i * 1024, 1024 + i * 1024 + 1023,
                             subscriber_id=i)
    for i in range(1, 40):
        pkt = bytearray(build_ipv4(
            "aa:00:00:00:00:01", "02:00:00:00:00:01",
            ip2u32("10.0.0.0") + i, ip2u32("93.184.216.34"),
            proto=17, sport=40000 + i, dport=53, payload=b"x" * 22))
        assert g.dp.nat44_egress(pkt) == abi.FWD
    recs = g.export_nat_sessions()
    assert len(recs) == 39

    # exactly the bench crafting: template + field splice from records
    t = np.frombuffer(build_ipv4(
        "02:00:00:00:00:01", "aa:00:00:00:00:00",
        ip2u32("93.184.216.34"), ip2u32("203.0.113.1"),
        proto=17, sport=53, dport=1024, payload=b"\x00" * 22),
        dtype=np.uint8)
    ret = np.zeros((39, 512), dtype=np.uint8)
    idxs = np.arange(39)
    ret[:, :64] = t
    ret[:, 26:30] = recs["dst_ip"][idxs].astype(">u4") \
        .view(np.uint8).reshape(-1, 4)
    ret[:, 30:34] = recs["nat_ip"][idxs].astype(">u4") \
        .view(np.uint8).reshape(-1, 4)
    ret[:, 34:36] = recs["dst_port"][idxs].astype(">u2") \
        .view(np.uint8).reshape(-1, 2)
    ret[:, 36:38] = recs["nat_port"][idxs].astype(">u2") \
        .view(np.uint8).reshape(-1, 2)
    for i in range(39):
        fb = bytearray(ret[i, :64].tobytes())
        assert g.dp.nat44_ingress(fb) == abi.FWD, f"row {i}"
        assert int.from_bytes(fb[30:34], "big") == \
            int(recs["src_ip"][i]), f"row {i} dst"
