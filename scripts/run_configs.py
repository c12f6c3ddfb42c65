#!/usr/bin/env python3
"""Measure the BASELINE.json config list (1-5) and print one JSON line
per config.  Configs 2/3/5 need a GPU (run under gpurun); config 1 runs
anywhere; config 4's 8-GPU form is the driver's SCALE run — its 1-GPU
slice is reported here."""
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np


def config1():
    """Standalone `bng run --pool-network 10.0.1.0/24` Go-analog slow-path
    DHCP on CPU, 16 subscribers, no GPU."""
    from bng_amd.cli.main import BNG, build_parser
    from bng_amd.dataplane.packets import mac_bytes
    from bng_amd.dhcp import message as dm
    from bng_amd.utils.loadtest import DHCPLoadTester
    app = BNG(build_parser().parse_args(
        ["run", "--pool-network", "10.0.1.0/24", "--gpu", "off"])).start()
    try:
        for i in range(16):
            mac = mac_bytes(f"aa:bb:cc:00:00:{i:02x}")
            offer = app.dhcp_server.handle(dm.build_request(mac, dm.DISCOVER))
            app.dhcp_server.handle(dm.build_request(
                mac, dm.REQUEST, requested_ip=offer.yiaddr))
        assert len(app.dhcp_server.leases) == 16

        def handler(mac, renew):
            mt = dm.REQUEST if renew else dm.DISCOVER
            return app.dhcp_server.handle(dm.build_request(mac, mt)) is not None

        t = DHCPLoadTester(handler, unique_macs=16, renewal_ratio=0.9,
                           concurrency=4, warmup=16)
        res = t.run(5000)
        return {"config": 1, "desc": "slow-path DHCP, 16 subs, CPU only",
                "rps": round(res.rps), "p50_us": round(res.p50 * 1e6, 1),
                "p99_us": round(res.p99 * 1e6, 1), "errors": res.errors}
    finally:
        app.stop()


def _gpu_launcher(n_subs):
    import bench
    import torch
    from bng_amd.dataplane.launcher import HipLauncher
    from bng_amd.dataplane.packets import ip2u32
    base = max(18, (n_subs - 1).bit_length() + 1)
    l = HipLauncher("cuda:0", sub_log2=base, sess_log2=base + 1,
                    eim_log2=base, subnat_log2=base, qos_log2=base,
                    binding_log2=base)
    l.set_server_config(b"\x02\x00\x00\x00\x00\x01", ip2u32("10.255.255.1"))
    l.add_pool(1, ip2u32("10.0.0.0"), 8, ip2u32("10.255.255.1"),
               ip2u32("8.8.8.8"))
    bench.build_tables(l, 0, 1, n_subs, 1_700_000_000)
    return l


def _run_uplink(l, n_subs, batch, dhcp_frac, steps=10, warmup=3):
    import bench
    import torch
    NOW = 1_700_000_000
    d_np, ln = bench.gen_batch(batch, n_subs, dhcp_frac, 512, 11)
    p = torch.from_numpy(d_np).cuda()
    w = torch.empty_like(p)
    lt = torch.from_numpy(ln.view(np.int16)).cuda()
    ns = NOW * 10**9
    for k in range(warmup):
        w.copy_(p)
        l.uplink(w, lt, now_ns=ns + k * 10**6, now_sec=NOW,
                 sort_by_type=True)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for k in range(steps):
        w.copy_(p)
        l.uplink(w, lt, now_ns=ns + (warmup + k) * 10**6, now_sec=NOW,
                 sort_by_type=True)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    return batch * steps / dt / 1e6


def _dhcp_latency(l, n_subs, batch=2048, reps=60):
    import bench
    import torch
    NOW = 1_700_000_000
    d_np, ln = bench.gen_batch(batch, n_subs, 1.0, 512, 7)
    p = torch.from_numpy(d_np).cuda()
    w = torch.empty_like(p)
    lt = torch.from_numpy(ln.view(np.int16)).cuda()
    lats = []
    for _ in range(reps):
        torch.cuda.synchronize()
        t = time.perf_counter()
        w.copy_(p)
        l.dhcp_fastpath(w, lt, now_sec=NOW)
        torch.cuda.synchronize()
        lats.append((time.perf_counter() - t) * 1e6)
    lats.sort()
    return lats[len(lats) // 2], lats[int(len(lats) * 0.99)]


def _svc_latency(l, n_subs, batch=256, reps=120):
    """DHCP OFFER latency through the persistent service kernel — the
    serving path (round-1 VERDICT tasks 4/10: the launched path missed
    the <100us P99 bar under load; the resident waves don't)."""
    import bench
    from bng_amd.dataplane.launcher import DhcpService
    NOW = 1_700_000_000
    d_np, ln = bench.gen_batch(batch, n_subs, 1.0, 512, 9)
    lats = []
    with DhcpService(l, n_slots=max(256, batch),
                 idle_exit_k=400_000) as svc:
        svc.serve(d_np, ln, NOW)
        for _ in range(reps):
            t = time.perf_counter()
            svc.serve(d_np, ln, NOW)
            lats.append((time.perf_counter() - t) * 1e6)
    lats.sort()
    return lats[len(lats) // 2], lats[int(len(lats) * 0.99)]


def config2():
    """DHCP fast path, 100k-entry subscriber table, DISCOVER/REQUEST
    flood, 1x MI355X."""
    l = _gpu_launcher(100_000)
    mpps = _run_uplink(l, 100_000, 1 << 20, 1.0)
    p50, p99 = _dhcp_latency(l, 100_000)
    s50, s99 = _svc_latency(l, 100_000)
    st = l.get_stats()
    hit = st["fastpath_hits"] / max(1, st["total_requests"])
    return {"config": 2, "desc": "DHCP flood, 100k subs, 1 GPU",
            "mpps": round(mpps, 1), "p50_us": round(p50, 1),
            "p99_us": round(p99, 1),
            "svc_p50_us": round(s50, 1), "svc_p99_us": round(s99, 1),
            "hit_rate": round(hit, 4)}


def config3():
    """NAT44/CGNAT + antispoof (+QoS), 1M concurrent flows, 64B IPv4 mix."""
    l = _gpu_launcher(1_000_000)
    mpps = _run_uplink(l, 1_000_000, 1 << 20, 0.0)
    ns = l.nat_get_stats()
    return {"config": 3, "desc": "NAT44+antispoof+QoS, 1M flows, 64B",
            "mpps": round(mpps, 1), "sessions": ns["sessions_created"],
            "snat": ns["packets_snat"]}


def config4():
    """QoS token-bucket slice on 1 GPU (the 8-GPU sharded form is the
    driver's SCALE run over bench.py)."""
    import bench
    import torch
    l = _gpu_launcher(1_000_000)
    NOW = 1_700_000_000
    d_np, ln = bench.gen_batch(1 << 20, 1_000_000, 0.0, 512, 13)
    p = torch.from_numpy(d_np).cuda()
    lt = torch.from_numpy(ln.view(np.int16)).cuda()
    ns = NOW * 10**9
    for k in range(3):
        l.qos(p, lt, egress=False, now_ns=ns + k)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for k in range(10):
        l.qos(p, lt, egress=False, now_ns=ns + 10**6 * k)
    torch.cuda.synchronize()
    mpps = (1 << 20) * 10 / (time.perf_counter() - t0) / 1e6
    return {"config": 4, "desc": "QoS token bucket, 1M subs, 1-GPU slice "
            "(8-GPU sharded = driver SCALE run)", "mpps": round(mpps, 1)}


def config5():
    """Full BNG (DHCP+NAT44+QoS+antispoof [+PPPoE control-plane]) at 8M
    subscribers."""
    l = _gpu_launcher(8_000_000)
    mpps = _run_uplink(l, 8_000_000, 1 << 20, 0.1)
    p50, p99 = _dhcp_latency(l, 8_000_000)
    s50, s99 = _svc_latency(l, 8_000_000)
    return {"config": 5, "desc": "full BNG, 8M subs, 64B mix",
            "mpps": round(mpps, 1), "p50_us": round(p50, 1),
            "p99_us": round(p99, 1),
            "svc_p50_us": round(s50, 1), "svc_p99_us": round(s99, 1)}


def main():
    gpu = "--cpu-only" not in sys.argv
    out = [config1()]
    if gpu:
        out += [config2(), config3(), config4(), config5()]
    for r in out:
        print(json.dumps(r))


if __name__ == "__main__":
    main()
