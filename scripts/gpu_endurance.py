#!/usr/bin/env python3
"""Combined endurance: 1500 overlapped-style steps with EVERYTHING live
at once — periodic NAT sweeps, QoS policy churn, compliance-ring and
spoof-ring drains, stat reads — the closest single-GPU approximation of
production steady state.  One JSON line at the end."""
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np


def main():
    import bench
    import torch
    from bng_amd.dataplane.launcher import HipLauncher
    from bng_amd.dataplane.packets import ip2u32
    NOW = 1_700_000_000
    ns = NOW * 10**9
    l = HipLauncher("cuda:0", sub_log2=21, sess_log2=22, eim_log2=21,
                    subnat_log2=21, qos_log2=21, binding_log2=21)
    l.set_server_config(b"\x02\x00\x00\x00\x00\x01", ip2u32("10.255.255.1"))
    l.add_pool(1, ip2u32("10.0.0.0"), 8, ip2u32("10.255.255.1"),
               ip2u32("8.8.8.8"))
    bench.build_tables(l, 0, 1, 1_000_000, NOW)
    N = 1 << 20
    d_np, ln = bench.gen_batch(N, 1_000_000, 0.1, 512, 31)
    p = torch.from_numpy(d_np).cuda()
    w = torch.empty_like(p)
    lt = torch.from_numpy(ln.view(np.int16)).cuda()
    ips = (np.uint64(ip2u32("10.0.0.0") + 2) +
           np.arange(1_000_000, dtype=np.uint64)).astype(np.uint32)
    drained = {"nat": 0, "spoof": 0}
    for k in range(5):
        w.copy_(p)
        l.uplink(w, lt, now_ns=ns + k, now_sec=NOW, sort_by_type=True)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    steps = 1500
    rates = []
    tband = time.perf_counter()
    for k in range(steps):
        w.copy_(p)
        l.uplink(w, lt, now_ns=ns + (k + 5) * 10**6, now_sec=NOW,
                 sort_by_type=True)
        if k % 97 == 50:
            l.sweep_nat(now_ns=ns + (k + 5) * 10**6)
        if k % 53 == 20:
            ip = int(ips[(k * 977) % 1_000_000])
            l.set_qos_policy(ip, 10**9, 4 << 20, direction="ingress",
                             now_ns=ns + k * 10**6)
        if k % 211 == 100:
            drained["nat"] += len(l.drain_nat_log())
            drained["spoof"] += len(l.drain_spoof_events())
            l.get_stats(); l.nat_get_stats()
        if k % 300 == 299:
            torch.cuda.synchronize()
            now = time.perf_counter()
            rates.append(round(N * 300 / (now - tband) / 1e6, 1))
            tband = now
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    st = l.nat_get_stats()
    print(json.dumps({
        "phase": "endurance1500",
        "mpps": round(N * steps / dt / 1e6, 1),
        "per_300_step_bands": rates,
        "sessions_created": st["sessions_created"],
        "sessions_expired": st["sessions_expired"],
        "log_records_drained": drained["nat"]}), flush=True)


if __name__ == "__main__":
    main()
