#!/usr/bin/env python3
"""DHCP OFFER latency distribution: 1000 submit->replies-ready samples
per batch size, quiesced GPU (the latency half of the BASELINE metric,
deeper than bench.py's p50/p99)."""
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
    l = HipLauncher("cuda:0", sub_log2=21, sess_log2=22, eim_log2=21,
                    subnat_log2=21, qos_log2=21, binding_log2=21)
    l.set_server_config(b"\x02\x00\x00\x00\x00\x01", ip2u32("10.255.255.1"))
    l.add_pool(1, ip2u32("10.0.0.0"), 8, ip2u32("10.255.255.1"),
               ip2u32("8.8.8.8"))
    bench.build_tables(l, 0, 1, 1_000_000, NOW)
    for batch in (256, 2048, 16384):
        d_np, ln = bench.gen_batch(batch, 1_000_000, 1.0, 512, 7)
        p = torch.from_numpy(d_np).cuda()
        w = torch.empty_like(p)
        lt = torch.from_numpy(ln.view(np.int16)).cuda()
        lats = []
        for r in range(1000):
            torch.cuda.synchronize()
            t = time.perf_counter()
            w.copy_(p)
            l.dhcp_fastpath(w, lt, now_sec=NOW)
            torch.cuda.synchronize()
            lats.append((time.perf_counter() - t) * 1e6)
        lats.sort()
        n = len(lats)
        print(json.dumps({
            "batch": batch,
            "p50_us": round(lats[n // 2], 1),
            "p90_us": round(lats[int(n * .9)], 1),
            "p99_us": round(lats[int(n * .99)], 1),
            "p999_us": round(lats[int(n * .999)], 1),
            "max_us": round(lats[-1], 1),
            "per_pkt_ns_p50": round(lats[n // 2] * 1000 / batch, 1)}),
            flush=True)


if __name__ == "__main__":
    main()
