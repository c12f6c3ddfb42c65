#!/usr/bin/env python3
"""Secondary-path soaks: (1) downlink return-path 300 steps against
live session state, (2) hipGraph captured-uplink replay 500x.  One JSON
line each."""
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
    d_np, ln = bench.gen_batch(N, 1_000_000, 0.0, 512, 11)
    up = torch.from_numpy(d_np).cuda()
    w = torch.empty_like(up)
    lt = torch.from_numpy(ln.view(np.int16)).cuda()
    # create sessions, learn SNAT, build return batch
    w.copy_(up)
    l.uplink(w, lt, now_ns=ns, now_sec=NOW, sort_by_type=False)
    torch.cuda.synchronize()
    h = w.cpu().numpy()
    ret = np.zeros_like(d_np)
    ret[:, :64] = d_np[:, :64]
    ret[:, 0:6] = d_np[:, 6:12]
    ret[:, 6:12] = d_np[:, 0:6]
    ret[:, 26:30] = d_np[:, 30:34]
    ret[:, 30:34] = h[:, 26:30]
    ret[:, 34:36] = d_np[:, 36:38]
    ret[:, 36:38] = h[:, 34:36]
    d_ret = torch.from_numpy(ret).cuda()
    w_ret = torch.empty_like(d_ret)
    for k in range(3):
        w_ret.copy_(d_ret)
        l.downlink(w_ret, lt, now_ns=ns + k)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    steps = 300
    for k in range(steps):
        w_ret.copy_(d_ret)
        l.downlink(w_ret, lt, now_ns=ns + (k + 3) * 10**6)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    st = l.nat_get_stats()
    print(json.dumps({"phase": "downlink-soak300",
                      "mpps": round(N * steps / dt / 1e6, 1),
                      "dnat": st["packets_dnat"]}), flush=True)

    # hipGraph replay soak
    g = l.capture_uplink(65536, 512, sort_by_type=True)
    d2_np, l2_np = bench.gen_batch(65536, 1_000_000, 0.1, 512, 5)
    d2 = torch.from_numpy(d2_np).cuda()
    l2 = torch.from_numpy(l2_np.view(np.int16)).cuda()
    for k in range(5):
        g.run(d2, l2, ns + k)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    reps = 500
    for k in range(reps):
        g.run(d2, l2, ns + (k + 5) * 10**6)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    print(json.dumps({"phase": "hipgraph-replay500",
                      "mpps": round(65536 * reps / dt / 1e6, 1),
                      "us_per_replay": round(dt / reps * 1e6, 1)}),
          flush=True)


if __name__ == "__main__":
    main()
