#!/usr/bin/env python3
"""Minimal uplink-pipeline driver for rocprofv3 PMC collection —
round-1 VERDICT task 5: measure TA_ADDR_STALLED_BY_* attribution to
close the kernel-plateau question (profiles/ROUND2_NOTES.md: six
experiments showed a ~0.6 ns/pkt service floor with TA/TCP ~100% busy;
the missing datum is whether TA stalls are caused by TC (L2 return
path) or TD (texture data path)).

Run under:  rocprofv3 --pmc <counters> -d <dir> -- python scripts/ta_probe.py
Keep it SHORT: PMC multiplexing replays kernels.
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np


def main():
    import torch
    import bench
    from scripts.run_configs import _gpu_launcher

    steps = int(sys.argv[1]) if len(sys.argv) > 1 else 6
    l = _gpu_launcher(1_000_000)
    NOW = 1_700_000_000
    d_np, ln = bench.gen_batch(1 << 20, 1_000_000, 0.1, 512, 11)
    p = torch.from_numpy(d_np).cuda()
    w = torch.empty_like(p)
    lt = torch.from_numpy(ln.view(np.int16)).cuda()
    ns = NOW * 10**9
    for k in range(2):
        w.copy_(p)
        l.uplink(w, lt, now_ns=ns + k * 10**6, now_sec=NOW,
                 sort_by_type=True)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for k in range(steps):
        w.copy_(p)
        l.uplink(w, lt, now_ns=ns + (2 + k) * 10**6, now_sec=NOW,
                 sort_by_type=True)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    print(f"[ta_probe] {steps} steps, "
          f"{(1 << 20) * steps / dt / 1e6:.0f} Mpps")


if __name__ == "__main__":
    main()
