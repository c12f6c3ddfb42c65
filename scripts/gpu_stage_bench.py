#!/usr/bin/env python3
"""Stage decomposition of the uplink pipeline: time each dataplane kernel
separately on the same batch to locate the cost (run under gpurun)."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

import bench
from bng_amd.dataplane.launcher import HipLauncher
from bng_amd.dataplane.packets import ip2u32

N = int(os.environ.get("N", 524288))
NOW = 1_700_000_000


def timeit(fn, reps=10):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(reps):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / reps * 1e6


def main():
    l = HipLauncher("cuda:0", sub_log2=21, sess_log2=22, eim_log2=21,
                    subnat_log2=21, qos_log2=21, binding_log2=21)
    l.set_server_config(b"\x02\x00\x00\x00\x00\x01", ip2u32("10.255.255.1"))
    l.add_pool(1, ip2u32("10.0.0.0"), 8, ip2u32("10.255.255.1"),
               ip2u32("8.8.8.8"))
    bench.build_tables(l, 0, 1, 1_000_000, NOW)

    # pure-data batch (64B UDP) and pure-DHCP batch
    data_np, lens_np = bench.gen_batch(N, 1_000_000, 0.0, 512, 11)
    dhcp_np, dhcp_lens_np = bench.gen_batch(N // 8, 1_000_000, 1.0, 512, 12)
    mix_np, mix_lens_np = bench.gen_batch(N, 1_000_000, 0.1, 512, 13)

    def dev(a, ln):
        return (torch.from_numpy(a).cuda(),
                torch.from_numpy(ln.view(np.int16)).cuda())

    d_data, l_data = dev(data_np, lens_np)
    d_dhcp, l_dhcp = dev(dhcp_np, dhcp_lens_np)
    d_mix, l_mix = dev(mix_np, mix_lens_np)
    w = torch.empty_like(d_data)
    w_dhcp = torch.empty_like(d_dhcp)
    w_mix = torch.empty_like(d_mix)

    ns = [NOW * 10**9]

    def tick():
        ns[0] += 10**6
        return ns[0]

    res = {}
    res["copy_268MB"] = timeit(lambda: w.copy_(d_data))
    res["antispoof"] = timeit(
        lambda: l.antispoof(d_data, l_data, now_ns=tick()))
    # nat needs writable batch: copy+nat fused timing minus copy
    def nat_step():
        w.copy_(d_data)
        l.nat44(w, l_data, egress=True, now_ns=tick())
    res["copy+nat44"] = timeit(nat_step)
    res["qos"] = timeit(lambda: l.qos(d_data, l_data, egress=False,
                                      now_ns=tick()))
    def dhcp_step():
        w_dhcp.copy_(d_dhcp)
        l.dhcp_fastpath(w_dhcp, l_dhcp, now_sec=NOW)
    res[f"copy+dhcp_{N//8}"] = timeit(dhcp_step)
    def up(sort):
        w_mix.copy_(d_mix)
        l.uplink(w_mix, l_mix, now_ns=tick(), now_sec=NOW,
                 sort_by_type=sort)
    res["copy+uplink_nosort"] = timeit(lambda: up(False))
    res["copy+uplink_sorted"] = timeit(lambda: up(True))
    cls = torch.empty(N, dtype=torch.uint8, device="cuda")
    res["classify"] = timeit(lambda: l.ext.pkt_class(d_mix, l_mix, cls))
    res["argsort"] = timeit(
        lambda: torch.argsort(cls, stable=True).to(torch.int32))

    # downlink: learn each flow's SNAT (ip, port) from the rewritten
    # uplink batch, build the return traffic, DNAT+QoS-egress it
    w.copy_(d_data)
    l.uplink(w, l_data, now_ns=tick(), now_sec=NOW, sort_by_type=False)
    torch.cuda.synchronize()
    up_host = w.cpu().numpy()
    nat_ip = up_host[:, 26:30].copy()
    nat_port = up_host[:, 34:36].copy()
    ret = np.zeros_like(data_np)
    ret[:, :64] = data_np[:, :64]
    ret[:, 0:6] = data_np[:, 6:12]          # eth swap
    ret[:, 6:12] = data_np[:, 0:6]
    ret[:, 26:30] = data_np[:, 30:34]       # ip src = orig dst
    ret[:, 30:34] = nat_ip                  # ip dst = SNAT public
    ret[:, 34:36] = data_np[:, 36:38]       # sport = orig dport
    ret[:, 36:38] = nat_port                # dport = SNAT port
    ret[:, 24:26] = 0     # ip csum zeroed: kernels update incrementally,
    ret[:, 40:42] = 0     # throughput timing does not verify checksums
    d_ret = torch.from_numpy(ret).cuda()
    w_ret = torch.empty_like(d_ret)

    def down():
        w_ret.copy_(d_ret)
        l.downlink(w_ret, l_data, now_ns=tick())
    res["copy+downlink"] = timeit(down)

    for k, v in res.items():
        print(f"{k:24s} {v:9.1f} us  ({N / v:.0f} pkt/us)" if v else k)
    print(f"nat44 alone ~= {res['copy+nat44'] - res['copy_268MB']:.1f} us")
    st = l.nat_get_stats()
    print("nat stats:", {k: v for k, v in st.items() if v})


if __name__ == "__main__":
    main()
