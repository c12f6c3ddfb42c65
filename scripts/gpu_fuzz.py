#!/usr/bin/env python3
"""Adversarial differential fuzz for the fused uplink kernel.

Three oracles that need no hand-written routing model:
 1. sorted == unsorted: type-sorted dispatch must not change any
    verdict or rewritten byte (fresh launchers, identical tables).
 2. batch-split invariance: one batch vs the same frames in two
    halves gives identical results (cross-packet isolation; one flow
    per subscriber keeps the port rotor deterministic).
 3. golden DHCP differential: structurally-valid DHCP frames with
    mutations confined to the options region (routing stays stable)
    must match the CPU model byte-for-byte.
Prints one JSON line; exit 1 on any mismatch."""
import json
import os
import random
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np

from bng_amd.dataplane import abi
from bng_amd.dataplane.launcher import GoldenLauncher, HipLauncher
from bng_amd.dataplane.packets import (build_dhcp_request, build_ipv4,
                                       ip2u32, mac_bytes)

NOW = 1_700_000_000
NOW_NS = NOW * 10**9


def fresh(golden=False):
    l = GoldenLauncher() if golden else HipLauncher(
        "cuda:0", sub_log2=14, sess_log2=16, eim_log2=15, subnat_log2=14,
        qos_log2=14, binding_log2=14, n_pools=64)
    if golden:
        l.dp.now_ns = NOW_NS
    l.set_server_config(mac_bytes("02:00:00:00:00:01"),
                        ip2u32("10.0.0.1"))
    l.add_pool(1, ip2u32("10.0.1.0"), 24, ip2u32("10.0.1.1"),
               ip2u32("8.8.8.8"), ip2u32("1.1.1.1"), 3600)
    l.set_antispoof_config(default_mode=abi.AS_DISABLED)
    # 4096 NAT subscribers: UNIQUE private IP per data frame in a batch
    # keeps the port rotor deterministic (same rule as the unit
    # differentials); 250 DHCP subscribers for the fast-path slice
    for k in range(250):
        l.add_subscriber(mac_bytes(f"aa:bb:00:00:00:{k:02x}"), 1,
                         ip2u32(f"10.0.1.{k + 2}"), NOW + 600)
    for k in range(4096):
        priv = ip2u32("10.0.4.0") + k
        l.add_subscriber_nat(priv, ip2u32("203.0.113.9"),
                             1024 + (k % 500) * 128,
                             1024 + (k % 500) * 128 + 127,
                             subscriber_id=k)
        l.set_qos_policy(priv, 0, 0, direction="ingress", now_ns=NOW_NS)
    return l


def gen(rng, n):
    frames = []
    for i in range(n):
        r = rng.random()
        if r < 0.25:
            f = bytearray(build_dhcp_request(
                f"aa:bb:00:00:00:{i % 250:02x}",
                rng.choice([1, 3]), xid=rng.randrange(1 << 32)))
            for _ in range(rng.randrange(0, 6)):
                f[rng.randrange(46, len(f))] ^= 1 << rng.randrange(8)
            frames.append(bytes(f))
        elif r < 0.55:
            f = bytearray(build_ipv4(
                f"aa:cc:00:00:{(i >> 8) & 0xFF:02x}:{i & 0xFF:02x}",
                "02:00:00:00:00:01",
                ip2u32("10.0.4.0") + i,       # unique subscriber/frame
                ip2u32("93.184.216.34"),
                proto=rng.choice([17, 6, 1, 47]),
                sport=1024 + i % 60000, dport=rng.choice([53, 80, 5060]),
                payload=bytes(rng.randrange(256)
                              for _ in range(rng.randrange(0, 24)))))
            frames.append(bytes(f))
        elif r < 0.75:
            base = build_dhcp_request("aa:bb:00:00:00:01", 1)
            frames.append(base[:rng.randrange(14, len(base))])
        else:
            frames.append(bytes(rng.randrange(256)
                                for _ in range(rng.randrange(14, 200))))
    return frames


def run(l, frames, sort):
    d, lens = l.make_batch(frames, stride=512)
    v, ol = l.uplink(d, lens, now_ns=NOW_NS, now_sec=NOW,
                     sort_by_type=sort)
    return (v.cpu().tolist(), ol.cpu().numpy().view(np.uint16).tolist(),
            d.cpu().numpy())


def main():
    rng = random.Random(int(os.environ.get("FUZZ_SEED", "20260913")))
    rounds = int(os.environ.get("FUZZ_ROUNDS", 12))
    bsz = 4096
    bad = {"sorted_vs_unsorted": 0, "split": 0, "dhcp_golden": 0}
    total = 0
    for rd in range(rounds):
        frames = gen(rng, bsz)
        total += bsz
        v1, o1, h1 = run(fresh(), frames, sort=False)
        v2, o2, h2 = run(fresh(), frames, sort=True)
        if v1 != v2 or o1 != o2 or not (h1 == h2).all():
            bad["sorted_vs_unsorted"] += 1
        la, lb = fresh(), fresh()
        va, oa, ha = run(la, frames, sort=True)
        half = bsz // 2
        vb1, ob1, hb1 = run(lb, frames[:half], sort=True)
        vb2, ob2, hb2 = run(lb, frames[half:], sort=True)
        if va != vb1 + vb2 or oa != ob1 + ob2 or \
                not (ha == np.concatenate([hb1, hb2])).all():
            bad["split"] += 1
    # golden DHCP differential: valid + options-mutated DHCP only
    g = fresh(golden=True)
    k = fresh()
    dh = [f for f in gen(rng, 8192)
          if len(f) > 240 and f[12:14] == b"\x08\x00" and f[23] == 17]
    res = g.process_dhcp(dh, now_sec=NOW)
    d, lens = k.make_batch(dh, stride=512)
    v, ol = k.dhcp_fastpath(d, lens, now_sec=NOW)
    v = v.cpu().tolist()
    ol = ol.cpu().numpy().view(np.uint16).tolist()
    host = d.cpu().numpy()
    for i in range(len(dh)):
        vc, out = res[i]
        if v[i] != vc or bytes(host[i][:ol[i]]) != out:
            bad["dhcp_golden"] += 1
    total += len(dh)
    print(json.dumps({"frames": total, "mismatch_rounds": bad}),
          flush=True)
    sys.exit(1 if any(bad.values()) else 0)


if __name__ == "__main__":
    main()
