"""Multi-process (gloo, world_size=2) tests for the shard-steering path —
covers the distributed exchange on CPU so the RCCL path is correct by
construction (the driver runs the real multi-GPU bench)."""
import os

import numpy as np
import pytest
import torch
import torch.multiprocessing as mp

from bng_amd.dataplane import abi
from bng_amd.parallel.hashring import (RendezvousRing, nexus_hash_ip,
                                       owner_of_ip, owner_of_mac)


def test_owner_matches_numpy_mix():
    import bench
    ips = np.array([ip for ip in range(1000, 1100)], dtype=np.uint64)
    owners = bench.mix64_np(ips) % np.uint64(4)
    for i, ip in enumerate(ips):
        assert owner_of_ip(int(ip), 4) == int(owners[i])


def test_rendezvous_ring_stability_and_fallback():
    ring = RendezvousRing(["a", "b", "c"])
    owners = {k: ring.owner(f"sub{k}") for k in range(100)}
    # stable under repeat
    assert owners == {k: ring.owner(f"sub{k}") for k in range(100)}
    # all nodes used
    assert set(owners.values()) == {"a", "b", "c"}
    # health-aware fallback (ref pool/peer.go:242-268)
    ring.set_healthy(owners[0], False)
    new_owner = ring.owner("sub0")
    assert new_owner != owners[0]
    ring.set_healthy(owners[0], True)
    assert ring.owner("sub0") == owners[0]
    # removing a node only remaps its keys
    before = {k: ring.owner(f"sub{k}") for k in range(100)}
    ring.remove_node("c")
    after = {k: ring.owner(f"sub{k}") for k in range(100)}
    for k in range(100):
        if before[k] != "c":
            assert after[k] == before[k]


def test_nexus_hash_deterministic():
    ip1 = nexus_hash_ip("sub-001", 0x0A000100, 256)
    assert ip1 == nexus_hash_ip("sub-001", 0x0A000100, 256)
    assert 0x0A000102 <= ip1 < 0x0A000100 + 256


def _exchange_worker(rank, world, rendezvous_file, q):
    import torch.distributed as dist
    dist.init_process_group(
        "gloo", init_method=f"file://{rendezvous_file}",
        rank=rank, world_size=world)
    from bng_amd.parallel.sharding import exchange

    n, stride = 64, 32
    rng = np.random.default_rng(100 + rank)
    data = rng.integers(0, 255, size=(n, stride), dtype=np.uint8)
    # stamp rank + index for provenance; owner in byte 2
    owners = rng.integers(0, world, size=n, dtype=np.int64)
    data[:, 0] = rank
    data[:, 1] = np.arange(n, dtype=np.uint8)
    data[:, 2] = owners.astype(np.uint8)
    lens = np.full(n, stride, dtype=np.uint16)

    d = torch.from_numpy(data)
    l = torch.from_numpy(lens.view(np.int16))
    o = torch.from_numpy(owners)
    d2, l2 = exchange(d, l, o)
    got = d2.numpy()
    # every received packet belongs to this rank
    assert (got[:, 2] == rank).all()
    q.put((rank, got[:, 0].tolist(), got[:, 1].tolist()))
    dist.barrier()
    dist.destroy_process_group()


def test_exchange_gloo_world2(tmp_path):
    world = 2
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    rv = str(tmp_path / "rdv")
    procs = [ctx.Process(target=_exchange_worker, args=(r, world, rv, q))
             for r in range(world)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(world):
        rank, srcs, idxs = q.get(timeout=120)
        results[rank] = list(zip(srcs, idxs))
    for p in procs:
        p.join(timeout=120)
        assert p.exitcode == 0
    # conservation: 64 packets per source rank, none lost or duplicated
    all_pkts = [p for r in results.values() for p in r]
    assert len(all_pkts) == world * 64
    assert len(set(all_pkts)) == world * 64


def test_bench_rss_arrival_model():
    """bench.gen_batch(world>1): data packets must arrive pre-steered to
    their IP-shard owner (the NIC-RSS analog); DHCP may land anywhere."""
    import numpy as np
    import bench
    from bng_amd.dataplane.packets import ip2u32
    world = 4
    for rank in range(2):
        data, lens = bench.gen_batch(4096, 100_000, 0.1, 512,
                                     seed=5, rank=rank, world=world)
        is_dhcp = lens > 64
        ips = data[:, 26:30].astype(np.uint32)
        ip_u32 = ((ips[:, 0] << 24) | (ips[:, 1] << 16) |
                  (ips[:, 2] << 8) | ips[:, 3]).astype(np.uint64)
        owners = bench.mix64_np(ip_u32) % np.uint64(world)
        assert (owners[~is_dhcp] == rank).all()
        # full-shuffle mode really shuffles
        data2, lens2 = bench.gen_batch(4096, 100_000, 0.0, 512,
                                       seed=5, rank=rank, world=world,
                                       steer_all=True)
        ips2 = data2[:, 26:30].astype(np.uint32)
        ip2_u32 = ((ips2[:, 0] << 24) | (ips2[:, 1] << 16) |
                   (ips2[:, 2] << 8) | ips2[:, 3]).astype(np.uint64)
        own2 = bench.mix64_np(ip2_u32) % np.uint64(world)
        assert len(set(own2.tolist())) == world


def _pipeline_order_worker(rank, world, rendezvous_file, q):
    """Replicates bench.py's overlapped-step CALL SEQUENCE: prep(0),
    then for each step k the processing of batch k followed by prep(k+1)
    — i.e., exchange() invoked one batch AHEAD of consumption, the
    collective-order pattern the multi-GPU SCALE run relies on.  gloo
    deadlocks (and the test times out) if any rank's order diverges."""
    import torch.distributed as dist
    dist.init_process_group(
        "gloo", init_method=f"file://{rendezvous_file}",
        rank=rank, world_size=world)
    from bng_amd.parallel.sharding import exchange
    import bench

    steps, nbuf = 6, 4
    n, stride = 256, 512
    total_received = 0
    batches = [None] * nbuf

    def prep(k):
        d_np, l_np = bench.gen_batch(n, 10_000, 0.25, stride,
                                     seed=1000 + rank * 100 + k,
                                     rank=rank, world=world)
        # RSS model: DHCP frames may be foreign, data frames are local
        from bng_amd.parallel.hashring import owner_of_ip, owner_of_mac
        owners = []
        for i in range(n):
            f = d_np[i]
            if l_np[i] > 64:      # DHCP: owner by chaddr MAC
                owners.append(owner_of_mac(
                    bytes(f[70:76]), world))
            else:                 # data: owner by src ip
                owners.append(owner_of_ip(
                    int.from_bytes(bytes(f[26:30]), "big"), world))
        d = torch.from_numpy(d_np.reshape(n, stride))
        l = torch.from_numpy(l_np.view(np.int16))
        o = torch.tensor(owners, dtype=torch.int64)
        batches[k % nbuf] = exchange(d, l, o)

    prep(0)
    for k in range(steps):
        d, l = batches[k % nbuf]
        total_received += l.numel()
        prep(k + 1)
    dist.barrier()
    q.put((rank, total_received))
    dist.destroy_process_group()


def test_bench_step_order_symmetry_gloo(tmp_path):
    """The SCALE-run collective pattern completes without deadlock and
    conserves packets across ranks."""
    world = 2
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    rv = str(tmp_path / "rdv2")
    procs = [ctx.Process(target=_pipeline_order_worker,
                         args=(r, world, rv, q)) for r in range(world)]
    for p in procs:
        p.start()
    got = {}
    for _ in range(world):
        rank, n = q.get(timeout=180)
        got[rank] = n
    for p in procs:
        p.join(timeout=180)
        assert p.exitcode == 0
    assert sum(got.values()) == world * 6 * 256   # conservation


def _prealloc_worker(rank, world, rendezvous_file, q, skew):
    """Steady-state exchange through pre-allocated ExchangeBuffers over
    several steps with UNEVEN splits: 'skew' concentrates ownership so
    per-rank recv counts differ wildly step to step (the shape the first
    real SCALE run sees; round-1 VERDICT task 2)."""
    import torch.distributed as dist
    dist.init_process_group(
        "gloo", init_method=f"file://{rendezvous_file}",
        rank=rank, world_size=world)
    from bng_amd.parallel.sharding import (ExchangeBuffers,
                                           ExchangeOverflow, exchange)
    n, stride, steps = 128, 64, 5
    bufs = ExchangeBuffers(n * world, stride, "cpu", world)
    total, overflowed = 0, False
    for s in range(steps):
        rng = np.random.default_rng(7000 + rank * 100 + s)
        data = rng.integers(0, 255, size=(n, stride), dtype=np.uint8)
        if skew == "all_to_zero":
            owners = np.zeros(n, dtype=np.int64)
        elif skew == "rotating":       # rank r sends all to (r+s) % world
            owners = np.full(n, (rank + s) % world, dtype=np.int64)
        else:                          # uneven random
            owners = np.minimum(
                rng.integers(0, world * 2, size=n), world - 1
            ).astype(np.int64)
        data[:, 2] = owners.astype(np.uint8)
        lens = np.full(n, stride, dtype=np.uint16)
        d2, l2 = exchange(torch.from_numpy(data),
                          torch.from_numpy(lens.view(np.int16)),
                          torch.from_numpy(owners), bufs=bufs)
        assert (d2.numpy()[:, 2] == rank).all()
        assert l2.dtype == torch.int16
        total += l2.numel()
    # overflow must raise, not corrupt: tiny capacity, ring ownership so
    # EVERY rank receives n > 4 and raises after the counts exchange but
    # before posting the data all-to-all (no deadlock — symmetric abort)
    tiny = ExchangeBuffers(4, stride, "cpu", world)
    owners = np.full(n, (rank + 1) % world, dtype=np.int64)
    try:
        exchange(torch.from_numpy(data), torch.from_numpy(lens.view(np.int16)),
                 torch.from_numpy(owners), bufs=tiny)
    except ExchangeOverflow:
        overflowed = True
    dist.barrier()
    q.put((rank, total, overflowed))
    dist.destroy_process_group()


def _run_prealloc(world, tmp_path, skew, expected_total):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    rv = str(tmp_path / f"rdv_{world}_{skew}")
    procs = [ctx.Process(target=_prealloc_worker,
                         args=(r, world, rv, q, skew)) for r in range(world)]
    for p in procs:
        p.start()
    got = {}
    for _ in range(world):
        rank, n, ovf = q.get(timeout=240)
        got[rank] = (n, ovf)
    for p in procs:
        p.join(timeout=240)
        assert p.exitcode == 0
    assert sum(n for n, _ in got.values()) == expected_total
    assert all(ovf for _, ovf in got.values()), \
        "every rank must see ExchangeOverflow on tiny capacity"
    return got


def test_exchange_prealloc_world4_uneven(tmp_path):
    got = _run_prealloc(4, tmp_path, "uneven", 4 * 5 * 128)
    # uneven skew: rank 3 owns ~9/16 of traffic
    assert got[3][0] > got[1][0]


def test_exchange_prealloc_world4_all_to_zero(tmp_path):
    got = _run_prealloc(4, tmp_path, "all_to_zero", 4 * 5 * 128)
    assert got[0][0] == 4 * 5 * 128         # rank 0 got everything
    assert got[1][0] == 0


def test_exchange_prealloc_world8_rotating(tmp_path):
    """World-8 rehearsal — the driver's SCALE shape — with per-step
    rotating hot-spot ownership (each step one rank receives 8x)."""
    got = _run_prealloc(8, tmp_path, "rotating", 8 * 5 * 128)


def _steer_all_worker(rank, world, rendezvous_file, q):
    """bench.py --steer-all rehearsal: every packet crosses the exchange,
    owners from the real shard_owner hash over generated traffic."""
    import torch.distributed as dist
    dist.init_process_group(
        "gloo", init_method=f"file://{rendezvous_file}",
        rank=rank, world_size=world)
    from bng_amd.parallel.sharding import ExchangeBuffers, exchange
    from bng_amd.parallel.hashring import owner_of_ip, owner_of_mac
    import bench
    n, stride = 512, 512
    d_np, l_np = bench.gen_batch(n, 10_000, 0.1, stride, seed=31 + rank,
                                 rank=rank, world=world, steer_all=True)
    owners = np.empty(n, dtype=np.int64)
    for i in range(n):
        if l_np[i] > 64:
            owners[i] = owner_of_mac(bytes(d_np[i, 70:76]), world)
        else:
            owners[i] = owner_of_ip(
                int.from_bytes(bytes(d_np[i, 26:30]), "big"), world)
    bufs = ExchangeBuffers(2 * n, stride, "cpu", world)
    d2, l2 = exchange(torch.from_numpy(d_np),
                      torch.from_numpy(l_np.view(np.int16)),
                      torch.from_numpy(owners), bufs=bufs)
    # every received data packet's src IP must hash to this rank
    got = d2.numpy()
    l2 = l2.numpy().view(np.uint16)
    for i in range(got.shape[0]):
        if l2[i] == 64:
            assert owner_of_ip(
                int.from_bytes(bytes(got[i, 26:30]), "big"), world) == rank
        else:
            assert owner_of_mac(bytes(got[i, 70:76]), world) == rank
    dist.barrier()
    q.put((rank, got.shape[0]))
    dist.destroy_process_group()


def test_steer_all_world4_preallocated(tmp_path):
    world = 4
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    rv = str(tmp_path / "rdv_sa")
    procs = [ctx.Process(target=_steer_all_worker, args=(r, world, rv, q))
             for r in range(world)]
    for p in procs:
        p.start()
    tot = 0
    for _ in range(world):
        _, n = q.get(timeout=240)
        tot += n
    for p in procs:
        p.join(timeout=240)
        assert p.exitcode == 0
    assert tot == world * 512


def _full_sequence_worker(rank, world, rendezvous_file, q):
    """The COMPLETE bench collective sequence at world N over gloo:
    overlapped prep/exchange steps, end barrier+MAX-allreduce, the
    rank-0-only latency phase while others hold at a barrier, final
    barrier — the exact order the 8-GPU SCALE run executes."""
    import torch.distributed as dist
    dist.init_process_group(
        "gloo", init_method=f"file://{rendezvous_file}",
        rank=rank, world_size=world)
    from bng_amd.parallel.hashring import owner_of_ip, owner_of_mac
    from bng_amd.parallel.sharding import ExchangeBuffers, exchange
    import bench

    steps, nbuf, n, stride = 5, 4, 128, 512
    bufs = [ExchangeBuffers(4 * n, stride, "cpu", world)
            for _ in range(nbuf)]
    batches = [None] * nbuf

    def prep(k):
        d_np, l_np = bench.gen_batch(n, 5_000, 0.2, stride,
                                     seed=4000 + rank * 100 + k,
                                     rank=rank, world=world)
        owners = []
        for i in range(n):
            if l_np[i] > 64:
                owners.append(owner_of_mac(bytes(d_np[i, 70:76]), world))
            else:
                owners.append(owner_of_ip(
                    int.from_bytes(bytes(d_np[i, 26:30]), "big"), world))
        batches[k % nbuf] = exchange(
            torch.from_numpy(d_np), torch.from_numpy(l_np.view(np.int16)),
            torch.tensor(owners, dtype=torch.int64), bufs=bufs[k % nbuf])

    prep(0)
    total = 0
    for k in range(steps):
        d, l = batches[k % nbuf]
        total += l.numel()
        prep(k + 1)
    dist.barrier()
    e = torch.tensor([float(rank + 1)])
    dist.all_reduce(e, op=dist.ReduceOp.MAX)     # elapsed MAX pattern
    assert e.item() == world
    if rank == 0:
        pass                                      # latency phase (rank 0)
    dist.barrier()                                # others hold here
    dist.barrier()                                # post-svc/host-io hold
    q.put((rank, total))
    dist.destroy_process_group()


def test_full_bench_sequence_world4(tmp_path):
    world = 4
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    rv = str(tmp_path / "rdv_full4")
    procs = [ctx.Process(target=_full_sequence_worker,
                         args=(r, world, rv, q)) for r in range(world)]
    for p in procs:
        p.start()
    tot = 0
    for _ in range(world):
        _, t = q.get(timeout=240)
        tot += t
    for p in procs:
        p.join(timeout=240)
        assert p.exitcode == 0
    assert tot == world * 5 * 128
