"""Packet-batch steering across GPU shards via torch.distributed
all-to-all — RCCL over xGMI on MI355X, gloo on CPU for tests.

The BNG analog of expert/sequence all-to-all exchange: every rank ingests
an arbitrary traffic mix, buckets each packet by its owning subscriber
shard (MAC hashring — parallel/hashring.py, matching the reference's
nexus hashring nexus/client.go:542-575), exchanges batches so each GPU
processes only its own subscribers' state, then processes locally.

xGMI note: MI355X GPUs are connected point-to-point (7 links/GPU), so
all_to_all is per-link bound, not switch-bound; batch buckets are
exchanged as one contiguous all_to_all_single per tensor to keep message
count minimal.
"""
from __future__ import annotations

from typing import Tuple


def bucket_by_owner(data, lens, owner, world_size: int):
    """Reorder a batch so packets are grouped by owning shard.

    Returns (data_sorted, lens_sorted, counts, perm) — counts[i] is the
    number of packets owned by shard i.
    """
    import torch
    perm = torch.argsort(owner, stable=True)
    counts = torch.bincount(owner, minlength=world_size)
    return data[perm], lens[perm], counts, perm


class ExchangeBuffers:
    """Pre-allocated recv-side buffers for the steering all-to-all.

    The first real multi-GPU run must not allocate fresh recv tensors on
    the prep stream every step (caching-allocator churn + cross-stream
    lifetime hazards once RCCL is in the loop — round-1 VERDICT task 2).
    Capacity is fixed at construction; exchange(..., bufs=) writes into
    these tensors and returns narrow views.  Reuse across in-flight
    batches is the caller's job (bench keeps one set per pipeline slot,
    gated by its work-free event)."""

    def __init__(self, capacity: int, stride: int, device, world: int):
        import torch
        assert stride % 8 == 0, "packet slot stride must be a multiple of 8"
        self.capacity, self.stride, self.world = capacity, stride, world
        self.data = torch.empty((capacity, stride), dtype=torch.uint8,
                                device=device)
        self.lens64 = torch.empty(capacity, dtype=torch.int64, device=device)
        self.lens16 = torch.empty(capacity, dtype=torch.int16, device=device)
        # classify/sort scratch for the type-sorted pipeline on the
        # exchanged batch (same lifetime rules as data/lens)
        self.cls = torch.empty(capacity, dtype=torch.uint8, device=device)
        self.order = torch.empty(capacity, dtype=torch.int32, device=device)


class ExchangeOverflow(RuntimeError):
    """Received more packets than the pre-allocated exchange capacity.

    Size ExchangeBuffers for the worst shard skew you admit (bench uses
    2x the per-rank batch; the RSS arrival model keeps the expected
    crossing at the DHCP fraction, and hashring uniformity over >=100k
    subscribers bounds data skew far below 2x)."""


def exchange(data, lens, owner, group=None,
             bufs: "ExchangeBuffers" = None) -> Tuple["object", "object"]:
    """All-to-all steer packets to their owning rank.

    data: [N, stride] uint8, lens: [N] int16, owner: [N] int32/int64.
    Returns (data_recv [M, stride], lens_recv [M]) on this rank — views
    into `bufs` when given (steady-state path), fresh tensors otherwise.
    Works over nccl(RCCL) with device tensors and gloo with CPU tensors.
    """
    import torch
    import torch.distributed as dist
    world = dist.get_world_size(group)
    if world == 1:
        return data, lens
    stride = data.size(1)
    assert stride % 8 == 0, "packet slot stride must be a multiple of 8"
    data_s, lens_s, send_counts, _ = bucket_by_owner(data, lens, owner, world)

    # exchange counts
    recv_counts = torch.zeros_like(send_counts)
    dist.all_to_all_single(recv_counts, send_counts, group=group)

    in_splits = send_counts.tolist()
    out_splits = recv_counts.tolist()
    m = sum(out_splits)

    # ship packet bytes as int64 words (gloo rejects uint8/int16; identical
    # bytes either way) — one contiguous all_to_all_single per tensor
    words = stride // 8
    if bufs is not None:
        if m > bufs.capacity:
            raise ExchangeOverflow(
                f"recv {m} packets > exchange capacity {bufs.capacity}")
        assert stride == bufs.stride
        data_recv = bufs.data[:m]
        lens_recv64 = bufs.lens64[:m]
    else:
        data_recv = data.new_empty((m, stride))
        lens_recv64 = torch.empty(m, dtype=torch.int64, device=lens.device)
    dist.all_to_all_single(data_recv.view(-1).view(torch.int64),
                           data_s.contiguous().view(-1).view(torch.int64),
                           [c * words for c in out_splits],
                           [c * words for c in in_splits], group=group)
    dist.all_to_all_single(lens_recv64,
                           lens_s.contiguous().to(torch.int64), out_splits,
                           in_splits, group=group)
    if bufs is not None and lens.dtype == torch.int16:
        lens_recv = bufs.lens16[:m]
        lens_recv.copy_(lens_recv64.to(torch.int16))
        return data_recv, lens_recv
    return data_recv, lens_recv64.to(lens.dtype)


def allreduce_stats(*stat_tensors, group=None):
    """Sum per-GPU stats blocks across shards (the per-CPU-map merge the
    reference does in userspace, done over RCCL instead)."""
    import torch.distributed as dist
    for t in stat_tensors:
        dist.all_reduce(t, group=group)
