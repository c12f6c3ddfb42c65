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


def exchange(data, lens, owner, group=None) -> Tuple["object", "object"]:
    """All-to-all steer packets to their owning rank.

    data: [N, stride] uint8, lens: [N] int16, owner: [N] int32/int64.
    Returns (data_recv [M, stride], lens_recv [M]) on this rank.
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
    data_recv = data.new_empty((m, stride))
    dist.all_to_all_single(data_recv.view(-1).view(torch.int64),
                           data_s.contiguous().view(-1).view(torch.int64),
                           [c * words for c in out_splits],
                           [c * words for c in in_splits], group=group)
    lens_recv64 = torch.empty(m, dtype=torch.int64, device=lens.device)
    dist.all_to_all_single(lens_recv64,
                           lens_s.contiguous().to(torch.int64), out_splits,
                           in_splits, group=group)
    return data_recv, lens_recv64.to(lens.dtype)


def allreduce_stats(*stat_tensors, group=None):
    """Sum per-GPU stats blocks across shards (the per-CPU-map merge the
    reference does in userspace, done over RCCL instead)."""
    import torch.distributed as dist
    for t in stat_tensors:
        dist.all_reduce(t, group=group)
