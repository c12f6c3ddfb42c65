"""Subscriber -> shard ownership hashing.

The multi-GPU analog of the reference's two sharding mechanisms:
  * the Nexus deterministic FNV hashring (pkg/nexus/client.go:542-575,
    AllocateIPForSubscriber -> fnv(subscriberID) mod hosts), and
  * rendezvous/HRW hashing over the peer set (pkg/pool/peer.go:721-760).

owner_of_* must match the device-side shard_owner_kernel bit-for-bit
(bng_kernels.hip): owner = mix64(key) % n_shards.
"""
from __future__ import annotations

from typing import Dict, List, Optional, Sequence

from ..dataplane.abi import fnv1a64, mac_to_u64, mix64


def owner_of_mac(mac, n_shards: int) -> int:
    key = mac if isinstance(mac, int) else mac_to_u64(bytes(mac))
    return mix64(key) % n_shards


def owner_of_ip(ip: int, n_shards: int) -> int:
    return mix64(ip) % n_shards


def nexus_hash_ip(subscriber_id: str, network: int, n_hosts: int) -> int:
    """Deterministic subscriber->IP inside a pool: FNV(subscriberID) mod
    usable hosts, skipping network/broadcast/gateway — the reference's
    RADIUS-time allocation invariant (nexus/client.go:542-575)."""
    if n_hosts <= 3:
        raise ValueError("pool too small")
    idx = fnv1a64(subscriber_id.encode()) % (n_hosts - 3)
    return network + 2 + idx     # skip .0 (network), .1 (gateway)


class RendezvousRing:
    """Highest-random-weight (HRW) node selection with health-aware
    fallback (ref pool/peer.go:721-760 + :242-268)."""

    def __init__(self, nodes: Sequence[str]):
        self.nodes: List[str] = list(nodes)
        self.healthy: Dict[str, bool] = {n: True for n in self.nodes}

    def add_node(self, node: str):
        if node not in self.nodes:
            self.nodes.append(node)
            self.healthy[node] = True

    def remove_node(self, node: str):
        if node in self.nodes:
            self.nodes.remove(node)
            self.healthy.pop(node, None)

    def set_healthy(self, node: str, ok: bool):
        if node in self.healthy:
            self.healthy[node] = ok

    @staticmethod
    def _weight(key: str, node: str) -> int:
        return mix64(fnv1a64(f"{key}|{node}".encode()))

    def ranked(self, key: str) -> List[str]:
        return sorted(self.nodes, key=lambda n: self._weight(key, n),
                      reverse=True)

    def owner(self, key: str, only_healthy: bool = True) -> Optional[str]:
        for n in self.ranked(key):
            if not only_healthy or self.healthy.get(n, False):
                return n
        return self.ranked(key)[0] if self.nodes else None
