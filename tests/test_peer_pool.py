"""Peer-pool tests: HRW ownership, HTTP forward, health fallback
(ref pkg/pool/peer_test.go pattern — several PeerPools on localhost)."""
import time

import pytest

from bng_amd.pool.peer import PeerPool


def make_mesh(n=3, cidr="10.7.0.0/24"):
    pools = [PeerPool(f"n{i}", {}, cidr).start() for i in range(n)]
    urls = {p.node_id: p.url for p in pools}
    for p in pools:
        p.peer_urls = {k: v for k, v in urls.items() if k != p.node_id}
        for k in urls:
            p.ring.add_node(k)
    return pools


class TestPeerPool:
    def test_owner_agreement(self):
        pools = make_mesh()
        try:
            for sid in (f"sub-{i}" for i in range(20)):
                owners = {p.owner_of(sid) for p in pools}
                assert len(owners) == 1
        finally:
            for p in pools:
                p.stop()

    def test_forwarded_allocation_lands_on_owner(self):
        pools = make_mesh()
        try:
            sid = "sub-42"
            owner_id = pools[0].owner_of(sid)
            owner = next(p for p in pools if p.node_id == owner_id)
            other = next(p for p in pools if p.node_id != owner_id)
            ip = other.allocate(sid)
            assert owner.lookup(sid) == ip
            # repeated allocation is stable
            assert other.allocate(sid) == ip
            other.release(sid)
            assert owner.lookup(sid) is None
        finally:
            for p in pools:
                p.stop()

    def test_health_fallback_local(self):
        pools = make_mesh(2)
        try:
            sid = next(s for s in (f"sub-{i}" for i in range(50))
                       if pools[0].owner_of(s) == "n1")
            pools[1].stop()      # owner n1 dies
            pools[0].health_threshold = 1
            ip = pools[0].allocate(sid)   # falls back to local allocation
            assert pools[0].lookup(sid) == ip
            assert not pools[0].ring.healthy.get("n1", True)
        finally:
            pools[0].stop()

    def test_probe_marks_down_then_up(self):
        pools = make_mesh(2)
        try:
            pools[0].health_threshold = 2
            pools[1].stop()
            pools[0].probe_once()
            assert pools[0].ring.healthy["n1"]      # 1 failure < threshold
            pools[0].probe_once()
            assert not pools[0].ring.healthy["n1"]  # 2 failures
        finally:
            pools[0].stop()
