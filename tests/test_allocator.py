"""Tests for allocator: bitmap, epoch bitmap, distributed (ref
pkg/allocator, incl. distributed_integration_test.go scenarios)."""
import json

import pytest

from bng_amd.allocator.bitmap import BitmapAllocator, PoolExhaustedError
from bng_amd.allocator.distributed import (MODE_LEASE, MODE_SESSION,
                                           DistributedAllocator)
from bng_amd.allocator.epoch_bitmap import EpochBitmapAllocator, NotFoundError
from bng_amd.nexus.store import MemoryStore


class TestBitmap:
    def test_ipv4_allocate_release(self):
        a = BitmapAllocator("10.0.0.0/29", 32)   # 8 addrs, net+bcast skip
        ips = [a.allocate(f"s{i}") for i in range(6)]
        assert len(set(ips)) == 6
        assert all(p.endswith("/32") for p in ips)
        with pytest.raises(PoolExhaustedError):
            a.allocate("s7")
        a.release("s0")
        assert a.allocate("s7") == ips[0]       # reuse

    def test_idempotent_per_subscriber(self):
        a = BitmapAllocator("10.0.0.0/24", 32)
        assert a.allocate("x") == a.allocate("x")

    def test_reserved_head_tail(self):
        a = BitmapAllocator("10.0.0.0/28", 32, reserve_head=3,
                            reserve_tail=2)
        got = {a.allocate(f"s{i}") for i in range(16 - 3 - 2 - 1)}
        # .0 net, .15 bcast are inside head/tail reservations here
        assert "10.0.0.0/32" not in got and "10.0.0.1/32" not in got
        assert "10.0.0.2/32" not in got
        assert "10.0.0.14/32" not in got and "10.0.0.15/32" not in got

    def test_ipv6_prefix_delegation(self):
        a = BitmapAllocator("2001:db8::/48", 56)    # 256 delegable /56s
        p1 = a.allocate("cpe-1")
        p2 = a.allocate("cpe-2")
        assert p1.endswith("/56") and p1 != p2
        assert a.lookup_by_prefix(p1) == "cpe-1"

    def test_json_roundtrip(self):
        a = BitmapAllocator("10.0.0.0/24", 32)
        ip = a.allocate("s1")
        b = BitmapAllocator.from_json(a.to_json())
        assert b.lookup("s1") == ip
        assert b.allocate("s1") == ip
        assert b.allocate("s2") != ip


class TestEpochBitmap:
    def test_allocate_renew_expire(self):
        a = EpochBitmapAllocator("10.0.0.0/28", 32, grace_period=1)
        ip = a.allocate("s1")
        assert a.lookup("s1") == ip
        a.advance_epoch()                # still in grace
        assert a.lookup("s1") == ip
        a.advance_epoch()                # expired
        assert a.lookup("s1") is None
        # the expired slot is reallocatable: filling the whole pool
        # succeeds only if s1's old slot is reclaimed
        ips = {a.allocate(f"s{i}") for i in range(2, 16)}
        assert len(ips) == 14 and ip in ips

    def test_renew_extends(self):
        a = EpochBitmapAllocator("10.0.0.0/28", 32)
        ip = a.allocate("s1")
        for _ in range(5):
            a.advance_epoch()
            a.renew("s1")
        assert a.lookup("s1") == ip

    def test_release_immediate(self):
        a = EpochBitmapAllocator("10.0.0.0/28", 32)
        ip = a.allocate("s1")
        a.release("s1")
        assert a.lookup("s1") is None
        assert a.allocate("s2") == ip

    def test_lookup_by_ip(self):
        a = EpochBitmapAllocator("10.0.0.0/24", 32)
        ip = a.allocate("s1")
        assert a.lookup_by_ip(ip) == "s1"
        assert a.lookup_by_ip("10.0.0.250") is None

    def test_o1_epoch_advance_memory(self):
        a = EpochBitmapAllocator("10.0.0.0/16", 32)
        assert len(a.generations) == 65536 // 4   # 16KB per /16 (ref doc)

    def test_json_roundtrip(self):
        a = EpochBitmapAllocator("10.0.0.0/24", 32)
        ip = a.allocate("s1")
        b = EpochBitmapAllocator.from_json(a.to_json())
        assert b.lookup("s1") == ip
        assert b.current_epoch == a.current_epoch


class TestDistributed:
    def test_two_allocators_one_store_converge(self):
        """ref distributed_integration_test.go:52-333: two allocators over
        one shared store see each other's allocations."""
        store = MemoryStore()
        a1 = DistributedAllocator(store, "p", "10.5.0.0/24", MODE_SESSION,
                                  node_id="n1")
        a2 = DistributedAllocator(store, "p", "10.5.0.0/24", MODE_SESSION,
                                  node_id="n2")
        p1 = a1.allocate("sub-1")
        assert a2.lookup("sub-1") == p1
        # n2 must not double-assign sub-1's address
        p2 = a2.allocate("sub-2")
        assert p2 != p1
        a1.close(); a2.close()

    def test_session_mode_renewal_is_read(self):
        store = MemoryStore()
        a = DistributedAllocator(store, "p", "10.5.0.0/24", MODE_SESSION)
        p = a.allocate("sub-1")
        rec_before = store.get("alloc/p/sub-1")
        assert a.renew("sub-1") == p
        assert store.get("alloc/p/sub-1") == rec_before   # no write
        a.close()

    def test_lease_mode_epoch_expiry_cleans_store(self):
        store = MemoryStore()
        a = DistributedAllocator(store, "p", "10.5.0.0/24", MODE_LEASE,
                                 grace_period=1)
        a.allocate("sub-1")
        assert store.get("alloc/p/sub-1") is not None
        a.advance_epoch()
        a.advance_epoch()     # beyond grace: lazily cleaned
        assert store.get("alloc/p/sub-1") is None
        a.close()

    def test_lease_mode_renew_bumps_epoch(self):
        store = MemoryStore()
        a = DistributedAllocator(store, "p", "10.5.0.0/24", MODE_LEASE)
        p = a.allocate("sub-1")
        a.advance_epoch()
        assert a.renew("sub-1") == p
        a.advance_epoch()
        assert a.lookup("sub-1") == p       # renewed => survives
        a.close()

    def test_persistence_via_store_reload(self):
        store = MemoryStore()
        a = DistributedAllocator(store, "p", "10.5.0.0/24", MODE_SESSION)
        p = a.allocate("sub-1")
        a.close()
        b = DistributedAllocator(store, "p", "10.5.0.0/24", MODE_SESSION)
        assert b.lookup("sub-1") == p
        assert b.allocate("sub-2") != p
        b.close()


class TestDistributedIntegration:
    """Watch notification / concurrency / IPv6 (ref
    distributed_integration_test.go)."""

    def test_watch_notification_on_allocate_release(self):
        store = MemoryStore()
        events = []
        store.watch("alloc/p/", lambda ev: events.append(
            (ev.type, ev.key)))
        a = DistributedAllocator(store, "p", "10.5.0.0/24", MODE_SESSION)
        a.allocate("sub-1")
        a.release("sub-1")
        a.close()
        types = [t for t, _ in events]
        assert "put" in types and "delete" in types
        assert all(k.startswith("alloc/p/") for _, k in events)

    def test_concurrent_allocations_unique(self):
        import threading
        store = MemoryStore()
        allocs = [DistributedAllocator(store, "p", "10.5.0.0/24",
                                       MODE_SESSION, node_id=f"n{k}")
                  for k in range(2)]
        got, errs = {}, []

        def worker(k):
            try:
                got[k] = allocs[k % 2].allocate(f"sub-{k}")
            except Exception as e:      # pragma: no cover
                errs.append(e)
        ts = [threading.Thread(target=worker, args=(k,))
              for k in range(40)]
        [t.start() for t in ts]
        [t.join() for t in ts]
        assert not errs
        assert len(set(got.values())) == 40   # no duplicate addresses
        for a in allocs:
            a.close()

    def test_ipv6_prefix_mode(self):
        store = MemoryStore()
        a = DistributedAllocator(store, "p6", "2001:db8:100::/48",
                                 MODE_SESSION)
        p1 = a.allocate("sub-1")
        p2 = a.allocate("sub-2")
        assert p1 != p2 and p1.startswith("2001:db8:100:")
        assert a.lookup("sub-1") == p1
        b = DistributedAllocator(store, "p6", "2001:db8:100::/48",
                                 MODE_SESSION, node_id="n2")
        assert b.lookup("sub-1") == p1
        a.close(); b.close()
