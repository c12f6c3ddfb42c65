"""Chaos/partition scenarios (ref pkg/resilience/partition_test.go:112-810
named scenarios: NetworkPartition, PartitionRecovery, SplitBrainConflicts,
GracefulDegradation, RequestQueuing, PoolExhaustion, RADIUSCachedAuth,
AccountingBuffering, ReauthenticationAfterRecovery)."""
import time

import pytest

from bng_amd.radius.client import Client as RadiusClient
from bng_amd.radius.server import RadiusServer
from bng_amd.resilience.conflict_detector import (Allocation,
                                                  ConflictDetector)
from bng_amd.resilience.manager import (STATE_ONLINE, STATE_PARTITIONED,
                                        STATE_RECOVERING, Manager)
from bng_amd.resilience.pool_monitor import (LEVEL_CRITICAL, LEVEL_EXHAUSTED,
                                             LEVEL_OK, LEVEL_WARNING,
                                             PoolMonitor)
from bng_amd.resilience.radius_handler import (MODE_ALLOW, MODE_CACHED,
                                               MODE_REJECT, ResilientRadius)
from bng_amd.resilience.request_queue import RequestQueue


class TestNetworkPartition:
    def test_partition_and_recovery_transitions(self):
        healthy = {"v": True}
        m = Manager(lambda: healthy["v"], failure_threshold=2,
                    recovery_checks=2)
        transitions = []
        m.on_transition(lambda a, b: transitions.append((a, b)))
        assert m.check_once() == STATE_ONLINE
        healthy["v"] = False
        m.check_once()
        assert m.state == STATE_ONLINE         # 1 failure < threshold
        m.check_once()
        assert m.state == STATE_PARTITIONED
        assert m.is_partitioned and m.partition_duration() >= 0
        healthy["v"] = True
        m.check_once()
        assert m.state == STATE_PARTITIONED    # 1 ok < recovery_checks
        m.check_once()
        assert m.state == STATE_RECOVERING
        m.check_once()
        assert m.state == STATE_ONLINE
        assert transitions == [(STATE_ONLINE, STATE_PARTITIONED),
                               (STATE_PARTITIONED, STATE_RECOVERING),
                               (STATE_RECOVERING, STATE_ONLINE)]


class TestRADIUSDegradation:
    def make(self, mode):
        srv = RadiusServer(b"sec", users={
            "alice": {"password": "pw", "policy": "gold"}}).start()
        client = RadiusClient([srv.addr], b"sec", timeout=0.2, retries=1)
        return srv, ResilientRadius(client, mode=mode)

    def test_cached_auth_during_partition(self):
        """RADIUSCachedAuth scenario."""
        srv, rr = self.make(MODE_CACHED)
        try:
            res = rr.authenticate("alice", "pw")
            assert res.success
            srv.drop_requests = True           # partition
            res2 = rr.authenticate("alice", "pw")
            assert res2.success                # served from cache
            assert res2.policy_name == "gold"
            assert rr.stats["cache_answers"] == 1
            # unknown user still rejected
            assert not rr.authenticate("mallory", "x").success
        finally:
            srv.stop()

    def test_reject_mode(self):
        srv, rr = self.make(MODE_REJECT)
        try:
            srv.drop_requests = True
            assert not rr.authenticate("alice", "pw").success
            assert rr.stats["rejects"] == 1
        finally:
            srv.stop()

    def test_allow_mode_graceful_degradation(self):
        """GracefulDegradation scenario."""
        srv, rr = self.make(MODE_ALLOW)
        try:
            srv.drop_requests = True
            assert rr.authenticate("anyone", "x").success
            assert rr.stats["allow_answers"] == 1
        finally:
            srv.stop()

    def test_accounting_buffered_and_replayed(self):
        """AccountingBuffering scenario."""
        from bng_amd.radius import packet as rp
        srv, rr = self.make(MODE_CACHED)
        try:
            srv.drop_requests = True
            assert not rr.send_accounting(rp.ACCT_START, "sess-1", "alice")
            assert rr.stats["acct_buffered"] == 1
            srv.drop_requests = False
            assert rr.replay_buffered() == 1
            assert len(srv.acct_records) == 1
        finally:
            srv.stop()

    def test_reauthentication_after_recovery(self):
        """ReauthenticationAfterRecovery: after the partition heals, a
        fresh auth goes to the real server again (and refreshes cache)."""
        srv, rr = self.make(MODE_CACHED)
        try:
            rr.authenticate("alice", "pw")
            srv.drop_requests = True
            rr.authenticate("alice", "pw")     # cached
            srv.drop_requests = False
            srv.users["alice"]["policy"] = "silver"
            res = rr.authenticate("alice", "pw")
            assert res.policy_name == "silver"
            assert rr.cache["alice"].result.policy_name == "silver"
        finally:
            srv.stop()


class TestPoolPressure:
    def test_thresholds_and_short_lease(self):
        """PoolExhaustion scenario."""
        u = {"v": 0.5}
        events = []
        pm = PoolMonitor(lambda: u["v"], normal_lease=3600, short_lease=60)
        pm.on_level_change(lambda a, b, x: events.append(b))
        assert pm.check() == LEVEL_OK
        assert pm.effective_lease_time() == 3600
        u["v"] = 0.85
        assert pm.check() == LEVEL_WARNING
        u["v"] = 0.95
        assert pm.check() == LEVEL_CRITICAL
        assert pm.effective_lease_time() == 60
        u["v"] = 0.99
        assert pm.check() == LEVEL_EXHAUSTED
        assert events == [LEVEL_WARNING, LEVEL_CRITICAL, LEVEL_EXHAUSTED]


class TestRequestQueue:
    def test_queue_and_drain(self):
        """RequestQueuing scenario."""
        q = RequestQueue(max_size=3)
        up = {"v": False}
        results = []

        def op(name):
            def fn():
                if up["v"]:
                    results.append(name)
                    return True
                return False
            return fn

        assert q.enqueue("a", op("a"))
        assert q.enqueue("b", op("b"))
        assert q.drain() == 0 and len(q) == 2   # still down, requeued
        up["v"] = True
        assert q.drain() == 2
        assert sorted(results) == ["a", "b"]

    def test_bounded_and_retry_budget(self):
        q = RequestQueue(max_size=1)
        assert q.enqueue("x", lambda: False, max_attempts=2)
        assert not q.enqueue("y", lambda: False)   # full
        q.drain()
        q.drain()
        assert len(q) == 0 and q.stats["gave_up"] == 1


class TestSplitBrain:
    def test_conflict_detection_keep_oldest(self):
        """SplitBrainConflicts + ConflictResolution scenarios."""
        seen = []
        det = ConflictDetector(on_conflict=lambda c: seen.append(c.ip))
        allocs = [
            Allocation("10.0.1.5", "sub-a", "node-1", 100.0),
            Allocation("10.0.1.5", "sub-b", "node-2", 200.0),  # conflict
            Allocation("10.0.1.6", "sub-c", "node-1", 100.0),
            Allocation("10.0.1.6", "sub-c", "node-2", 150.0),  # same sub: ok
        ]
        conflicts = det.scan(allocs)
        assert len(conflicts) == 1
        c = conflicts[0]
        assert c.ip == "10.0.1.5"
        assert c.keeper.subscriber_id == "sub-a"       # oldest wins
        assert [a.subscriber_id for a in c.evicted] == ["sub-b"]
        assert seen == ["10.0.1.5"]
