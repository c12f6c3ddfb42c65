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


class TestSiteConflictDetector:
    """Site-aware conflict detection + the reference's resolution
    policy (ref conflict_detector.go:121-330, manager.go:430-527)."""

    def _d(self):
        from bng_amd.resilience.conflict_detector import (
            IPAllocation, SiteConflictDetector)
        d = SiteConflictDetector("site-a")
        return d, IPAllocation

    def test_same_subscriber_is_no_conflict(self):
        d, A = self._d()
        d.record("10.0.1.5", "aa:01", "sub-1", allocated_at=100)
        remote = [A("10.0.1.5", "AA:01", "sub-1", "site-b", 90)]
        assert d.detect(remote) == []

    def test_cross_site_conflict_and_policy(self):
        from bng_amd.resilience.conflict_detector import (
            R_LOCAL_WINS, R_REMOTE_WINS, SiteConflictDetector)
        d, A = self._d()
        # local pre-partition vs remote during-partition -> local wins
        d.record("10.0.1.5", "aa:01", "sub-1", allocated_at=100)
        c = d.detect([A("10.0.1.5", "bb:02", "sub-2", "site-b", 200,
                        is_partition=True)])[0]
        d.resolve(c)
        assert c.resolution == R_LOCAL_WINS
        assert c.affected_mac == "bb:02"
        # local during-partition vs remote pre-partition -> remote wins
        d2 = SiteConflictDetector("site-a")
        d2.record("10.0.2.5", "aa:01", "sub-1", allocated_at=300,
                  is_partition=True)
        c2 = d2.detect([A("10.0.2.5", "bb:02", "sub-2", "site-b",
                          100)])[0]
        d2.resolve(c2)
        assert c2.resolution == R_REMOTE_WINS
        assert c2.affected_mac == "aa:01"
        # both partition-era -> most recent wins
        d3 = SiteConflictDetector("site-a")
        d3.record("10.0.3.5", "aa:01", "sub-1", allocated_at=500,
                  is_partition=True)
        c3 = d3.detect([A("10.0.3.5", "bb:02", "sub-2", "site-b", 400,
                          is_partition=True)])[0]
        d3.resolve(c3)
        assert c3.resolution == R_LOCAL_WINS

    def test_validate_raises_on_foreign_holder(self):
        import pytest as _pt
        from bng_amd.resilience.conflict_detector import ConflictError
        d, A = self._d()
        d.record("10.0.1.5", "aa:01", "sub-1")
        d.validate("10.0.1.5", "AA:01")        # same MAC ok
        with _pt.raises(ConflictError):
            d.validate("10.0.1.5", "bb:02")
        d.remove("10.0.1.5")
        d.validate("10.0.1.5", "bb:02")        # freed
        # export/import roundtrip
        d.record("10.0.9.9", "cc:03", "sub-3", is_partition=True)
        from bng_amd.resilience.conflict_detector import \
            SiteConflictDetector
        d2 = SiteConflictDetector("site-a")
        d2.import_allocations(d.export_allocations())
        assert d2.get("10.0.9.9").mac == "cc:03"
        assert len(d2.partition_allocations()) == 1
        d2.clear_partition_flags()
        assert d2.partition_allocations() == []


class TestReconciler:
    """Partition-heal pipeline (ref manager.go:342-427)."""

    def test_full_reconciliation(self):
        from bng_amd.radius.client import AuthResult, RadiusTimeout
        from bng_amd.resilience.conflict_detector import (
            IPAllocation, SiteConflictDetector)
        from bng_amd.resilience.manager import Reconciler
        from bng_amd.resilience.radius_handler import ResilientRadius
        from bng_amd.resilience.request_queue import RequestQueue

        class FlappyClient:
            def __init__(self):
                self.down = True
                self.acct = []

            def authenticate(self, user, pw, **kw):
                if self.down:
                    raise RadiusTimeout()
                return AuthResult(True, policy_name="gold")

            def send_accounting(self, *a, **kw):
                if self.down:
                    raise RadiusTimeout()
                self.acct.append(a)
                return True

        client = FlappyClient()
        rr = ResilientRadius(client, mode="allow")
        # partition: degraded admits + buffered accounting
        assert rr.authenticate("alice", "pw").success
        assert rr.authenticate("bob", "pw").success
        rr.send_accounting("alice", 100)
        assert rr.stats["acct_buffered"] == 1
        assert sorted(rr.degraded_sessions()) == ["alice", "bob"]
        det = SiteConflictDetector("site-a")
        det.record("10.0.1.5", "aa:01", "alice", allocated_at=10,
                   is_partition=True)
        q = RequestQueue()
        drained = []
        q.enqueue("update", lambda: drained.append(1) or True)
        conflicts_seen = []
        rec = Reconciler(det, rr, q,
                         on_conflict=conflicts_seen.append)
        # heal
        client.down = False
        remote = [IPAllocation("10.0.1.5", "bb:02", "carol", "site-b",
                               5)]
        result = rec.reconcile(remote,
                               credentials=lambda u: ("pw", {}))
        assert result["conflicts_found"] == 1
        assert result["conflicts_resolved"] == 1
        assert conflicts_seen[0].resolution == "remote_wins"
        assert result["reauths_queued"] == 2
        assert result["reauths_completed"] == 2
        assert result["acct_records_synced"] == 1
        assert result["requests_drained"] == 1
        assert drained == [1]
        assert rr.degraded_sessions() == []
        assert det.partition_allocations() == []

    def test_short_lease_policy(self):
        from bng_amd.resilience.manager import (Manager, ShortLeasePolicy,
                                                STATE_PARTITIONED)
        m = Manager(health_check=lambda: False, failure_threshold=1)
        pol = ShortLeasePolicy(m, short_lease=300, normal_lease=86400)
        assert pol.lease_time() == 86400
        m.check_once()                          # -> partitioned
        assert m.state == STATE_PARTITIONED
        assert pol.should_use_short_lease()
        assert pol.lease_time() == 300
        assert pol.short_leases_issued == 1
