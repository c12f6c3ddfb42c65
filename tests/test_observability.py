"""Metrics / audit / intercept / state-store / subscriber-manager tests."""
import gzip
import os
import time

import pytest

from bng_amd.audit.logger import (CAT_AUTH, CAT_SESSION, Event, FileExporter,
                                  Logger, MemoryStorage, SyslogExporter)
from bng_amd.dataplane.launcher import GoldenLauncher
from bng_amd.intercept.manager import (JSONExporter, Manager as Intercept,
                                       X1Exporter, X2Exporter)
from bng_amd.metrics.metrics import Metrics
from bng_amd.state.store import StateStore
from bng_amd.subscriber.manager import Manager as SubMgr


class TestMetrics:
    def test_collect_from_launcher_and_render(self):
        l = GoldenLauncher()
        l.dp.dhcp_stats[1] = 95      # hits
        l.dp.dhcp_stats[2] = 5       # misses
        m = Metrics()
        m.collect_once(launcher=l)
        text = m.render().decode()
        assert "bng_dataplane_fastpath_hits_total 95.0" in text
        assert "bng_dataplane_cache_hit_rate 0.95" in text
        assert 'bng_dataplane_stat{module="nat"' in text

    def test_http_endpoint(self):
        import requests
        m = Metrics().serve(port=0)
        try:
            r = requests.get(f"http://127.0.0.1:{m.port}/metrics", timeout=3)
            assert r.status_code == 200 and b"bng_" in r.content
            assert requests.get(f"http://127.0.0.1:{m.port}/health",
                                timeout=3).status_code == 200
        finally:
            m.stop()


class TestAudit:
    def test_log_query(self):
        log = Logger().start()
        log.log("session_start", subscriber="sub-1", ip="10.0.1.5")
        log.log("auth_reject", category=CAT_AUTH, subscriber="sub-2",
                outcome="failure")
        log.stop()
        assert len(log.storage.query(category=CAT_AUTH)) == 1
        assert log.storage.query(subscriber="sub-1")[0].action == \
            "session_start"

    def test_file_exporter_rotation_gzip(self, tmp_path):
        path = str(tmp_path / "audit.log")
        ex = FileExporter(path, rotate_bytes=200, retention=2)
        log = Logger(exporters=[ex])
        for i in range(20):
            log.log("event", subscriber=f"s{i}")
        log.flush()
        ex.close()
        files = os.listdir(tmp_path)
        assert any(f.endswith(".gz") for f in files)
        assert sum(1 for f in files if ".log." in f) <= 2

    def test_retention_respects_legal_hold(self):
        st = MemoryStorage()
        log = Logger(storage=st)
        log.set_legal_hold("sub-held")
        log.log("old_event", subscriber="sub-held")
        log.log("old_event", subscriber="sub-free")
        log.flush()
        for e in st.all():
            e.timestamp -= 10_000
        removed = st.apply_retention(max_age=3600)
        assert removed == 1
        assert st.all()[0].subscriber == "sub-held"

    def test_syslog_exporter(self):
        lines = []
        log = Logger(exporters=[SyslogExporter(lines.append)])
        log.log("session_start", subscriber="s1", ip="1.2.3.4")
        log.flush()
        assert lines and "session_start" in lines[0]


class TestIntercept:
    def test_warrant_lifecycle_and_records(self):
        x1, x2, js = X1Exporter(), X2Exporter(), JSONExporter()
        m = Intercept(exporters=[x2, js], admin_exporter=x1)
        w = m.add_warrant("sub-1", authority="LEA", case_reference="C-9")
        assert m.is_target("sub-1") and not m.is_target("sub-2")
        m.on_session_start("sub-1", "10.0.1.5")
        m.on_session_start("sub-2", "10.0.1.6")   # not a target: no record
        m.on_nat_event("sub-1", "203.0.113.1", port_start=1024)
        assert len(m.records) == 2
        assert len(x2.records) == 2 and "session_start" in x2.records[0]
        assert len(js.lines) == 2
        assert "activate" in x1.records[0]
        m.revoke_warrant(w.id)
        m.on_session_stop("sub-1")
        assert len(m.records) == 2                # revoked: no new records
        assert "deactivate" in x1.records[-1]

    def test_expired_warrant(self):
        m = Intercept()
        w = m.add_warrant("sub-1", duration=0.01)
        time.sleep(0.05)
        assert not m.is_target("sub-1")


class TestStateStore:
    def test_persistence_roundtrip(self, tmp_path):
        path = str(tmp_path / "state.json")
        s = StateStore(path)
        s.put("leases", "aa:bb", {"ip": "10.0.1.5"})
        s.put("sessions", "s1", {"subscriber": "sub-1"})
        s.save()
        s2 = StateStore(path)
        assert s2.get("leases", "aa:bb")["ip"] == "10.0.1.5"
        assert len(s2.list("sessions")) == 1


class TestSubscriberManager:
    def test_lifecycle_and_events(self):
        events = []
        m = SubMgr()
        m.on_event(lambda ev, s: events.append((ev, s.subscriber_id)))
        s = m.create_session("sub-1", mac="aa:bb", ip="10.0.1.5")
        assert s.state == "active"
        assert m.get_by_ip("10.0.1.5").id == s.id
        assert m.get_by_subscriber("sub-1").id == s.id
        # idempotent create
        assert m.create_session("sub-1").id == s.id
        assert m.terminate_session(s.id, reason="admin")
        assert m.count() == 0
        assert events == [("session_start", "sub-1"),
                          ("session_stop", "sub-1")]

    def test_auth_failure(self):
        class Deny:
            def authenticate(self, sid, cred):
                return False
        m = SubMgr(authenticator=Deny())
        assert m.create_session("sub-1") is None
        assert m.stats["auth_failed"] == 1


class TestRuntimeStore:
    """Typed operational DB behaviors (ref pkg/state/store_test.go)."""

    def make(self, **kw):
        from bng_amd.state.runtime import Config, RuntimeStore
        return RuntimeStore(Config(**kw) if kw else None)

    def test_subscriber_crud_and_indexes(self):
        from bng_amd.state.runtime import StateError, Subscriber
        st = self.make()
        st.create_subscriber(Subscriber("s1", mac="aa:bb:cc:00:00:01",
                                        nte_id="nte-1"))
        assert st.get_subscriber_by_mac("aa:bb:cc:00:00:01").id == "s1"
        assert st.get_subscriber_by_nte("nte-1").id == "s1"
        with pytest.raises(StateError):      # duplicate MAC
            st.create_subscriber(Subscriber("s2",
                                            mac="aa:bb:cc:00:00:01"))
        st.delete_subscriber("s1")
        assert st.get_subscriber_by_mac("aa:bb:cc:00:00:01") is None

    def test_find_pool_priority_isp_class_capacity(self):
        from bng_amd.state.runtime import Pool, StateError, Subscriber
        st = self.make()
        st.create_pool(Pool("p-low", priority=1, total_addresses=10))
        st.create_pool(Pool("p-high", priority=9, total_addresses=10))
        st.create_pool(Pool("p-full", priority=99, total_addresses=10,
                            allocated_addresses=10))
        st.create_pool(Pool("p-isp", priority=50, total_addresses=10,
                            isp_ids=["isp-2"]))
        st.create_pool(Pool("p-class", priority=60, total_addresses=10,
                            subscriber_class=["gold"]))
        sub = Subscriber("s1", isp_id="isp-1", klass="silver")
        assert st.find_pool_for_subscriber(sub).id == "p-high"
        gold = Subscriber("s2", isp_id="isp-2", klass="gold")
        assert st.find_pool_for_subscriber(gold).id == "p-class"
        with pytest.raises(StateError):
            st.find_pool_for_subscriber(sub, version=6)

    def test_lease_lifecycle_and_cleanup(self):
        from bng_amd.state.runtime import Lease
        st = self.make()
        now = time.time()
        st.create_lease(Lease("l1", mac="aa:00:00:00:00:01",
                              ipv4="10.0.0.5", expires_at=now + 100))
        st.create_lease(Lease("l2", mac="aa:00:00:00:00:02",
                              ipv4="10.0.0.6", expires_at=now - 1))
        assert st.get_lease_by_ip("10.0.0.5").id == "l1"
        assert st.get_lease_by_mac("aa:00:00:00:00:02").id == "l2"
        st.renew_lease("l1", 3600)
        assert st.cleanup_expired_leases(now=now) == 1
        assert st.get_lease("l2") is None
        assert st.get_lease_by_ip("10.0.0.6") is None
        assert st.get_lease("l1") is not None     # renewed survives

    def test_session_idle_and_hard_timeout(self):
        from bng_amd.state.runtime import Session
        st = self.make()
        now = time.time()
        st.create_session(Session("a", mac="m1", ipv4="10.0.0.1",
                                  idle_timeout=30,
                                  last_activity=now - 60))
        st.create_session(Session("b", mac="m2", ipv4="10.0.0.2",
                                  session_timeout=100,
                                  started_at=now - 200,
                                  last_activity=now))
        st.create_session(Session("c", mac="m3", ipv4="10.0.0.3",
                                  idle_timeout=30, last_activity=now))
        st.update_session_activity("c", bytes_in=100, bytes_out=50)
        assert st.cleanup_idle_sessions(now=now) == 2
        assert st.get_session("c").bytes_in == 100
        assert st.get_session_by_ip("10.0.0.1") is None
        assert st.get_session_by_mac("m3").id == "c"

    def test_nat_binding_index_and_cleanup(self):
        from bng_amd.state.runtime import NATBinding
        st = self.make()
        now = time.time()
        st.create_nat_binding(NATBinding("n1", private_ip="10.0.0.1",
                                         private_port=4000, protocol=17,
                                         expires_at=now + 60))
        st.create_nat_binding(NATBinding("n2", private_ip="10.0.0.2",
                                         private_port=4001, protocol=6,
                                         expires_at=now - 1))
        b = st.get_nat_binding_by_private("10.0.0.1", 4000, 17)
        assert b.id == "n1"
        assert st.cleanup_expired_nat(now=now) == 1
        assert st.get_nat_binding_by_private("10.0.0.2", 4001, 6) is None

    def test_max_limits(self):
        from bng_amd.state.runtime import (Lease, LimitExceeded, Session,
                                           Subscriber)
        st = self.make(max_subscribers=1, max_leases=1, max_sessions=1)
        st.create_subscriber(Subscriber("s1"))
        with pytest.raises(LimitExceeded):
            st.create_subscriber(Subscriber("s2"))
        st.create_lease(Lease("l1"))
        with pytest.raises(LimitExceeded):
            st.create_lease(Lease("l2"))
        st.create_session(Session("x"))
        with pytest.raises(LimitExceeded):
            st.create_session(Session("y"))
        s = st.stats()
        assert s["subscribers"] == 1 and s["sessions"] == 1

    def test_start_stop_loops(self):
        st = self.make(lease_cleanup_interval=0.01,
                       session_cleanup_interval=0.01,
                       nat_cleanup_interval=0.01)
        st.start()
        time.sleep(0.05)
        st.stop()


class TestSecurityAuditor:
    """TLS/cert/mTLS event stream (ref pkg/audit/security_test.go)."""

    def test_security_events(self):
        from bng_amd.audit.logger import (CAT_TLS, Logger, MemoryStorage,
                                          SecurityAuditor)
        st = MemoryStorage()
        lg = Logger(storage=st).start()
        sec = SecurityAuditor(lg)
        sec.log_tls_handshake("192.0.2.9", True, tls_version="1.3",
                              cipher="TLS_AES_128_GCM_SHA256")
        sec.log_tls_handshake("192.0.2.9", False, error="bad cert")
        sec.log_certificate_expiring("CN=bng", "2027-01-01", 20)
        sec.log_certificate_expired("CN=bng", "2025-01-01")
        sec.log_certificate_invalid("CN=bng", "hostname mismatch")
        sec.log_certificate_pin_failed("192.0.2.9", "ab" * 32)
        sec.log_certificate_renewed("CN=bng", "2028-01-01")
        sec.log_mtls_auth("nte-1", True, subject="CN=nte-1")
        sec.log_mtls_auth("nte-2", False, error="unknown CA")
        lg.stop()
        evs = st.query(category=CAT_TLS)
        acts = [e.action for e in evs]
        assert "tls_handshake" in acts and \
            "certificate_pin_failed" in acts
        fails = [e for e in evs if e.outcome == "failure"]
        assert len(fails) == 4
        auth = [e for e in st.all() if e.action == "mtls_auth"]
        assert {e.outcome for e in auth} == {"success", "failure"}
        assert auth[0].subscriber == "nte-1"


class TestSubscriberManagerLifecycle:
    """Staged lifecycle + walled garden + cleanup (ref
    pkg/subscriber/manager_test.go)."""

    def test_staged_create_auth_assign_activate(self):
        class Alloc:
            def allocate(self, sid):
                return "10.9.0.1"

            def release(self, sid):
                pass
        m = SubMgr(allocator=Alloc())
        s = m.open_session("sub-1", mac="aa:01", isp_id="isp-a",
                           metadata={"circuit": "pon0/1"})
        assert s.state == "created"
        assert s.attributes["circuit"] == "pon0/1"
        assert m.authenticate(s.id)
        assert s.state == "authenticated"
        assert m.assign_address(s.id) == "10.9.0.1"
        assert s.state == "addressed"
        assert m.activate_session(s.id)
        assert s.state == "active"
        assert m.get_by_mac("aa:01").id == s.id
        assert m.get_by_ip("10.9.0.1").id == s.id
        # idempotent open returns the same session
        assert m.open_session("sub-1").id == s.id

    def test_walled_garden_transitions(self):
        m = SubMgr()
        s = m.open_session("sub-1")
        m.activate_session(s.id, walled=True)
        assert s.state == "walled_garden"
        assert not m.set_walled_garden(s.id)      # already walled
        assert m.clear_walled_garden(s.id)
        assert s.state == "active"
        assert not m.clear_walled_garden(s.id)    # already clear
        assert m.set_walled_garden(s.id)
        assert not m.set_walled_garden("missing")

    def test_cleanup_idle_and_session_timeout(self):
        m = SubMgr(idle_timeout=30, session_timeout=1000)
        a = m.create_session("sub-a", ip="10.0.0.1")
        b = m.create_session("sub-b", ip="10.0.0.2")
        c = m.create_session("sub-c", ip="10.0.0.3")
        now = time.time()
        a.last_activity = now - 60                 # idle
        b.started_at = now - 2000                  # hard timeout
        assert m.cleanup(now=now) == 2
        assert m.count() == 1 and m.get(c.id) is not None
        assert c.attributes.get("terminate_reason") is None

    def test_list_by_isp_and_multiple_handlers(self):
        got1, got2 = [], []
        m = SubMgr()
        m.on_event(lambda ev, s: got1.append(ev))
        m.on_event(lambda ev, s: got2.append(ev))
        sa = m.open_session("a", isp_id="isp-1")
        m.open_session("b", isp_id="isp-2")
        m.open_session("c", isp_id="isp-1")
        assert {s.subscriber_id for s in m.list_by_isp("isp-1")} == \
            {"a", "c"}
        assert m.list_by_isp("isp-9") == []
        assert len(m.list_sessions()) == 3
        m.activate_session(sa.id)
        m.terminate_session(sa.id)
        assert "session_created" in got1 and "session_stop" in got2

    def test_max_sessions_staged(self):
        m = SubMgr(max_sessions=1)
        assert m.open_session("a") is not None
        assert m.open_session("b") is None
        assert m.stats["rejected_capacity"] == 1


class TestInterceptWarrants:
    """Pending warrants / validation / types (ref
    pkg/intercept/manager_test.go)."""

    def test_pending_warrant_activates_at_valid_from(self):
        m = Intercept()
        w = m.add_warrant("sub-1", valid_from=time.time() + 100)
        assert m.warrant_status(w.id) == "pending"
        assert not m.is_target("sub-1")
        w.start_time = time.time() - 1            # time passes
        assert m.warrant_status(w.id) == "active"
        assert m.is_target("sub-1")

    def test_validation_and_types(self):
        m = Intercept()
        with pytest.raises(ValueError):
            m.add_warrant("")
        with pytest.raises(ValueError):
            m.add_warrant("sub-1", intercept_type="metadata")
        for t in ("iri", "cc", "iri+cc"):
            assert m.add_warrant("sub-t", intercept_type=t)

    def test_status_transitions(self):
        m = Intercept()
        w = m.add_warrant("sub-1", duration=0.01)
        assert m.warrant_status(w.id) == "active"
        time.sleep(0.05)
        assert m.warrant_status(w.id) == "expired"
        m.revoke_warrant(w.id)
        assert m.warrant_status(w.id) == "revoked"
        assert m.warrant_status("nope") == "unknown"


class TestIPFIXAndKafkaExporters:
    """Binary IPFIX framing + topic routing (ref pkg/audit/export.go)."""

    def _nat_event(self, **d):
        from bng_amd.audit.logger import CAT_NAT, Event
        base = {"private_ip": 0x0A000105, "public_ip": 0xCB007101,
                "private_port": 40000, "public_port": 2048,
                "protocol": 17}
        base.update(d)
        return Event(id="e1", category=CAT_NAT, action="nat_mapping",
                     timestamp=1700000000.0, subscriber="sub-1",
                     ip="10.0.1.5", outcome="success",
                     details={k: str(v) for k, v in base.items()})

    def test_ipfix_message_framing(self):
        import struct as st
        from bng_amd.audit.logger import CAT_AUTH, Event, IPFIXExporter
        msgs = []
        x = IPFIXExporter(msgs.append, observation_domain=7)
        assert x.export(self._nat_event())
        # non-NAT events are skipped
        assert not x.export(Event(id="e2", category=CAT_AUTH,
                                  action="login", timestamp=0.0,
                                  subscriber="", ip="", outcome="ok",
                                  details={}))
        ver, length, t, seq, dom = st.unpack_from(">HHIII", msgs[0], 0)
        assert ver == 10 and dom == 7 and seq == 0
        assert length == len(msgs[0])
        # first message carries the template set (set id 2)
        set_id, set_len = st.unpack_from(">HH", msgs[0], 16)
        assert set_id == 2
        tid, nfields = st.unpack_from(">HH", msgs[0], 20)
        assert tid == 256 and nfields == 6
        # data set follows, record decodes to the NAT 5-tuple
        off = 16 + set_len
        dsid, dlen = st.unpack_from(">HH", msgs[0], off)
        assert dsid == 256
        priv, pub, pport, natport, proto, ts = st.unpack_from(
            ">IIHHBI", msgs[0], off + 4)
        assert (priv, pub, pport, natport, proto) == \
            (0x0A000105, 0xCB007101, 40000, 2048, 17)
        # second export: no template, sequence advanced
        x.export(self._nat_event())
        set_id2, _ = st.unpack_from(">HH", msgs[1], 16)
        assert set_id2 == 256
        assert st.unpack_from(">HHIII", msgs[1], 0)[3] == 1

    def test_kafka_topic_routing(self):
        from bng_amd.audit.logger import KafkaExporter
        got = []
        x = KafkaExporter(lambda t, k, v: got.append((t, k)),
                          topic_by_category=True)
        x.export(self._nat_event())
        assert got == [("bng-audit-nat", "sub-1")]
        x2 = KafkaExporter(lambda t, k, v: got.append((t, k)))
        x2.export(self._nat_event())
        assert got[-1] == ("bng-audit", "sub-1")
        assert x.exported == 1 and x2.exported == 1


class TestAuditSeverityRetention:
    """Severity model + RetentionManager (ref pkg/audit
    types.go:254-403, retention.go:9-357)."""

    def test_action_severity_and_category(self):
        from bng_amd.audit import retention as rt
        assert rt.action_severity("mac_spoof") == rt.CRITICAL
        assert rt.action_severity("brute_force_detected") == rt.ALERT
        assert rt.action_severity("auth_failure") == rt.WARNING
        assert rt.action_severity("session_start") == rt.INFO
        assert rt.action_severity("nat_mapping") == rt.DEBUG
        assert rt.action_category("mac_spoof") == "security"
        assert rt.action_category("certificate_expired") == "tls"
        assert rt.action_category("dhcp_ack") == "dhcp"
        assert rt.action_category("whatever") == "other"
        assert rt.severity_name(rt.EMERGENCY) == "EMERGENCY"

    def test_logger_min_severity_and_disabled_category(self):
        from bng_amd.audit.logger import Logger, MemoryStorage
        from bng_amd.audit import retention as rt
        log = Logger(MemoryStorage(), min_severity=rt.WARNING,
                     disabled_categories={"nat"})
        log.log("session_start", subscriber="s1")       # INFO -> filtered
        log.log("auth_failure", category="auth")        # WARNING -> kept
        log.log("mac_spoof", category="nat")            # category disabled
        log.flush()
        assert log.filtered == 2 and log.logged == 1
        evs = log.storage.all()
        assert len(evs) == 1 and evs[0].action == "auth_failure"
        assert evs[0].severity == rt.WARNING
        st = log.stats()
        assert st["stored"] == 1 and st["filtered"] == 2

    def test_retention_per_category_with_action_override(self):
        from bng_amd.audit.retention import RetentionManager
        rm = RetentionManager()
        assert rm.get_retention("admin") == 730
        assert rm.get_retention("system") == 30
        assert rm.get_retention("unknown") == 365       # default
        assert rm.get_retention_for_action("dhcp_ack") == 90
        rm.set_action_retention("dhcp_ack", 7)
        assert rm.get_retention_for_action("dhcp_ack") == 7
        rm.set_category_retention("system", 60)
        assert rm.policy_summary()["system"] == 60

    def test_legal_hold_criteria_all_must_match(self):
        import time as _t
        from bng_amd.audit.logger import Event
        from bng_amd.audit.retention import LegalHold
        now = _t.time()
        ev = Event(id="e1", category="session", action="session_stop",
                   timestamp=now, subscriber="alice", ip="10.0.1.5",
                   details={"mac": "aa:bb:cc:00:00:01",
                            "session_id": "sess-9"})
        assert LegalHold(subscribers=["alice"]).matches(ev)
        assert not LegalHold(subscribers=["bob"]).matches(ev)
        assert LegalHold(subscribers=["alice"],
                         ips=["10.0.1.5"]).matches(ev)
        assert not LegalHold(subscribers=["alice"],
                             ips=["10.9.9.9"]).matches(ev)
        assert LegalHold(macs=["aa:bb:cc:00:00:01"]).matches(ev)
        assert LegalHold(sessions=["sess-9"]).matches(ev)
        assert not LegalHold(actions=["auth_failure"]).matches(ev)
        # time-window criterion
        assert not LegalHold(start_time=now + 10).matches(ev)
        assert LegalHold(start_time=now - 10,
                         end_time=now + 10).matches(ev)

    def test_expiry_honors_holds_and_hold_expiry(self):
        import time as _t
        from bng_amd.audit.logger import Event, Logger, MemoryStorage
        from bng_amd.audit.retention import LegalHold, RetentionManager
        rm = RetentionManager()
        now = _t.time()
        old = now - 200 * 86400                          # 200 days old
        held = Event(id="h", category="dhcp", action="dhcp_ack",
                     timestamp=old, subscriber="alice")
        loose = Event(id="l", category="dhcp", action="dhcp_ack",
                      timestamp=old, subscriber="bob")
        fresh = Event(id="f", category="dhcp", action="dhcp_ack",
                      timestamp=now, subscriber="bob")
        rm.add_hold(LegalHold(subscribers=["alice"]))
        # dhcp retention is 90 days: old events expired unless held
        assert not rm.expired(held, now)
        assert rm.expired(loose, now)
        assert not rm.expired(fresh, now)
        # wired through the logger + storage sweep
        st = MemoryStorage()
        for e in (held, loose, fresh):
            st.store(e)
        log = Logger(st, retention=rm)
        assert log.cleanup_expired(now) == 1
        assert {e.id for e in st.all()} == {"h", "f"}
        # an expired hold stops protecting
        hid = rm.get_holds()[0].id
        rm.holds[hid].expires_at = now - 1
        assert rm.cleanup_expired_holds() == 1
        assert rm.expired(held, now)

    def test_storage_query_limit_severity_delete(self):
        from bng_amd.audit.logger import Event, MemoryStorage
        from bng_amd.audit import retention as rt
        st = MemoryStorage()
        for i in range(10):
            st.store(Event(id=f"e{i}", category="auth",
                           action="auth_failure", timestamp=float(i),
                           severity=rt.WARNING if i % 2 else rt.INFO))
        assert len(st.query(min_severity=rt.WARNING)) == 5
        assert len(st.query(limit=3)) == 3
        assert st.count() == 10
        assert st.delete(["e0", "e1", "nope"]) == 2
        assert st.count() == 8

    def test_syslog_priority_encodes_severity(self):
        from bng_amd.audit.logger import Event, SyslogExporter
        from bng_amd.audit import retention as rt
        lines = []
        ex = SyslogExporter(lines.append)
        ex.export(Event(id="a", category="security", action="mac_spoof",
                        timestamp=0.0, severity=rt.CRITICAL))
        ex.export(Event(id="b", category="session",
                        action="session_start", timestamp=0.0,
                        severity=rt.INFO))
        # facility 13: CRITICAL -> level 2 -> PRI 106; INFO -> 6 -> 110
        assert lines[0].startswith("<106>")
        assert lines[1].startswith("<110>")


class TestInterceptDepth:
    """Multi-criteria warrants, intercept sessions, filtered CC capture,
    ETSI HI2/HI3 PDUs (ref pkg/intercept manager.go:260-460,
    exporter.go:191-318)."""

    def test_match_session_any_criterion_dedup(self):
        m = Intercept()
        w1 = m.add_warrant("alice", target_mac="AA:BB:CC:00:00:01",
                           target_ipv4="10.0.1.5")
        w2 = m.add_warrant("", target_ipv4="10.0.1.5")
        # same warrant reachable via two criteria -> one result
        got = m.match_session(subscriber="alice",
                              mac="aa:bb:cc:00:00:01")
        assert [w.id for w in got] == [w1.id]
        got = m.match_session(ipv4="10.0.1.5")
        assert {w.id for w in got} == {w1.id, w2.id}
        assert m.match_session(subscriber="bob") == []
        with pytest.raises(ValueError):
            m.add_warrant()                     # no criterion at all

    def test_intercept_session_lifecycle_and_stats(self):
        ex = JSONExporter()
        m = Intercept(exporters=[ex])
        w = m.add_warrant("alice", intercept_type="iri+cc",
                          liid="LIID-CASE-9")
        s = m.start_intercept(w, "sess-1", "alice",
                              ipv4="10.0.1.5")
        assert s.liid == "LIID-CASE-9"
        assert w.sessions_matched == 1
        assert m.get_intercept("sess-1") is s
        ok = m.record_cc("sess-1", "up", "10.0.1.5", "9.9.9.9",
                         5555, 443, 6, b"x" * 100)
        assert ok and s.cc_records == 1 and s.bytes_captured == 100
        assert w.bytes_intercepted == 100
        st = m.stats()
        assert st["active_interceptions"] == 1
        assert st["total_cc_records"] == 1
        assert st["total_bytes_delivered"] == 100
        m.stop_intercept("sess-1")
        assert m.get_intercept("sess-1") is None
        assert m.stats()["active_interceptions"] == 0
        # session_start + session_stop IRI both exported
        import json as _json
        kinds = [_json.loads(l)["record_type"] for l in ex.lines]
        assert kinds.count("session_start") == 1
        assert kinds.count("session_stop") == 1

    def test_cc_filters_every_axis(self):
        m = Intercept()
        w = m.add_warrant("alice", intercept_type="cc",
                          filter_dest_ports=[443],
                          filter_protocols=[6],
                          filter_dest_ips=["9.9.9.9"])
        m.start_intercept(w, "s", "alice")
        ok = m.record_cc("s", "up", "10.0.1.5", "9.9.9.9",
                         1, 443, 6, b"y")
        assert ok
        assert not m.record_cc("s", "up", "10.0.1.5", "9.9.9.9",
                               1, 80, 6, b"y")       # port filtered
        assert not m.record_cc("s", "up", "10.0.1.5", "9.9.9.9",
                               1, 443, 17, b"y")     # proto filtered
        assert not m.record_cc("s", "up", "10.0.1.5", "8.8.8.8",
                               1, 443, 6, b"y")      # dest filtered
        # an IRI-only warrant never captures content
        w2 = m.add_warrant("bob", intercept_type="iri")
        m.start_intercept(w2, "s2", "bob")
        assert not m.record_cc("s2", "up", "1.1.1.1", "2.2.2.2",
                               1, 2, 6, b"z")

    def test_etsi_pdu_roundtrip_and_sequences(self):
        from bng_amd.intercept.etsi import (ETSIExporter, decode_pdu,
                                            split_stream, HI2, HI3)
        frames = []
        ex = ETSIExporter(send=frames.append, country_code="DE")
        m = Intercept(exporters=[ex])
        w = m.add_warrant("alice", intercept_type="iri+cc",
                          liid="LIID-7")
        m.start_intercept(w, "sess-9", "alice", ipv4="10.0.1.5")
        m.record_cc("sess-9", "down", "9.9.9.9", "10.0.1.5",
                    443, 5555, 6, b"PAYLOAD")
        assert ex.sent_iri == 1 and ex.sent_cc == 1
        iri = decode_pdu(frames[0])
        assert iri["handover"] == HI2 and iri["liid"] == "LIID-7"
        assert iri["seq"] == 0
        assert iri["iri"]["event_type"] == "session_start"
        assert iri["iri"]["session_id"] == "sess-9"
        assert iri["iri"]["country_code"] == "DE"
        cc = decode_pdu(frames[1])
        assert cc["handover"] == HI3 and cc["seq"] == 1   # per-LIID seq
        assert cc["direction"] == "down"
        assert cc["src_ip"] == "9.9.9.9" and cc["dst_port"] == 5555
        assert cc["protocol"] == 6 and cc["payload"] == b"PAYLOAD"
        # stream framing: concatenated PDUs split cleanly
        assert split_stream(frames[0] + frames[1]) == frames
        # a second LIID starts its own sequence space
        ex.send(ex.build_hi2("LIID-8", "session_start", "x"))
        assert decode_pdu(frames[2])["seq"] == 0


class TestSubscriberManagerDepth:
    """Rich authentication applying RADIUS attributes, per-session
    timeouts, traffic accounting (ref pkg/subscriber/manager.go
    179-296, 535-552, 648-690)."""

    def _mgr(self, auth=None, **kw):
        from bng_amd.subscriber.manager import Manager
        return Manager(authenticator=auth, **kw)

    def test_authenticate_full_applies_attributes(self):
        from bng_amd.subscriber.manager import S_AUTHENTICATED

        class RadiusLike:
            def authenticate_session(self, session, credentials):
                assert credentials["password"] == "pw"
                return {"success": True, "subscriber_id": "sub-real",
                        "isp_id": "isp-a", "radius_session_id": "R1",
                        "session_timeout": 3600,
                        "idle_timeout": 300,
                        "download_rate_bps": 100_000_000,
                        "upload_rate_bps": 20_000_000,
                        "qos_policy_id": "gold"}

        m = self._mgr(RadiusLike())
        s = m.open_session("mac-tmp", mac="aa:bb:cc:00:00:01")
        r = m.authenticate_full(s.id, {"password": "pw"})
        assert r["success"]
        assert s.state == S_AUTHENTICATED
        assert s.subscriber_id == "sub-real"       # identity rebound
        assert m.get_by_subscriber("sub-real") is s
        assert s.session_timeout == 3600 and s.idle_timeout == 300
        assert s.download_rate_bps == 100_000_000
        assert s.qos_policy_id == "gold"
        assert m.manager_stats()["auth_successes"] == 1

    def test_authenticate_full_walled_garden(self):
        from bng_amd.subscriber.manager import S_WALLED

        class Waller:
            def authenticate_session(self, session, credentials):
                return {"success": True, "walled_garden": True,
                        "walled_reason": "payment_overdue"}

        m = self._mgr(Waller())
        s = m.open_session("sub-1")
        assert m.authenticate_full(s.id)["success"]
        assert s.state == S_WALLED
        assert s.walled_reason == "payment_overdue"
        assert m.manager_stats()["walled_garden_sessions"] == 1

    def test_authenticate_full_failure_restores_state(self):
        class Rejector:
            def authenticate_session(self, session, credentials):
                return {"success": False, "error": "bad password"}

        m = self._mgr(Rejector())
        s = m.open_session("sub-1")
        old = s.state
        r = m.authenticate_full(s.id, {})
        assert not r["success"]
        assert s.state == old and s.state_reason == "bad password"
        assert m.manager_stats()["auth_failures"] == 1
        # boolean-protocol fallback still works
        m2 = self._mgr()
        s2 = m2.open_session("sub-2")
        assert m2.authenticate_full(s2.id)["success"]

    def test_per_session_timeout_overrides_global(self):
        import time as _t
        m = self._mgr(idle_timeout=10_000.0)
        a = m.open_session("sub-a")
        b = m.open_session("sub-b")
        a.idle_timeout = 1.0                    # RADIUS-applied
        a.last_activity = _t.time() - 5
        b.last_activity = _t.time() - 5
        assert m.cleanup() == 1                 # only a reaped
        assert m.get_by_subscriber("sub-a") is None
        assert m.get_by_subscriber("sub-b") is not None

    def test_update_activity_accumulates(self):
        m = self._mgr()
        s = m.open_session("sub-1")
        assert m.update_activity(s.id, bytes_in=100, bytes_out=50,
                                 packets_in=2, packets_out=1)
        assert m.update_activity(s.id, bytes_in=10)
        assert s.input_octets == 110 and s.output_octets == 50
        assert s.packets_in == 2 and s.packets_out == 1
        st = m.manager_stats()
        assert st["total_bytes_in"] == 110
        assert not m.update_activity("nope")


class TestMetricsFamilies:
    """The full reference metric inventory renders (ref
    pkg/metrics/metrics.go:16-84)."""

    def test_all_reference_families_present(self):
        from bng_amd.metrics.metrics import Metrics
        m = Metrics()
        m.session_duration.observe(120)
        m.session_bytes_in.labels("dhcp").inc(1000)
        m.nat_translations.labels("egress").inc()
        m.nat_ports_used.labels("203.0.113.1").set(512)
        m.radius_latency.observe(0.01)
        m.radius_timeouts.labels("10.0.0.9").inc()
        m.pppoe_sessions.set(3)
        m.pppoe_negotiations.labels("lcp", "ok").inc()
        m.routes_active.labels("bgp").set(100)
        m.bgp_peers_up.set(2)
        m.subscriber_total.set(5000)
        m.subscriber_by_isp.labels("isp-a").set(4000)
        m.circuit_id_collisions.inc()
        m.pool_available.labels("p1").set(200)
        m.table_entries.labels("subscribers").set(65536)
        text = m.render().decode()
        for family in ("bng_session_duration_seconds",
                       "bng_session_bytes_in_total",
                       "bng_nat_translations_total",
                       "bng_nat_ports_used",
                       "bng_radius_request_duration_seconds",
                       "bng_radius_timeouts_total",
                       "bng_pppoe_sessions_active",
                       "bng_pppoe_negotiations_total",
                       "bng_routes_active", "bng_bgp_peers_up",
                       "bng_subscribers_total",
                       "bng_subscribers_by_isp",
                       "bng_circuit_id_collisions_total",
                       "bng_pool_available",
                       "bng_dataplane_table_entries"):
            assert family in text, family
        assert 'bng_subscribers_by_isp{isp="isp-a"} 4000' in text
