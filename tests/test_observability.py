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
