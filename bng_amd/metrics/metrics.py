"""Prometheus metrics (ref pkg/metrics/metrics.go:16-135): ~30 bng_*
instruments across dhcp/dataplane/pool/session/nat/radius/qos/antispoof,
a periodic collector pulling GPU dataplane stats (the reference pulls
eBPF stats every 5s, cmd/bng/main.go:1241), and the /metrics + /health
HTTP endpoint."""
from __future__ import annotations

import threading
from typing import Optional

from prometheus_client import (CollectorRegistry, Counter, Gauge, Histogram,
                               generate_latest)


class Metrics:
    def __init__(self, registry: Optional[CollectorRegistry] = None):
        self.registry = registry or CollectorRegistry()
        r = self.registry
        # DHCP (ref metrics.go bng_dhcp_*)
        self.dhcp_requests = Counter(
            "bng_dhcp_requests_total", "DHCP requests by type",
            ["type"], registry=r)
        self.dhcp_responses = Counter(
            "bng_dhcp_responses_total", "DHCP responses by type",
            ["type"], registry=r)
        self.dhcp_latency = Histogram(
            "bng_dhcp_request_duration_seconds", "slow-path latency",
            buckets=(1e-5, 1e-4, 1e-3, 5e-3, 1e-2, 5e-2, 1e-1, 1),
            registry=r)
        # dataplane (the bng_ebpf_* analog)
        self.fastpath_hits = Gauge(
            "bng_dataplane_fastpath_hits_total", "GPU fast-path hits",
            registry=r)
        self.fastpath_misses = Gauge(
            "bng_dataplane_fastpath_misses_total", "GPU fast-path misses",
            registry=r)
        self.fastpath_hit_rate = Gauge(
            "bng_dataplane_cache_hit_rate", "hit / (hit+miss)", registry=r)
        self.dataplane_stat = Gauge(
            "bng_dataplane_stat", "raw dataplane counters",
            ["module", "name"], registry=r)
        # pools
        self.pool_allocated = Gauge(
            "bng_pool_allocated", "allocated addresses", ["pool"],
            registry=r)
        self.pool_utilization = Gauge(
            "bng_pool_utilization", "pool utilization 0..1", ["pool"],
            registry=r)
        # sessions
        self.sessions_active = Gauge(
            "bng_sessions_active", "active subscriber sessions",
            ["access_type"], registry=r)
        self.sessions_total = Counter(
            "bng_sessions_created_total", "sessions created",
            ["access_type"], registry=r)
        # NAT
        self.nat_sessions = Gauge(
            "bng_nat_sessions_active", "active NAT sessions", registry=r)
        self.nat_port_exhaustion = Gauge(
            "bng_nat_port_exhaustion_total", "port exhaustion events",
            registry=r)
        # RADIUS
        self.radius_requests = Counter(
            "bng_radius_requests_total", "RADIUS ops", ["op", "result"],
            registry=r)
        # QoS
        self.qos_policies = Gauge(
            "bng_qos_active_policies", "installed QoS buckets", registry=r)
        self.qos_dropped = Gauge(
            "bng_qos_packets_dropped_total", "rate-limited drops",
            registry=r)
        # antispoof
        self.antispoof_violations = Gauge(
            "bng_antispoof_violations_total", "uRPF violations",
            registry=r)
        # remaining reference families (ref metrics.go:16-84)
        self.pool_available = Gauge(
            "bng_pool_available", "free addresses", ["pool"], registry=r)
        self.circuit_id_collisions = Counter(
            "bng_circuit_id_collisions_total",
            "option-82 circuit-id hash collisions", registry=r)
        self.session_duration = Histogram(
            "bng_session_duration_seconds", "session lifetime",
            buckets=(60, 300, 1800, 3600, 14400, 86400, 604800),
            registry=r)
        self.session_bytes_in = Counter(
            "bng_session_bytes_in_total", "subscriber ingress bytes",
            ["access_type"], registry=r)
        self.session_bytes_out = Counter(
            "bng_session_bytes_out_total", "subscriber egress bytes",
            ["access_type"], registry=r)
        self.nat_translations = Counter(
            "bng_nat_translations_total", "NAT translations",
            ["direction"], registry=r)
        self.nat_ports_used = Gauge(
            "bng_nat_ports_used", "allocated NAT ports",
            ["public_ip"], registry=r)
        self.radius_latency = Histogram(
            "bng_radius_request_duration_seconds", "RADIUS round-trip",
            buckets=(1e-3, 5e-3, 1e-2, 5e-2, 1e-1, 5e-1, 1, 5),
            registry=r)
        self.radius_timeouts = Counter(
            "bng_radius_timeouts_total", "RADIUS timeouts", ["server"],
            registry=r)
        self.pppoe_sessions = Gauge(
            "bng_pppoe_sessions_active", "established PPPoE sessions",
            registry=r)
        self.pppoe_negotiations = Counter(
            "bng_pppoe_negotiations_total", "PPPoE phase outcomes",
            ["phase", "result"], registry=r)
        self.routes_active = Gauge(
            "bng_routes_active", "installed routes", ["proto"],
            registry=r)
        self.bgp_peers_up = Gauge(
            "bng_bgp_peers_up", "established BGP peers", registry=r)
        self.subscriber_total = Gauge(
            "bng_subscribers_total", "known subscribers", registry=r)
        self.subscriber_by_isp = Gauge(
            "bng_subscribers_by_isp", "subscribers per ISP", ["isp"],
            registry=r)
        self.table_entries = Gauge(
            "bng_dataplane_table_entries",
            "GPU table occupancy (the eBPF map-entries analog)",
            ["table"], registry=r)
        self._stop = threading.Event()
        self._collector: Optional[threading.Thread] = None
        self._extra_collectors = []
        self._httpd = None

    # ---------------------------------------------------------- collector
    def add_collector(self, fn):
        """Extra per-cycle collection hook (e.g. RoutingMetrics.collect);
        runs inside the 5s collector loop."""
        self._extra_collectors.append(fn)

    def start_collector(self, launcher=None, dhcp_server=None,
                        session_manager=None, interval: float = 5.0):
        """Periodic pull from the GPU dataplane (ref main.go:1241)."""
        def loop():
            while not self._stop.wait(interval):
                self.collect_once(launcher, dhcp_server, session_manager)
        self._collector = threading.Thread(target=loop, daemon=True)
        self._collector.start()
        return self

    def collect_once(self, launcher=None, dhcp_server=None,
                     session_manager=None):
        for fn in self._extra_collectors:
            try:
                fn()
            except Exception:
                pass
        if launcher is not None:
            st = launcher.get_stats()
            hits = st.get("fastpath_hits", 0)
            misses = st.get("fastpath_misses", 0)
            self.fastpath_hits.set(hits)
            self.fastpath_misses.set(misses)
            if hits + misses:
                self.fastpath_hit_rate.set(hits / (hits + misses))
            for name, v in st.items():
                self.dataplane_stat.labels("dhcp", name).set(v)
            for name, v in launcher.nat_get_stats().items():
                self.dataplane_stat.labels("nat", name).set(v)
            self.nat_port_exhaustion.set(
                launcher.nat_get_stats().get("port_exhaustion", 0))
            for name, v in launcher.qos_get_stats().items():
                self.dataplane_stat.labels("qos", name).set(v)
            self.qos_dropped.set(
                launcher.qos_get_stats().get("packets_dropped", 0))
            for name, v in launcher.antispoof_get_stats().items():
                self.dataplane_stat.labels("antispoof", name).set(v)
            self.antispoof_violations.set(
                launcher.antispoof_get_stats().get("ipv4_violations", 0))
        if dhcp_server is not None:
            for pool_stats in dhcp_server.pools.all_stats():
                pid = str(pool_stats["pool_id"])
                self.pool_allocated.labels(pid).set(pool_stats["allocated"])
                total = pool_stats["allocated"] + pool_stats["available"]
                if total:
                    self.pool_utilization.labels(pid).set(
                        pool_stats["allocated"] / total)
        if session_manager is not None:
            self.sessions_active.labels("dhcp").set(session_manager.count())

    # --------------------------------------------------------------- HTTP
    def serve(self, host: str = "127.0.0.1", port: int = 9090):
        """/metrics + /health (ref main.go:1220-1237)."""
        from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
        reg = self.registry

        class Handler(BaseHTTPRequestHandler):
            def log_message(self, *a):
                pass

            def do_GET(self):
                if self.path == "/metrics":
                    body = generate_latest(reg)
                    self.send_response(200)
                    self.send_header("Content-Type",
                                     "text/plain; version=0.0.4")
                    self.send_header("Content-Length", str(len(body)))
                    self.end_headers()
                    self.wfile.write(body)
                elif self.path == "/health":
                    self.send_response(200)
                    self.send_header("Content-Length", "2")
                    self.end_headers()
                    self.wfile.write(b"ok")
                else:
                    self.send_response(404)
                    self.send_header("Content-Length", "0")
                    self.end_headers()

        self._httpd = ThreadingHTTPServer((host, port), Handler)
        self.port = self._httpd.server_address[1]
        threading.Thread(target=self._httpd.serve_forever,
                         daemon=True).start()
        return self

    def stop(self):
        self._stop.set()
        if self._httpd:
            self._httpd.shutdown()
            self._httpd.server_close()

    def render(self) -> bytes:
        return generate_latest(self.registry)
