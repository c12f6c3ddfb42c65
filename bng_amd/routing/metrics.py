"""Routing-specific Prometheus instruments (ref pkg/routing/metrics.go:
bng_routing_* series: subscriber routes, BGP neighbors/prefixes/state
changes, BFD peers/packets/state changes, FRR command stats) plus a
collect() that pulls live numbers from the BGP controller and
subscriber-route manager."""
from __future__ import annotations

from typing import Optional

from prometheus_client import CollectorRegistry, Counter, Gauge, Histogram


class RoutingMetrics:
    def __init__(self, registry: Optional[CollectorRegistry] = None):
        self.registry = registry or CollectorRegistry()
        r = self.registry
        self.subscriber_routes_active = Gauge(
            "bng_routing_subscriber_routes_active",
            "currently injected /32 subscriber routes", registry=r)
        self.routes_injected = Counter(
            "bng_routing_subscriber_routes_injected_total",
            "route injections", registry=r)
        self.routes_withdrawn = Counter(
            "bng_routing_subscriber_routes_withdrawn_total",
            "route withdrawals", registry=r)
        self.injection_latency = Histogram(
            "bng_routing_route_injection_latency_seconds",
            "inject latency", registry=r,
            buckets=(1e-4, 1e-3, 1e-2, 1e-1, 1, 5))
        self.withdrawal_latency = Histogram(
            "bng_routing_route_withdrawal_latency_seconds",
            "withdraw latency", registry=r,
            buckets=(1e-4, 1e-3, 1e-2, 1e-1, 1, 5))
        self.injection_errors = Counter(
            "bng_routing_route_injection_errors_total",
            "inject failures", registry=r)
        self.withdrawal_errors = Counter(
            "bng_routing_route_withdrawal_errors_total",
            "withdraw failures", registry=r)
        self.bgp_neighbors_total = Gauge(
            "bng_routing_bgp_neighbors_total", "configured neighbors",
            registry=r)
        self.bgp_neighbors_established = Gauge(
            "bng_routing_bgp_neighbors_established",
            "neighbors in Established", registry=r)
        self.bgp_prefixes_announced = Gauge(
            "bng_routing_bgp_prefixes_announced", "announced prefixes",
            registry=r)
        self.bgp_prefixes_received = Gauge(
            "bng_routing_bgp_prefixes_received", "received prefixes",
            registry=r)
        self.bgp_state_changes = Counter(
            "bng_routing_bgp_session_state_changes_total",
            "BGP session transitions", ["neighbor", "state"], registry=r)
        self.bfd_peers_total = Gauge(
            "bng_routing_bfd_peers_total", "configured BFD peers",
            registry=r)
        self.bfd_peers_up = Gauge(
            "bng_routing_bfd_peers_up", "BFD peers up", registry=r)
        self.bfd_peers_down = Gauge(
            "bng_routing_bfd_peers_down", "BFD peers down", registry=r)
        self.bfd_state_changes = Counter(
            "bng_routing_bfd_state_changes_total", "BFD transitions",
            ["peer", "state"], registry=r)
        self.bfd_packets_tx = Counter(
            "bng_routing_bfd_packets_tx_total", "BFD control tx",
            registry=r)
        self.bfd_packets_rx = Counter(
            "bng_routing_bfd_packets_rx_total", "BFD control rx",
            registry=r)
        self.frr_commands = Counter(
            "bng_routing_frr_commands_total", "vtysh invocations",
            ["result"], registry=r)
        self.frr_command_latency = Histogram(
            "bng_routing_frr_command_latency_seconds", "vtysh latency",
            registry=r, buckets=(1e-3, 1e-2, 1e-1, 1, 5))

    def record_route_injection(self, seconds: float = 0.0,
                               ok: bool = True):
        if ok:
            self.routes_injected.inc()
            self.injection_latency.observe(seconds)
        else:
            self.injection_errors.inc()

    def record_route_withdrawal(self, seconds: float = 0.0,
                                ok: bool = True):
        if ok:
            self.routes_withdrawn.inc()
            self.withdrawal_latency.observe(seconds)
        else:
            self.withdrawal_errors.inc()

    def collect(self, bgp=None, route_manager=None, bfd=None):
        """Pull gauges from the live controllers (the reference's
        periodic UpdateFromState, metrics.go:203-260)."""
        if route_manager is not None:
            self.subscriber_routes_active.set(len(route_manager.installed))
        if bgp is not None:
            self.bgp_neighbors_total.set(len(bgp.neighbors))
            self.bgp_neighbors_established.set(
                sum(1 for n in bgp.neighbors.values() if n.established))
            self.bgp_prefixes_announced.set(len(bgp.announced))
        if bfd is not None:
            self.bfd_peers_total.set(len(bfd.peers))
            up = sum(1 for p in bfd.peers.values() if p.up)
            self.bfd_peers_up.set(up)
            self.bfd_peers_down.set(len(bfd.peers) - up)
