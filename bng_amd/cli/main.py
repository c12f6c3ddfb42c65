"""bng CLI (ref cmd/bng/main.go): `run` wires every subsystem in the
reference's order (main.go:441-1298) with reverse-order cleanup
(:1300-1379); `demo` simulates the ONT->walled-garden->activation
lifecycle with no dataplane (demo.go:46-60); `stats`; `version`.

Flags mirror the reference's surface (~110 cobra flags; the load-bearing
ones are implemented, YAML config supplies the rest); file values are
applied only to flags not explicitly set on the command line
(ref main.go:1420-1457), and secrets support --*-file indirection
(ref resolveSecret :1567-1592).
"""
from __future__ import annotations

import argparse
import json
import logging
import os
import signal
import sys
import time
from typing import List, Optional

from .. import __version__


def parse_duration(v) -> int:
    """Accept Go-style duration strings from flags/YAML (the reference
    uses time.Duration flags): '300', '300s', '5m', '24h' -> seconds."""
    if isinstance(v, (int, float)):
        return int(v)
    s = str(v).strip().lower()
    mult = 1
    for suffix, m in (("ms", 0.001), ("s", 1), ("m", 60), ("h", 3600),
                      ("d", 86400)):
        if s.endswith(suffix):
            s = s[:-len(suffix)]
            mult = m
            break
    return int(float(s) * mult)


def build_parser() -> argparse.ArgumentParser:
    p = argparse.ArgumentParser(prog="bng",
                                description="MI355X-native BNG")
    sub = p.add_subparsers(dest="command")

    run = sub.add_parser("run", help="run the BNG")
    g = run.add_argument_group("core")
    g.add_argument("--config", help="YAML config file")
    g.add_argument("--log-level", default="info",
                   choices=["debug", "info", "warn", "error"])
    g.add_argument("--node-id", default="bng-1")
    g.add_argument("--interface", default="eth0")
    g.add_argument("--server-ip", default="10.0.0.1")
    g.add_argument("--server-mac", default="02:00:00:00:00:01")
    g.add_argument("--gpu", default="auto",
                   help="auto|off|cuda:N — dataplane device")
    g.add_argument("--pktio", default="off",
                   choices=["off", "afpacket", "afxdp"],
                   help="NIC edge: attach a packet pump to --interface "
                        "(afxdp = UMEM rings + XDP redirect, the ref's "
                        "loader.go:294-315 attach; afpacket = raw-socket "
                        "fallback, the ref's generic mode)")
    g.add_argument("--pktio-batch", type=int, default=8192)
    g.add_argument("--pktio-max-wait", type=float, default=0.0005,
                   help="batch deadline seconds (latency/throughput "
                        "trade at the NIC edge)")
    g.add_argument("--dhcp-serve", default="batched",
                   choices=["batched", "persistent"],
                   help="persistent = route DHCP through the resident "
                        "service kernel (~31us p50) instead of the "
                        "batched pipeline (GPU only)")
    g.add_argument("--bpf-path", default="",
                   help="accepted for reference CLI compatibility; the "
                        "MI355X dataplane compiles in-tree HIP kernels "
                        "instead of loading .bpf.o")
    g.add_argument("--metrics-addr", default="",
                   help="host:port for /metrics (reference spelling of "
                        "--metrics-port)")
    g = run.add_argument_group("dhcp")
    g.add_argument("--pool-network", default="")
    g.add_argument("--pool-gateway", default="")
    g.add_argument("--pool-dns", default="")
    g.add_argument("--lease-time", type=parse_duration, default=3600,
                   help="seconds or Go duration (24h, 5m)")
    g.add_argument("--dhcp-listen", action="store_true",
                   help="bind UDP :67 (off in tests)")
    g = run.add_argument_group("radius")
    g.add_argument("--radius-server", action="append", default=[])
    g.add_argument("--radius-servers", default="",
                   help="comma list (reference spelling)")
    g.add_argument("--radius-enabled", action="store_true")
    g.add_argument("--radius-nas-id", default="bng")
    g.add_argument("--radius-timeout", type=parse_duration, default=3)
    g.add_argument("--radius-secret", default="")
    g.add_argument("--radius-secret-file", default="")
    g.add_argument("--radius-auth-mode", default="none",
                   choices=["none", "mac"])
    g.add_argument("--radius-coa-port", type=int, default=0)
    g.add_argument("--radius-partition-mode", default="cached",
                   choices=["off", "reject", "cached", "allow"])
    g = run.add_argument_group("resilience")
    g.add_argument("--health-check-interval", type=float, default=5.0)
    g.add_argument("--short-lease-enable", "--short-lease-enabled",
                   action="store_true")
    g.add_argument("--short-lease-threshold", type=float, default=0.9)
    g.add_argument("--short-lease-duration", type=parse_duration,
                   default=60)
    g.add_argument("--health-check-retries", type=int, default=3)
    g.add_argument("--pool-mode", default="static",
                   choices=["static", "session", "lease"])
    g.add_argument("--epoch-period", type=parse_duration, default=300)
    g.add_argument("--epoch-grace", type=int, default=1)
    g = run.add_argument_group("nexus")
    g.add_argument("--nexus-url", default="")
    g.add_argument("--nexus-auth", default="none",
                   choices=["none", "psk", "mtls"])
    g.add_argument("--nexus-psk", default="")
    g.add_argument("--nexus-pool", default="default")
    g.add_argument("--clset-data-dir", default="",
                   help="run an embedded CLSet CRDT replica persisted "
                        "here (ref crdt_backend.go badger store)")
    g.add_argument("--clset-peer", action="append", default=[],
                   help="CLSet peer sync URL (repeatable)")
    g.add_argument("--clset-listen-port", type=int, default=0)
    g = run.add_argument_group("deviceauth")
    g.add_argument("--auth-mode", default="",
                   choices=["", "none", "psk", "mtls"],
                   help="device->Nexus auth (reference spelling of "
                        "--nexus-auth)")
    g.add_argument("--auth-psk", default="")
    g.add_argument("--auth-psk-file", default="")
    g.add_argument("--auth-mtls-cert", default="")
    g.add_argument("--auth-mtls-key", default="")
    g.add_argument("--auth-mtls-ca", default="")
    g.add_argument("--auth-mtls-server-name", default="",
                   help="accepted for compatibility; SNI follows the "
                        "--nexus-url host")
    g.add_argument("--auth-mtls-insecure", action="store_true")
    g = run.add_argument_group("peers")
    g.add_argument("--peer", action="append", default=[],
                   help="node_id=url")
    g.add_argument("--peers", default="",
                   help="comma list of node_id=url (reference spelling)")
    g.add_argument("--peer-discovery", default="static",
                   choices=["static", "dns"])
    g.add_argument("--peer-service", default="",
                   help="DNS SRV name for --peer-discovery dns")
    g.add_argument("--peer-listen", default=":8081")
    g.add_argument("--peer-dns-server", default="",
                   help="DNS server for --peer-discovery dns "
                        "(host:port)")
    g = run.add_argument_group("nat")
    g.add_argument("--nat-enable", "--nat-enabled", action="store_true")
    g.add_argument("--nat-public-ip", action="append", default=[])
    g.add_argument("--nat-ports-per-subscriber", type=int, default=1024)
    g.add_argument("--nat-log-path", default="")
    g.add_argument("--nat-log-format", default="json",
                   choices=["json", "csv", "syslog", "nel"])
    g.add_argument("--nat-bulk-logging", action="store_true")
    g.add_argument("--nat-bpf-path", default="",
                   help="accepted for compatibility (see --bpf-path)")
    g.add_argument("--nat-public-ips", default="",
                   help="comma list (reference spelling)")
    g.add_argument("--nat-ports-per-sub", type=int, default=0,
                   help="reference spelling of --nat-ports-per-subscriber")
    g.add_argument("--nat-log-enabled", action="store_true",
                   help="accepted for compatibility; logging activates "
                        "when --nat-log-path is set")
    g.add_argument("--nat-inside-interface", default="",
                   help="accepted for compatibility; the uplink pump "
                        "attaches to --interface (the access NIC)")
    g.add_argument("--nat-outside-interface", default="")
    g.add_argument("--nat-eim", default=True, type=lambda v: v not in
                   ("false", "0", "no", False))
    g.add_argument("--nat-eif", default=True, type=lambda v: v not in
                   ("false", "0", "no", False))
    g.add_argument("--nat-hairpin", default=True, type=lambda v: v not in
                   ("false", "0", "no", False))
    g.add_argument("--nat-alg-ftp", default=True, type=lambda v: v not in
                   ("false", "0", "no", False))
    g.add_argument("--nat-alg-sip", default=False, type=lambda v: v in
                   ("true", "1", "yes", True))
    g = run.add_argument_group("qos")
    g.add_argument("--qos-policy", action="append", default=[],
                   help="name:down_mbps:up_mbps")
    g.add_argument("--qos-enabled", action="store_true",
                   help="accepted for compatibility; the QoS manager "
                        "is always wired, policies activate it")
    g.add_argument("--qos-bpf-path", default="",
                   help="accepted for compatibility (see --bpf-path)")
    g.add_argument("--qos-default-policy", default="")
    g = run.add_argument_group("antispoof")
    g.add_argument("--antispoof-mode", default="disabled",
                   choices=["disabled", "strict", "loose", "log_only"])
    g = run.add_argument_group("pppoe")
    g.add_argument("--pppoe-enable", "--pppoe-enabled", action="store_true")
    g.add_argument("--pppoe-ac-name", default="bng-amd")
    g.add_argument("--pppoe-auth", "--pppoe-auth-type", default="chap",
                   choices=["chap", "pap", "none"])
    g.add_argument("--pppoe-interface", default="",
                   help="defaults to --interface")
    g.add_argument("--pppoe-service-name", default="")
    g.add_argument("--pppoe-session-timeout", type=parse_duration,
                   default=1800)
    g = run.add_argument_group("ipv6")
    g.add_argument("--dhcpv6-enable", "--dhcpv6-enabled", action="store_true")
    g.add_argument("--dhcpv6-na-pool", default="2001:db8:1::/64")
    g.add_argument("--dhcpv6-pd-pool", default="2001:db8:f000::/40")
    g.add_argument("--slaac-enable", "--slaac-enabled", action="store_true")
    g.add_argument("--slaac-prefix", default="")
    g.add_argument("--dhcpv6-address-pool", default="",
                   help="reference spelling of --dhcpv6-na-pool")
    g.add_argument("--dhcpv6-prefix-pool", default="",
                   help="reference spelling of --dhcpv6-pd-pool")
    g.add_argument("--dhcpv6-dns", default="", help="comma list")
    g.add_argument("--dhcpv6-domain-search", default="",
                   help="comma list")
    g.add_argument("--slaac-prefixes", default="",
                   help="comma list (reference spelling)")
    g.add_argument("--slaac-managed", action="store_true")
    g.add_argument("--slaac-other", action="store_true")
    g.add_argument("--slaac-dns", default="", help="comma RDNSS list")
    g.add_argument("--slaac-dns-domains", default="",
                   help="comma DNSSL list")
    g.add_argument("--slaac-min-interval", type=parse_duration,
                   default=200)
    g.add_argument("--slaac-max-interval", type=parse_duration,
                   default=600)
    g = run.add_argument_group("routing")
    g.add_argument("--bgp-enable", "--bgp-enabled", action="store_true")
    g.add_argument("--bgp-local-as", type=int, default=65000)
    g.add_argument("--bgp-neighbor", action="append", default=[],
                   help="addr:remote_as")
    g.add_argument("--bgp-announce-subscribers", action="store_true")
    g.add_argument("--bgp-router-id", default="")
    g.add_argument("--bgp-neighbors", default="",
                   help="comma list of addr:remote_as (reference "
                        "spelling)")
    g.add_argument("--bgp-bfd-enabled", action="store_true")
    g = run.add_argument_group("ha")
    g.add_argument("--ha-role", default="",
                   choices=["", "active", "standby"])
    g.add_argument("--ha-partner-url", default="")
    g.add_argument("--ha-listen-port", type=int, default=0)
    g.add_argument("--ha-peer", default="",
                   help="reference spelling of --ha-partner-url")
    g.add_argument("--ha-listen", default="",
                   help="host:port bind for the sync server")
    g.add_argument("--ha-auth-token", default="",
                   help="shared secret for /sync/* (required for "
                        "non-loopback --ha-listen)")
    g.add_argument("--ha-tls-cert", default="")
    g.add_argument("--ha-tls-key", default="")
    g.add_argument("--ha-tls-ca", default="")
    g.add_argument("--ha-tls-skip-verify", action="store_true")
    g = run.add_argument_group("observability")
    g.add_argument("--metrics-port", type=int, default=9090)
    g.add_argument("--metrics-enable", action="store_true")
    g.add_argument("--audit-log-path", default="")
    g = run.add_argument_group("walledgarden")
    g.add_argument("--walled-garden", action="store_true")
    g.add_argument("--walled-garden-portal", default="")

    demo = sub.add_parser("demo", help="simulated subscriber lifecycle")
    demo.add_argument("--subscribers", type=int, default=3)

    st = sub.add_parser("stats", help="fetch stats from a running bng")
    st.add_argument("--metrics-url", default="http://127.0.0.1:9090")
    sub.add_parser("version", help="print version")
    sub.add_parser("verify", help="verify the dataplane: extension "
                   "build, struct ABI, golden self-test (the "
                   "cmd/verify-bpf analog)")
    return p


def load_yaml_over_args(args: argparse.Namespace,
                        parser: argparse.ArgumentParser,
                        argv: List[str]) -> argparse.Namespace:
    """Merge YAML config under explicit flags (ref main.go:1420-1457):
    file values apply only to flags NOT set on the command line."""
    if not getattr(args, "config", None):
        return args
    import yaml
    with open(args.config) as f:
        cfg = yaml.safe_load(f) or {}
    explicit = set()
    for tok in argv:
        if tok.startswith("--"):
            explicit.add(tok.split("=")[0][2:].replace("-", "_"))
    for key, value in cfg.items():
        attr = key.replace("-", "_")
        if attr in explicit or not hasattr(args, attr):
            continue
        cur = getattr(args, attr)
        if isinstance(cur, list) and not isinstance(value, list):
            value = [value]
        setattr(args, attr, value)
    return args


def apply_env_overrides(args: argparse.Namespace,
                        argv: List[str],
                        environ: Optional[dict] = None
                        ) -> argparse.Namespace:
    """BNG_<FLAG> environment overrides (ref FEATURES.md config
    sources: flags > env > YAML): applied to any flag not given
    explicitly on the command line, coerced to the current attr's
    type; lists split on commas."""
    env = environ if environ is not None else os.environ
    explicit = {tok.split("=")[0][2:].replace("-", "_")
                for tok in argv if tok.startswith("--")}
    for name, raw in env.items():
        if not name.startswith("BNG_"):
            continue
        attr = name[4:].lower()
        if attr in explicit or not hasattr(args, attr):
            continue
        cur = getattr(args, attr)
        try:
            if isinstance(cur, bool):
                value = raw.lower() in ("1", "true", "yes", "on")
            elif isinstance(cur, int):
                value = parse_duration(raw)
            elif isinstance(cur, float):
                value = float(raw)
            elif isinstance(cur, list):
                value = [x for x in raw.split(",") if x]
            else:
                value = raw
        except (ValueError, TypeError):
            continue
        setattr(args, attr, value)
    return args


def resolve_secret(value: str, file_value: str) -> str:
    """ref main.go:1567-1592 resolveSecret."""
    if file_value:
        with open(file_value) as f:
            return f.read().strip()
    return value


class BNG:
    """The wired application (runBNG analog).  Subsystems start in the
    reference's order and stop in reverse."""

    def __init__(self, args: argparse.Namespace):
        self.args = args
        self.log = logging.getLogger("bng")
        self._cleanup: List = []
        self.launcher = None
        self.dhcp_server = None
        self.metrics = None
        self.audit = None
        self.bgp = None
        self.sub_routes = None

    def _defer(self, fn):
        self._cleanup.append(fn)

    @staticmethod
    def _normalize(a):
        """Fold reference-spelling flags into their canonical attrs
        (comma lists, -enabled aliases, host:port forms)."""
        if getattr(a, "radius_servers", ""):
            a.radius_server += [x for x in a.radius_servers.split(",") if x]
        if getattr(a, "nat_public_ips", ""):
            a.nat_public_ip += [x for x in a.nat_public_ips.split(",") if x]
        if getattr(a, "nat_ports_per_sub", 0):
            a.nat_ports_per_subscriber = a.nat_ports_per_sub
        if getattr(a, "bgp_neighbors", ""):
            a.bgp_neighbor += [x for x in a.bgp_neighbors.split(",") if x]
        if getattr(a, "peers", ""):
            a.peer += [x for x in a.peers.split(",") if x]
        if getattr(a, "slaac_prefixes", ""):
            if not a.slaac_prefix:
                a.slaac_prefix = a.slaac_prefixes.split(",")[0]
        if getattr(a, "dhcpv6_address_pool", ""):
            a.dhcpv6_na_pool = a.dhcpv6_address_pool
        if getattr(a, "dhcpv6_prefix_pool", ""):
            a.dhcpv6_pd_pool = a.dhcpv6_prefix_pool
        if getattr(a, "ha_peer", "") and not a.ha_partner_url:
            a.ha_partner_url = a.ha_peer
        if getattr(a, "metrics_addr", ""):
            try:
                a.metrics_port = int(a.metrics_addr.rsplit(":", 1)[-1])
            except ValueError:
                pass
        if getattr(a, "auth_mode", ""):
            a.nexus_auth = a.auth_mode
        if getattr(a, "auth_psk", "") and not a.nexus_psk:
            a.nexus_psk = a.auth_psk
        if getattr(a, "auth_psk_file", "") and not a.nexus_psk:
            a.nexus_psk = resolve_secret("", a.auth_psk_file)
        if getattr(a, "radius_enabled", False) and not a.radius_server:
            raise SystemExit(
                "--radius-enabled requires --radius-server(s)")
        return a

    def start(self):
        a = self._normalize(self.args)
        from ..dataplane.launcher import GoldenLauncher, HipLauncher
        from ..dataplane.packets import ip2u32, mac_bytes

        # 1. dataplane launcher (the eBPF loader analog, main.go:498)
        use_gpu = False
        if a.gpu != "off":
            try:
                import torch
                use_gpu = torch.cuda.is_available()
            except Exception:
                use_gpu = False
        if use_gpu:
            self.launcher = HipLauncher(
                a.gpu if a.gpu not in ("auto",) else "cuda:0")
            self.log.info("GPU dataplane on %s", self.launcher.device)
        else:
            self.launcher = GoldenLauncher()
            self.log.info("CPU (golden-model) dataplane")

        # 2. antispoof (main.go:532)
        from ..antispoof.manager import Manager as AntispoofMgr
        self.antispoof = AntispoofMgr(self.launcher,
                                      default_mode=a.antispoof_mode)

        # 3. walled garden (main.go:556)
        from ..walledgarden.manager import Manager as WGMgr
        portal = a.walled_garden_portal or \
            ("10.255.255.1:8080" if a.walled_garden else "")
        p_ip, _, p_port = portal.partition(":")
        self.walledgarden = WGMgr(
            portal_ip=p_ip,
            portal_port=int(p_port) if p_port else 8080).start()
        self._defer(self.walledgarden.stop)

        # 4. pools + DHCP server (main.go:567-594, :1244)
        from ..dhcp.pool import PoolConfig, PoolManager
        from ..dhcp.server import DHCPServer
        self.pool_manager = PoolManager(self.launcher)
        if a.pool_network:
            self.pool_manager.add_pool(PoolConfig(
                1, a.pool_network, gateway=a.pool_gateway,
                dns=[d for d in a.pool_dns.split(",") if d],
                lease_time=a.lease_time))
        self.dhcp_server = DHCPServer(self.pool_manager, a.server_ip,
                                      mac_bytes(a.server_mac),
                                      lease_time=a.lease_time)
        self.dhcp_server.set_launcher(self.launcher)
        self.dhcp_server.set_walled_garden(self.walledgarden)

        # 5. nexus (main.go:653-689); an embedded CLSet replica is the
        # reference's CRDT-backed store mode (crdt_backend.go)
        if a.clset_data_dir or a.clset_peer:
            from ..nexus.clset import CLSetHTTPServer, CLSetStore
            from ..nexus.client import Client as NexusClient
            self.clset = CLSetStore(
                a.node_id, data_dir=a.clset_data_dir or None)
            self.clset_http = CLSetHTTPServer(
                self.clset, port=a.clset_listen_port).start()
            self.clset.advertise_url = self.clset_http.url
            for u in a.clset_peer:
                self.clset.add_peer_url(u)
            self.clset.start()
            self._defer(self.clset_http.stop)
            self._defer(self.clset.close)
            self.nexus_client = NexusClient(self.clset,
                                            node_id=a.node_id)
            self.dhcp_server.set_nexus(client=self.nexus_client)
            self.log.info("embedded CLSet replica at %s",
                          self.clset_http.url)
        if a.nexus_url:
            from ..nexus.http_allocator import HTTPAllocator
            headers = None
            if a.nexus_auth == "psk" and a.nexus_psk:
                from ..deviceauth.authenticator import PSKAuthenticator
                headers = PSKAuthenticator(a.nexus_psk).headers(a.node_id)
            self.nexus_allocator = HTTPAllocator(
                a.nexus_url, auth_headers=headers,
                client_cert=a.auth_mtls_cert, client_key=a.auth_mtls_key,
                ca_cert=a.auth_mtls_ca, insecure=a.auth_mtls_insecure)
            self.dhcp_server.set_nexus(allocator=self.nexus_allocator)

        # 6. peer pool (main.go:719-756); DNS-SRV peer discovery
        # (ref peer-discovery=dns + peer-service): each SRV record
        # target:port becomes a peer URL, node id = target host
        if a.peer_discovery == "dns" and a.peer_service:
            from ..dns.resolver import resolve_srv
            try:
                srvs = resolve_srv(a.peer_service,
                                   getattr(a, "peer_dns_server",
                                           "") or "127.0.0.1:53")
                for _pri, _w, port, target in sorted(srvs):
                    if target and target != a.node_id:
                        a.peer.append(f"{target}=http://{target}:{port}")
                self.log.info("DNS-SRV discovered %d peer(s)", len(srvs))
            except OSError as e:
                self.log.warning("DNS-SRV discovery failed: %s", e)
        if a.peer and a.pool_network:
            from ..pool.peer import PeerPool
            peers = dict(p.split("=", 1) for p in a.peer)
            listen_port = 0
            if a.peer_listen:
                try:
                    listen_port = int(a.peer_listen.rsplit(":", 1)[-1])
                except ValueError:
                    pass
            self.peer_pool = PeerPool(a.node_id, peers, a.pool_network,
                                      listen_port=listen_port).start()
            self._defer(self.peer_pool.stop)
            self.dhcp_server.set_peer_pool(self.peer_pool)

        # 6b. store-replicated allocator modes (ref modes.go:14-92):
        # pool-mode session|lease runs a DistributedAllocator over the
        # embedded CLSet (or a process-local store without one) with
        # the configured epoch period/grace
        if a.pool_mode != "static" and a.pool_network:
            from ..allocator.distributed import DistributedAllocator
            from ..nexus.store import MemoryStore
            store = getattr(self, "clset", None) or MemoryStore()
            self.distributed_alloc = DistributedAllocator(
                store, a.nexus_pool, a.pool_network, mode=a.pool_mode,
                grace_period=a.epoch_grace,
                epoch_interval=(float(a.epoch_period)
                                if a.pool_mode == "lease" else 0.0),
                node_id=a.node_id)
            self._defer(self.distributed_alloc.close)
            self.dhcp_server.set_distributed(self.distributed_alloc)
            self.log.info("pool-mode %s (epoch %ss, grace %d)",
                          a.pool_mode, a.epoch_period, a.epoch_grace)

        # 7. HA (main.go:826-881)
        if a.ha_role:
            from ..ha.failover import FailoverController
            from ..ha.health_monitor import HealthMonitor
            from ..ha.sync import HASyncer
            ha_host, ha_port = "127.0.0.1", a.ha_listen_port
            if a.ha_listen:
                h, _, pt = a.ha_listen.rpartition(":")
                ha_host = h or "0.0.0.0"
                ha_port = int(pt or 0)
            self.ha = HASyncer(a.node_id, a.ha_role,
                               listen_port=ha_port,
                               listen_host=ha_host,
                               auth_token=a.ha_auth_token,
                               partner_url=a.ha_partner_url,
                               tls_cert=a.ha_tls_cert,
                               tls_key=a.ha_tls_key,
                               tls_ca=a.ha_tls_ca,
                               tls_skip_verify=a.ha_tls_skip_verify).start()
            self._defer(self.ha.stop)
            from ..ha import nat_glue, session_glue
            session_glue.attach(self.dhcp_server, self.ha,
                                nat_mgr=lambda: getattr(self, "nat", None))
            # NAT flow replication rides the same syncer (task: a
            # promoted standby keeps established NAT sessions)
            self._nat_ha = nat_glue.NatHaGlue(self.launcher,
                                              self.ha).start()
            self._defer(self._nat_ha.stop)
            if a.ha_partner_url:
                self.ha_monitor = HealthMonitor(a.ha_partner_url).start()
                self._defer(self.ha_monitor.stop)

                def _role_change(r):
                    if r == "active":
                        session_glue.promote(
                            self.dhcp_server, self.ha,
                            getattr(self, "qos", None),
                            getattr(self, "nat", None))
                        nat_glue.promote_nat(self.launcher, self.ha)
                    else:
                        self.ha.demote()
                self.ha_failover = FailoverController(
                    a.node_id, a.ha_role, monitor=self.ha_monitor,
                    role_change_callback=_role_change)

        # 8. routing (main.go:901-939)
        if a.bgp_enable:
            from ..routing.bgp import BGPController
            from ..routing.frr import FakeExecutor, VtyshExecutor
            exe = VtyshExecutor() if os.path.exists("/usr/bin/vtysh") \
                else FakeExecutor()
            self.bgp = BGPController(exe, a.bgp_local_as,
                                     router_id=a.bgp_router_id).start()
            if a.bgp_bfd_enabled:
                from ..routing.bgp import BFDManager
                self.bfd = BFDManager(exe)
            for n in a.bgp_neighbor:
                addr, _, ras = n.partition(":")
                self.bgp.add_neighbor(addr, int(ras or 65000))
                if a.bgp_bfd_enabled:
                    self.bfd.add_peer(addr)
            if a.bgp_announce_subscribers:
                from ..dataplane.packets import u32_to_ip
                from ..routing.manager import (SessionRouteIntegration,
                                               SubscriberRouteManager)
                self.sub_routes = SubscriberRouteManager(self.bgp).start()
                self._defer(self.sub_routes.stop)
                self.route_integ = SessionRouteIntegration(self.sub_routes)

                def _lease_to_route(event, lease):
                    sid = "dhcp-" + lease.mac.hex()
                    if event == "add":
                        self.route_integ.on_session_activate(
                            sid, lease.mac.hex(), u32_to_ip(lease.ip))
                    elif event == "delete":
                        self.route_integ.on_session_terminate(
                            sid, reason="lease-removed")
                self.dhcp_server.on_lease_event.append(_lease_to_route)

        # 9. RADIUS + policies + QoS (main.go:951-985)
        from ..radius.policy import Policy, PolicyManager
        self.policy_manager = PolicyManager()
        for spec in a.qos_policy:
            name, down, up = spec.split(":")
            self.policy_manager.add_policy(Policy(
                name, int(float(down) * 1e6), int(float(up) * 1e6)))
        if a.qos_default_policy:
            self.policy_manager.default_policy = \
                self.policy_manager.get(a.qos_default_policy)
        from ..qos.manager import Manager as QoSMgr
        self.qos = QoSMgr(self.launcher, self.policy_manager)
        self.dhcp_server.set_qos_manager(self.qos)
        self.dhcp_server.set_policy_manager(self.policy_manager)
        if a.radius_server:
            from ..radius.accounting import AccountingManager
            from ..radius.client import Client as RadiusClient
            secret = resolve_secret(a.radius_secret, a.radius_secret_file)
            self.radius = RadiusClient(a.radius_server, secret.encode(),
                                       nas_identifier=a.radius_nas_id,
                                       timeout=a.radius_timeout)
            if a.radius_partition_mode != "off":
                # partition degradation modes (ref resilience wiring
                # main.go:1182-1211, radius_handler.go:52)
                from ..resilience.radius_handler import ResilientRadius
                self.radius_resilient = ResilientRadius(
                    self.radius, mode=a.radius_partition_mode)
                self.dhcp_server.set_radius(self.radius_resilient,
                                            a.radius_auth_mode)
            else:
                self.dhcp_server.set_radius(self.radius,
                                            a.radius_auth_mode)
            self.accounting = AccountingManager(self.radius).start()
            self._defer(self.accounting.stop)
            self.dhcp_server.set_accounting(self.accounting)
            if a.radius_coa_port:
                from ..radius.coa import CoAProcessor, CoAServer
                proc = CoAProcessor(
                    session_lookup=self._coa_lookup,
                    terminate=self._coa_terminate,
                    qos_updater=lambda lease, pol:
                        self.qos.update_subscriber_policy(lease.ip, pol))
                self.coa = CoAServer(secret.encode(),
                                     port=a.radius_coa_port,
                                     handler=proc).start()
                self._defer(self.coa.stop)

        # 9b. resilience: partition FSM + short-lease under pool
        # pressure (ref main.go:1182-1211)
        from ..resilience.manager import Manager as ResilienceMgr

        def _health() -> bool:
            alloc = getattr(self, "nexus_allocator", None)
            if alloc is None:
                return True
            try:
                return bool(alloc.health_check())
            except Exception:
                return False
        self.resilience = ResilienceMgr(
            _health, check_interval=a.health_check_interval,
            failure_threshold=a.health_check_retries).start()
        self._defer(self.resilience.stop)
        if getattr(self, "radius_resilient", None) is not None:
            self.resilience.on_transition(
                lambda old, new: new == "online" and
                self.radius_resilient.replay_buffered())
        if a.short_lease_enable:
            from ..resilience.pool_monitor import PoolMonitor

            def _util() -> float:
                stats = self.dhcp_server.pools.all_stats()
                alloc = sum(x["allocated"] for x in stats)
                total = sum(x["allocated"] + x["available"]
                            for x in stats)
                return alloc / total if total else 0.0
            self.pool_monitor = PoolMonitor(
                _util, critical=a.short_lease_threshold,
                normal_lease=a.lease_time,
                short_lease=a.short_lease_duration)
            self.dhcp_server.lease_time_provider = \
                self.pool_monitor.effective_lease_time

        # 10. NAT (main.go:1001-1040); mode flags mirror the
        # reference's nat-eim/eif/hairpin/alg-* booleans
        if a.nat_enable:
            from ..dataplane import abi as _abi
            from ..nat.logging import ComplianceLogger
            from ..nat.manager import Manager as NATMgr
            logger = ComplianceLogger(a.nat_log_path or None,
                                      fmt=a.nat_log_format,
                                      bulk_mode=a.nat_bulk_logging)
            natflags = 0
            alg_ports = []
            if a.nat_eim:
                natflags |= _abi.NAT_FLAG_EIM
            if a.nat_eif:
                natflags |= _abi.NAT_FLAG_EIF
            if a.nat_hairpin:
                natflags |= _abi.NAT_FLAG_HAIRPIN
            if a.nat_alg_ftp:
                natflags |= _abi.NAT_FLAG_ALG_FTP
                alg_ports.append((21, 6))
            if a.nat_alg_sip:
                natflags |= _abi.NAT_FLAG_ALG_SIP
                alg_ports += [(5060, 17), (5060, 6)]
            self.nat = NATMgr(self.launcher,
                              ports_per_subscriber=a.nat_ports_per_subscriber,
                              flags=natflags, logger=logger,
                              alg_ports=alg_ports)
            for ip in a.nat_public_ip:
                self.nat.add_public_ip(ip)
            self.nat.start()
            self._defer(self.nat.stop)
            self.dhcp_server.set_nat_manager(self.nat)

        # 11. PPPoE / DHCPv6 / SLAAC (main.go:1063-1180)
        if a.pppoe_enable:
            from ..pppoe.server import PPPoEServer
            from ..dataplane.packets import mac_bytes as mb, u32_to_ip
            self.pppoe = PPPoEServer(
                mb(a.server_mac), ac_name=a.pppoe_ac_name,
                service_name=a.pppoe_service_name, auth=a.pppoe_auth,
                session_timeout=float(a.pppoe_session_timeout))
            if getattr(self, "radius", None):
                self.pppoe.radius = self.radius
            if a.pool_network:
                # PPPoE IPCP addresses from the same pool family
                from ..allocator.bitmap import BitmapAllocator
                ppp_alloc = BitmapAllocator(a.pool_network, 32,
                                            reserve_head=2)
                self.pppoe.allocator = \
                    lambda user: ppp_alloc.allocate(user).split("/")[0]
                self.pppoe.releaser = ppp_alloc.release

            # provision the dataplane for opened PPPoE sessions
            # (antispoof binding + QoS + NAT + HA), like the DHCP ACK path
            def _ppp_open(sess):
                self.antispoof.add_binding(sess.client_mac,
                                           ipv4=u32_to_ip(sess.ip))
                self.qos.apply_policy(sess.ip, sess.policy_name)
                if getattr(self, "nat", None):
                    self.nat.allocate_nat(sess.ip, sess.username)
                if getattr(self, "ha", None):
                    from ..ha.protocol import SessionState
                    self.ha.publish_add(SessionState(
                        session_id=f"pppoe-{sess.session_id}",
                        subscriber_id=sess.username,
                        mac=sess.client_mac.hex(),
                        ip=u32_to_ip(sess.ip), access_type="pppoe",
                        policy_name=sess.policy_name))

            def _ppp_close(sess):
                self.antispoof.remove_binding(sess.client_mac)
                self.qos.remove_policy(sess.ip)
                if getattr(self, "nat", None):
                    self.nat.release_nat(sess.ip)
                if getattr(self, "ha", None):
                    self.ha.publish_delete(f"pppoe-{sess.session_id}")

            self.pppoe.on_session_open = _ppp_open
            self.pppoe.on_session_close = _ppp_close
        if a.dhcpv6_enable:
            from ..dhcpv6.server import DHCPv6Server
            self.dhcpv6 = DHCPv6Server(
                na_pool=a.dhcpv6_na_pool, pd_pool=a.dhcpv6_pd_pool,
                dns=[d for d in a.dhcpv6_dns.split(",") if d],
                domains=[d for d in a.dhcpv6_domain_search.split(",")
                         if d])
        if a.slaac_enable and (a.slaac_prefix or a.slaac_prefixes):
            from ..slaac.radvd import PrefixConfig, RAConfig, Server
            prefixes = ([p for p in a.slaac_prefixes.split(",") if p]
                        if a.slaac_prefixes else [a.slaac_prefix])
            self.slaac = Server(RAConfig(
                prefixes=[PrefixConfig(p) for p in prefixes],
                managed=a.slaac_managed, other_config=a.slaac_other,
                rdnss=[d for d in a.slaac_dns.split(",") if d],
                dnssl=[d for d in a.slaac_dns_domains.split(",") if d],
                min_interval=float(a.slaac_min_interval),
                max_interval=float(a.slaac_max_interval))).start()
            self._defer(self.slaac.stop)

        # 12. metrics (main.go:1214-1241)
        if a.metrics_enable:
            from ..metrics.metrics import Metrics
            self.metrics = Metrics()
            if self.bgp is not None:
                from ..routing.metrics import RoutingMetrics
                self.routing_metrics = RoutingMetrics(
                    registry=self.metrics.registry)
                self.metrics.add_collector(
                    lambda: self.routing_metrics.collect(
                        bgp=self.bgp, route_manager=self.sub_routes))
            self.metrics.start_collector(self.launcher, self.dhcp_server)
            try:
                self.metrics.serve(port=a.metrics_port)
                self._defer(self.metrics.stop)
            except OSError:
                self.log.warning("metrics port busy; collector only")

        # 12b. audit trail: rotating JSON-lines file + the DHCP
        # server's session events (severity/retention defaults apply)
        if a.audit_log_path:
            from ..audit.logger import FileExporter, Logger as AuditLog
            from ..audit.retention import RetentionManager
            self.audit = AuditLog(
                exporters=[FileExporter(a.audit_log_path)],
                retention=RetentionManager()).start()
            self.dhcp_server.audit = self.audit
            self._defer(self.audit.stop)
            self.log.info("audit trail at %s", a.audit_log_path)

        # 13. DHCP serve loop last (main.go:1244)
        self.dhcp_server.start(serve=a.dhcp_listen)
        self._defer(self.dhcp_server.stop)

        # 13b. NIC-edge packet pump (XDP attach analog,
        # loader.go:294-315: afxdp = driver-ish path, afpacket =
        # generic fallback)
        if a.pktio != "off":
            from ..dataplane.pktio import AFPacketIO, Pump
            if a.pktio == "afxdp":
                from ..dataplane.afxdp import XskSocket
                io = XskSocket(a.interface, mode="auto")
                self.log.info("AF_XDP %s mode on %s", io.mode, a.interface)
            else:
                io = AFPacketIO(a.interface)
                self.log.info("AF_PACKET raw socket on %s", a.interface)
            self.pktio = io
            svc = None
            if a.dhcp_serve == "persistent" and use_gpu:
                from ..dataplane.launcher import DhcpService
                svc = DhcpService(self.launcher,
                                  n_slots=max(256, a.pktio_batch))
                self._defer(svc.stop)
                self.dhcp_service = svc
                self.log.info("persistent DHCP service attached")
            self.pump = Pump(self.launcher, io, io,
                             slow_path=self._frame_slow_path,
                             batch=a.pktio_batch,
                             max_wait=a.pktio_max_wait,
                             dhcp_service=svc).start()
            self._defer(io.close)
            self._defer(self.pump.stop)
            # Optional second pump on the core-side NIC running the
            # downlink pipeline (ref tc_egress on the NAT outside
            # interface, tc.c / loader.go) — DNAT'd return traffic is
            # forwarded back out the access NIC.
            dn_if = a.nat_outside_interface
            if dn_if and dn_if != a.interface:
                if a.pktio == "afxdp":
                    from ..dataplane.afxdp import XskSocket
                    dio = XskSocket(dn_if, mode="auto")
                else:
                    dio = AFPacketIO(dn_if)
                self.log.info("downlink pump on %s", dn_if)
                self.pktio_downlink = dio
                self.pump_downlink = Pump(
                    self.launcher, dio, io, direction="downlink",
                    batch=a.pktio_batch,
                    max_wait=a.pktio_max_wait).start()
                self._defer(dio.close)
                self._defer(self.pump_downlink.stop)
            # PPPoE on its own access NIC (ref --pppoe-interface,
            # main.go pppoe raw socket on a dedicated interface):
            # a second uplink pump sharing the launcher + slow path
            pe_if = a.pppoe_interface
            if pe_if and pe_if != a.interface and a.pppoe_enable:
                pio = AFPacketIO(pe_if)
                self.log.info("PPPoE pump on %s", pe_if)
                self.pktio_pppoe = pio
                self.pump_pppoe = Pump(
                    self.launcher, pio, pio,
                    slow_path=self._frame_slow_path,
                    batch=a.pktio_batch,
                    max_wait=a.pktio_max_wait).start()
                self._defer(pio.close)
                self._defer(self.pump_pppoe.stop)
        return self

    def _frame_slow_path(self, frame: bytes):
        """PASS-verdict frames -> slow-path servers: PPPoE discovery/
        session frames to the PPPoE server (which may answer with
        several frames), DHCP cache misses to the DHCP server
        (ref XDP_PASS -> server4 / AF_PACKET -> pppoe)."""
        from ..dataplane.packets import parse_dhcp_frame
        from ..dhcp import message as dm
        if len(frame) >= 14:
            et = int.from_bytes(frame[12:14], "big")
            if et in (0x8863, 0x8864) and \
                    getattr(self, "pppoe", None) is not None:
                return self.pppoe.handle_frame(frame) or None
            if et == 0x0806:
                return self._arp_reply(frame)
        try:
            p = parse_dhcp_frame(frame)
        except (AssertionError, IndexError, ValueError):
            return None
        off = 14 + p.vlan_offset + 20 + 8
        try:
            msg = dm.DHCPMessage.decode(frame[off:])
        except Exception:
            return None
        resp = self.dhcp_server.handle(msg)
        return resp.encode() if resp else None

    def _arp_reply(self, frame: bytes):
        """Answer who-has for our server IP (the reference leans on
        the kernel stack for ARP; a userspace NIC edge must answer it
        itself or no subscriber traffic ever arrives)."""
        import struct as st
        from ..dataplane.packets import ip2u32, mac_bytes
        if len(frame) < 42:
            return None
        htype, ptype, hlen, plen, op = st.unpack_from(">HHBBH", frame, 14)
        if (htype, ptype, hlen, plen, op) != (1, 0x0800, 6, 4, 1):
            return None
        sha = frame[22:28]
        spa = frame[28:32]
        tpa = frame[38:42]
        if int.from_bytes(tpa, "big") != ip2u32(self.args.server_ip):
            return None
        our_mac = mac_bytes(self.args.server_mac)
        return (sha + our_mac + b"\x08\x06" +
                st.pack(">HHBBH", 1, 0x0800, 6, 4, 2) +
                our_mac + tpa + sha + spa)

    def _coa_lookup(self, req):
        for lease in self.dhcp_server.leases.values():
            from ..dataplane.packets import u32_to_ip
            if req.framed_ip and u32_to_ip(lease.ip) == req.framed_ip:
                return lease
            if req.session_id and getattr(lease, "_acct_id", "") == \
                    req.session_id:
                return lease
        return None

    def _coa_terminate(self, lease):
        from ..dhcp import message as dm
        msg = dm.DHCPMessage()
        msg.op = 1
        msg.chaddr = lease.mac
        msg.set_option(dm.OPT_MSG_TYPE, bytes([dm.RELEASE]))
        self.dhcp_server.handle_release(msg)
        return True

    def stop(self):
        for fn in reversed(self._cleanup):
            try:
                fn()
            except Exception:
                pass
        self._cleanup.clear()

    def reload(self, argv: Optional[List[str]] = None) -> dict:
        """SIGHUP hot reload (ref FEATURES.md Hot Reload): re-read the
        YAML config and apply the session-safe subset without
        restarting — log level, lease time, QoS policy definitions and
        the default policy.  The file is parsed and validated before
        anything is applied; a bad config changes nothing."""
        argv = argv if argv is not None else []
        parser = build_parser()
        base = parser.parse_args(["run"] + argv)
        base.config = self.args.config
        if not base.config:
            return {"reloaded": False, "reason": "no config file"}
        try:
            fresh = load_yaml_over_args(base, parser, argv)
            policies = []
            for spec in fresh.qos_policy:
                name, down, up = spec.split(":")
                policies.append((name, int(float(down) * 1e6),
                                 int(float(up) * 1e6)))
        except Exception as e:
            self.log.warning("reload rejected: %s", e)
            return {"reloaded": False, "reason": str(e)}
        from ..radius.policy import Policy
        applied = {"reloaded": True, "changed": []}
        if fresh.log_level != self.args.log_level:
            logging.getLogger().setLevel(getattr(
                logging, fresh.log_level.upper().replace(
                    "WARN", "WARNING")))
            self.args.log_level = fresh.log_level
            applied["changed"].append("log_level")
        if fresh.lease_time != self.args.lease_time:
            self.args.lease_time = fresh.lease_time
            self.dhcp_server.lease_time = fresh.lease_time
            applied["changed"].append("lease_time")
        for name, down, up in policies:
            cur = self.policy_manager.get(name)
            if cur is None or cur.download_rate_bps != down or \
                    cur.upload_rate_bps != up:
                self.policy_manager.add_policy(Policy(name, down, up))
                applied["changed"].append(f"qos_policy:{name}")
        if fresh.qos_default_policy:
            self.policy_manager.default_policy = \
                self.policy_manager.get(fresh.qos_default_policy)
        self.log.info("config reloaded: %s", applied["changed"])
        return applied

    def stats(self) -> dict:
        out = {"dhcp": dict(self.dhcp_server.stats),
               "leases": len(self.dhcp_server.leases),
               "fastpath": self.launcher.get_stats()}
        if getattr(self, "nat", None):
            out["nat"] = self.nat.get_stats()
        if getattr(self, "walledgarden", None):
            out["walledgarden"] = self.walledgarden.get_stats()
        if getattr(self, "resilience", None):
            out["resilience"] = {
                "state": self.resilience.state,
                "partition_duration":
                    self.resilience.partition_duration()}
        if getattr(self, "audit", None):
            out["audit"] = self.audit.stats()
        if getattr(self, "pump", None):
            out["pktio"] = dict(self.pump.stats)
            if getattr(self, "pump_downlink", None):
                out["pktio_downlink"] = dict(self.pump_downlink.stats)
        if getattr(self, "distributed_alloc", None):
            n, total, util = self.distributed_alloc.local.stats()
            out["pool_mode"] = {"mode": self.distributed_alloc.mode,
                                "allocated": n, "usable": total,
                                "utilization": util}
        return out


def cmd_demo(args) -> int:
    """ref cmd/bng/demo.go:46-60: full subscriber lifecycle with
    in-memory Nexus and no dataplane."""
    from ..nexus.client import Client
    from ..nexus.model import IPPool, ISPConfig, Subscriber
    from ..nexus.store import MemoryStore
    from ..pon.manager import Manager as PONMgr, QoSProfile
    from ..qinq.mapper import Mapper
    from ..walledgarden.manager import Manager as WGMgr

    store = MemoryStore()
    nexus = Client(store)
    nexus.pools.put("pool-1", IPPool("pool-1", "10.0.1.0/24").to_dict())
    nexus.isps.put("isp-1", ISPConfig("isp-1",
                                      ipv4_pools=["pool-1"]).to_dict())
    qm = Mapper()
    qm.add_range(100)
    pon = PONMgr(store, vlan_mapper=qm)
    pon.add_profile(QoSProfile("residential", 1000, 200))
    wg = WGMgr(portal_ip="10.0.0.10")

    print(f"bng demo — {args.subscribers} subscriber(s)")
    for i in range(args.subscribers):
        serial = f"ONT{i:04d}"
        mac = f"aa:bb:cc:00:{i >> 8:02x}:{i & 0xFF:02x}"
        nte = pon.ont_discovered(serial, f"pon0/{i}")
        print(f"  [{serial}] ONT discovered on {nte.pon_port}")
        wg.add(mac, "0.0.0.0", reason="new ONT")
        print(f"  [{serial}] quarantined in walled garden")
        nte = pon.provision(nte.id, profile="residential")
        print(f"  [{serial}] provisioned s_tag={nte.s_tag} "
              f"c_tag={nte.c_tag}")
        sub = Subscriber(f"sub-{serial}", nte_id=nte.id, isp_id="isp-1",
                         s_tag=nte.s_tag, c_tag=nte.c_tag, mac=mac)
        nexus.save_subscriber(sub)
        ip = nexus.allocate_ip_for_subscriber(sub.id)
        print(f"  [{serial}] RADIUS-time allocation -> {ip}")
        wg.activate(mac)
        print(f"  [{serial}] activated; DHCP is now a pure read: "
              f"{nexus.lookup_subscriber_ip(sub.id)}")
    print("demo complete")
    return 0


def cmd_verify() -> int:
    """Dataplane verification (ref cmd/verify-bpf: load every BPF
    object through the kernel verifier).  Here: hipcc-build the gfx950
    extension, check every struct's sizeof/offsetof against the ctypes
    mirrors, and run a golden-model self-test batch through all four
    program analogs."""
    import ctypes as C
    failures = 0
    try:
        from ..dataplane.build import build
        so = build()
        print(f"[verify] extension built: {so}")
    except Exception as e:
        print(f"[verify] BUILD FAILED: {e}")
        return 1
    try:
        from ..dataplane import abi
        from ..dataplane.build import get_ext
        ext = get_ext()
        rep = ext.layout_report() if ext is not None else None
        if rep is None:
            print("[verify] extension not importable (no GPU runtime?) "
                  "— layout check skipped")
        else:
            for name, (cls, size) in abi.EXPECTED_SIZES.items():
                got = rep.get(name)
                ok = got == C.sizeof(cls) == size
                print(f"[verify] {name}: device={got} host="
                      f"{C.sizeof(cls)} expected={size} "
                      f"{'OK' if ok else 'MISMATCH'}")
                failures += 0 if ok else 1
    except Exception as e:
        print(f"[verify] ABI check failed: {e}")
        failures += 1
    try:
        from ..dataplane.launcher import GoldenLauncher
        from ..dataplane.packets import (build_dhcp_request, build_ipv4,
                                         ip2u32, mac_bytes)
        g = GoldenLauncher()
        g.set_server_config(mac_bytes("02:00:00:00:00:01"),
                            ip2u32("10.0.0.1"))
        g.add_pool(1, ip2u32("10.0.1.0"), 24, ip2u32("10.0.1.1"))
        g.add_subscriber(mac_bytes("aa:00:00:00:00:01"), 1,
                         ip2u32("10.0.1.50"), 1 << 40)
        g.add_subscriber_nat(ip2u32("10.0.1.50"), ip2u32("203.0.113.1"),
                             1024, 2047)
        from ..dataplane import abi as _abi
        v, out = g.process_dhcp([build_dhcp_request(
            mac_bytes("aa:00:00:00:00:01"), 1)])[0]
        assert v == _abi.TX and len(out) > 240, "dhcp fast path self-test"
        v2, _ = g.process_nat44([build_ipv4(
            "aa:00:00:00:00:01", "02:00:00:00:00:01",
            ip2u32("10.0.1.50"), ip2u32("93.184.216.34"), proto=17,
            sport=40000, dport=53)])[0]
        assert v2 == _abi.FWD, "nat44 self-test"
        print("[verify] golden self-test: dhcp OFFER + nat44 SNAT OK")
    except Exception as e:
        print(f"[verify] golden self-test failed: {e}")
        failures += 1
    print(f"[verify] {'PASS' if failures == 0 else f'{failures} FAILURES'}")
    return 0 if failures == 0 else 1


def main(argv: Optional[List[str]] = None) -> int:
    argv = argv if argv is not None else sys.argv[1:]
    parser = build_parser()
    args = parser.parse_args(argv)
    if args.command == "version":
        print(f"bng {__version__} (MI355X gfx950 dataplane)")
        return 0
    if args.command == "verify":
        return cmd_verify()
    if args.command == "demo":
        return cmd_demo(args)
    if args.command == "run":
        logging.basicConfig(
            level=getattr(logging, args.log_level.upper().replace(
                "WARN", "WARNING")),
            format="%(asctime)s %(levelname)s %(name)s %(message)s")
        args = load_yaml_over_args(args, parser, argv)
        args = apply_env_overrides(args, argv)
        app = BNG(args).start()
        stop = {"flag": False}

        def on_sig(*_):
            stop["flag"] = True
        signal.signal(signal.SIGINT, on_sig)
        signal.signal(signal.SIGTERM, on_sig)
        # SIGHUP = hot reload of the YAML config (ref FEATURES.md)
        signal.signal(signal.SIGHUP, lambda *_: app.reload(argv[1:]))
        print(f"bng running (node {args.node_id}); ^C to stop",
              flush=True)
        try:
            while not stop["flag"]:
                time.sleep(0.2)
        finally:
            app.stop()
        return 0
    if args.command == "stats":
        import requests
        try:
            r = requests.get(f"{args.metrics_url}/metrics", timeout=3)
        except Exception as e:
            print(json.dumps({"error": str(e)}))
            return 1
        out = {}
        for line in r.text.splitlines():
            if line.startswith("bng_") and " " in line:
                k, v = line.rsplit(" ", 1)
                try:
                    out[k] = float(v)
                except ValueError:
                    pass
        print(json.dumps(out, indent=2, sort_keys=True))
        return 0
    parser.print_help()
    return 2


if __name__ == "__main__":
    sys.exit(main())
