"""DHCP slow-path server — the control-plane hub
(ref pkg/dhcp/server.go: handleDiscover :398, handleRequest :556,
handleRelease :864, updateFastPathCache :1057).

Handles the DORA cycle for packets the GPU fast path PASSed up:
new/unknown subscribers, expired leases, RELEASE/DECLINE/INFORM.  On a
successful REQUEST it performs the full provisioning chain the reference
does at :595-834: RADIUS auth -> IP allocation (nexus lookup -> nexus
allocator -> peer pool -> local pool priority) -> lease + circuit-ID
index -> GPU fast-path cache insert -> QoS policy -> NAT port block ->
accounting start.

Collaborators are injected via setters (ref server.go:140-177) so every
piece is optional — `bng demo` runs with none of them.
"""
from __future__ import annotations

import socket
import struct
import threading
import time
from dataclasses import dataclass, field
from typing import Dict, Optional

from ..dataplane.packets import ip2u32, u32_to_ip
from . import message as dm
from .pool import Pool, PoolExhaustedError, PoolManager


@dataclass
class Lease:
    mac: bytes
    ip: int
    pool_id: int
    expiry: float
    circuit_id: bytes = b""
    hostname: str = ""
    subscriber_id: str = ""
    policy_name: str = ""
    walled_garden: bool = False
    created: float = field(default_factory=time.time)


class DHCPServer:
    def __init__(self, pool_manager: PoolManager, server_ip: str,
                 server_mac: bytes = b"\x02\x00\x00\x00\x00\x01",
                 lease_time: int = 3600, authoritative: bool = True):
        self.pools = pool_manager
        self.server_ip = ip2u32(server_ip)
        self.server_mac = server_mac
        self.lease_time = lease_time
        # optional override (resilience short-lease mode under pool
        # pressure, ref resilience/pool_monitor + types.go:69-100)
        self.lease_time_provider = None
        self.authoritative = authoritative
        self.leases: Dict[bytes, Lease] = {}          # by MAC
        self.leases_by_circuit: Dict[bytes, Lease] = {}
        self._lock = threading.RLock()
        # collaborators (ref server.go:140-177 setters)
        self.launcher = None          # GPU dataplane (pkg/ebpf analog)
        self.radius = None            # radius.Client
        self.qos_mgr = None           # qos.Manager
        self.nat_mgr = None           # nat.Manager
        self.nexus_allocator = None   # nexus.HTTPAllocator
        self.nexus_client = None      # nexus.Client
        self.peer_pool = None         # pool.PeerPool
        self.distributed = None       # allocator.DistributedAllocator
        self.walled_garden = None     # walledgarden.Manager
        self.accounting = None        # radius.AccountingManager
        self.policy_mgr = None        # radius.PolicyManager
        self.intercept = None         # intercept.Manager
        self.audit = None             # audit.Logger
        self.metrics = None
        self.auth_mode = "none"       # none|mac (RADIUS auth w/ MAC creds)
        self.on_lease_event = []      # [(event, Lease)] listeners (HA sync)
        self.stats = {k: 0 for k in (
            "discover", "request", "release", "decline", "inform",
            "offer", "ack", "nak", "auth_reject", "walled_garden",
            "expired_swept")}
        self._stop = threading.Event()
        self._sweeper: Optional[threading.Thread] = None
        self._sock = None

    # ----------------------------------------------------------- wiring
    def set_launcher(self, l):
        self.launcher = l

    def _lease_seconds(self) -> float:
        if self.lease_time_provider is not None:
            return min(self.lease_time, self.lease_time_provider())
        return self.lease_time

    def set_radius(self, c, auth_mode: str = "mac"):
        self.radius = c
        self.auth_mode = auth_mode

    def set_qos_manager(self, m):
        self.qos_mgr = m

    def set_nat_manager(self, m):
        self.nat_mgr = m

    def set_nexus(self, allocator=None, client=None):
        self.nexus_allocator = allocator
        self.nexus_client = client

    def set_peer_pool(self, p):
        self.peer_pool = p

    def set_distributed(self, alloc):
        """Store-replicated allocator (ref modes.go framework): slots
        into the chain after the peer pool; pool-mode session/lease
        semantics live in the allocator itself."""
        self.distributed = alloc

    def set_walled_garden(self, w):
        self.walled_garden = w

    def set_accounting(self, a):
        self.accounting = a

    def set_policy_manager(self, pm):
        self.policy_mgr = pm

    # --------------------------------------------------------- lifecycle
    def start(self, bind: str = "0.0.0.0", port: int = 67,
              serve: bool = False):
        self._sweeper = threading.Thread(target=self._sweep_loop,
                                         daemon=True)
        self._sweeper.start()
        if serve:
            self._sock = socket.socket(socket.AF_INET, socket.SOCK_DGRAM)
            self._sock.setsockopt(socket.SOL_SOCKET, socket.SO_BROADCAST, 1)
            self._sock.bind((bind, port))
            threading.Thread(target=self._serve_loop, daemon=True).start()
        # publish server config to the fast path (ref server.go:259-279)
        if self.launcher is not None:
            self.launcher.set_server_config(self.server_mac, self.server_ip)
        return self

    def stop(self):
        self._stop.set()
        if self._sock:
            self._sock.close()

    def _serve_loop(self):
        self._sock.settimeout(0.2)
        while not self._stop.is_set():
            try:
                data, addr = self._sock.recvfrom(2048)
            except socket.timeout:
                continue
            except OSError:
                break
            try:
                req = dm.DHCPMessage.decode(data)
                resp = self.handle(req)
            except Exception:
                continue
            if resp is not None:
                dest = self.reply_dest(req, resp)
                try:
                    self._sock.sendto(resp.encode(), dest)
                except OSError:
                    break

    @staticmethod
    def reply_dest(req: dm.DHCPMessage,
                   resp: dm.DHCPMessage) -> tuple:
        """RFC 2131 §4.1 reply addressing (round-1 advisor: renewing
        clients were getting broadcast ACKs): relay (giaddr) wins;
        then unicast to ciaddr for a renewing client; then honor the
        BROADCAST flag; else unicast to the offered yiaddr (delivered
        at L2 by chaddr — the GPU fast path makes the same decision in
        setup_reply_l2_headers, ref dhcp_fastpath.c:436-482)."""
        if req.giaddr != 0:
            return (u32_to_ip(req.giaddr), 67)
        if req.ciaddr != 0:
            return (u32_to_ip(req.ciaddr), 68)
        if req.flags & 0x8000 or resp.yiaddr == 0:
            return ("255.255.255.255", 68)
        return (u32_to_ip(resp.yiaddr), 68)

    # ----------------------------------------------------------- handler
    def handle(self, req: dm.DHCPMessage) -> Optional[dm.DHCPMessage]:
        if req.op != 1:
            return None
        t = req.msg_type
        if t == dm.DISCOVER:
            return self.handle_discover(req)
        if t == dm.REQUEST:
            return self.handle_request(req)
        if t == dm.RELEASE:
            return self.handle_release(req)
        if t == dm.DECLINE:
            return self.handle_decline(req)
        if t == dm.INFORM:
            return self.handle_inform(req)
        return None

    # ---------------------------------------------------------- discover
    def handle_discover(self, req: dm.DHCPMessage) -> Optional[dm.DHCPMessage]:
        """ref server.go:398 handleDiscover."""
        self.stats["discover"] += 1
        mac = req.client_mac
        lease = self._find_lease(mac, req.circuit_id())
        if lease is None or lease.expiry <= time.time():
            lease = self._provision(req)
            if lease is None:
                return None     # auth failed / exhausted: stay silent
        self.stats["offer"] += 1
        return self._reply(req, dm.OFFER, lease)

    # ----------------------------------------------------------- request
    def handle_request(self, req: dm.DHCPMessage) -> Optional[dm.DHCPMessage]:
        """ref server.go:556 handleRequest — the full provisioning chain."""
        self.stats["request"] += 1
        mac = req.client_mac
        lease = self._find_lease(mac, req.circuit_id())
        if lease is None or lease.expiry <= time.time():
            lease = self._provision(req)
            if lease is None:
                if self.authoritative:
                    self.stats["nak"] += 1
                    return self._nak(req)
                return None
        wanted = req.requested_ip or req.ciaddr
        if wanted and wanted != lease.ip:
            # client asks for a different IP than its lease: NAK
            self.stats["nak"] += 1
            return self._nak(req)
        lease.expiry = time.time() + self._lease_seconds()
        self._post_ack(lease, req)
        self.stats["ack"] += 1
        self._emit_lease("add", lease)
        return self._reply(req, dm.ACK, lease)

    # ----------------------------------------------------------- release
    def handle_release(self, req: dm.DHCPMessage) -> None:
        """ref server.go:864 handleRelease."""
        self.stats["release"] += 1
        mac = req.client_mac
        with self._lock:
            lease = self.leases.pop(mac, None)
            if lease and lease.circuit_id:
                self.leases_by_circuit.pop(lease.circuit_id, None)
        if lease is None:
            return None
        self._teardown(lease)
        return None

    def handle_decline(self, req: dm.DHCPMessage) -> None:
        """DECLINE blacklists the IP (ref pool decline handling)."""
        self.stats["decline"] += 1
        ip = req.requested_ip
        if not ip:
            return None
        pool = self.pools.find_pool_for_ip(ip)
        if pool is not None:
            pool.mark_unavailable(ip)
        with self._lock:
            lease = self.leases.pop(req.client_mac, None)
        if lease is not None:
            self._remove_fastpath(lease)
        return None

    def handle_inform(self, req: dm.DHCPMessage) -> Optional[dm.DHCPMessage]:
        self.stats["inform"] += 1
        pool = self.pools.classify_client(req.client_mac, req.vendor_class)
        if pool is None:
            return None
        resp = self._base_reply(req, dm.ACK)
        resp.yiaddr = 0
        self._add_net_options(resp, pool)
        return resp

    # -------------------------------------------------------- provision
    def _provision(self, req: dm.DHCPMessage) -> Optional[Lease]:
        """Auth + allocate: nexus lookup -> nexus client -> peer pool ->
        local pool (ref server.go:595-705)."""
        mac = req.client_mac
        mac_str = ":".join(f"{b:02x}" for b in mac)
        sub_id = mac_str
        policy_name = ""
        framed_ip = 0

        # RADIUS authentication for new sessions (ref :595-627)
        if self.radius is not None and self.auth_mode != "none":
            try:
                res = self.radius.authenticate(mac_str, mac_str, mac=mac_str)
            except Exception:
                res = None
            if res is not None and not res.success:
                self.stats["auth_reject"] += 1
                if self.audit:
                    self.audit.log("auth_reject", subscriber=mac_str)
                return None
            if res is not None:
                policy_name = res.policy_name
                if res.framed_ip:
                    framed_ip = ip2u32(res.framed_ip)

        walled = False
        ip = framed_ip
        pool = None
        # 1. central Nexus lookup (pure read; ref :431 LookupIPv4)
        if not ip and self.nexus_allocator is not None:
            from ..nexus.http_allocator import NoAllocationError
            try:
                got, _pool = self.nexus_allocator.lookup_ipv4(sub_id)
                ip = ip2u32(got)
            except NoAllocationError:
                # unknown to Nexus: walled-garden quarantine
                walled = True
            except Exception:
                pass
        # 2. nexus client hashring
        if not ip and not walled and self.nexus_client is not None:
            try:
                sub = self.nexus_client.get_subscriber_by_mac(mac_str)
                if sub is not None:
                    ip = ip2u32(
                        self.nexus_client.allocate_ip_for_subscriber(sub.id))
                    sub_id = sub.id
            except Exception:
                pass
        # 3. peer pool (HRW-distributed)
        if not ip and self.peer_pool is not None:
            try:
                ip = ip2u32(self.peer_pool.allocate(sub_id))
            except Exception:
                pass
        # 3b. store-replicated distributed allocator (pool-mode
        # session|lease)
        if not ip and self.distributed is not None:
            try:
                ip = ip2u32(
                    self.distributed.allocate(sub_id).split("/")[0])
            except Exception:
                pass
        # 4. local pool
        if not ip:
            pool = self.pools.classify_client(mac, req.vendor_class)
            if pool is None:
                return None
            try:
                ip = pool.allocate(mac)
            except PoolExhaustedError:
                return None
        if pool is None:
            pool = self.pools.find_pool_for_ip(ip) or \
                self.pools.classify_client(mac, req.vendor_class)
        if pool is None:
            return None

        if walled and self.walled_garden is not None:
            self.walled_garden.add(mac_str, u32_to_ip(ip))
            self.stats["walled_garden"] += 1

        lease = Lease(mac=mac, ip=ip, pool_id=pool.cfg.pool_id,
                      expiry=time.time() + self._lease_seconds(),
                      circuit_id=req.circuit_id(), subscriber_id=sub_id,
                      policy_name=policy_name, walled_garden=walled)
        with self._lock:
            self.leases[mac] = lease
            if lease.circuit_id:
                self.leases_by_circuit[lease.circuit_id] = lease
        return lease

    def _post_ack(self, lease: Lease, req: dm.DHCPMessage):
        """Provisioning side effects on ACK (ref server.go:708-834)."""
        self._update_fastpath(lease)
        # QoS policy (ref :774-794)
        if self.qos_mgr is not None and not lease.walled_garden:
            name = lease.policy_name
            self.qos_mgr.apply_policy(lease.ip, name)
        # NAT port block (ref :797-814)
        if self.nat_mgr is not None and not lease.walled_garden:
            self.nat_mgr.allocate_nat(lease.ip, lease.subscriber_id)
        # accounting start (ref :817-834, async in the reference)
        if self.accounting is not None:
            if not getattr(lease, "_acct_id", None):
                lease._acct_id = self.accounting.start_session(
                    lease.subscriber_id, mac=lease.subscriber_id,
                    framed_ip=u32_to_ip(lease.ip))
        if self.intercept is not None:
            self.intercept.on_session_start(lease.subscriber_id,
                                            u32_to_ip(lease.ip))
        if self.audit is not None:
            self.audit.log("session_start", subscriber=lease.subscriber_id,
                           ip=u32_to_ip(lease.ip))

    def _emit_lease(self, event: str, lease: Lease):
        for cb in self.on_lease_event:
            try:
                cb(event, lease)
            except Exception:
                pass

    def restore_lease(self, lease: Lease):
        """HA promotion: install a replicated lease (shadow -> live),
        including the GPU fast-path entry."""
        with self._lock:
            self.leases[lease.mac] = lease
            if lease.circuit_id:
                self.leases_by_circuit[lease.circuit_id] = lease
        self._update_fastpath(lease)

    def _teardown(self, lease: Lease):
        self._emit_lease("delete", lease)
        self._remove_fastpath(lease)
        pool = self.pools.get_pool(lease.pool_id)
        if pool is not None:
            pool.release(lease.ip)
        if self.peer_pool is not None:
            try:
                self.peer_pool.release(lease.subscriber_id)
            except Exception:
                pass
        if self.distributed is not None:
            try:
                self.distributed.release(lease.subscriber_id)
            except Exception:
                pass
        if self.qos_mgr is not None:
            self.qos_mgr.remove_policy(lease.ip)
        if self.nat_mgr is not None:
            self.nat_mgr.release_nat(lease.ip)
        if self.accounting is not None and getattr(lease, "_acct_id", None):
            self.accounting.stop_session(lease._acct_id)
        if self.audit is not None:
            self.audit.log("session_stop", subscriber=lease.subscriber_id,
                           ip=u32_to_ip(lease.ip))

    # ------------------------------------------------- fast-path mirror
    def _update_fastpath(self, lease: Lease):
        """ref server.go:1057 updateFastPathCache + circuit maps
        (:716-770): next DISCOVER/REQUEST is answered on the GPU."""
        if self.launcher is None:
            return
        self.launcher.add_subscriber(lease.mac, lease.pool_id, lease.ip,
                                     int(lease.expiry))
        if lease.circuit_id:
            self.launcher.add_circuit_subscriber(
                lease.circuit_id, lease.pool_id, lease.ip,
                int(lease.expiry))

    def _remove_fastpath(self, lease: Lease):
        if self.launcher is None:
            return
        self.launcher.remove_subscriber(lease.mac)
        if lease.circuit_id:
            self.launcher.remove_circuit_subscriber(lease.circuit_id)

    # ----------------------------------------------------------- replies
    def _find_lease(self, mac: bytes, circuit_id: bytes) -> Optional[Lease]:
        with self._lock:
            lease = self.leases.get(mac)
            if lease is None and circuit_id:
                lease = self.leases_by_circuit.get(circuit_id)
            return lease

    def _base_reply(self, req: dm.DHCPMessage,
                    msg_type: int) -> dm.DHCPMessage:
        resp = dm.DHCPMessage()
        resp.op = 2
        resp.xid = req.xid
        resp.flags = req.flags
        resp.giaddr = req.giaddr
        resp.chaddr = req.chaddr
        resp.siaddr = self.server_ip
        resp.set_option(dm.OPT_MSG_TYPE, bytes([msg_type]))
        resp.set_option(dm.OPT_SERVER_ID, struct.pack(">I", self.server_ip))
        return resp

    def _add_net_options(self, resp: dm.DHCPMessage, pool: Pool):
        mask = int(pool.net.netmask)
        resp.set_option(dm.OPT_SUBNET_MASK, struct.pack(">I", mask))
        resp.set_option(dm.OPT_ROUTER, struct.pack(">I",
                                                   ip2u32(pool.gateway)))
        if pool.dns:
            resp.set_option(dm.OPT_DNS, b"".join(
                struct.pack(">I", ip2u32(d)) for d in pool.dns[:2]))

    def _reply(self, req: dm.DHCPMessage, msg_type: int,
               lease: Lease) -> dm.DHCPMessage:
        resp = self._base_reply(req, msg_type)
        resp.yiaddr = lease.ip
        pool = self.pools.get_pool(lease.pool_id)
        lt = pool.cfg.lease_time if pool else self.lease_time
        if self.lease_time_provider is not None:
            lt = min(lt, self.lease_time_provider())
        resp.set_option(dm.OPT_LEASE_TIME, struct.pack(">I", lt))
        resp.set_option(dm.OPT_RENEWAL_TIME, struct.pack(">I", lt // 2))
        resp.set_option(dm.OPT_REBIND_TIME, struct.pack(">I", lt * 7 // 8))
        if pool is not None:
            self._add_net_options(resp, pool)
        return resp

    def _nak(self, req: dm.DHCPMessage) -> dm.DHCPMessage:
        return self._base_reply(req, dm.NAK)

    # ------------------------------------------------------------ sweeper
    def _sweep_loop(self):
        """Per-minute lease expiry sweeper (ref server.go lease sweeper)."""
        while not self._stop.wait(1.0):
            self.sweep_expired()

    def sweep_expired(self, now: Optional[float] = None) -> int:
        now = now or time.time()
        dead = []
        with self._lock:
            for mac, lease in list(self.leases.items()):
                if lease.expiry <= now:
                    dead.append(lease)
                    del self.leases[mac]
                    if lease.circuit_id:
                        self.leases_by_circuit.pop(lease.circuit_id, None)
        for lease in dead:
            self._teardown(lease)
            self.stats["expired_swept"] += 1
        return len(dead)
