"""NAT44/CGNAT manager — userspace side of the GPU NAT kernel
(ref pkg/nat/manager.go): public-IP pool, deterministic per-subscriber
RFC 6431 port-block allocation (block index = subscriber counter,
port_start = range_start + idx * ports_per_sub, manager.go:398-496),
device table population, ALG configuration, hairpin IPs, and the
compliance-log drain feeding the logging pipeline (pkg/nat/logging.go
analog in bng_amd/nat/logging.py)."""
from __future__ import annotations

import threading
import time
from dataclasses import dataclass
from typing import Dict, List, Optional, Tuple

from ..dataplane import abi
from ..dataplane.packets import ip2u32, u32_to_ip


class NATExhaustedError(Exception):
    pass


@dataclass
class NATAllocation:
    private_ip: int
    public_ip: int
    port_start: int
    port_end: int
    subscriber_id: int
    allocated_at: float


class Manager:
    PORT_RANGE_START = 1024
    PORT_RANGE_END = 65535

    def __init__(self, launcher=None, ports_per_subscriber: int = 1024,
                 flags: int = abi.NAT_FLAG_EIM, logger=None,
                 alg_ports=None):
        self.launcher = launcher
        self.ports_per_sub = ports_per_subscriber
        self.flags = flags
        self.logger = logger          # nat.logging.ComplianceLogger
        self.public_ips: List[int] = []
        self.allocations: Dict[int, NATAllocation] = {}  # by private ip
        self._sub_counter = 0
        self._lock = threading.RLock()
        self._alg_ports: List[Tuple[int, int]] = list(alg_ports or [])
        self._hairpin = False
        self._stop = threading.Event()
        self._drain_thread: Optional[threading.Thread] = None

    # ------------------------------------------------------------ config
    def add_public_ip(self, ip: str):
        """ref manager.go AddPublicIP."""
        with self._lock:
            v = ip2u32(ip)
            if v not in self.public_ips:
                self.public_ips.append(v)

    def configure_alg(self, ftp: bool = True, sip: bool = False):
        """ref manager.go:542-561: FTP 21/tcp, SIP 5060/udp+tcp punts."""
        self._alg_ports = []
        if ftp:
            self.flags |= abi.NAT_FLAG_ALG_FTP
            self._alg_ports.append((21, 6))
        if sip:
            self.flags |= abi.NAT_FLAG_ALG_SIP
            self._alg_ports.append((5060, 17))
            self._alg_ports.append((5060, 6))
        self._push_config()

    def enable_hairpin(self, enabled: bool = True):
        self._hairpin = enabled
        if enabled:
            self.flags |= abi.NAT_FLAG_HAIRPIN
        else:
            self.flags &= ~abi.NAT_FLAG_HAIRPIN
        self._push_config()

    def _push_config(self):
        if self.launcher is None:
            return
        self.launcher.set_nat_config(
            flags=self.flags,
            port_range=(self.PORT_RANGE_START, self.PORT_RANGE_END),
            ports_per_sub=self.ports_per_sub,
            alg_ports=self._alg_ports)
        if self._hairpin:
            self.launcher.set_hairpin_ips(self.public_ips)

    def start(self, drain_interval: float = 1.0):
        """Attach (the TC-attach analog is table init; ref :563-654) and
        start the compliance-log drain loop."""
        self._push_config()
        if self.launcher is not None and self.logger is not None:
            self._drain_thread = threading.Thread(
                target=self._drain_loop, args=(drain_interval,), daemon=True)
            self._drain_thread.start()
        return self

    def stop(self):
        self._stop.set()
        self.drain_logs()

    # -------------------------------------------------------- allocation
    def allocate_nat(self, private_ip: int, subscriber_id: str = "") -> NATAllocation:
        """Deterministic port-block allocation (ref AllocateNAT
        manager.go:398-496): block index from a monotonic subscriber
        counter; blocks tile the port range across the public IPs."""
        with self._lock:
            existing = self.allocations.get(private_ip)
            if existing is not None:
                return existing
            if not self.public_ips:
                raise NATExhaustedError("no public IPs configured")
            blocks_per_ip = (self.PORT_RANGE_END + 1 -
                             self.PORT_RANGE_START) // self.ports_per_sub
            idx = self._sub_counter
            self._sub_counter += 1
            ip_idx = (idx // blocks_per_ip) % len(self.public_ips)
            blk = idx % blocks_per_ip
            if idx >= blocks_per_ip * len(self.public_ips):
                # all blocks taken: reuse round-robin (oversubscription)
                ip_idx = idx % len(self.public_ips)
                blk = (idx // len(self.public_ips)) % blocks_per_ip
            port_start = self.PORT_RANGE_START + blk * self.ports_per_sub
            alloc = NATAllocation(
                private_ip=private_ip, public_ip=self.public_ips[ip_idx],
                port_start=port_start,
                port_end=port_start + self.ports_per_sub - 1,
                subscriber_id=idx + 1, allocated_at=time.time())
            self.allocations[private_ip] = alloc
        if self.launcher is not None:
            self.launcher.add_subscriber_nat(
                alloc.private_ip, alloc.public_ip, alloc.port_start,
                alloc.port_end, alloc.subscriber_id)
        if self.logger is not None:
            self.logger.log_event({
                "timestamp": time.time_ns(),
                "event_type": abi.LOG_PB_ASSIGN,
                "subscriber_id": alloc.subscriber_id,
                "private_ip": alloc.private_ip,
                "public_ip": alloc.public_ip,
                "private_port": alloc.port_start,
                "public_port": alloc.port_end,
                "dest_ip": 0, "dest_port": 0, "protocol": 0, "flags": 0})
        return alloc

    def restore_nat(self, private_ip: int, public_ip: int,
                    port_start: int, port_end: int,
                    subscriber_id: int = 0) -> "NATAllocation":
        """Exact-block restore at HA promotion: the standby must keep
        the ACTIVE's block assignment, not allocate a fresh one —
        otherwise restored sessions' external ports land inside other
        subscribers' new blocks (round-1 VERDICT task 3)."""
        with self._lock:
            alloc = NATAllocation(
                private_ip=private_ip, public_ip=public_ip,
                port_start=port_start, port_end=port_end,
                subscriber_id=subscriber_id, allocated_at=time.time())
            self.allocations[private_ip] = alloc
            # keep the counter ahead of restored block indices so new
            # allocations never collide with restored ones
            blocks_per_ip = (self.PORT_RANGE_END + 1 -
                             self.PORT_RANGE_START) // self.ports_per_sub
            blk = (port_start - self.PORT_RANGE_START) // self.ports_per_sub
            try:
                ip_idx = self.public_ips.index(public_ip)
            except ValueError:
                ip_idx = 0
            idx = ip_idx * blocks_per_ip + blk
            if idx >= self._sub_counter:
                self._sub_counter = idx + 1
        if self.launcher is not None:
            self.launcher.add_subscriber_nat(
                alloc.private_ip, alloc.public_ip, alloc.port_start,
                alloc.port_end, alloc.subscriber_id)
        return alloc

    def release_nat(self, private_ip: int):
        with self._lock:
            alloc = self.allocations.pop(private_ip, None)
        if alloc is not None and self.logger is not None:
            self.logger.log_event({
                "timestamp": time.time_ns(),
                "event_type": abi.LOG_PB_RELEASE,
                "subscriber_id": alloc.subscriber_id,
                "private_ip": alloc.private_ip,
                "public_ip": alloc.public_ip,
                "private_port": alloc.port_start,
                "public_port": alloc.port_end,
                "dest_ip": 0, "dest_port": 0, "protocol": 0, "flags": 0})

    def get_allocation(self, private_ip: int) -> Optional[NATAllocation]:
        with self._lock:
            return self.allocations.get(private_ip)

    # --------------------------------------------------------- log drain
    def _drain_loop(self, interval: float):
        while not self._stop.wait(interval):
            self.drain_logs()

    def drain_logs(self) -> int:
        """Pull kernel-side compliance events (the BPF ring-buffer drain,
        ref nat/logging.go:293) into the logger."""
        if self.launcher is None or self.logger is None:
            return 0
        events = self.launcher.drain_nat_log()
        for e in events:
            self.logger.log_event(e)
        return len(events)

    def get_stats(self) -> Dict[str, int]:
        if self.launcher is not None:
            return self.launcher.nat_get_stats()
        return {}

    def sweep_sessions(self):
        if self.launcher is not None:
            self.launcher.sweep_nat()
