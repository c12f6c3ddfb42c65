"""Local DHCP pool + pool manager (ref pkg/dhcp/pool.go).

Pool: free-list with reserved head/tail, decline blacklist
(MarkUnavailable), MAC-sticky allocation.  PoolManager: pool CRUD that
mirrors pools into the GPU dataplane tables on AddPool
(ref pool.go:250-294), plus ClassifyClient (:323).
"""
from __future__ import annotations

import ipaddress
import threading
import time
from dataclasses import dataclass
from typing import Dict, List, Optional

from ..dataplane.packets import ip2u32, u32_to_ip


class PoolExhaustedError(Exception):
    pass


@dataclass
class PoolConfig:
    pool_id: int
    network: str                 # CIDR
    gateway: str = ""
    dns: List[str] = None
    lease_time: int = 3600
    reserved_start: int = 2      # skip .0 and gateway .1 by default
    reserved_end: int = 1        # skip broadcast


class Pool:
    def __init__(self, cfg: PoolConfig):
        self.cfg = cfg
        self.net = ipaddress.IPv4Network(cfg.network, strict=False)
        self.gateway = cfg.gateway or str(self.net.network_address + 1)
        self.dns = cfg.dns or []
        base = int(self.net.network_address)
        n = self.net.num_addresses
        self.available: List[int] = [
            base + i for i in range(cfg.reserved_start, n - cfg.reserved_end)
            if base + i != ip2u32(self.gateway)]
        self.allocated: Dict[int, bytes] = {}     # ip -> mac
        self.by_mac: Dict[bytes, int] = {}
        self.unavailable: set = set()             # declined IPs
        self._lock = threading.RLock()

    def allocate(self, mac: bytes) -> int:
        """MAC-sticky first-free allocation (ref pool.go:146)."""
        with self._lock:
            if mac in self.by_mac:
                return self.by_mac[mac]
            while self.available:
                ip = self.available.pop(0)
                if ip in self.unavailable:
                    continue
                self.allocated[ip] = mac
                self.by_mac[mac] = ip
                return ip
            raise PoolExhaustedError(self.cfg.network)

    def release(self, ip: int):
        with self._lock:
            mac = self.allocated.pop(ip, None)
            if mac is not None:
                self.by_mac.pop(mac, None)
                self.available.append(ip)

    def contains(self, ip: int) -> bool:
        return ipaddress.IPv4Address(ip) in self.net

    def mark_unavailable(self, ip: int):
        """DHCP DECLINE blacklist (ref pool.go:191)."""
        with self._lock:
            self.unavailable.add(ip)
            mac = self.allocated.pop(ip, None)
            if mac is not None:
                self.by_mac.pop(mac, None)

    def stats(self) -> dict:
        with self._lock:
            return {
                "pool_id": self.cfg.pool_id,
                "network": self.cfg.network,
                "allocated": len(self.allocated),
                "available": len([i for i in self.available
                                  if i not in self.unavailable]),
                "declined": len(self.unavailable),
            }


class PoolManager:
    """ref pool.go:241-369; `launcher` is the GPU dataplane (pkg/ebpf
    loader analog) — may be None for CPU-only operation."""

    def __init__(self, launcher=None):
        self.launcher = launcher
        self.pools: Dict[int, Pool] = {}
        self.default_pool_id: Optional[int] = None
        self._lock = threading.RLock()

    def add_pool(self, cfg: PoolConfig) -> Pool:
        pool = Pool(cfg)
        with self._lock:
            self.pools[cfg.pool_id] = pool
            if self.default_pool_id is None:
                self.default_pool_id = cfg.pool_id
        if self.launcher is not None:
            # push pool metadata into the device ip_pools table
            # (ref pool.go:250-294 AddPool -> eBPF sync)
            self.launcher.add_pool(
                cfg.pool_id, int(pool.net.network_address),
                pool.net.prefixlen, ip2u32(pool.gateway),
                ip2u32(pool.dns[0]) if pool.dns else 0,
                ip2u32(pool.dns[1]) if len(pool.dns) > 1 else 0,
                cfg.lease_time)
        return pool

    def remove_pool(self, pool_id: int):
        with self._lock:
            self.pools.pop(pool_id, None)
        if self.launcher is not None:
            self.launcher.remove_pool(pool_id)

    def get_pool(self, pool_id: int) -> Optional[Pool]:
        with self._lock:
            return self.pools.get(pool_id)

    def set_default_pool(self, pool_id: int):
        with self._lock:
            if pool_id not in self.pools:
                raise KeyError(pool_id)
            self.default_pool_id = pool_id

    def classify_client(self, mac: bytes,
                        vendor_class: str = "") -> Optional[Pool]:
        """ref pool.go:323 ClassifyClient: default pool, else first."""
        with self._lock:
            if self.default_pool_id in self.pools:
                return self.pools[self.default_pool_id]
            for p in self.pools.values():
                return p
        return None

    def find_pool_for_ip(self, ip: int) -> Optional[Pool]:
        with self._lock:
            for p in self.pools.values():
                if p.contains(ip):
                    return p
        return None

    def all_stats(self) -> List[dict]:
        with self._lock:
            return [p.stats() for p in self.pools.values()]
