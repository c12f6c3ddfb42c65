"""Nexus data model (ref pkg/nexus/store.go:211-291): Subscriber, NTE,
ISPConfig, IPPool, Device — plain dataclasses with JSON round-trip."""
from __future__ import annotations

import time
from dataclasses import asdict, dataclass, field
from typing import List


def _now() -> float:
    return time.time()


@dataclass
class Subscriber:
    id: str
    # physical layer (NetCo) — stable
    nte_id: str = ""
    device_id: str = ""
    s_tag: int = 0
    c_tag: int = 0
    netco_id: str = ""
    # service layer (ISPCo)
    isp_id: str = ""
    radius_realm: str = ""
    # IP allocation (done at RADIUS-auth time — the core design invariant)
    ipv4_pool: str = ""
    ipv4_addr: str = ""
    ipv6_pool: str = ""
    ipv6_addr: str = ""
    # state: walledgarden | active | blocked (ref walledgarden/manager.go)
    state: str = "walledgarden"
    mac: str = ""
    circuit_id: str = ""
    updated_at: float = field(default_factory=_now)

    def to_dict(self):
        return asdict(self)

    @classmethod
    def from_dict(cls, d):
        return cls(**{k: v for k, v in d.items()
                      if k in cls.__dataclass_fields__})


@dataclass
class NTE:
    id: str
    device_id: str = ""
    serial_number: str = ""
    pon_port: str = ""
    s_tag: int = 0
    c_tag: int = 0
    state: str = "discovered"
    first_seen: float = field(default_factory=_now)
    last_seen: float = field(default_factory=_now)
    provisioned: bool = False

    def to_dict(self):
        return asdict(self)

    @classmethod
    def from_dict(cls, d):
        return cls(**{k: v for k, v in d.items()
                      if k in cls.__dataclass_fields__})


@dataclass
class ISPConfig:
    id: str
    radius_servers: List[str] = field(default_factory=list)
    radius_secret: str = ""
    radius_realm: str = ""
    ipv4_pools: List[str] = field(default_factory=list)
    ipv6_pools: List[str] = field(default_factory=list)
    default_gateway: str = ""
    dns_servers: List[str] = field(default_factory=list)

    def to_dict(self):
        return asdict(self)

    @classmethod
    def from_dict(cls, d):
        return cls(**{k: v for k, v in d.items()
                      if k in cls.__dataclass_fields__})


@dataclass
class IPPool:
    id: str
    cidr: str
    isp_id: str = ""
    type: str = "residential"

    def to_dict(self):
        return asdict(self)

    @classmethod
    def from_dict(cls, d):
        return cls(**{k: v for k, v in d.items()
                      if k in cls.__dataclass_fields__})


@dataclass
class Device:
    id: str
    serial_number: str = ""
    model: str = ""
    firmware: str = ""
    mac: str = ""
    state: str = "registered"
    last_seen: float = field(default_factory=_now)
    capabilities: List[str] = field(default_factory=list)

    def to_dict(self):
        return asdict(self)

    @classmethod
    def from_dict(cls, d):
        return cls(**{k: v for k, v in d.items()
                      if k in cls.__dataclass_fields__})
