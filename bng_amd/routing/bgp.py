"""BGP controller over FRR (ref pkg/routing/bgp.go:219-553): neighbor
CRUD, prefix announce/withdraw, ECMP maximum-paths, per-neighbor
route-maps; and BFD peer management (ref bfd.go:153-629)."""
from __future__ import annotations

import threading
from dataclasses import dataclass, field
from typing import Dict, List, Optional

from .frr import FRRExecutor


@dataclass
class Neighbor:
    address: str
    remote_as: int
    description: str = ""
    password: str = ""
    bfd: bool = False
    route_map_in: str = ""
    route_map_out: str = ""
    established: bool = False


class BGPController:
    def __init__(self, executor: FRRExecutor, local_as: int,
                 router_id: str = "", ecmp_paths: int = 1):
        self.exe = executor
        self.local_as = local_as
        self.router_id = router_id
        self.ecmp_paths = ecmp_paths
        self.neighbors: Dict[str, Neighbor] = {}
        self.announced: Dict[str, str] = {}       # prefix -> next-hop/""
        self._lock = threading.RLock()
        self.started = False

    def _cfg(self, inner: List[str]) -> None:
        self.exe.run(["configure terminal", f"router bgp {self.local_as}"]
                     + inner + ["end"])

    def start(self):
        """Base BGP config incl. ECMP (ref bgp.go Start + maximum-paths)."""
        inner = []
        if self.router_id:
            inner.append(f"bgp router-id {self.router_id}")
        if self.ecmp_paths > 1:
            inner.append(f"maximum-paths {self.ecmp_paths}")
        self._cfg(inner)
        self.started = True
        return self

    # ---------------------------------------------------------- neighbors
    def add_neighbor(self, address: str, remote_as: int, *,
                     description: str = "", password: str = "",
                     bfd: bool = False, route_map_in: str = "",
                     route_map_out: str = ""):
        n = Neighbor(address, remote_as, description, password, bfd,
                     route_map_in, route_map_out)
        inner = [f"neighbor {address} remote-as {remote_as}"]
        if description:
            inner.append(f"neighbor {address} description {description}")
        if password:
            inner.append(f"neighbor {address} password {password}")
        if bfd:
            inner.append(f"neighbor {address} bfd")
        if route_map_in:
            inner.append(
                f"neighbor {address} route-map {route_map_in} in")
        if route_map_out:
            inner.append(
                f"neighbor {address} route-map {route_map_out} out")
        self._cfg(inner)
        with self._lock:
            self.neighbors[address] = n

    def remove_neighbor(self, address: str):
        self._cfg([f"no neighbor {address}"])
        with self._lock:
            self.neighbors.pop(address, None)

    # ----------------------------------------------------------- prefixes
    def announce_prefix(self, prefix: str, next_hop: str = ""):
        """ref bgp.go:323 AnnouncePrefix."""
        self._cfg([f"network {prefix}"])
        with self._lock:
            self.announced[prefix] = next_hop

    def withdraw_prefix(self, prefix: str):
        self._cfg([f"no network {prefix}"])
        with self._lock:
            self.announced.pop(prefix, None)

    def announced_prefixes(self) -> List[str]:
        with self._lock:
            return sorted(self.announced)

    def announce_prefix_with_options(self, prefix: str,
                                     next_hop: str = "",
                                     community: str = "",
                                     local_pref: int = 0,
                                     med: int = 0) -> dict:
        """Announcement with path attributes via a per-prefix
        route-map (ref AnnouncePrefixWithOptions bgp.go:328-361 —
        the reference caches the options; applying them in FRR needs
        a route-map, which we emit too)."""
        rm = f"BNG-{prefix.replace('/', '-').replace('.', '-')}"
        lines = []
        if community or local_pref or med:
            lines += [f"route-map {rm} permit 10"]
            if community:
                lines += [f"set community {community}"]
            if local_pref:
                lines += [f"set local-preference {local_pref}"]
            if med:
                lines += [f"set metric {med}"]
            lines += ["exit"]
        self.exe.run(["configure terminal"] + lines + [
            f"router bgp {self.local_as}",
            "address-family ipv4 unicast",
            f"network {prefix}" + (f" route-map {rm}"
                                   if lines else ""),
            "exit-address-family", "end"])
        ann = {"prefix": prefix, "next_hop": next_hop,
               "community": community, "local_pref": local_pref,
               "med": med}
        with self._lock:
            self.announced[prefix] = ann
        return ann

    def enable_max_paths(self, max_paths: int):
        """ECMP across equal BGP paths (ref EnableMaxPaths
        bgp.go:431-448)."""
        if not (1 <= max_paths <= 128):
            raise ValueError("max_paths out of range")
        self._cfg([f"maximum-paths {max_paths}"])

    def configure_bfd_for_neighbor(self, address: str):
        """Tie the neighbor to BFD liveness (ref ConfigureBFD
        bgp.go:451-470)."""
        self.exe.run(["configure terminal",
                      f"router bgp {self.local_as}",
                      f"neighbor {address} bfd", "end"])


@dataclass
class BFDPeer:
    address: str
    interval_ms: int = 50
    multiplier: int = 3
    up: bool = False


class BFDManager:
    """BFD peers through FRR (ref bfd.go:153-629); ~50ms detection
    driving the ~200ms failover timeline."""

    def __init__(self, executor: FRRExecutor):
        self.exe = executor
        self.peers: Dict[str, BFDPeer] = {}
        self._lock = threading.RLock()
        self._listeners: List = []

    def on_state_change(self, cb):
        self._listeners.append(cb)

    def add_peer(self, address: str, interval_ms: int = 50,
                 multiplier: int = 3):
        self.exe.run(["configure terminal", "bfd",
                      f"peer {address}",
                      f"receive-interval {interval_ms}",
                      f"transmit-interval {interval_ms}",
                      f"detect-multiplier {multiplier}",
                      "no shutdown", "end"])
        with self._lock:
            self.peers[address] = BFDPeer(address, interval_ms, multiplier)

    def remove_peer(self, address: str):
        self.exe.run(["configure terminal", "bfd",
                      f"no peer {address}", "end"])
        with self._lock:
            self.peers.pop(address, None)

    def handle_state_change(self, address: str, up: bool):
        """Called by the FRR event feed (or tests)."""
        with self._lock:
            p = self.peers.get(address)
            if p is None or p.up == up:
                return
            p.up = up
        for cb in self._listeners:
            try:
                cb(address, up)
            except Exception:
                pass
