"""HIP dataplane launcher — the MI355X analog of the reference's eBPF
loader (pkg/ebpf/loader.go): owns the HBM-resident tables for one GPU
shard and exposes the same CRUD surface (AddSubscriber / AddPool /
SetServerConfig / GetStats / ..., loader.go:349-661) plus the batched
packet-processing entry points.

Every table mutation is enqueued on the current HIP stream, so the
dataplane kernels that follow it on that stream observe a consistent
table snapshot — the BPF-map semantics the reference gets from the
kernel, for free from stream ordering.

A CPU-backed launcher (GoldenLauncher) offers the same API over the
golden model for CPU-only tests and `bng demo` (the analog of running
the reference with no eBPF, cmd/bng/demo.go:46-60).
"""
from __future__ import annotations

import ctypes
import time
from typing import Dict, List, Optional, Sequence, Tuple

from . import abi
from .golden import (BindingRec, GoldenDataplane, PoolRecord, QosBucketRec,
                     SubnatRec, SubRecord)

# NAT timeouts (ref nat44.c:50-53)
UDP_TIMEOUT_NS = 120 * 10**9
EIM_TIMEOUT_NS = 2 * 3600 * 10**9     # idle EIM mapping lifetime
TCP_TRANSIENT_TIMEOUT_NS = 240 * 10**9
TCP_EST_TIMEOUT_NS = 7200 * 10**9
ICMP_TIMEOUT_NS = 60 * 10**9


class CircuitCollisionError(Exception):
    """Two distinct circuit-IDs hashed to the same key (the analog of the
    reference's FNV-1a collision detection, pkg/ebpf/loader.go:519-608)."""


def _struct_bytes(s: ctypes.Structure) -> bytes:
    return ctypes.string_at(ctypes.addressof(s), ctypes.sizeof(s))


class HipLauncher:
    """GPU-backed dataplane for one shard."""

    def __init__(self, device: str = "cuda:0", *,
                 sub_log2: int = 18, sess_log2: int = 20, eim_log2: int = 19,
                 subnat_log2: int = 18, qos_log2: int = 18,
                 binding_log2: int = 18, n_pools: int = 1024,
                 svc_cus: int = 0):
        import torch
        from .build import get_ext
        self.torch = torch
        self.ext = get_ext(required=True)
        # CU partition: reserve svc_cus CUs for the resident DHCP
        # service; pipeline launches go out masked off those CUs (the
        # saturated-flood latency lever — costs the pipeline
        # svc_cus/256 of peak)
        self.masked_compute = False
        if svc_cus > 0:
            self.ext.set_cu_partition(svc_cus)
            self.masked_compute = True
        if self.ext is None:
            raise RuntimeError(
                "bng dataplane extension not built — run "
                "python -m bng_amd.dataplane.build (no silent CPU fallback)")
        self.device = torch.device(device)
        z = lambda nbytes: torch.zeros(nbytes, dtype=torch.uint8,
                                       device=self.device)
        self.subs = z((1 << sub_log2) * 32)
        self.pools = z(n_pools * 28)        # bng_ip_pool is 28 B, packed
        self.n_pools = n_pools
        self.server_cfg = z(16)
        self.dhcp_stats = torch.zeros(abi.DHCP_NSTATS, dtype=torch.int64,
                                      device=self.device)
        self.sessions = z((1 << sess_log2) * 128)
        self.reverse = z((1 << sess_log2) * 48)
        self.eim = z((1 << eim_log2) * 48)
        # merged per-subscriber uplink context (NAT port block + ingress
        # token bucket in one 64-B entry; subnat_log2 names the kwarg the
        # reference's subscriber_nat map sizing maps to)
        self.subctx = z((1 << subnat_log2) * 64)
        self.nat_cfg = z(ctypes.sizeof(abi.NatConfig))
        self.hairpin = torch.zeros(abi.MAX_HAIRPIN_IPS, dtype=torch.int32,
                                   device=self.device)
        self.n_hairpin = 0
        self.nat_stats = torch.zeros(abi.NAT_NSTATS, dtype=torch.int64,
                                     device=self.device)
        self.nat_log_ring = z((1 << abi.LOG_RING_LOG2) * 40)
        self.nat_log_hdr = z(16)
        self._nat_log_ridx = 0
        self.qos_egress = z((1 << qos_log2) * 64)
        self.qos_stats = torch.zeros(abi.QOS_NSTATS, dtype=torch.int64,
                                     device=self.device)
        self.bindings = z((1 << binding_log2) * 32)
        self.as_cfg = z(ctypes.sizeof(abi.AntispoofConfig))
        self.as_stats = torch.zeros(abi.AS_NSTATS, dtype=torch.int64,
                                    device=self.device)
        self.spoof_ring = z((1 << abi.SPOOF_RING_LOG2) * 56)
        self.spoof_hdr = z(16)
        self._spoof_ridx = 0
        # host-side circuit-id collision registry (ref loader.go:519-608)
        self._circuit_ids: Dict[int, bytes] = {}
        # defaults
        self.set_nat_config()
        self.set_antispoof_config()

    # ------------------------------------------------------------ helpers
    def _to_dev(self, raw: bytes):
        import numpy as np
        t = self.torch.from_numpy(np.frombuffer(bytearray(raw), dtype=np.uint8))
        return t.to(self.device, non_blocking=True)

    # =================================================== DHCP (loader.go)
    def add_subscriber(self, mac, pool_id: int, ip: int, lease_expiry: int,
                       vlan_id: int = 0, client_class: int = 0,
                       flags: int = 0):
        """ref loader.go:352 AddSubscriber."""
        key = mac if isinstance(mac, int) else abi.mac_to_u64(bytes(mac))
        self._add_sub_entry(key, pool_id, ip, lease_expiry, vlan_id,
                            client_class, flags)

    def add_vlan_subscriber(self, s_tag: int, c_tag: int, pool_id: int,
                            ip: int, lease_expiry: int, **kw):
        self._add_sub_entry(abi.vlan_key(s_tag, c_tag), pool_id, ip,
                            lease_expiry, kw.get("vlan_id", s_tag),
                            kw.get("client_class", 0), kw.get("flags", 0))

    def add_circuit_subscriber(self, circuit_id: bytes, pool_id: int,
                               ip: int, lease_expiry: int, **kw):
        """Circuit-ID subscribers with FNV collision detection
        (ref loader.go:519-608 AddCircuitIDSubscriberWithCollisionCheck)."""
        cid = (bytes(circuit_id)[:32] + b"\x00" * 32)[:32]
        key = abi.circuit_key(cid)
        prev = self._circuit_ids.get(key)
        if prev is not None and prev != cid:
            raise CircuitCollisionError(
                f"circuit-id hash collision: {prev!r} vs {cid!r}")
        self._circuit_ids[key] = cid
        self._add_sub_entry(key, pool_id, ip, lease_expiry,
                            kw.get("vlan_id", 0), kw.get("client_class", 0),
                            kw.get("flags", 0))

    def _add_sub_entry(self, key, pool_id, ip, lease_expiry, vlan_id,
                       client_class, flags):
        e = abi.SubEntry(key=key, pool_id=pool_id, allocated_ip=ip,
                         lease_expiry=lease_expiry, vlan_id=vlan_id,
                         client_class=client_class, flags=flags)
        batch = self._to_dev(_struct_bytes(e))
        rc = self.torch.zeros(1, dtype=self.torch.int32, device=self.device)
        self.ext.sub_upsert(self.subs, batch, rc)

    def remove_subscriber(self, mac):
        key = mac if isinstance(mac, int) else abi.mac_to_u64(bytes(mac))
        self._del_sub_key(key)

    def remove_vlan_subscriber(self, s_tag: int, c_tag: int):
        self._del_sub_key(abi.vlan_key(s_tag, c_tag))

    def remove_circuit_subscriber(self, circuit_id: bytes):
        cid = (bytes(circuit_id)[:32] + b"\x00" * 32)[:32]
        key = abi.circuit_key(cid)
        self._circuit_ids.pop(key, None)
        self._del_sub_key(key)

    def _del_sub_key(self, key: int):
        import numpy as np
        keys = self.torch.from_numpy(
            np.array([key], dtype=np.uint64).view(np.int64)).to(self.device)
        self.ext.sub_delete(self.subs, keys)

    def add_pool(self, pool_id: int, network: int, prefix_len: int,
                 gateway: int, dns_primary: int = 0, dns_secondary: int = 0,
                 lease_time: int = 3600):
        """ref loader.go AddPool -> ip_pools map."""
        assert 0 <= pool_id < self.n_pools, "pool_id out of range"
        p = abi.IpPool(network=network, gateway=gateway,
                       dns_primary=dns_primary, dns_secondary=dns_secondary,
                       lease_time=lease_time, prefix_len=prefix_len, valid=1)
        raw = self._to_dev(_struct_bytes(p))
        self.pools[pool_id * 28:(pool_id + 1) * 28] = raw

    def remove_pool(self, pool_id: int):
        self.pools[pool_id * 28:(pool_id + 1) * 28] = 0

    def set_server_config(self, server_mac, server_ip: int,
                          if_index: int = 0):
        """ref loader.go:485 SetServerConfig."""
        mac = bytes(server_mac)
        c = abi.ServerConfig(server_ip=server_ip, if_index=if_index)
        for i in range(6):
            c.server_mac[i] = mac[i]
        self.server_cfg.copy_(self._to_dev(_struct_bytes(c)))

    def get_stats(self) -> Dict[str, int]:
        """ref loader.go:459 GetStats."""
        v = self.dhcp_stats.cpu().tolist()
        return dict(zip(abi.DHCP_STAT_NAMES, v))

    # ====================================================== NAT (manager)
    def set_nat_config(self, flags: int = abi.NAT_FLAG_EIM,
                       port_range=(1024, 65535), ports_per_sub: int = 1024,
                       private_ranges: Optional[Sequence[Tuple[int, int]]] = None,
                       alg_ports: Sequence[Tuple[int, int]] = ()):
        from .golden import DEFAULT_PRIVATE_RANGES
        pr = list(private_ranges) if private_ranges is not None \
            else list(DEFAULT_PRIVATE_RANGES)
        iv = abi.prefixes_to_intervals(pr)
        c = abi.NatConfig(flags=flags, port_range_start=port_range[0],
                          port_range_end=port_range[1],
                          default_ports_per_sub=ports_per_sub,
                          n_private_ranges=len(iv), n_alg_ports=len(alg_ports))
        for i, (lo, hi) in enumerate(iv):
            c.priv_lo[i] = lo
            c.priv_hi[i] = hi
        for i, (port, proto) in enumerate(alg_ports):
            c.alg_key[i] = (port << 16) | proto
        self.nat_cfg.copy_(self._to_dev(_struct_bytes(c)))

    def add_subscriber_nat(self, private_ip: int, public_ip: int,
                           port_start: int, port_end: int,
                           subscriber_id: int = 0):
        """ref nat/manager.go:398 AllocateNAT -> subscriber_nat map write."""
        e = abi.SubCtx(key_ip=private_ip, subscriber_id=subscriber_id,
                       public_ip=public_ip, port_start=port_start,
                       port_end=port_end, next_port=port_start)
        batch = self._to_dev(_struct_bytes(e))
        rc = self.torch.zeros(1, dtype=self.torch.int32, device=self.device)
        self.ext.subctx_upsert(self.subctx, batch, abi.CTX_SET_NAT, rc)

    def set_hairpin_ips(self, ips: Sequence[int]):
        import numpy as np
        arr = np.zeros(abi.MAX_HAIRPIN_IPS, dtype=np.uint32)
        arr[:len(ips)] = ips
        self.hairpin.copy_(self.torch.from_numpy(arr.view(np.int32))
                           .to(self.device))
        self.n_hairpin = len(ips)

    def nat_get_stats(self) -> Dict[str, int]:
        return dict(zip(abi.NAT_STAT_NAMES, self.nat_stats.cpu().tolist()))

    def drain_nat_log(self) -> List[dict]:
        """Drain the device->host compliance-log ring (the BPF ring-buffer
        analog, ref nat44.c:294-298 + nat/logging.go:293)."""
        return self._drain_ring(self.nat_log_hdr, self.nat_log_ring,
                                abi.NatLogEntry, 40, abi.LOG_RING_LOG2,
                                "_nat_log_ridx")

    def _drain_ring(self, hdr, ring, cls, esize, ring_log2, ridx_attr):
        import numpy as np
        widx = int(hdr.cpu().numpy().view(np.uint32)[0])
        ridx = getattr(self, ridx_attr)
        cap = 1 << ring_log2
        n = widx - ridx
        out: List[dict] = []
        if n <= 0:
            return out
        if n > cap:      # overrun: oldest records lost
            ridx = widx - cap
        raw = ring.cpu().numpy().tobytes()
        for i in range(ridx, widx):
            off = (i & (cap - 1)) * esize
            rec = cls.from_buffer_copy(raw[off:off + esize])
            out.append({f[0]: getattr(rec, f[0]) for f in cls._fields_
                        if not f[0].startswith("_")})
        setattr(self, ridx_attr, widx)
        return out

    def sweep_nat(self, now_ns: Optional[int] = None):
        """Expire timed-out sessions (the LRU/timeout sweeper; the reference
        gets eviction from BPF LRU maps)."""
        self.ext.nat_sweep(self.sessions, self.reverse, self.subctx,
                           self.eim, EIM_TIMEOUT_NS,
                           now_ns or time.time_ns(), UDP_TIMEOUT_NS,
                           TCP_EST_TIMEOUT_NS, TCP_TRANSIENT_TIMEOUT_NS,
                           ICMP_TIMEOUT_NS, self.nat_stats)

    # ============================================================== QoS
    def set_qos_policy(self, ip: int, rate_bps: int, burst_bytes: int,
                       priority: int = 0, direction: str = "egress",
                       now_ns: Optional[int] = None):
        """ref qos/manager.go:248 SetSubscriberPolicy."""
        now = now_ns if now_ns is not None else time.time_ns()
        rc = self.torch.zeros(1, dtype=self.torch.int32, device=self.device)
        if direction == "egress":
            b = abi.QosBucket(key_ip=ip, valid=1, priority=priority,
                              rate_bps=rate_bps, tokens=burst_bytes,
                              last_update=now, burst_bytes=burst_bytes)
            self.ext.qos_upsert(self.qos_egress, self._to_dev(
                _struct_bytes(b)), rc)
        else:
            e = abi.SubCtx(key_ip=ip, priority=priority, rate_bps=rate_bps,
                           tokens=burst_bytes, last_update=now,
                           burst_bytes=burst_bytes)
            self.ext.subctx_upsert(self.subctx, self._to_dev(
                _struct_bytes(e)), abi.CTX_SET_QOS, rc)

    def remove_qos_policy(self, ip: int, direction: str = "egress"):
        rc = self.torch.zeros(1, dtype=self.torch.int32, device=self.device)
        if direction == "egress":
            b = abi.QosBucket(key_ip=ip, valid=0)
            self.ext.qos_upsert(self.qos_egress, self._to_dev(
                _struct_bytes(b)), rc)
        else:
            e = abi.SubCtx(key_ip=ip)
            self.ext.subctx_upsert(self.subctx, self._to_dev(
                _struct_bytes(e)), abi.CTX_CLR_QOS, rc)

    def qos_get_stats(self) -> Dict[str, int]:
        return dict(zip(abi.QOS_STAT_NAMES, self.qos_stats.cpu().tolist()))

    # ======================================================== antispoof
    def set_antispoof_config(self, default_mode: int = abi.AS_DISABLED,
                             log_violations: bool = False,
                             allowed_ranges: Sequence[Tuple[int, int]] = ()):
        iv = abi.prefixes_to_intervals(list(allowed_ranges))
        c = abi.AntispoofConfig(default_mode=default_mode,
                                log_violations=1 if log_violations else 0,
                                n_allowed_ranges=len(iv))
        for i, (lo, hi) in enumerate(iv):
            c.allowed_lo[i] = lo
            c.allowed_hi[i] = hi
        self.as_cfg.copy_(self._to_dev(_struct_bytes(c)))

    def add_binding(self, mac, ipv4: int = 0, ipv6: bytes = b"",
                    mode: int = abi.AS_STRICT):
        """ref antispoof/manager.go:200 AddBinding."""
        key = mac if isinstance(mac, int) else abi.mac_to_u64(bytes(mac))
        b = abi.BindingEntry(key_mac=key, ipv4_addr=ipv4,
                             ipv4_valid=1 if ipv4 else 0,
                             ipv6_valid=1 if ipv6 else 0, mode=mode)
        for i, by in enumerate((ipv6 or b"")[:16]):
            b.ipv6_addr[i] = by
        batch = self._to_dev(_struct_bytes(b))
        rc = self.torch.zeros(1, dtype=self.torch.int32, device=self.device)
        self.ext.binding_upsert(self.bindings, batch, rc)

    def remove_binding(self, mac):
        import numpy as np
        key = mac if isinstance(mac, int) else abi.mac_to_u64(bytes(mac))
        keys = self.torch.from_numpy(
            np.array([key], dtype=np.uint64).view(np.int64)).to(self.device)
        self.ext.binding_delete(self.bindings, keys)

    def antispoof_get_stats(self) -> Dict[str, int]:
        return dict(zip(abi.AS_STAT_NAMES, self.as_stats.cpu().tolist()))

    def drain_spoof_events(self) -> List[dict]:
        return self._drain_ring(self.spoof_hdr, self.spoof_ring,
                                abi.SpoofEvent, 56, abi.SPOOF_RING_LOG2,
                                "_spoof_ridx")

    # ================================================ batched processing
    def make_batch(self, frames: Sequence[bytes], stride: int = 512):
        """Pack frames into a device batch (data[n,stride], len[n])."""
        import numpy as np
        n = len(frames)
        data = np.zeros((n, stride), dtype=np.uint8)
        lens = np.zeros(n, dtype=np.uint16)
        for i, f in enumerate(frames):
            L = min(len(f), stride)
            data[i, :L] = np.frombuffer(f, dtype=np.uint8)[:L]
            lens[i] = L
        d = self.torch.from_numpy(data).to(self.device)
        l = self.torch.from_numpy(lens.view(np.int16)).to(self.device)
        return d, l

    def _outs(self, n):
        v = self.torch.zeros(n, dtype=self.torch.uint8, device=self.device)
        o = self.torch.zeros(n, dtype=self.torch.int16, device=self.device)
        return v, o

    def dhcp_fastpath(self, data, lens, now_sec: Optional[int] = None):
        n = lens.numel()
        verdict, out_len = self._outs(n)
        self.ext.dhcp_fastpath(data, lens, out_len, verdict, self.subs,
                               self.pools, self.server_cfg, self.dhcp_stats,
                               now_sec if now_sec is not None
                               else int(time.time()))
        return verdict, out_len

    def nat44(self, data, lens, egress: bool = True,
              now_ns: Optional[int] = None):
        n = lens.numel()
        verdict = self.torch.zeros(n, dtype=self.torch.uint8,
                                   device=self.device)
        self.ext.nat44(data, lens, verdict, egress, self.sessions,
                       self.reverse, self.eim, self.subctx, self.nat_cfg,
                       self.hairpin, self.n_hairpin, self.nat_stats,
                       self.nat_log_ring, self.nat_log_hdr,
                       now_ns if now_ns is not None else time.time_ns())
        return verdict

    def qos(self, data, lens, egress: bool = True,
            now_ns: Optional[int] = None):
        n = lens.numel()
        verdict = self.torch.zeros(n, dtype=self.torch.uint8,
                                   device=self.device)
        # ingress policies live in the merged subscriber context
        table = self.qos_egress if egress else self.subctx
        self.ext.qos(data, lens, verdict, egress, table, self.qos_stats,
                     now_ns if now_ns is not None else time.time_ns())
        return verdict

    def antispoof(self, data, lens, now_ns: Optional[int] = None):
        n = lens.numel()
        verdict = self.torch.zeros(n, dtype=self.torch.uint8,
                                   device=self.device)
        self.ext.antispoof(data, lens, verdict, self.bindings, self.as_cfg,
                           self.as_stats, self.spoof_ring, self.spoof_hdr,
                           now_ns if now_ns is not None else time.time_ns())
        return verdict

    def uplink(self, data, lens, now_ns: Optional[int] = None,
               now_sec: Optional[int] = None, sort_by_type: bool = False,
               order=None):
        """Fused antispoof -> NAT44 SNAT -> QoS-ingress + DHCP fast path.

        sort_by_type=True first classifies packets on-device and feeds the
        kernel a type-sorted index order, so each wave's 64 lanes run ONE
        of the two pipelines instead of serializing both (the wave-
        divergence fix; packet data itself is not moved)."""
        n = lens.numel()
        verdict, out_len = self._outs(n)
        now = now_ns if now_ns is not None else time.time_ns()
        if order is None and sort_by_type and n > 64:
            cls = self.torch.empty(n, dtype=self.torch.uint8,
                                   device=self.device)
            self.ext.pkt_class(data, lens, cls)
            order = self.torch.argsort(cls, stable=True).to(self.torch.int32)
        self.ext.uplink_pipeline(
            data, lens, out_len, verdict, self.subs, self.pools,
            self.server_cfg, self.dhcp_stats, self.bindings, self.as_cfg,
            self.as_stats, self.spoof_ring, self.spoof_hdr, self.sessions,
            self.reverse, self.eim, self.subctx, self.nat_cfg, self.hairpin,
            self.n_hairpin, self.nat_stats, self.nat_log_ring,
            self.nat_log_hdr, self.qos_egress, self.qos_stats, now,
            now_sec if now_sec is not None else now // 10**9, order=order,
            masked=self.masked_compute)
        return verdict, out_len

    def downlink(self, data, lens, now_ns: Optional[int] = None):
        """Fused NAT44 DNAT -> QoS-egress return path."""
        n = lens.numel()
        verdict, out_len = self._outs(n)
        now = now_ns if now_ns is not None else time.time_ns()
        self.ext.uplink_pipeline(
            data, lens, out_len, verdict, self.subs, self.pools,
            self.server_cfg, self.dhcp_stats, self.bindings, self.as_cfg,
            self.as_stats, self.spoof_ring, self.spoof_hdr, self.sessions,
            self.reverse, self.eim, self.subctx, self.nat_cfg, self.hairpin,
            self.n_hairpin, self.nat_stats, self.nat_log_ring,
            self.nat_log_hdr, self.qos_egress, self.qos_stats, now,
            now // 10**9, order=None, downlink=True,
            masked=self.masked_compute)
        return verdict

    # ------------------------------------------- HA table snapshotting
    def export_subscribers(self) -> List[dict]:
        """Download the subscriber fast-path table for HA sync (the GPU
        analog of the reference's session-store snapshot feeding
        HASyncer full syncs)."""
        import numpy as np
        raw = self.subs.cpu().numpy()
        arr = raw.view([("key", "<u8"), ("pool_id", "<u4"),
                        ("ip", "<u4"), ("lease", "<u8"),
                        ("vlan", "<u2"), ("cc", "u1"), ("fl", "u1"),
                        ("pad", "<u4")])
        live = arr[(arr["key"] != 0) &
                   (arr["key"] != 0xFFFFFFFFFFFFFFFF)]
        return [{"key": int(e["key"]), "pool_id": int(e["pool_id"]),
                 "ip": int(e["ip"]), "lease_expiry": int(e["lease"]),
                 "vlan_id": int(e["vlan"]),
                 "client_class": int(e["cc"]), "flags": int(e["fl"])}
                for e in live]

    def import_subscribers(self, entries: Sequence[dict]) -> int:
        """Bulk-load a snapshot (standby promotion: shadow state becomes
        the live fast path)."""
        import numpy as np
        if not entries:
            return 0
        n = len(entries)
        arr = np.zeros(n, dtype=[("key", "<u8"), ("pool_id", "<u4"),
                                 ("ip", "<u4"), ("lease", "<u8"),
                                 ("vlan", "<u2"), ("cc", "u1"),
                                 ("fl", "u1"), ("pad", "<u4")])
        for i, e in enumerate(entries):
            arr[i] = (e["key"], e.get("pool_id", 0), e.get("ip", 0),
                      e.get("lease_expiry", 0), e.get("vlan_id", 0),
                      e.get("client_class", 0), e.get("flags", 0), 0)
        batch = self.torch.from_numpy(arr.view(np.uint8)).to(self.device)
        rc = self.torch.zeros(n, dtype=self.torch.int32,
                              device=self.device)
        self.ext.sub_upsert(self.subs, batch.flatten(), rc)
        return n - int((rc != 0).sum().item())

    def export_nat_sessions(self, since_ns: int = 0):
        """Download live NAT sessions for HA sync as a structured numpy
        array (SESS_EXPORT_DTYPE records).  since_ns>0 = delta export
        (sessions seen since then); the EIM flag is set when EIM mode is
        on so promotion restores the mapping too.  Ref ha/sync.go
        session replication; round-1 VERDICT task 3."""
        import numpy as np
        blob = self.sessions.cpu().numpy()
        arr = blob.view([("sig", "<u8"), ("src_ip", "<u4"),
                         ("dst_ip", "<u4"), ("src_port", "<u2"),
                         ("dst_port", "<u2"), ("protocol", "u1"),
                         ("_kp", "3u1"), ("nat_ip", "<u4"),
                         ("nat_port", "<u2"), ("orig_port", "<u2"),
                         ("orig_ip", "<u4"), ("state", "u1"),
                         ("is_hairpin", "u1"), ("ready", "u1"),
                         ("_p", "u1"), ("last_seen", "<u8"),
                         ("created", "<u8"), ("_acct", "9u8")])
        live = arr[(arr["sig"] != 0) &
                   (arr["sig"] != 0xFFFFFFFFFFFFFFFF) &
                   (arr["ready"] != 0) &
                   (arr["last_seen"] >= np.uint64(since_ns))]
        out = np.zeros(len(live), dtype=abi.SESS_EXPORT_DTYPE)
        for f in ("src_ip", "dst_ip", "src_port", "dst_port", "protocol",
                  "state", "is_hairpin", "nat_ip", "nat_port", "created",
                  "last_seen"):
            out[f] = live[f]
        # EIM restore flag: the create path stores the SAME host-order
        # allocated port in session.nat_port and eim.external_port
        eim_on = bool(abi.NatConfig.from_buffer_copy(
            self.nat_cfg.cpu().numpy().tobytes()).flags & abi.NAT_FLAG_EIM)
        if eim_on:
            out["flags"] = 1
            out["eim_port"] = live["nat_port"]
        return out

    def import_nat_sessions(self, records) -> int:
        """Bulk-restore exported NAT sessions (+ reverse + EIM) on a
        standby at promotion (sess_import_kernel)."""
        import numpy as np
        records = np.asarray(records, dtype=abi.SESS_EXPORT_DTYPE)
        if len(records) == 0:
            return 0
        batch = self.torch.from_numpy(records.view(np.uint8)) \
            .to(self.device)
        rc = self.torch.zeros(len(records), dtype=self.torch.int32,
                              device=self.device)
        self.ext.sess_import(self.sessions, self.reverse, self.eim,
                             batch.flatten(), rc)
        return len(records) - int((rc != 0).sum().item())

    # ------------------------------------------- hipGraph steady state
    def capture_uplink(self, n: int, stride: int = 512,
                       sort_by_type: bool = True):
        """Capture the steady-state forwarding sequence
        [RX-copy -> classify -> type-sort -> uplink] into a hipGraph
        (SURVEY blueprint: hipGraph-captured steady-state forwarding).
        Returns a CapturedUplink; batch time advances through a device
        now-buffer so replays see fresh timestamps."""
        return CapturedUplink(self, n, stride, sort_by_type)

    def capture_dhcp(self, n: int, stride: int = 512):
        return CapturedDHCP(self, n, stride)

    def shard_owner(self, data, lens, n_shards: int):
        n = lens.numel()
        owner = self.torch.zeros(n, dtype=self.torch.int32,
                                 device=self.device)
        self.ext.shard_owner(data, lens, owner, n_shards)
        return owner


class DhcpService:
    """Persistent-kernel DHCP fast path: one resident workgroup polls a
    pinned-host doorbell and serves request batches with no kernel
    launch in the loop.  Because its waves permanently own their CU, a
    saturating data flood cannot starve it (round-1 measured flood p99
    453 us for the launched path; the reference's in-IRQ XDP has the
    same always-resident property, dhcp_fastpath.c:619).

    Latency path only — the batched dhcp_fastpath/uplink kernels remain
    the throughput path."""

    # Pinned-coherent buffers are allocated ONCE per (n_slots, stride)
    # and cached for the process lifetime: hipHostMalloc/hipHostFree
    # can implicitly synchronize the device, which deadlocks against a
    # resident kernel (measured as a hang at service start on hardware)
    # — never allocate or free host-mapped memory while services run.
    _buf_cache: Dict[tuple, dict] = {}

    @classmethod
    def _buffers(cls, launcher, n_slots: int, stride: int) -> dict:
        torch = launcher.torch
        key = (str(launcher.device), n_slots, stride)
        b = cls._buf_cache.get(key)
        if b is None:
            alloc = launcher.ext.alloc_pinned_coherent
            b = {"ctrl": alloc(64),
                 "req": alloc(n_slots * stride).view(n_slots, stride),
                 "in_len": alloc(n_slots * 2).view(torch.int16),
                 "out_len": alloc(n_slots * 2).view(torch.int16),
                 "verdict": alloc(n_slots),
                 "scratch": torch.empty((n_slots, stride),
                                        dtype=torch.uint8,
                                        device=launcher.device),
                 "ctrs": torch.zeros(8, dtype=torch.int32,
                                     device=launcher.device)}
            cls._buf_cache[key] = b
        return b

    _active: Optional["DhcpService"] = None

    def __init__(self, launcher: "HipLauncher", n_slots: int = 2048,
                 stride: int = 512, idle_exit_k: int = 1_000_000,
                 n_blocks: int = 4):
        import numpy as np
        # one service at a time per process: the service stream is
        # global, so a second resident kernel would queue behind the
        # first and its doorbell would look dead
        if DhcpService._active is not None and \
                not DhcpService._active.c.exited:
            raise RuntimeError("another DhcpService is still running; "
                               "stop() it first")
        self.l = launcher
        self.n_slots, self.stride = n_slots, stride
        b = self._buffers(launcher, n_slots, stride)
        self.ctrl_t = b["ctrl"]
        self.req = b["req"]
        self.in_len = b["in_len"]
        self.out_len = b["out_len"]
        self.verdict = b["verdict"]
        self.scratch = b["scratch"]
        self.ctrs = b["ctrs"]
        # reset reused state (doorbell + completion counters)
        self.ctrl_t.zero_()
        self.ctrs.zero_()
        self.c = abi.SvcCtrl.from_address(self.ctrl_t.data_ptr())
        self.c.stride = stride
        # idle self-exit: ~1us/poll (PCIe acquire) -> default reaps
        # after ~17 min (the box-safety bound if the owner dies
        # without stop())
        self.c.idle_exit_k = idle_exit_k
        self.c.run = 1
        launcher.torch.cuda.synchronize(launcher.device)  # ctrs zeroed
        launcher.ext.dhcp_service_start(
            self.ctrl_t, self.req, self.in_len, self.out_len,
            self.verdict, self.scratch, n_slots, n_blocks, self.ctrs,
            launcher.subs, launcher.pools, launcher.server_cfg,
            launcher.dhcp_stats)
        DhcpService._active = self
        self._np = np

    @property
    def running(self) -> bool:
        return bool(self.c.run)

    def serve(self, data_np, lens_np, now_sec: int, timeout: float = 1.0):
        """One request batch through the resident kernel; returns
        (verdict, out_len, reply_bytes) numpy views — copy before the
        next serve() if they must survive."""
        np = self._np
        n = len(lens_np)
        assert n <= self.n_slots
        rq = self.req.numpy()
        rq[:n, :data_np.shape[1]] = data_np
        self.in_len.numpy()[:n] = lens_np.view(np.int16)
        self.c.n_pkts = n
        self.c.now_sec = now_sec
        target = (self.c.head + 1) & 0xFFFFFFFF
        self.c.head = target             # doorbell (x86 TSO publishes)
        deadline = time.perf_counter() + timeout
        spins = 0
        while self.c.tail != target:
            spins += 1
            if (spins & 0x3FFF) == 0:
                if self.c.exited:
                    raise RuntimeError("dhcp service kernel exited")
                if time.perf_counter() > deadline:
                    raise TimeoutError("dhcp service timeout")
        return (self.verdict.numpy()[:n], self.out_len.numpy()[:n],
                rq[:n])

    def stats(self) -> Dict[str, int]:
        return {"served": int(self.c.served),
                "batches": int(self.c.batches)}

    def stop(self, timeout: float = 3.0):
        """Never blocks unboundedly: wait for the kernel's exit ack,
        and if the doorbell is dead leave the kernel to its idle
        self-exit instead of hanging on a stream sync."""
        self.c.run = 0
        deadline = time.perf_counter() + timeout
        while not self.c.exited:
            if time.perf_counter() > deadline:
                raise RuntimeError(
                    "dhcp service kernel did not ack stop; doorbell "
                    "dead — kernel left to idle self-exit")
            time.sleep(0.001)
        self.l.ext.dhcp_service_join()
        if DhcpService._active is self:
            DhcpService._active = None

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        try:
            self.stop()
        except RuntimeError:
            if exc[0] is None:     # don't mask the original exception
                raise


class GoldenLauncher:
    """CPU launcher over the golden model: same API, for CPU tests/demo."""

    def __init__(self, **_kw):
        self.dp = GoldenDataplane(now_ns=time.time_ns())
        self._circuit_ids: Dict[int, bytes] = {}

    # DHCP
    def add_subscriber(self, mac, pool_id, ip, lease_expiry, vlan_id=0,
                       client_class=0, flags=0):
        key = mac if isinstance(mac, int) else abi.mac_to_u64(bytes(mac))
        self.dp.subscribers[key] = SubRecord(pool_id, ip, lease_expiry,
                                             vlan_id, client_class, flags)

    def add_vlan_subscriber(self, s_tag, c_tag, pool_id, ip, lease_expiry,
                            **kw):
        self.dp.subscribers[abi.vlan_key(s_tag, c_tag)] = SubRecord(
            pool_id, ip, lease_expiry, kw.get("vlan_id", s_tag))

    def add_circuit_subscriber(self, circuit_id, pool_id, ip, lease_expiry,
                               **kw):
        cid = (bytes(circuit_id)[:32] + b"\x00" * 32)[:32]
        key = abi.circuit_key(cid)
        prev = self._circuit_ids.get(key)
        if prev is not None and prev != cid:
            raise CircuitCollisionError(
                f"circuit-id hash collision: {prev!r} vs {cid!r}")
        self._circuit_ids[key] = cid
        self.dp.subscribers[key] = SubRecord(pool_id, ip, lease_expiry)

    def remove_subscriber(self, mac):
        key = mac if isinstance(mac, int) else abi.mac_to_u64(bytes(mac))
        self.dp.subscribers.pop(key, None)

    def remove_vlan_subscriber(self, s_tag, c_tag):
        self.dp.subscribers.pop(abi.vlan_key(s_tag, c_tag), None)

    def remove_circuit_subscriber(self, circuit_id):
        cid = (bytes(circuit_id)[:32] + b"\x00" * 32)[:32]
        key = abi.circuit_key(cid)
        self._circuit_ids.pop(key, None)
        self.dp.subscribers.pop(key, None)

    def add_pool(self, pool_id, network, prefix_len, gateway, dns_primary=0,
                 dns_secondary=0, lease_time=3600):
        self.dp.pools[pool_id] = PoolRecord(network, prefix_len, gateway,
                                            dns_primary, dns_secondary,
                                            lease_time)

    def remove_pool(self, pool_id):
        self.dp.pools.pop(pool_id, None)

    def set_server_config(self, server_mac, server_ip, if_index=0):
        self.dp.server_mac = bytes(server_mac)
        self.dp.server_ip = server_ip

    def get_stats(self):
        return dict(zip(abi.DHCP_STAT_NAMES, self.dp.dhcp_stats))

    # NAT
    def set_nat_config(self, flags=abi.NAT_FLAG_EIM, port_range=(1024, 65535),
                       ports_per_sub=1024, private_ranges=None, alg_ports=()):
        self.dp.nat_flags = flags
        if private_ranges is not None:
            self.dp.private_ranges = list(private_ranges)
        self.dp.alg_ports = set(alg_ports)

    def add_subscriber_nat(self, private_ip, public_ip, port_start, port_end,
                           subscriber_id=0):
        self.dp.subnat[private_ip] = SubnatRec(
            public_ip, port_start, port_end, port_start, subscriber_id)

    def set_hairpin_ips(self, ips):
        self.dp.hairpin_ips = set(ips)

    def nat_get_stats(self):
        return dict(zip(abi.NAT_STAT_NAMES, self.dp.nat_stats))

    def drain_nat_log(self):
        out, self.dp.nat_log = self.dp.nat_log, []
        return out

    def sweep_nat(self, now_ns=None):
        now = now_ns or time.time_ns()
        dead = []
        for key, s in self.dp.nat_sessions.items():
            to = UDP_TIMEOUT_NS
            if s.protocol == 6:
                to = TCP_EST_TIMEOUT_NS if s.state == abi.NAT_ESTABLISHED \
                    else TCP_TRANSIENT_TIMEOUT_NS
            elif s.protocol == 1:
                to = ICMP_TIMEOUT_NS
            if now - s.last_seen >= to or (
                    s.state == abi.NAT_CLOSING and
                    now - s.last_seen >= TCP_TRANSIENT_TIMEOUT_NS):
                dead.append((key, s))
        for key, s in dead:
            rev = (s.dest_ip, s.nat_ip, s.dest_port, s.nat_port, s.protocol)
            self.dp.nat_reverse.pop(rev, None)
            del self.dp.nat_sessions[key]
            blk = self.dp.subnat.get(s.orig_ip)
            if blk:
                blk.sessions_active -= 1
            self.dp.nat_stats[abi.NS_SESS_EXPIRED] += 1
        # EIM idle expiry (mirrors the GPU sweep's LRU analog)
        for ek in [k for k, e in self.dp.eim.items()
                   if now - e.last_used >= EIM_TIMEOUT_NS]:
            del self.dp.eim[ek]

    # QoS
    def set_qos_policy(self, ip, rate_bps, burst_bytes, priority=0,
                       direction="egress", now_ns=None):
        table = self.dp.qos_egress if direction == "egress" \
            else self.dp.qos_ingress
        table[ip] = QosBucketRec(rate_bps, burst_bytes, burst_bytes,
                                 now_ns or self.dp.now_ns, priority)

    def remove_qos_policy(self, ip, direction="egress"):
        table = self.dp.qos_egress if direction == "egress" \
            else self.dp.qos_ingress
        table.pop(ip, None)

    def qos_get_stats(self):
        return dict(zip(abi.QOS_STAT_NAMES, self.dp.qos_stats))

    # antispoof
    def set_antispoof_config(self, default_mode=abi.AS_DISABLED,
                             log_violations=False, allowed_ranges=()):
        self.dp.as_default_mode = default_mode
        self.dp.as_log_violations = 1 if log_violations else 0
        self.dp.allowed_ranges = list(allowed_ranges)

    def add_binding(self, mac, ipv4=0, ipv6=b"", mode=abi.AS_STRICT):
        key = mac if isinstance(mac, int) else abi.mac_to_u64(bytes(mac))
        self.dp.bindings[key] = BindingRec(
            ipv4, 1 if ipv4 else 0,
            (ipv6 + b"\x00" * 16)[:16] if ipv6 else b"\x00" * 16,
            1 if ipv6 else 0, mode)

    def remove_binding(self, mac):
        key = mac if isinstance(mac, int) else abi.mac_to_u64(bytes(mac))
        self.dp.bindings.pop(key, None)

    def antispoof_get_stats(self):
        return dict(zip(abi.AS_STAT_NAMES, self.dp.as_stats))

    def drain_spoof_events(self):
        out, self.dp.spoof_events = self.dp.spoof_events, []
        return out

    # processing (frames as list[bytearray]); mirrors HipLauncher outputs
    def process_dhcp(self, frames, now_sec=None):
        if now_sec is not None:
            self.dp.now_ns = now_sec * 10**9
        results = []
        for f in frames:
            fb = bytearray(f)
            v, L = self.dp.dhcp_fastpath(fb)
            results.append((v, bytes(fb[:L])))
        return results

    def export_subscribers(self):
        from . import abi as _abi
        return [{"key": k, "pool_id": r.pool_id, "ip": r.allocated_ip,
                 "lease_expiry": int(r.lease_expiry),
                 "vlan_id": r.vlan_id, "client_class": r.client_class,
                 "flags": r.flags}
                for k, r in self.dp.subscribers.items()]

    def import_subscribers(self, entries):
        for e in entries:
            self.dp.subscribers[e["key"]] = SubRecord(
                e.get("pool_id", 0), e.get("ip", 0),
                e.get("lease_expiry", 0), e.get("vlan_id", 0),
                e.get("client_class", 0), e.get("flags", 0))
        return len(entries)

    def export_nat_sessions(self, since_ns: int = 0):
        """Golden-model NAT session export, same record layout as the
        GPU path (SESS_EXPORT_DTYPE)."""
        import numpy as np
        recs = [(k, s) for k, s in self.dp.nat_sessions.items()
                if s.last_seen >= since_ns]
        out = np.zeros(len(recs), dtype=abi.SESS_EXPORT_DTYPE)
        eim_on = bool(self.dp.nat_flags & abi.NAT_FLAG_EIM)
        for i, (k, s) in enumerate(recs):
            src, dst, sp, dp_, pr = k
            out[i] = (src, dst, sp, dp_, pr, s.state, s.is_hairpin,
                      1 if eim_on else 0, s.nat_ip, s.nat_port,
                      s.nat_port if eim_on else 0, s.created,
                      s.last_seen, 0)
        return out

    def import_nat_sessions(self, records) -> int:
        import numpy as np
        from .golden import EimRec, NatSessionRec
        records = np.asarray(records, dtype=abi.SESS_EXPORT_DTYPE)
        for r in records:
            key = (int(r["src_ip"]), int(r["dst_ip"]), int(r["src_port"]),
                   int(r["dst_port"]), int(r["protocol"]))
            self.dp.nat_sessions[key] = NatSessionRec(
                int(r["nat_ip"]), int(r["nat_port"]), int(r["src_port"]),
                int(r["src_ip"]), int(r["dst_ip"]), int(r["dst_port"]),
                int(r["state"]), int(r["is_hairpin"]),
                int(r["last_seen"]), int(r["created"]), 0, 0, 0, 0,
                int(r["protocol"]))
            self.dp.nat_reverse[(int(r["dst_ip"]), int(r["nat_ip"]),
                                 int(r["dst_port"]), int(r["nat_port"]),
                                 int(r["protocol"]))] = key
            if int(r["flags"]) & 1:
                self.dp.eim[(int(r["src_ip"]), int(r["src_port"]),
                             int(r["protocol"]))] = EimRec(
                    int(r["nat_ip"]), int(r["eim_port"]),
                    int(r["created"]), int(r["last_seen"]), 1)
        return len(records)

    def process_nat44(self, frames, egress=True, now_ns=None):
        if now_ns is not None:
            self.dp.now_ns = now_ns
        results = []
        for f in frames:
            fb = bytearray(f)
            v = self.dp.nat44_egress(fb) if egress else self.dp.nat44_ingress(fb)
            results.append((v, bytes(fb)))
        return results


def make_launcher(prefer_gpu: bool = True, **kw):
    """GPU launcher when a device is present, else the golden-model CPU
    launcher.  On a GPU host the HIP path is mandatory (no silent
    fallback): if torch sees a device but the extension is missing, raise."""
    if prefer_gpu:
        try:
            import torch
            has = torch.cuda.is_available()
        except Exception:
            has = False
        if has:
            return HipLauncher(**kw)
    return GoldenLauncher(**kw)


class CapturedUplink:
    """hipGraph replay wrapper for the fused uplink pipeline."""

    def __init__(self, launcher: HipLauncher, n: int, stride: int,
                 sort_by_type: bool):
        import torch
        L = self.l = launcher
        self.n, self.stride = n, stride
        dev = L.device
        self.src = torch.zeros((n, stride), dtype=torch.uint8, device=dev)
        self.work = torch.empty_like(self.src)
        self.lens = torch.zeros(n, dtype=torch.int16, device=dev)
        self.verdict = torch.zeros(n, dtype=torch.uint8, device=dev)
        self.out_len = torch.zeros(n, dtype=torch.int16, device=dev)
        self.now_buf = torch.zeros(2, dtype=torch.int64, device=dev)
        self.cls = torch.zeros(n, dtype=torch.uint8, device=dev)
        self.sort = sort_by_type

        def body():
            self.work.copy_(self.src)
            order = None
            if self.sort:
                L.ext.pkt_class(self.work, self.lens, self.cls)
                order = torch.argsort(self.cls, stable=True).to(torch.int32)
            L.ext.uplink_pipeline(
                self.work, self.lens, self.out_len, self.verdict, L.subs,
                L.pools, L.server_cfg, L.dhcp_stats, L.bindings, L.as_cfg,
                L.as_stats, L.spoof_ring, L.spoof_hdr, L.sessions,
                L.reverse, L.eim, L.subctx, L.nat_cfg, L.hairpin,
                L.n_hairpin, L.nat_stats, L.nat_log_ring, L.nat_log_hdr,
                L.qos_egress, L.qos_stats, 0, 0, order=order,
                downlink=False, now_buf=self.now_buf)

        # warmup on a side stream, then capture
        s = torch.cuda.Stream(device=dev)
        s.wait_stream(torch.cuda.current_stream(dev))
        with torch.cuda.stream(s):
            for _ in range(2):
                body()
        torch.cuda.current_stream(dev).wait_stream(s)
        torch.cuda.synchronize(dev)
        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            body()

    def run(self, data, lens, now_ns: int):
        """Copy the batch into the captured buffers, bump time, replay."""
        import torch
        self.src.copy_(data)
        self.lens.copy_(lens)
        self.now_buf.copy_(torch.tensor(
            [now_ns, now_ns // 10**9], dtype=torch.int64))
        self.graph.replay()
        return self.verdict, self.out_len


class CapturedDHCP:
    """hipGraph replay wrapper for the DHCP fast path (latency path)."""

    def __init__(self, launcher: HipLauncher, n: int, stride: int):
        import torch
        L = self.l = launcher
        dev = L.device
        self.src = torch.zeros((n, stride), dtype=torch.uint8, device=dev)
        self.work = torch.empty_like(self.src)
        self.lens = torch.zeros(n, dtype=torch.int16, device=dev)
        self.verdict = torch.zeros(n, dtype=torch.uint8, device=dev)
        self.out_len = torch.zeros(n, dtype=torch.int16, device=dev)
        self.now_buf = torch.zeros(2, dtype=torch.int64, device=dev)

        def body():
            self.work.copy_(self.src)
            L.ext.dhcp_fastpath(self.work, self.lens, self.out_len,
                                self.verdict, L.subs, L.pools,
                                L.server_cfg, L.dhcp_stats, 0,
                                now_buf=self.now_buf)

        s = torch.cuda.Stream(device=dev)
        s.wait_stream(torch.cuda.current_stream(dev))
        with torch.cuda.stream(s):
            for _ in range(2):
                body()
        torch.cuda.current_stream(dev).wait_stream(s)
        torch.cuda.synchronize(dev)
        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            body()

    def run(self, now_sec: int):
        """Replay on whatever is in self.src (caller filled it)."""
        import torch
        self.now_buf.copy_(torch.tensor(
            [now_sec * 10**9, now_sec], dtype=torch.int64))
        self.graph.replay()
        return self.verdict, self.out_len
