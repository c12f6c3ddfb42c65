"""HA <-> dataplane glue: replicate DHCP leases (and their GPU fast-path
entries) through the HASyncer, and rebuild the live state on standby
promotion — the reference's session_integration analog with the GPU
table as the final sink (GPU-table snapshot/delta sync
feeding the HASyncer protocol)."""
from __future__ import annotations

import time

from ..dataplane.packets import u32_to_ip, ip2u32
from ..dhcp.server import DHCPServer, Lease
from .protocol import SessionState
from .sync import HASyncer


def lease_to_session(lease: Lease) -> SessionState:
    return SessionState(
        session_id=f"dhcp-{lease.mac.hex()}",
        subscriber_id=lease.subscriber_id,
        mac=lease.mac.hex(),
        ip=u32_to_ip(lease.ip),
        access_type="dhcp",
        policy_name=lease.policy_name,
        lease_expiry=lease.expiry,
        vlan=0)


def session_to_lease(s: SessionState, pool_id: int = 1) -> Lease:
    return Lease(mac=bytes.fromhex(s.mac), ip=ip2u32(s.ip),
                 pool_id=pool_id, expiry=s.lease_expiry or
                 time.time() + 3600, subscriber_id=s.subscriber_id,
                 policy_name=s.policy_name)


def attach(dhcp_server: DHCPServer, syncer: HASyncer,
           nat_mgr=None) -> None:
    """Active side: every lease add/delete becomes a sync delta.  With
    a NAT manager the delta carries the exact port-block assignment, so
    promotion restores the SAME block (ref SessionState carries
    NATPortStart/End, ha/protocol.go:76-111)."""

    def on_event(event: str, lease: Lease):
        if event == "add":
            s = lease_to_session(lease)
            # nat_mgr may be a callable for late binding (CLI wires HA
            # before the NAT manager exists)
            mgr = nat_mgr() if callable(nat_mgr) else nat_mgr
            if mgr is not None:
                alloc = mgr.allocations.get(lease.ip)
                if alloc is not None:
                    s.nat_public_ip = u32_to_ip(alloc.public_ip)
                    s.nat_port_start = alloc.port_start
                    s.nat_port_end = alloc.port_end
            syncer.publish_add(s)
        else:
            syncer.publish_delete(f"dhcp-{lease.mac.hex()}")

    dhcp_server.on_lease_event.append(on_event)


def promote(dhcp_server: DHCPServer, syncer: HASyncer,
            qos_mgr=None, nat_mgr=None) -> int:
    """Standby -> active: the shadow session store becomes authoritative.
    Rebuild leases, the GPU fast-path table, QoS buckets and NAT blocks
    from the replicated state (ref SURVEY §3.5 failover call stack)."""
    syncer.promote()
    n = 0
    for s in syncer.store.all():
        if s.access_type != "dhcp" or not s.mac:
            continue
        lease = session_to_lease(s)
        dhcp_server.restore_lease(lease)
        if qos_mgr is not None and s.policy_name:
            qos_mgr.apply_policy(lease.ip, s.policy_name)
        if nat_mgr is not None:
            try:
                if s.nat_public_ip:   # exact-block restore (see attach)
                    nat_mgr.restore_nat(lease.ip, ip2u32(s.nat_public_ip),
                                        s.nat_port_start, s.nat_port_end)
                else:
                    nat_mgr.allocate_nat(lease.ip, s.subscriber_id)
            except Exception:
                pass
        n += 1
    return n
