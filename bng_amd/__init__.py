"""bng_amd — MI355X-native Broadband Network Gateway.

A from-scratch rebuild of the capabilities of codelaboratoryltd/bng
(eBPF/XDP BNG for ISP edge) as an AMD MI355X-native framework:

  * dataplane/  — hand-written CDNA4 HIP kernels (DHCP fast path, NAT44,
                  QoS token bucket, antispoof uRPF) over HBM-resident hash
                  tables, plus the host launcher (the pkg/ebpf analog) and
                  a CPU golden model.
  * parallel/   — subscriber sharding across GPUs: MAC hashring ownership,
                  packet-batch steering via RCCL all-to-all over xGMI.
  * everything else — the control plane: DHCP/DHCPv6/SLAAC/PPPoE servers,
                  RADIUS, NAT/QoS/antispoof managers, nexus/pool/allocator
                  distributed state, HA, resilience, metrics, audit, ...

Reference layer map and component inventory: SURVEY.md.
"""

__version__ = "0.1.0"
