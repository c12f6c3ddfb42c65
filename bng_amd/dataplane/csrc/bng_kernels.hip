/* bng_kernels.hip — CDNA4 (gfx950) BNG dataplane kernels.
 *
 * MI355X-native re-design of the reference eBPF dataplane:
 *   dhcp_fastpath_kernel  <- bpf/dhcp_fastpath.c:619-813 (XDP)
 *   nat44_egress_kernel   <- bpf/nat44.c:565-802 (TC egress SNAT)
 *   nat44_ingress_kernel  <- bpf/nat44.c:805-948 (TC ingress DNAT)
 *   qos_kernel            <- bpf/qos_ratelimit.c:126-222 (TC token bucket)
 *   antispoof_kernel      <- bpf/antispoof.c:189-293 (TC uRPF)
 *   uplink_pipeline_kernel — fused antispoof+NAT44+QoS+DHCP single pass
 *                            (the per-packet program chain the reference
 *                            runs as four separate TC/XDP hooks)
 *
 * Execution model: one thread per packet, grid-stride, 256-thread blocks
 * (4 waves).  Packet batches are fixed-stride slots in HBM; every lookup
 * table is an open-addressing HBM hash table (bng_abi.h).  Per-packet
 * ktime becomes one host-supplied batch timestamp.  Stats are wave-
 * aggregated before one device-scope atomic per wave (bng_device.h).
 *
 * Differential-tested against bng_amd/dataplane/golden.py on random
 * batches (tests/test_kernels_gpu.py).
 */
#include <hip/hip_runtime.h>
#include "bng_abi.h"
#include "bng_params.h"
#include "bng_device.h"

/* ==================================================== table primitives */

typedef __attribute__((address_space(1))) uint64_t gu64_t;

BNG_DEV uint64_t rlx_load64(const void* ptr) {
  return __hip_atomic_load((const uint64_t*)ptr, __ATOMIC_RELAXED,
                           __HIP_MEMORY_SCOPE_AGENT);
}
BNG_DEV void rlx_store64(void* ptr, uint64_t v) {
  __hip_atomic_store((uint64_t*)ptr, v, __ATOMIC_RELAXED,
                     __HIP_MEMORY_SCOPE_AGENT);
}

/* Table-probe load: 16 B of a hash-table entry.  BNG_SC1_PROBES=1 issues
 * it as two relaxed agent-scope (sc1) loads — L2-served, bypassing L1 —
 * because random probes have ~0 L1 hit rate and only evict the packet
 * lines the parser needs (same reasoning as the session hit path). */
#ifndef BNG_SC1_PROBES
#define BNG_SC1_PROBES 1
#endif
BNG_DEV uint4 ld_probe16(const void* p) {
#if BNG_SC1_PROBES
  const uint64_t* q = (const uint64_t*)p;
  uint64_t a = rlx_load64(q), b = rlx_load64(q + 1);
  uint4 v;
  v.x = (uint32_t)a; v.y = (uint32_t)(a >> 32);
  v.z = (uint32_t)b; v.w = (uint32_t)(b >> 32);
  return v;
#else
  return *(const uint4*)p;
#endif
}

BNG_DEV const bng_sub_entry* sub_lookup(const bng_sub_entry* t, uint32_t mask,
                                        uint64_t key) {
  uint32_t slot = (uint32_t)bng_mix64(key) & mask;
  for (int i = 0; i < BNG_MAX_PROBE; ++i) {
    const bng_sub_entry* e = &t[(slot + i) & mask];
    /* one 16-B vector load covers key+pool_id+allocated_ip (entry is
     * 32-B aligned); most probes end after this single transaction */
    uint4 v = ld_probe16(e);
    uint64_t k = ((uint64_t)v.y << 32) | v.x;
    if (k == key) return e;
    if (k == BNG_KEY_EMPTY) return nullptr;
  }
  return nullptr;
}

/* Merged subscriber-context probe: ONE table walk serves both the NAT
 * port-block and ingress-QoS stages (the reference walks subscriber_nat
 * nat44.c:157-164 AND qos_ingress qos_ratelimit.c:44-50 separately —
 * two random HBM touches per packet; this is one). */
BNG_DEV bng_subctx* subctx_lookup_hint(bng_subctx* t,
                                       uint32_t mask, uint32_t ip,
                                       uint32_t slot, uint4 first) {
  if (first.x == ip) return &t[slot];
  if (first.x == 0) return nullptr;
  for (int i = 1; i < BNG_MAX_PROBE; ++i) {
    bng_subctx* e = &t[(slot + i) & mask];
    uint4 v = ld_probe16(e);
    if (v.x == ip) return e;
    if (v.x == 0) return nullptr;
  }
  return nullptr;
}

BNG_DEV bng_subctx* subctx_lookup(bng_subctx* t, uint32_t mask,
                                  uint32_t ip) {
  if (ip == 0) return nullptr;
  uint32_t slot = (uint32_t)bng_mix64(ip) & mask;
  for (int i = 0; i < BNG_MAX_PROBE; ++i) {
    bng_subctx* e = &t[(slot + i) & mask];
    uint4 v = ld_probe16(e);    /* key + pub_ip + ports + valid flags */
    if (v.x == ip) return e;
    if (v.x == 0) return nullptr;
  }
  return nullptr;
}

BNG_DEV bng_qos_bucket* qos_lookup(bng_qos_bucket* t, uint32_t mask,
                                   uint32_t ip, uint64_t* rate_out) {
  if (ip == 0) return nullptr;
  uint32_t slot = (uint32_t)bng_mix64(ip) & mask;
  for (int i = 0; i < BNG_MAX_PROBE; ++i) {
    bng_qos_bucket* e = &t[(slot + i) & mask];
    uint4 v = ld_probe16(e);    /* key_ip+valid+prio | rate in 16B */
    if (v.x == ip && (v.y & 0xFF)) {
      *rate_out = ((uint64_t)v.w << 32) | v.z;
      return e;
    }
    if (v.x == 0) return nullptr;
  }
  return nullptr;
}

BNG_DEV const bng_binding_entry* binding_lookup_hint(
    const bng_binding_entry* t, uint32_t mask, uint64_t mac,
    uint32_t slot, uint4 first) {
  uint64_t k = ((uint64_t)first.y << 32) | first.x;
  if (k == mac) return &t[slot];
  if (k == BNG_KEY_EMPTY) return nullptr;
  for (int i = 1; i < BNG_MAX_PROBE; ++i) {
    const bng_binding_entry* e = &t[(slot + i) & mask];
    uint4 v = ld_probe16(e);
    k = ((uint64_t)v.y << 32) | v.x;
    if (k == mac) return e;
    if (k == BNG_KEY_EMPTY) return nullptr;
  }
  return nullptr;
}

BNG_DEV const bng_binding_entry* binding_lookup(const bng_binding_entry* t,
                                                uint32_t mask, uint64_t mac) {
  if (mac == 0) return nullptr;
  uint32_t slot = (uint32_t)bng_mix64(mac) & mask;
  for (int i = 0; i < BNG_MAX_PROBE; ++i) {
    const bng_binding_entry* e = &t[(slot + i) & mask];
    uint4 v = ld_probe16(e);    /* key + ipv4 + flags in one load */
    uint64_t k = ((uint64_t)v.y << 32) | v.x;
    if (k == mac) return e;
    if (k == BNG_KEY_EMPTY) return nullptr;
  }
  return nullptr;
}

/* find-or-claim for device-created entries (sessions / EIM / reverse).
 * Returns entry and sets *claimed when this thread won the slot; on a
 * sig match the caller must bng_wait_ready() before trusting fields.
 * Tombstoned slots (sweep-expired sessions) are RECLAIMED: the probe
 * remembers the first tombstone and claims it once the full chain
 * rules out an existing entry — without this, long-uptime churn fills
 * the chain with tombstones (the reference gets reuse for free from
 * BPF LRU maps). */
/* Post-claim duplicate check (round-1 advisor finding): if a chain slot
 * is tombstoned concurrently (in-batch port-exhaustion release), one
 * lane can reclaim that earlier tombstone while another lane of the
 * SAME flow claims a later slot — leaking a duplicate session + NAT
 * port until sweep.  After winning a claim CAS, rescan the chain for a
 * concurrent same-sig insert and keep only the EARLIEST-position claim:
 * the later claimant re-tombstones its slot and adopts the earlier one.
 * The positional tie-break is what makes this safe for lockstep lanes —
 * a symmetric release-on-any-hit rule would make two same-wave lanes
 * release each other every retry.  Residual window: a later claimant
 * whose rescan completed before our CAS landed keeps a shadowed
 * duplicate; that reverts to the pre-fix behavior (sweep reclaims it,
 * EIM collision check keeps the port safe) in a window now measured in
 * the tens of nanoseconds instead of the whole create race. */
template <typename E>
BNG_DEV E* claim_dedup(E* t, uint32_t mask, uint64_t sig, uint32_t slot,
                       int my_pos, bool* claimed, bool* found) {
  for (int i = 0; i < BNG_MAX_PROBE; ++i) {
    if (i == my_pos) continue;
    E* e = &t[(slot + i) & mask];
    uint64_t k = __hip_atomic_load(&e->sig, __ATOMIC_RELAXED,
                                   __HIP_MEMORY_SCOPE_AGENT);
    if (k == sig && i < my_pos) {
      /* earlier claim wins: release ours, adopt theirs (caller does
       * bng_wait_ready before trusting fields) */
      __hip_atomic_store(&t[(slot + my_pos) & mask].sig,
                         BNG_KEY_TOMBSTONE, __ATOMIC_RELAXED,
                         __HIP_MEMORY_SCOPE_AGENT);
      *claimed = false; *found = true;
      return e;
    }
    if (k == BNG_KEY_EMPTY) break;
  }
  *claimed = true;
  return &t[(slot + my_pos) & mask];
}

template <typename E>
BNG_DEV E* sig_find_or_claim(E* t, uint32_t mask, uint64_t sig,
                             bool* claimed, bool* found) {
  *claimed = false; *found = false;
  uint32_t slot = (uint32_t)sig & mask;
  /* pass 1 — pure hit scan, identical cost to the pre-reclamation hot
   * path (hits dominate: established flows) */
  for (int i = 0; i < BNG_MAX_PROBE; ++i) {
    E* e = &t[(slot + i) & mask];
    uint64_t k = __hip_atomic_load(&e->sig, __ATOMIC_RELAXED,
                                   __HIP_MEMORY_SCOPE_AGENT);
    if (k == sig) { *found = true; return e; }
    if (k == BNG_KEY_EMPTY) break;
  }
  /* pass 2 — claim scan (session-create path only) */
  for (int attempt = 0; attempt < 4; ++attempt) {
    int first_tomb = -1, first_tomb_pos = -1;
    for (int i = 0; i < BNG_MAX_PROBE; ++i) {
      E* e = &t[(slot + i) & mask];
      uint64_t k = __hip_atomic_load(&e->sig, __ATOMIC_RELAXED,
                                     __HIP_MEMORY_SCOPE_AGENT);
      if (k == sig) { *found = true; return e; }
      if (k == BNG_KEY_TOMBSTONE) {
        if (first_tomb < 0) {
          first_tomb = (int)((slot + i) & mask);
          first_tomb_pos = i;
        }
        continue;
      }
      if (k == BNG_KEY_EMPTY) {
        E* d = first_tomb >= 0 ? &t[first_tomb] : e;
        int d_pos = first_tomb >= 0 ? first_tomb_pos : i;
        uint64_t want = first_tomb >= 0 ? BNG_KEY_TOMBSTONE
                                        : BNG_KEY_EMPTY;
        if (__hip_atomic_compare_exchange_strong(
                &d->sig, &want, sig, __ATOMIC_RELAXED, __ATOMIC_RELAXED,
                __HIP_MEMORY_SCOPE_AGENT)) {
          return claim_dedup(t, mask, sig, slot, d_pos, claimed, found);
        }
        if (want == sig) { *found = true; return d; }
        /* lost the slot to a DIFFERENT key: keep probing.  If it was
         * the remembered tombstone, forget it and retry this empty
         * slot; if it was the empty slot itself, scan past it. */
        if (first_tomb >= 0) { first_tomb = -1; first_tomb_pos = -1; --i; }
        continue;
      }
    }
    if (first_tomb < 0)
      return nullptr;            /* chain genuinely full of other keys */
    /* chain full but reclaimable: take the tombstone directly */
    E* d = &t[first_tomb];
    uint64_t want = BNG_KEY_TOMBSTONE;
    if (__hip_atomic_compare_exchange_strong(
            &d->sig, &want, sig, __ATOMIC_RELAXED, __ATOMIC_RELAXED,
            __HIP_MEMORY_SCOPE_AGENT)) {
      return claim_dedup(t, mask, sig, slot, first_tomb_pos, claimed,
                         found);
    }
    if (want == sig) { *found = true; return d; }
    /* raced: rescan */
  }
  return nullptr;
}

template <typename E>
BNG_DEV E* sig_lookup(E* t, uint32_t mask, uint64_t sig) {
  uint32_t slot = (uint32_t)sig & mask;
  for (int i = 0; i < BNG_MAX_PROBE; ++i) {
    E* e = &t[(slot + i) & mask];
    uint64_t k = __hip_atomic_load(&e->sig, __ATOMIC_RELAXED,
                                   __HIP_MEMORY_SCOPE_AGENT);
    if (k == sig) return e;
    if (k == BNG_KEY_EMPTY) return nullptr;
  }
  return nullptr;
}

BNG_DEV bool tuple_eq(const bng_nat_tuple& a, const bng_nat_tuple& b) {
  return a.src_ip == b.src_ip && a.dst_ip == b.dst_ip &&
         a.src_port == b.src_port && a.dst_port == b.dst_port &&
         a.protocol == b.protocol;
}

/* ===================================================== parsed context */

struct pktctx {
  uint8_t* p;        /* slot base */
  int len;
  int ip_off;        /* -1 if not IPv4 */
  int l4_off;
  uint8_t proto;
  uint32_t saddr, daddr;      /* big-endian byte order, as host ints */
  uint16_t sport, dport;      /* host ints of the BE fields */
  uint16_t icmp_id;
  uint8_t tcp_flags;
  /* vlan */
  int vlan_offset;
  uint16_t s_tag, c_tag;
  bool tagged;
  bool l4_ok;
  /* register window: bswapped words of bytes 0..47 when the fast parse
   * ran (untagged ihl=5).  TA (vector-memory address) occupancy is the
   * measured kernel bound — header rewrites happen on these registers
   * and flush as two dwordx4 stores instead of ~10 divergent byte ops */
  uint32_t W[12];
  bool regs;
};

/* constant-offset register accessors (offsets fold at compile time) */
#define RG_B(c, o) (((c).W[(o) >> 2] >> (24 - 8 * ((o) & 3))) & 0xFFu)
BNG_DEV uint16_t rg_ld16(const pktctx& c, int o) {
  return (uint16_t)((RG_B(c, o) << 8) | RG_B(c, o + 1));
}
BNG_DEV void rg_st8(pktctx& c, int o, uint32_t v) {
  int i = o >> 2, sh = 24 - 8 * (o & 3);
  c.W[i] = (c.W[i] & ~(0xFFu << sh)) | ((v & 0xFFu) << sh);
}
BNG_DEV void rg_st16(pktctx& c, int o, uint32_t v) {
  rg_st8(c, o, v >> 8); rg_st8(c, o + 1, v);
}
BNG_DEV void rg_st32(pktctx& c, int o, uint32_t v) {
  rg_st16(c, o, v >> 16); rg_st16(c, o + 2, v);
}
BNG_DEV void rg_flush16_47(pktctx& c) {
  uint4* q = (uint4*)c.p;
  q[1] = make_uint4(__builtin_bswap32(c.W[4]), __builtin_bswap32(c.W[5]),
                    __builtin_bswap32(c.W[6]), __builtin_bswap32(c.W[7]));
  q[2] = make_uint4(__builtin_bswap32(c.W[8]), __builtin_bswap32(c.W[9]),
                    __builtin_bswap32(c.W[10]),
                    __builtin_bswap32(c.W[11]));
}

/* Vectorized parse fast path: untagged IPv4 with ihl=5 (the 64B-mix hot
 * case).  Three dwordx4 loads replace ~30 dependent byte loads; fields
 * are extracted from registers with constant shifts.  Returns false to
 * fall back to the general byte parser (VLAN, options, short frames). */
BNG_DEV bool parse_pkt_fast(pktctx& c, uint8_t* p, int len) {
  if (len < 48 || ((uintptr_t)p & 15)) return false;
  /* no alias pointer to c.W: an escaping pointer would force the
   * register window into scratch memory */
  #pragma unroll
  for (int i = 0; i < 3; ++i) {
    uint4 v = ((const uint4*)p)[i];
    c.W[i * 4 + 0] = __builtin_bswap32(v.x);
    c.W[i * 4 + 1] = __builtin_bswap32(v.y);
    c.W[i * 4 + 2] = __builtin_bswap32(v.z);
    c.W[i * 4 + 3] = __builtin_bswap32(v.w);
  }
  #define BNG_B(o) ((c.W[(o) >> 2] >> (24 - 8 * ((o) & 3))) & 0xFFu)
  #define BNG_H(o) ((BNG_B(o) << 8) | BNG_B((o) + 1))
  uint32_t ethertype = BNG_H(12);
  if (ethertype != 0x0800) return false;        /* VLAN etc: slow path */
  if (BNG_B(14) != 0x45) return false;          /* options/ihl!=5 */
  c.p = p; c.len = len; c.vlan_offset = 0; c.s_tag = c.c_tag = 0;
  c.tagged = false; c.l4_ok = false;
  c.regs = true;
  c.ip_off = 14;
  c.proto = (uint8_t)BNG_B(23);
  c.saddr = (BNG_H(26) << 16) | BNG_H(28);
  c.daddr = (BNG_H(30) << 16) | BNG_H(32);
  c.l4_off = 34;
  if (c.proto == 6 && len >= 54) {
    c.sport = (uint16_t)BNG_H(34);
    c.dport = (uint16_t)BNG_H(36);
    c.tcp_flags = (uint8_t)BNG_B(47);
    c.l4_ok = true;
  } else if (c.proto == 17 && len >= 42) {
    c.sport = (uint16_t)BNG_H(34);
    c.dport = (uint16_t)BNG_H(36);
    c.l4_ok = true;
  } else if (c.proto == 1 && len >= 42) {
    c.icmp_id = (uint16_t)BNG_H(38);
    c.l4_ok = true;
  }
  #undef BNG_B
  #undef BNG_H
  return true;
}

/* Parse Ethernet [+VLAN/QinQ] + IPv4 + L4 ports.  NAT/QoS/antispoof paths
 * in the reference parse untagged frames only (nat44.c:573-581); the DHCP
 * path handles tags (dhcp_fastpath.c:352-428).  want_vlan selects. */
BNG_DEV bool parse_pkt_slow(pktctx& c, uint8_t* p, int len, bool want_vlan) {
  c.p = p; c.len = len; c.ip_off = -1; c.vlan_offset = 0;
  c.s_tag = c.c_tag = 0; c.tagged = false; c.l4_ok = false;
  c.regs = false;
  if (len < 14) return false;
  uint16_t proto = ld_u16be(p + 12);
  int off = 14;
  if (want_vlan && (proto == 0x8100 || proto == 0x88A8)) {
    if (len < off + 4) return false;
    c.tagged = true;
    c.s_tag = ld_u16be(p + off) & 0xFFF;
    proto = ld_u16be(p + off + 2);
    off += 4; c.vlan_offset = 4;
    if (proto == 0x8100) {
      if (len < off + 4) return false;
      c.c_tag = ld_u16be(p + off) & 0xFFF;
      proto = ld_u16be(p + off + 2);
      off += 4; c.vlan_offset = 8;
    }
  }
  if (proto != 0x0800 || len < off + 20) return false;
  c.ip_off = off;
  c.proto = p[off + 9];
  c.saddr = ld_u32be(p + off + 12);
  c.daddr = ld_u32be(p + off + 16);
  int ihl = (p[off] & 0xF) * 4;
  c.l4_off = off + ihl;
  if (c.proto == 6 && len >= c.l4_off + 20) {
    c.sport = ld_u16be(p + c.l4_off);
    c.dport = ld_u16be(p + c.l4_off + 2);
    c.tcp_flags = p[c.l4_off + 13];
    c.l4_ok = true;
  } else if (c.proto == 17 && len >= c.l4_off + 8) {
    c.sport = ld_u16be(p + c.l4_off);
    c.dport = ld_u16be(p + c.l4_off + 2);
    c.l4_ok = true;
  } else if (c.proto == 1 && len >= c.l4_off + 8) {
    c.icmp_id = ld_u16be(p + c.l4_off + 4);
    c.l4_ok = true;
  }
  return true;
}

BNG_DEV bool parse_pkt(pktctx& c, uint8_t* p, int len, bool want_vlan) {
  if (parse_pkt_fast(c, p, len)) return true;
  return parse_pkt_slow(c, p, len, want_vlan);
}

/* ================================================== DHCP fast path K1 */

struct dhcp_tables {
  const bng_sub_entry* subs; uint32_t sub_mask;
  const bng_ip_pool* pools;  uint32_t n_pools;
  const bng_server_config* cfg;
  unsigned long long* stats;
  uint64_t now_sec;
};

/* stats flags gathered per packet, ballot-aggregated by the caller */
struct dhcp_flags {
  bool vlan, total, hit, miss, expired, error, o82, bcast, ucast;
  BNG_DEV void clear() {
    vlan = total = hit = miss = expired = error = o82 = bcast = ucast = false;
  }
};

/* Scan DHCP options for msg-type (53) and option-82 circuit-id.  Full TLV
 * scan — the GPU has no verifier, so we upgrade the reference's
 * fixed-offset workaround (dhcp_fastpath.c:216-323) while keeping its
 * bounds (64 options / 312 bytes, maps.h:19-22). */
BNG_DEV void scan_dhcp_options(const uint8_t* p, int opt_off, int end,
                               uint8_t* msg_type, const uint8_t** cid,
                               int* cid_len) {
  *msg_type = 0; *cid = nullptr; *cid_len = 0;
  int i = opt_off;
  int limit = min(end, opt_off + 312);
  for (int iters = 0; iters < 64 && i < limit; ++iters) {
    uint8_t code = p[i];
    if (code == 0) { ++i; continue; }
    if (code == 255) break;
    if (i + 1 >= limit) break;
    uint8_t ln = p[i + 1];
    if (i + 2 + ln > limit) break;
    if (code == 53 && ln == 1) *msg_type = p[i + 2];
    else if (code == 82) {
      int j = i + 2, sub_end = i + 2 + ln;
      while (j + 2 <= sub_end) {
        uint8_t sc = p[j], sl = p[j + 1];
        if (j + 2 + sl > sub_end) break;
        if (sc == 1 && sl > 0 && sl <= 32) { *cid = p + j + 2; *cid_len = sl; }
        j += 2 + sl;
      }
    }
    i += 2 + ln;
  }
}

/* Per-packet DHCP fast path.  Returns verdict; *out_len set on TX. */
BNG_DEV int dhcp_process(uint8_t* p, int len, int stride,
                         const dhcp_tables& T, dhcp_flags& F,
                         uint16_t* out_len) {
  *out_len = (uint16_t)len;
  pktctx c;
  if (!parse_pkt(c, p, len, /*want_vlan=*/true)) {
    if (c.tagged) F.vlan = true;
    return BNG_PASS;
  }
  if (c.tagged) F.vlan = true;
  if (c.ip_off < 0 || c.proto != 17 || !c.l4_ok) return BNG_PASS;
  if (c.dport != 67) return BNG_PASS;
  int dhcp_off = c.l4_off + 8;
  if (len < dhcp_off + 240) return BNG_PASS;
  if (p[dhcp_off] != 1) return BNG_PASS;               /* BOOTREQUEST */
  if (ld_u32be(p + dhcp_off + 236) != 0x63825363u) return BNG_PASS;

  F.total = true;

  uint8_t msg_type; const uint8_t* cid; int cid_len;
  scan_dhcp_options(p, dhcp_off + 240, len, &msg_type, &cid, &cid_len);
  if (msg_type != 1 && msg_type != 3) { F.miss = true; return BNG_PASS; }

  /* 3-way lookup: VLAN -> circuit-id -> MAC (ref :647-687) */
  const bng_sub_entry* sub = nullptr;
  if (c.tagged) {
    uint64_t vk = BNG_KEY_VLAN | ((uint64_t)c.s_tag << 16) | c.c_tag;
    sub = sub_lookup(T.subs, T.sub_mask, vk);
  }
  if (!sub && cid) {
    uint8_t padded[32];
    #pragma unroll
    for (int i = 0; i < 32; ++i) padded[i] = (i < cid_len) ? cid[i] : 0;
    uint64_t ck = BNG_KEY_CIRCUIT | (bng_fnv1a64(padded, 32) >> 2);
    sub = sub_lookup(T.subs, T.sub_mask, ck);
    if (sub) F.o82 = true;
  }
  if (!sub) {
    uint64_t mac = 0;
    #pragma unroll
    for (int i = 0; i < 6; ++i) mac = (mac << 8) | p[dhcp_off + 28 + i];
    sub = sub_lookup(T.subs, T.sub_mask, mac);
  }
  if (!sub) { F.miss = true; return BNG_PASS; }
  if (T.now_sec > sub->lease_expiry) { F.expired = true; return BNG_PASS; }
  if (sub->pool_id >= T.n_pools) { F.error = true; return BNG_PASS; }
  const bng_ip_pool pool = T.pools[sub->pool_id];
  if (!pool.valid) { F.error = true; return BNG_PASS; }
  F.hit = true;

  const bng_server_config cfg = *T.cfg;
  uint8_t reply_type = (msg_type == 1) ? 2 : 5;     /* OFFER : ACK */
  uint32_t giaddr = ld_u32be(p + dhcp_off + 24);
  /* table IPs are stored as host ints equal to their wire (big-endian)
   * interpretation — the same values ip2u32()/ld_u32be() produce */
  uint32_t server_ip = cfg.server_ip ? cfg.server_ip : pool.gateway;

  int ip_off = c.ip_off, udp_off = c.l4_off;
  if (giaddr != 0) {   /* relayed: unicast to relay agent (ref :726-743) */
    #pragma unroll
    for (int i = 0; i < 6; ++i) p[i] = p[6 + i];
    #pragma unroll
    for (int i = 0; i < 6; ++i) p[6 + i] = cfg.server_mac[i];
    st_u32be(p + ip_off + 12, server_ip);
    st_u32be(p + ip_off + 16, giaddr);
    st_u16be(p + udp_off, 67); st_u16be(p + udp_off + 2, 67);
    F.ucast = true;
  } else {
    uint16_t flags = ld_u16be(p + dhcp_off + 10);
    uint32_t ciaddr = ld_u32be(p + dhcp_off + 12);
    bool use_bcast = (flags & 0x8000) || ciaddr == 0;
    if (use_bcast) {
      #pragma unroll
      for (int i = 0; i < 6; ++i) p[i] = 0xFF;
      F.bcast = true;
    } else {
      #pragma unroll
      for (int i = 0; i < 6; ++i) p[i] = p[dhcp_off + 28 + i];
      F.ucast = true;
    }
    #pragma unroll
    for (int i = 0; i < 6; ++i) p[6 + i] = cfg.server_mac[i];
    st_u32be(p + ip_off + 12, server_ip);
    st_u32be(p + ip_off + 16, 0xFFFFFFFFu);
    st_u16be(p + udp_off, 67); st_u16be(p + udp_off + 2, 68);
  }
  p[ip_off + 8] = 64;                 /* TTL */
  st_u16be(p + udp_off + 6, 0);       /* UDP csum 0 (ref :741,755) */

  p[dhcp_off] = 2;                    /* BOOTREPLY */
  p[dhcp_off + 3] = 0;                /* hops */
  st_u32be(p + dhcp_off + 16, sub->allocated_ip);
  st_u32be(p + dhcp_off + 20, server_ip);
  bng_zero(p + dhcp_off + 44, 192);   /* sname + file (ref :765-766) */

  /* options 53/54/51/1/3/6/58/59/255 (ref build_dhcp_options :519-602) */
  uint8_t* o = p + dhcp_off + 240;
  int w = 0;
  o[w++] = 53; o[w++] = 1; o[w++] = reply_type;
  o[w++] = 54; o[w++] = 4; st_u32be(o + w, server_ip); w += 4;
  o[w++] = 51; o[w++] = 4; st_u32be(o + w, pool.lease_time); w += 4;
  uint32_t mask32 = (pool.prefix_len == 0) ? 0
      : (pool.prefix_len >= 32) ? 0xFFFFFFFFu
      : (0xFFFFFFFFu << (32 - pool.prefix_len));
  o[w++] = 1;  o[w++] = 4; st_u32be(o + w, mask32); w += 4;
  o[w++] = 3;  o[w++] = 4; st_u32be(o + w, pool.gateway); w += 4;
  if (pool.dns_primary) {
    uint8_t dl = pool.dns_secondary ? 8 : 4;
    o[w++] = 6; o[w++] = dl;
    st_u32be(o + w, pool.dns_primary); w += 4;
    if (pool.dns_secondary) { st_u32be(o + w, pool.dns_secondary); w += 4; }
  }
  o[w++] = 58; o[w++] = 4; st_u32be(o + w, pool.lease_time / 2); w += 4;
  o[w++] = 59; o[w++] = 4; st_u32be(o + w, (pool.lease_time * 7) / 8); w += 4;
  o[w++] = 255;

  int dhcp_len = 240 + w;
  int udp_len = 8 + dhcp_len;
  int ip_len = 20 + udp_len;
  int total = 14 + c.vlan_offset + ip_len;
  st_u16be(p + ip_off + 2, (uint16_t)ip_len);
  st_u16be(p + udp_off + 4, (uint16_t)udp_len);
  st_u16be(p + ip_off + 10, 0);
  /* 10x u16 sum checksum (ref ip_checksum :488-503) */
  uint32_t s = 0;
  #pragma unroll
  for (int i = 0; i < 10; ++i) s += ld_u16be(p + ip_off + 2 * i);
  s = (s & 0xFFFF) + (s >> 16); s = (s & 0xFFFF) + (s >> 16);
  st_u16be(p + ip_off + 10, (uint16_t)~s);

  *out_len = (uint16_t)min(total, stride);
  return BNG_TX;
}

BNG_DEV void dhcp_commit_stats(const dhcp_flags& F, unsigned long long* st) {
  stat_inc(&st[BNG_ST_VLAN_PACKETS], F.vlan);
  stat_inc(&st[BNG_ST_TOTAL_REQUESTS], F.total);
  stat_inc(&st[BNG_ST_FASTPATH_HITS], F.hit);
  stat_inc(&st[BNG_ST_FASTPATH_MISSES], F.miss);
  stat_inc(&st[BNG_ST_CACHE_EXPIRED], F.expired);
  stat_inc(&st[BNG_ST_ERRORS], F.error);
  stat_inc(&st[BNG_ST_OPTION82_PRESENT], F.o82);
  stat_inc(&st[BNG_ST_BROADCAST_REPLIES], F.bcast);
  stat_inc(&st[BNG_ST_UNICAST_REPLIES], F.ucast);
}

__global__ void dhcp_fastpath_kernel(
    uint8_t* __restrict__ data, const uint16_t* __restrict__ in_len,
    uint16_t* __restrict__ out_len, uint8_t* __restrict__ verdict,
    int n, int stride,
    const bng_sub_entry* __restrict__ subs, uint32_t sub_mask,
    const bng_ip_pool* __restrict__ pools, uint32_t n_pools,
    const bng_server_config* __restrict__ cfg,
    unsigned long long* __restrict__ stats, uint64_t now_sec,
    const uint64_t* __restrict__ now_ptr) {
  if (now_ptr) now_sec = now_ptr[1];
  dhcp_tables T{subs, sub_mask, pools, n_pools, cfg, stats, now_sec};
  int tid = blockIdx.x * blockDim.x + threadIdx.x;
  int nthreads = gridDim.x * blockDim.x;
  for (int base = 0; base < n; base += nthreads) {
    int pid = base + tid;
    dhcp_flags F; F.clear();
    if (pid < n) {
      uint16_t ol = in_len[pid];
      int v = dhcp_process(data + (size_t)pid * stride, in_len[pid], stride,
                           T, F, &ol);
      verdict[pid] = (uint8_t)v;
      out_len[pid] = ol;
    }
    dhcp_commit_stats(F, stats);
  }
}

/* ------------------------------------------- persistent DHCP service.
 * One workgroup (256 threads, 1 CU of 256) stays resident and polls a
 * pinned-host doorbell.  Because its waves already occupy their CU, a
 * saturating 1M-packet data flood cannot starve it the way it starves
 * a freshly launched kernel (measured r1: flood p99 453 us vs <70 us
 * quiesced; a priority stream did not help because the launch itself
 * must wait for CU space).  Request path: host writes packets into the
 * pinned ring and bumps head; the kernel stages the batch to an HBM
 * scratch with wide coalesced PCIe reads, runs dhcp_process, copies
 * replies back, and release-stores tail.  Self-exits after
 * idle_exit_k*1024 empty polls so a crashed host can never wedge the
 * box. */
/* Multi-block service: each block owns a fixed slice of the batch and
 * polls the doorbell independently (no grid sync).  Spreading the
 * slices over N CUs multiplies the outstanding PCIe reads during
 * staging (the single-block form measured latency-bound at ~68 us for
 * a 256-pkt batch) and divides the per-CU work a saturating data flood
 * contends with.  Completion: the last block to finish a batch
 * (device-scope acq_rel counter) publishes tail; the last block to
 * exit publishes the exit ack. */
__global__ void __launch_bounds__(256, 1) dhcp_service_kernel(
    bng_svc_ctrl* __restrict__ ctrl,
    uint8_t* __restrict__ req,            /* pinned [n_slots, stride]  */
    const uint16_t* __restrict__ in_len,  /* pinned [n_slots]          */
    uint16_t* __restrict__ out_len,       /* pinned [n_slots]          */
    uint8_t* __restrict__ verdict,        /* pinned [n_slots]          */
    uint8_t* __restrict__ scratch,        /* device [n_slots, stride]  */
    int n_slots,
    uint32_t* __restrict__ ctrs,          /* device {done, served_lo,
                                             served_hi, exits}         */
    const bng_sub_entry* __restrict__ subs, uint32_t sub_mask,
    const bng_ip_pool* __restrict__ pools, uint32_t n_pools,
    const bng_server_config* __restrict__ cfg,
    unsigned long long* __restrict__ stats) {
  __shared__ uint32_t s_head, s_run, s_n;
  __shared__ uint64_t s_now;
  const int tid = threadIdx.x;
  const int nb = (int)gridDim.x, blk = (int)blockIdx.x;
  uint32_t done = __hip_atomic_load(&ctrl->tail, __ATOMIC_RELAXED,
                                    __HIP_MEMORY_SCOPE_SYSTEM);
  const int stride = (int)ctrl->stride;
  uint64_t idle = 0;
  const uint64_t idle_max = (uint64_t)ctrl->idle_exit_k * 1024u;
  uint64_t batches = 0;
  unsigned long long* served64 = (unsigned long long*)(ctrs + 4);
  for (;;) {
    if (tid == 0) {
      s_head = __hip_atomic_load(&ctrl->head, __ATOMIC_ACQUIRE,
                                 __HIP_MEMORY_SCOPE_SYSTEM);
      s_run = __hip_atomic_load(&ctrl->run, __ATOMIC_RELAXED,
                                __HIP_MEMORY_SCOPE_SYSTEM);
      s_n = ctrl->n_pkts;
      s_now = ctrl->now_sec;
    }
    __syncthreads();
    bool work = (s_head != done);
    if (!s_run || (!work && idle >= idle_max)) break;
    if (!work) {
      ++idle;
      __builtin_amdgcn_s_sleep(8);
      __syncthreads();           /* re-converge before the next poll */
      continue;
    }
    idle = 0;
    int n = (int)s_n;
    if (n > n_slots) n = n_slots;
    const int lo = (int)((long)n * blk / nb);
    const int hi = (int)((long)n * (blk + 1) / nb);
    /* stage this block's slice to HBM: wide coalesced PCIe reads */
    const size_t w0 = (size_t)lo * stride / 16;
    const int words = (hi - lo) * stride / 16;
    const uint4* src = (const uint4*)req + w0;
    uint4* dst = (uint4*)scratch + w0;
    for (int i = tid; i < words; i += blockDim.x) dst[i] = src[i];
    __syncthreads();
    dhcp_tables T{subs, sub_mask, pools, n_pools, cfg, stats, s_now};
    for (int pid = lo + tid; pid < hi; pid += blockDim.x) {
      dhcp_flags F; F.clear();
      uint16_t ol = in_len[pid];
      int v = dhcp_process(scratch + (size_t)pid * stride, in_len[pid],
                           stride, T, F, &ol);
      verdict[pid] = (uint8_t)v;
      out_len[pid] = ol;
      dhcp_commit_stats(F, stats);
    }
    __syncthreads();
    /* replies back to the pinned ring (posted PCIe writes) */
    for (int i = tid; i < words; i += blockDim.x)
      ((uint4*)req)[w0 + i] = dst[i];
    __syncthreads();
    ++done; ++batches;
    if (tid == 0) {
      atomicAdd(served64, (unsigned long long)(hi - lo));
      /* acq_rel: the winner observes every block's served add */
      uint32_t prev = __hip_atomic_fetch_add(&ctrs[0], 1u,
                                             __ATOMIC_ACQ_REL,
                                             __HIP_MEMORY_SCOPE_AGENT);
      if ((int)(prev % (uint32_t)nb) == nb - 1) {   /* last finisher */
        ctrl->served = *served64;
        ctrl->batches = batches;
        __hip_atomic_store(&ctrl->tail, done, __ATOMIC_RELEASE,
                           __HIP_MEMORY_SCOPE_SYSTEM);
      }
    }
    __syncthreads();
  }
  if (tid == 0) {
    uint32_t prev = __hip_atomic_fetch_add(&ctrs[1], 1u, __ATOMIC_ACQ_REL,
                                           __HIP_MEMORY_SCOPE_AGENT);
    if ((int)prev == nb - 1) {
      ctrl->served = *served64;
      /* exit ack: the host's stop() waits on this instead of a stream
       * sync, so a dead doorbell can never block the control plane */
      __hip_atomic_store(&ctrl->exited, 1u, __ATOMIC_RELEASE,
                         __HIP_MEMORY_SCOPE_SYSTEM);
    }
  }
}

/* ========================================================== NAT44  K2 */

struct nat_tables {
  bng_nat_session* sessions; uint32_t sess_mask;
  bng_nat_reverse* reverse;  uint32_t rev_mask;
  bng_eim_entry* eim;        uint32_t eim_mask;
  bng_subctx* ctx;           uint32_t ctx_mask;
  const bng_nat_config* cfg;
  const uint32_t* hairpin_ips; uint32_t n_hairpin;
  unsigned long long* stats;
  bng_nat_log_entry* log_ring; bng_ring_header* log_hdr;
  uint64_t now_ns;
};

struct nat_flags {
  bool snat, dnat, hairpin, dropped, passed, sess_created, sess_expired,
       port_exh, eim_hit, eim_miss, alg;
  BNG_DEV void clear() {
    snat = dnat = hairpin = dropped = passed = sess_created = sess_expired =
        port_exh = eim_hit = eim_miss = alg = false;
  }
};

/* membership over sorted disjoint intervals: binary search, O(log n)
 * (the LPM-trie equivalent for a membership query) */
BNG_DEV bool ip_in_intervals(const uint32_t* lo, const uint32_t* hi,
                             uint32_t n, uint32_t ip) {
  uint32_t l = 0, r = n;
  while (l < r) {
    uint32_t m = (l + r) >> 1;
    if (ip < lo[m]) r = m;
    else if (ip > hi[m]) l = m + 1;
    else return true;
  }
  return false;
}

BNG_DEV bool nat_is_private(const bng_nat_config* cfg, uint32_t ip) {
  return ip_in_intervals(cfg->priv_lo, cfg->priv_hi,
                         cfg->n_private_ranges, ip);
}

BNG_DEV bool nat_is_hairpin(const nat_tables& T, uint32_t ip) {
  for (uint32_t i = 0; i < T.n_hairpin; ++i)
    if (T.hairpin_ips[i] == ip) return true;
  return false;
}

BNG_DEV bool nat_is_alg(const bng_nat_config* cfg, uint16_t port,
                        uint8_t proto) {
  uint32_t key = ((uint32_t)port << 16) | proto;
  for (uint32_t i = 0; i < cfg->n_alg_ports; ++i)
    if (cfg->alg_key[i] == key) return true;
  return false;
}

BNG_DEV void nat_log_push(const nat_tables& T, uint32_t ev, uint32_t sub_id,
                          uint32_t priv_ip, uint32_t pub_ip,
                          uint16_t priv_port, uint16_t pub_port,
                          uint32_t dest_ip, uint16_t dest_port,
                          uint8_t proto, uint8_t flags) {
  if (!T.log_ring) return;
  uint32_t idx = ring_claim(T.log_hdr);
  bng_nat_log_entry* e = &T.log_ring[idx & ((1u << BNG_LOG_RING_LOG2) - 1)];
  e->timestamp = T.now_ns; e->event_type = ev; e->subscriber_id = sub_id;
  e->private_ip = priv_ip; e->public_ip = pub_ip;
  e->private_port = priv_port; e->public_port = pub_port;
  e->dest_ip = dest_ip; e->dest_port = dest_port;
  e->protocol = proto; e->flags = flags;
}

/* Port rotor (ref allocate_port_from_block nat44.c:408-466): atomic
 * next_port bump, wrap, optional RTP parity, EIM-collision heuristic.
 * The reference's wrap is a tolerated benign race; ours keeps it. */
BNG_DEV uint16_t nat_alloc_port(const nat_tables& T, bng_subctx* blk,
                                bool parity, uint16_t orig_port,
                                uint32_t internal_ip, uint8_t proto) {
  uint8_t orig_parity = orig_port & 1;
  for (int i = 0; i < 64; ++i) {
    uint32_t r = atomicAdd(&blk->next_port, 1u);
    uint16_t port = (uint16_t)r;
    if (port > blk->port_end) port = blk->port_start;
    if (r + 1 > blk->port_end)
      atomicCAS(&blk->next_port, r + 1, (uint32_t)blk->port_start);
    if (parity && (port & 1) != orig_parity) continue;
    uint64_t esig = bng_eim_sig(internal_ip, port, proto);
    bng_eim_entry* ex = sig_lookup(T.eim, T.eim_mask, esig);
    if (ex) {
      /* verify it really is this (ip,port,proto) */
      if (bng_wait_ready(&ex->ready, 4096) && ex->internal_ip == internal_ip
          && ex->internal_port == port && ex->protocol == proto)
        continue;
    }
    return port;
  }
  return 0;
}

/* incremental checksum helpers (ref update_csum nat44.c:378-398) */
BNG_DEV uint16_t csum_upd32(uint16_t csum, uint32_t oldv, uint32_t newv) {
  uint32_t s = (~csum) & 0xFFFF;
  s += (~oldv & 0xFFFF) + ((~oldv >> 16) & 0xFFFF);
  s += (newv & 0xFFFF) + (newv >> 16);
  s = (s & 0xFFFF) + (s >> 16);
  s = (s & 0xFFFF) + (s >> 16);
  return (uint16_t)~s;
}
BNG_DEV uint16_t csum_upd16(uint16_t csum, uint16_t oldv, uint16_t newv) {
  uint32_t s = (~csum) & 0xFFFF;
  s += (~oldv & 0xFFFF) + newv;
  s = (s & 0xFFFF) + (s >> 16);
  s = (s & 0xFFFF) + (s >> 16);
  return (uint16_t)~s;
}

/* SNAT (ref nat44_egress nat44.c:565-802); blk_pre/blk_pre_valid let the
 * fused pipeline supply an already-probed port block */
BNG_DEV int nat_egress_process(pktctx& c, const nat_tables& T, nat_flags& F,
                               bng_subctx* blk_pre = nullptr,
                               bool blk_pre_valid = false) {
  uint8_t* p = c.p;
  if (c.ip_off < 0) return BNG_FWD;
  const bng_nat_config* cfg = T.cfg;
  if (!nat_is_private(cfg, c.saddr)) return BNG_FWD;
  bng_subctx* blk = blk_pre_valid ? blk_pre
      : subctx_lookup(T.ctx, T.ctx_mask, c.saddr);
  if (!blk || !blk->nat_valid) { F.passed = true; return BNG_PASS; }

  uint16_t sport, dport;
  if (c.proto == 6) {
    if (!c.l4_ok) return BNG_FWD;
    sport = c.sport; dport = c.dport;
    if (cfg->flags & (BNG_NAT_FLAG_ALG_FTP | BNG_NAT_FLAG_ALG_SIP)) {
      if (nat_is_alg(cfg, dport, 6)) {
        F.alg = true;
        nat_log_push(T, BNG_LOG_ALG_TRIGGER, blk->subscriber_id, c.saddr, 0,
                     sport, 0, c.daddr, dport, 6, 0);
        return BNG_PASS;
      }
    }
  } else if (c.proto == 17) {
    if (!c.l4_ok) return BNG_FWD;
    sport = c.sport; dport = c.dport;
    if (cfg->flags & BNG_NAT_FLAG_ALG_SIP) {
      if (nat_is_alg(cfg, dport, 17)) {
        F.alg = true;
        nat_log_push(T, BNG_LOG_ALG_TRIGGER, blk->subscriber_id, c.saddr, 0,
                     sport, 0, c.daddr, dport, 17, 0);
        return BNG_PASS;
      }
    }
  } else if (c.proto == 1) {
    if (!c.l4_ok) return BNG_FWD;
    sport = c.icmp_id; dport = 0;
  } else {
    return BNG_FWD;
  }

  uint8_t is_hairpin = 0;
  if ((cfg->flags & BNG_NAT_FLAG_HAIRPIN) && nat_is_hairpin(T, c.daddr)) {
    is_hairpin = 1; F.hairpin = true;
  }

  uint64_t sig = bng_tuple_sig(c.saddr, c.daddr, sport, dport, c.proto);
  bool claimed, found;
  bng_nat_session* sess = sig_find_or_claim(T.sessions, T.sess_mask, sig,
                                            &claimed, &found);
  if (!sess) { F.passed = true; return BNG_PASS; }  /* table section full */

  uint32_t nat_ip; uint16_t nat_port;
  if (found) {
    /* HIT fast path: the producer plain-stored fields then issued an
     * agent release (L2 write-back) before the sc1 `ready` store, so
     * relaxed agent (sc1, L2-served) loads are always fresh — no
     * acquire fence (which would flush this CU's whole L1) needed. */
    const uint64_t* e64 = (const uint64_t*)sess;
    uint64_t k0 = rlx_load64(e64 + 1);   /* src_ip | dst_ip<<32 */
    uint64_t k1 = rlx_load64(e64 + 2);   /* sport | dport<<16 | proto<<32 */
    uint64_t t0 = rlx_load64(e64 + 3);   /* nat_ip | nat_port<<32 | orig_port<<48 */
    uint64_t t1 = rlx_load64(e64 + 4);   /* orig_ip | state<<32 | hairpin<<40 | ready<<48 */
    if (!((t1 >> 48) & 0xFF)) {          /* creation still in flight */
      if (!bng_wait_ready(&sess->ready, 8192)) {
        F.passed = true; return BNG_PASS;
      }
      k0 = rlx_load64(e64 + 1); k1 = rlx_load64(e64 + 2);
      t0 = rlx_load64(e64 + 3); t1 = rlx_load64(e64 + 4);
    }
    uint64_t ek0 = ((uint64_t)c.daddr << 32) | c.saddr;
    uint64_t ek1 = (uint64_t)sport | ((uint64_t)dport << 16) |
                   ((uint64_t)c.proto << 32);
    if (k0 != ek0 || (k1 & 0xFFFFFFFFFFull) != ek1) {
      /* sig collision with a different tuple: slow path (rare) */
      F.passed = true; return BNG_PASS;
    }
    nat_ip = (uint32_t)t0;
    nat_port = (uint16_t)(t0 >> 32);
    /* one fabric store per batch per session, not per packet */
    if (rlx_load64(e64 + 5) != T.now_ns)
      rlx_store64((void*)(e64 + 5), T.now_ns);
    /* one packed accounting atomic: bytes<<24 | packets (folded into the
     * u64 counters by the sweep kernel) */
    atomicAdd((unsigned long long*)(e64 + 11),
              ((unsigned long long)c.len << 24) | 1ull);
  } else {
    /* we claimed the slot: allocate mapping (EIM or fresh port) */
    bool parity = (cfg->flags & BNG_NAT_FLAG_PARITY) != 0;
    bool have = false;
    if (cfg->flags & BNG_NAT_FLAG_EIM) {
      uint64_t esig = bng_eim_sig(c.saddr, sport, c.proto);
      bool eclaimed, efound;
      bng_eim_entry* eim = sig_find_or_claim(T.eim, T.eim_mask, esig,
                                             &eclaimed, &efound);
      if (eim && efound) {
        if (bng_wait_ready(&eim->ready, 8192) && eim->internal_ip == c.saddr
            && eim->internal_port == sport && eim->protocol == c.proto) {
          __hip_atomic_store(&eim->last_used, T.now_ns, __ATOMIC_RELAXED,
                             __HIP_MEMORY_SCOPE_AGENT);
          atomicAdd(&eim->ref_count, 1u);
          F.eim_hit = true;
          nat_ip = eim->external_ip; nat_port = eim->external_port;
          have = true;
        }
      } else if (eim && eclaimed) {
        uint16_t ap = nat_alloc_port(T, blk, parity, sport, c.saddr, c.proto);
        if (ap == 0) {
          /* release the claimed EIM slot as tombstone, drop the packet */
          __hip_atomic_store(&eim->sig, BNG_KEY_TOMBSTONE, __ATOMIC_RELAXED,
                             __HIP_MEMORY_SCOPE_AGENT);
          __hip_atomic_store(&sess->sig, BNG_KEY_TOMBSTONE, __ATOMIC_RELAXED,
                             __HIP_MEMORY_SCOPE_AGENT);
          F.port_exh = true; F.dropped = true;
          nat_log_push(T, BNG_LOG_PORT_EXHAUSTION, blk->subscriber_id,
                       c.saddr, blk->public_ip, sport, 0, c.daddr, dport,
                       c.proto, 0);
          return BNG_DROP;
        }
        eim->internal_ip = c.saddr; eim->internal_port = sport;
        eim->protocol = c.proto;
        eim->external_ip = blk->public_ip;
        eim->external_port = ap;
        eim->created = T.now_ns; eim->last_used = T.now_ns;
        eim->ref_count = 1; eim->flags = 0;
        bng_publish_ready(&eim->ready);
        F.eim_miss = true;
        nat_ip = eim->external_ip;
        nat_port = ap;
        have = true;
      }
    }
    if (!have) {
      uint16_t ap = nat_alloc_port(T, blk, parity, sport, c.saddr, c.proto);
      if (ap == 0) {
        __hip_atomic_store(&sess->sig, BNG_KEY_TOMBSTONE, __ATOMIC_RELAXED,
                           __HIP_MEMORY_SCOPE_AGENT);
        F.port_exh = true; F.dropped = true;
        nat_log_push(T, BNG_LOG_PORT_EXHAUSTION, blk->subscriber_id, c.saddr,
                     blk->public_ip, sport, 0, c.daddr, dport, c.proto, 0);
        return BNG_DROP;
      }
      nat_ip = blk->public_ip;
      nat_port = ap;
    }
    /* fill the session (ref :711-730) */
    sess->key = bng_nat_tuple{c.saddr, c.daddr, sport, dport, c.proto, {0,0,0}};
    sess->nat_ip = nat_ip; sess->nat_port = nat_port; sess->orig_port = sport;
    sess->orig_ip = c.saddr; sess->state = BNG_NAT_NEW;
    sess->is_hairpin = is_hairpin;
    sess->last_seen = T.now_ns; sess->created = T.now_ns;
    sess->packets_out = 1; sess->packets_in = 0;
    sess->bytes_out = c.len; sess->bytes_in = 0;
    bng_publish_ready(&sess->ready);

    /* reverse mapping for DNAT (ref :732-740) */
    uint64_t rsig = bng_tuple_sig(c.daddr, nat_ip, dport, nat_port, c.proto);
    bool rclaimed, rfound;
    bng_nat_reverse* rev = sig_find_or_claim(T.reverse, T.rev_mask, rsig,
                                             &rclaimed, &rfound);
    if (rev && (rclaimed || rfound)) {
      if (rclaimed) {
        rev->key = bng_nat_tuple{c.daddr, nat_ip, dport, nat_port, c.proto,
                                 {0, 0, 0}};
        rev->orig = sess->key;
        bng_publish_ready(&rev->ready);
      }
    }
    atomicAdd(&blk->sessions_active, 1u);
    atomicAdd(&blk->sessions_total, 1u);
    F.sess_created = true;
    nat_log_push(T, BNG_LOG_SESSION_CREATE, blk->subscriber_id, c.saddr,
                 nat_ip, sport, nat_port, c.daddr, dport, c.proto,
                 is_hairpin);
  }

  /* SNAT rewrite + incremental checksums (ref :752-798).  When the
   * fast parse ran (c.regs: untagged ihl=5, fixed offsets 14/34) the
   * rewrite happens on the register window and flushes as two dwordx4
   * stores — the TA unit is the measured kernel bound and this path
   * replaces ~10 divergent byte R/W ops with 2 wide stores (the TCP
   * checksum at byte 50 stays in memory: outside the 48-B window). */
  uint32_t old_ip = c.saddr;
  uint16_t nat_port_host = nat_port;
  if (c.regs) {
    rg_st32(c, 26, nat_ip);
    rg_st16(c, 24, csum_upd32(rg_ld16(c, 24), old_ip, nat_ip));
    if (c.proto == 6) {
      uint16_t old_port = rg_ld16(c, 34);
      rg_st16(c, 34, nat_port_host);
      uint16_t ck = ld_u16be(p + 50);
      ck = csum_upd32(ck, old_ip, nat_ip);
      ck = csum_upd16(ck, old_port, nat_port_host);
      rg_flush16_47(c);
      st_u16be(p + 50, ck);
    } else if (c.proto == 17) {
      uint16_t old_port = rg_ld16(c, 34);
      rg_st16(c, 34, nat_port_host);
      uint16_t ck = rg_ld16(c, 40);
      if (ck != 0) {
        ck = csum_upd32(ck, old_ip, nat_ip);
        ck = csum_upd16(ck, old_port, nat_port_host);
        if (ck == 0) ck = 0xFFFF;
        rg_st16(c, 40, ck);
      }
      rg_flush16_47(c);
    } else {            /* ICMP: id at 38, checksum at 36 */
      uint16_t old_id = rg_ld16(c, 38);
      rg_st16(c, 38, nat_port_host);
      rg_st16(c, 36, csum_upd16(rg_ld16(c, 36), old_id, nat_port_host));
      rg_flush16_47(c);
    }
    F.snat = true;
    return BNG_FWD;
  }
  st_u32be(p + c.ip_off + 12, nat_ip);
  uint16_t ipck = ld_u16be(p + c.ip_off + 10);
  st_u16be(p + c.ip_off + 10, csum_upd32(ipck, old_ip, nat_ip));
  if (c.proto == 6) {
    uint16_t old_port = ld_u16be(p + c.l4_off);
    st_u16be(p + c.l4_off, nat_port_host);
    uint16_t ck = ld_u16be(p + c.l4_off + 16);
    ck = csum_upd32(ck, old_ip, nat_ip);
    ck = csum_upd16(ck, old_port, nat_port_host);
    st_u16be(p + c.l4_off + 16, ck);
  } else if (c.proto == 17) {
    uint16_t old_port = ld_u16be(p + c.l4_off);
    st_u16be(p + c.l4_off, nat_port_host);
    uint16_t ck = ld_u16be(p + c.l4_off + 6);
    if (ck != 0) {
      ck = csum_upd32(ck, old_ip, nat_ip);
      ck = csum_upd16(ck, old_port, nat_port_host);
      if (ck == 0) ck = 0xFFFF;
      st_u16be(p + c.l4_off + 6, ck);
    }
  } else if (c.proto == 1) {
    uint16_t old_id = ld_u16be(p + c.l4_off + 4);
    st_u16be(p + c.l4_off + 4, nat_port_host);
    uint16_t ck = ld_u16be(p + c.l4_off + 2);
    st_u16be(p + c.l4_off + 2, csum_upd16(ck, old_id, nat_port_host));
  }
  F.snat = true;
  return BNG_FWD;
}

/* DNAT (ref nat44_ingress nat44.c:805-948) */
BNG_DEV int nat_ingress_process(pktctx& c, const nat_tables& T, nat_flags& F) {
  uint8_t* p = c.p;
  if (c.ip_off < 0) return BNG_FWD;
  uint16_t sport, dport;
  if (c.proto == 6 || c.proto == 17) {
    if (!c.l4_ok) return BNG_FWD;
    sport = c.sport; dport = c.dport;
  } else if (c.proto == 1) {
    if (!c.l4_ok) return BNG_FWD;
    sport = 0; dport = c.icmp_id;
  } else return BNG_FWD;

  uint64_t rsig = bng_tuple_sig(c.saddr, c.daddr, sport, dport, c.proto);
  bng_nat_reverse* rev = sig_lookup(T.reverse, T.rev_mask, rsig);
  if (!rev) { F.passed = true; return BNG_FWD; }
  {
    const uint64_t* r64 = (const uint64_t*)rev;
    uint64_t rdy = rlx_load64(r64 + 5);      /* ready at byte 40 */
    if (!(rdy & 0xFF) && !bng_wait_ready(&rev->ready, 8192)) {
      F.passed = true; return BNG_FWD;
    }
    uint64_t k0 = rlx_load64(r64 + 1), k1 = rlx_load64(r64 + 2);
    uint64_t ek0 = ((uint64_t)c.daddr << 32) | c.saddr;
    uint64_t ek1 = (uint64_t)sport | ((uint64_t)dport << 16) |
                   ((uint64_t)c.proto << 32);
    if (k0 != ek0 || (k1 & 0xFFFFFFFFFFull) != ek1) {
      F.passed = true; return BNG_FWD;
    }
  }
  bng_nat_tuple orig;
  {
    const uint64_t* r64 = (const uint64_t*)rev;
    uint64_t o0 = rlx_load64(r64 + 3), o1 = rlx_load64(r64 + 4);
    orig.src_ip = (uint32_t)o0; orig.dst_ip = (uint32_t)(o0 >> 32);
    orig.src_port = (uint16_t)o1; orig.dst_port = (uint16_t)(o1 >> 16);
    orig.protocol = (uint8_t)(o1 >> 32);
  }
  uint64_t osig = bng_tuple_sig(orig.src_ip, orig.dst_ip, orig.src_port,
                                orig.dst_port, orig.protocol);
  bng_nat_session* sess = sig_lookup(T.sessions, T.sess_mask, osig);
  if (!sess || !bng_wait_ready(&sess->ready, 8192) ||
      !tuple_eq(sess->key, orig)) {
    /* session expired: clean the reverse entry (ref :871-875) */
    __hip_atomic_store(&rev->sig, BNG_KEY_TOMBSTONE, __ATOMIC_RELAXED,
                       __HIP_MEMORY_SCOPE_AGENT);
    F.sess_expired = true;
    return BNG_FWD;
  }
  (void)0;
  {
    const uint64_t* e64 = (const uint64_t*)sess;
    if (rlx_load64(e64 + 5) != T.now_ns)
      rlx_store64((void*)((uint64_t*)sess + 5), T.now_ns);
    atomicAdd((unsigned long long*)((uint64_t*)sess + 12),
              ((unsigned long long)c.len << 24) | 1ull);
  }
  if (c.proto == 6) {
    if (c.tcp_flags & 0x05)
      sess->state = BNG_NAT_CLOSING;
    else if (sess->state == BNG_NAT_NEW && (c.tcp_flags & 0x10))
      sess->state = BNG_NAT_ESTABLISHED;
  }

  uint32_t old_ip = c.daddr, new_ip = sess->orig_ip;
  st_u32be(p + c.ip_off + 16, new_ip);
  uint16_t ipck = ld_u16be(p + c.ip_off + 10);
  st_u16be(p + c.ip_off + 10, csum_upd32(ipck, old_ip, new_ip));
  uint16_t new_port = sess->orig_port;   /* host-read value at egress */
  if (c.proto == 6) {
    uint16_t old_port = ld_u16be(p + c.l4_off + 2);
    st_u16be(p + c.l4_off + 2, new_port);
    uint16_t ck = ld_u16be(p + c.l4_off + 16);
    ck = csum_upd32(ck, old_ip, new_ip);
    ck = csum_upd16(ck, old_port, new_port);
    st_u16be(p + c.l4_off + 16, ck);
  } else if (c.proto == 17) {
    uint16_t old_port = ld_u16be(p + c.l4_off + 2);
    st_u16be(p + c.l4_off + 2, new_port);
    uint16_t ck = ld_u16be(p + c.l4_off + 6);
    if (ck != 0) {
      ck = csum_upd32(ck, old_ip, new_ip);
      ck = csum_upd16(ck, old_port, new_port);
      if (ck == 0) ck = 0xFFFF;
      st_u16be(p + c.l4_off + 6, ck);
    }
  } else if (c.proto == 1) {
    uint16_t old_id = ld_u16be(p + c.l4_off + 4);
    st_u16be(p + c.l4_off + 4, new_port);
    uint16_t ck = ld_u16be(p + c.l4_off + 2);
    st_u16be(p + c.l4_off + 2, csum_upd16(ck, old_id, new_port));
  }
  F.dnat = true;
  return BNG_FWD;
}

BNG_DEV void nat_commit_stats(const nat_flags& F, unsigned long long* st) {
  stat_inc(&st[BNG_NS_SNAT], F.snat);
  stat_inc(&st[BNG_NS_DNAT], F.dnat);
  stat_inc(&st[BNG_NS_HAIRPIN], F.hairpin);
  stat_inc(&st[BNG_NS_DROPPED], F.dropped);
  stat_inc(&st[BNG_NS_PASSED], F.passed);
  stat_inc(&st[BNG_NS_SESS_CREATED], F.sess_created);
  stat_inc(&st[BNG_NS_SESS_EXPIRED], F.sess_expired);
  stat_inc(&st[BNG_NS_PORT_EXHAUSTION], F.port_exh);
  stat_inc(&st[BNG_NS_EIM_HITS], F.eim_hit);
  stat_inc(&st[BNG_NS_EIM_MISSES], F.eim_miss);
  stat_inc(&st[BNG_NS_ALG_TRIGGERS], F.alg);
}

__global__ void nat44_kernel(
    uint8_t* __restrict__ data, const uint16_t* __restrict__ in_len,
    uint8_t* __restrict__ verdict, int n, int stride, int is_egress,
    bng_nat_session* sessions, uint32_t sess_mask,
    bng_nat_reverse* reverse, uint32_t rev_mask,
    bng_eim_entry* eim, uint32_t eim_mask,
    bng_subctx* ctx, uint32_t ctx_mask,
    const bng_nat_config* cfg,
    const uint32_t* hairpin_ips, uint32_t n_hairpin,
    unsigned long long* stats,
    bng_nat_log_entry* log_ring, bng_ring_header* log_hdr,
    uint64_t now_ns) {
  nat_tables T{sessions, sess_mask, reverse, rev_mask, eim, eim_mask,
               ctx, ctx_mask, cfg, hairpin_ips, n_hairpin, stats,
               log_ring, log_hdr, now_ns};
  int tid = blockIdx.x * blockDim.x + threadIdx.x;
  int nthreads = gridDim.x * blockDim.x;
  for (int base = 0; base < n; base += nthreads) {
    int pid = base + tid;
    nat_flags F; F.clear();
    if (pid < n) {
      pktctx c;
      parse_pkt(c, data + (size_t)pid * stride, in_len[pid], false);
      int v = is_egress ? nat_egress_process(c, T, F)
                        : nat_ingress_process(c, T, F);
      verdict[pid] = (uint8_t)v;
    }
    nat_commit_stats(F, stats);
  }
}

/* ============================================================ QoS  K3 */

struct qos_flags { bool passed, dropped; uint32_t bytes; };

/* Token-bucket check (ref token_bucket_check qos_ratelimit.c:70-104),
 * consume-first form: one atomicAdd(-len) when tokens suffice (the
 * common case); only an INSUFFICIENT balance triggers the refill
 * (elapsed-credit since last_update, capped at burst, refill winner by
 * CAS on last_update) and one retry.  Verdict-equivalent to the
 * reference's refill-then-consume: unclaimed credit stays recoverable
 * because last_update only advances when a refill actually runs. */
BNG_DEV bool qos_consume(int64_t* tokens, uint32_t pkt_len) {
  long long old = (long long)atomicAdd(
      (unsigned long long*)tokens,
      (unsigned long long)(-(int64_t)pkt_len));
  if (old - (int64_t)pkt_len >= 0) return true;
  atomicAdd((unsigned long long*)tokens, (unsigned long long)pkt_len);
  return false;
}

/* field-pointer form so the standalone egress bucket (bng_qos_bucket)
 * and the merged uplink context (bng_subctx) share one implementation */
BNG_DEV bool qos_tb_check(int64_t* tokens, uint64_t* last_update,
                          uint32_t burst_bytes, uint32_t pkt_len,
                          uint64_t now_ns, uint64_t rate) {
  if (rate == 0) return true;
  if (qos_consume(tokens, pkt_len)) return true;
  /* insufficient: claim the refill, credit elapsed time, retry once */
  uint64_t last = __hip_atomic_load(last_update, __ATOMIC_RELAXED,
                                    __HIP_MEMORY_SCOPE_AGENT);
  if (last != now_ns &&
      __hip_atomic_compare_exchange_strong(
          last_update, &last, now_ns, __ATOMIC_RELAXED,
          __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT)) {
    uint64_t elapsed = now_ns - last;
    int64_t burst = (int64_t)burst_bytes;
    uint64_t add = (elapsed > 100000000000ull) ? (uint64_t)burst
        : (elapsed * (rate / 8)) / 1000000000ull;
    /* capped add, bounded CAS loop */
    for (int t = 0; t < 16; ++t) {
      int64_t cur = (int64_t)__hip_atomic_load(
          (uint64_t*)tokens, __ATOMIC_RELAXED,
          __HIP_MEMORY_SCOPE_AGENT);
      int64_t want = cur + (int64_t)add;
      if (want > burst) want = burst;
      if (want == cur) break;
      uint64_t expect = (uint64_t)cur;
      if (__hip_atomic_compare_exchange_strong(
              (uint64_t*)tokens, &expect, (uint64_t)want,
              __ATOMIC_RELAXED, __ATOMIC_RELAXED,
              __HIP_MEMORY_SCOPE_AGENT))
        break;
    }
  }
  return qos_consume(tokens, pkt_len);
}

BNG_DEV int qos_process(pktctx& c, bng_qos_bucket* table, uint32_t mask,
                        bool egress, uint64_t now_ns, qos_flags& F) {
  if (c.ip_off < 0) return BNG_FWD;
  uint32_t key = egress ? c.daddr : c.saddr;
  /* table keys are stored as the BE byte pattern read as host int */
  uint64_t rate = 0;
  bng_qos_bucket* tb = qos_lookup(table, mask, key, &rate);
  if (!tb) return BNG_FWD;
  bool ok = qos_tb_check(&tb->tokens, &tb->last_update, tb->burst_bytes,
                         c.len, now_ns, rate);
  F.bytes = c.len;
  if (ok) { F.passed = true; return BNG_FWD; }
  F.dropped = true;
  return BNG_DROP;
}

/* ingress QoS over the merged subscriber context — shares the entry the
 * NAT stage already probed, so the policy check costs no extra walk */
BNG_DEV int qos_process_ctx(pktctx& c, bng_subctx* e, uint64_t now_ns,
                            qos_flags& F) {
  if (c.ip_off < 0 || !e || !e->qos_valid) return BNG_FWD;
  bool ok = qos_tb_check(&e->tokens, &e->last_update, e->burst_bytes,
                         c.len, now_ns, e->rate_bps);
  F.bytes = c.len;
  if (ok) { F.passed = true; return BNG_FWD; }
  F.dropped = true;
  return BNG_DROP;
}

__global__ void qos_kernel(
    uint8_t* __restrict__ data, const uint16_t* __restrict__ in_len,
    uint8_t* __restrict__ verdict, int n, int stride, int is_egress,
    void* table, uint32_t mask,
    unsigned long long* stats, uint64_t now_ns) {
  int tid = blockIdx.x * blockDim.x + threadIdx.x;
  int nthreads = gridDim.x * blockDim.x;
  for (int base = 0; base < n; base += nthreads) {
    int pid = base + tid;
    qos_flags F{false, false, 0};
    if (pid < n) {
      pktctx c;
      parse_pkt(c, data + (size_t)pid * stride, in_len[pid], false);
      int v;
      if (is_egress) {
        v = qos_process(c, (bng_qos_bucket*)table, mask, true, now_ns, F);
      } else {
        bng_subctx* e = (c.ip_off >= 0)
            ? subctx_lookup((bng_subctx*)table, mask, c.saddr) : nullptr;
        v = qos_process_ctx(c, e, now_ns, F);
      }
      verdict[pid] = (uint8_t)v;
    }
    stat_inc(&stats[BNG_QS_PKT_PASSED], F.passed);
    stat_inc(&stats[BNG_QS_PKT_DROPPED], F.dropped);
    stat_add(&stats[BNG_QS_BYTES_PASSED], F.bytes, F.passed);
    stat_add(&stats[BNG_QS_BYTES_DROPPED], F.bytes, F.dropped);
  }
}

/* ====================================================== antispoof  K4 */

struct as_flags { bool allowed, dropped, logged, v4viol, v6viol; };

BNG_DEV void spoof_log_push(bng_spoof_event* ring, bng_ring_header* hdr,
                            const uint8_t* mac, uint8_t proto,
                            uint32_t spoofed, uint32_t allowed,
                            uint64_t now_ns) {
  if (!ring) return;
  uint32_t idx = ring_claim(hdr);
  bng_spoof_event* e = &ring[idx & ((1u << BNG_SPOOF_RING_LOG2) - 1)];
  e->timestamp = now_ns;
  #pragma unroll
  for (int i = 0; i < 6; ++i) e->src_mac[i] = mac[i];
  e->protocol = proto;
  e->spoofed_ip = spoofed; e->allowed_ip = allowed;
  #pragma unroll
  for (int i = 0; i < 16; ++i) { e->spoofed_ipv6[i] = 0; e->allowed_ipv6[i] = 0; }
}

/* uRPF source validation (ref antispoof_ingress antispoof.c:189-293),
 * quirks preserved (see golden.py docstring). */
BNG_DEV int antispoof_process_with(uint8_t* p, int len,
                              const bng_binding_entry* b,
                              const bng_antispoof_config* cfg,
                              bng_spoof_event* ring, bng_ring_header* hdr,
                              uint64_t now_ns, as_flags& F) {
  uint8_t mode = b ? b->mode : cfg->default_mode;
  if (mode == BNG_AS_DISABLED) { F.allowed = true; return BNG_FWD; }
  uint16_t proto = ld_u16be(p + 12);
  if (proto == 0x0800) {
    if (len < 34) return BNG_FWD;
    uint32_t src_ip = ld_u32be(p + 14 + 12);
    bool allowed = false;
    uint32_t bound_ip = 0;
    if (b && b->ipv4_valid) {
      bound_ip = b->ipv4_addr;
      if (mode == BNG_AS_STRICT || mode == BNG_AS_LOG_ONLY)
        allowed = (src_ip == bound_ip);
    } else if (mode == BNG_AS_LOOSE) {
      allowed = ip_in_intervals(cfg->allowed_lo, cfg->allowed_hi,
                                cfg->n_allowed_ranges, src_ip);
    }
    if (!allowed) {
      if (cfg->log_violations) {
        spoof_log_push(ring, hdr, p + 6, 4, src_ip, bound_ip, now_ns);
        F.logged = true;
      }
      if (mode == BNG_AS_LOG_ONLY) { F.allowed = true; return BNG_FWD; }
      F.dropped = true; F.v4viol = true;
      return BNG_DROP;
    }
    F.allowed = true; return BNG_FWD;
  }
  if (proto == 0x86DD) {
    if (len < 14 + 40) return BNG_FWD;
    bool allowed = false;
    if (b && b->ipv6_valid) {
      allowed = true;
      #pragma unroll
      for (int i = 0; i < 16; ++i)
        if (p[14 + 8 + i] != b->ipv6_addr[i]) { allowed = false; break; }
    } else if (mode == BNG_AS_LOOSE) {
      allowed = true;
    }
    if (!allowed && mode != BNG_AS_LOG_ONLY) {
      if (cfg->log_violations) {
        spoof_log_push(ring, hdr, p + 6, 6, 0, 0, now_ns);
        F.logged = true;
      }
      F.dropped = true; F.v6viol = true;
      return BNG_DROP;
    }
    F.allowed = true; return BNG_FWD;
  }
  F.allowed = true;
  return BNG_FWD;
}

BNG_DEV int antispoof_process(uint8_t* p, int len,
                              const bng_binding_entry* bindings,
                              uint32_t bmask,
                              const bng_antispoof_config* cfg,
                              bng_spoof_event* ring, bng_ring_header* hdr,
                              uint64_t now_ns, as_flags& F) {
  if (len < 14) return BNG_FWD;
  uint64_t mac = 0;
  #pragma unroll
  for (int i = 0; i < 6; ++i) mac = (mac << 8) | p[6 + i];
  const bng_binding_entry* b = binding_lookup(bindings, bmask, mac);
  return antispoof_process_with(p, len, b, cfg, ring, hdr, now_ns, F);
}

__global__ void antispoof_kernel(
    uint8_t* __restrict__ data, const uint16_t* __restrict__ in_len,
    uint8_t* __restrict__ verdict, int n, int stride,
    const bng_binding_entry* bindings, uint32_t bmask,
    const bng_antispoof_config* cfg,
    unsigned long long* stats,
    bng_spoof_event* ring, bng_ring_header* hdr, uint64_t now_ns) {
  int tid = blockIdx.x * blockDim.x + threadIdx.x;
  int nthreads = gridDim.x * blockDim.x;
  for (int base = 0; base < n; base += nthreads) {
    int pid = base + tid;
    as_flags F{false, false, false, false, false};
    if (pid < n) {
      verdict[pid] = (uint8_t)antispoof_process(
          data + (size_t)pid * stride, in_len[pid], bindings, bmask, cfg,
          ring, hdr, now_ns, F);
    }
    stat_inc(&stats[BNG_AS_ALLOWED], F.allowed);
    stat_inc(&stats[BNG_AS_DROPPED], F.dropped);
    stat_inc(&stats[BNG_AS_LOGGED], F.logged);
    stat_inc(&stats[BNG_AS_V4_VIOLATIONS], F.v4viol);
    stat_inc(&stats[BNG_AS_V6_VIOLATIONS], F.v6viol);
  }
}

/* ============================== fused uplink pipeline (the hot chain) */
/* One pass per packet over the chain the reference runs as four separate
 * kernel hooks: DHCP fast path for UDP:67, else antispoof -> NAT44 SNAT
 * -> QoS ingress.  One parse, one packet-data round trip — the fusion the
 * CDNA4 guide prescribes for HBM-bound pipelines. */
__global__ __launch_bounds__(256, 5)  /* cap VGPRs at 96 -> 5 waves/SIMD:
    the pipeline is latency-bound (SQ_WAIT ~95%), occupancy is the lever */
void uplink_pipeline_kernel(bng_uplink_params P) {
  if (P.now_ptr) { P.now_ns = P.now_ptr[0]; P.now_sec = P.now_ptr[1]; }
  dhcp_tables DT{P.subs, P.sub_mask, P.pools, P.n_pools, P.scfg,
                 P.dhcp_stats, P.now_sec};
  nat_tables NT{P.sessions, P.sess_mask, P.reverse, P.rev_mask, P.eim,
                P.eim_mask, P.subctx, P.subctx_mask, P.ncfg, P.hairpin_ips,
                P.n_hairpin, P.nat_stats, P.log_ring, P.log_hdr, P.now_ns};
  int tid = blockIdx.x * blockDim.x + threadIdx.x;
  int nthreads = gridDim.x * blockDim.x;
  for (int base = 0; base < P.n; base += nthreads) {
    int i = base + tid;
    dhcp_flags DF; DF.clear();
    nat_flags NF; NF.clear();
    as_flags AF{false, false, false, false, false};
    qos_flags QF{false, false, 0};
    if (i < P.n) {
      /* with a type-sorted order array, a wave's 64 packets share one
       * code path — divergence between the DHCP and data pipelines no
       * longer serializes both per wave */
      int pid = P.order ? P.order[i] : i;
      if ((unsigned)pid >= (unsigned)P.n) continue;  /* defensive */
      uint8_t* p = P.data + (size_t)pid * P.stride;
      int len = P.in_len[pid];
      uint16_t ol = (uint16_t)len;
      int v;
      pktctx c;
      /* PPPoE control/session ethertypes go to the host PPPoE server:
       * the reference's TC_ACT_OK continues the kernel stack where
       * AF_PACKET PPPoE sockets receive them — our FWD would hairpin
       * them back out the wire instead, so the pump equivalent of
       * "continue the stack" is PASS (cli _frame_slow_path). */
      uint16_t et = len >= 14 ? (uint16_t)((p[12] << 8) | p[13]) : 0;
      if (et == 0x8863 || et == 0x8864 || et == 0x0806) {
        P.verdict[pid] = (uint8_t)BNG_PASS;
        P.out_len[pid] = ol;
        continue;
      }
      bool ip_ok = parse_pkt(c, p, len, /*want_vlan=*/true);
      bool is_dhcp = ip_ok && c.ip_off >= 0 && c.proto == 17 && c.l4_ok &&
                     c.dport == 67;
      if (is_dhcp) {
        v = dhcp_process(p, len, P.stride, DT, DF, &ol);
      } else if (ip_ok && !c.tagged && len >= 14) {
        /* hot data path: the two independent first-probe loads
         * (binding / merged subscriber context) issue together; the
         * context entry then serves BOTH the NAT port-block and the
         * ingress-QoS stages — one random HBM touch where the
         * reference's hook chain pays two */
        uint64_t mac;
        if (c.regs) {
          mac = ((uint64_t)rg_ld16(c, 6) << 32) |
                ((uint64_t)rg_ld16(c, 8) << 16) | rg_ld16(c, 10);
        } else {
          mac = 0;
          #pragma unroll
          for (int j = 0; j < 6; ++j) mac = (mac << 8) | p[6 + j];
        }
        uint32_t s0 = (uint32_t)bng_mix64(mac) & P.bmask;
        uint32_t s1 = (uint32_t)bng_mix64(c.saddr) & P.subctx_mask;
        uint4 f0 = ld_probe16(&P.bindings[s0]);
        uint4 f1 = ld_probe16(&P.subctx[s1]);
        const bng_binding_entry* b =
            binding_lookup_hint(P.bindings, P.bmask, mac, s0, f0);
        v = antispoof_process_with(p, len, b, P.acfg, P.spoof_ring,
                                   P.spoof_hdr, P.now_ns, AF);
        if (v == BNG_FWD) {
          bng_subctx* ctxe = subctx_lookup_hint(
              P.subctx, P.subctx_mask, c.saddr, s1, f1);
          v = nat_egress_process(c, NT, NF, ctxe, true);
          if (v == BNG_FWD)
            v = qos_process_ctx(c, ctxe, P.now_ns, QF);
        }
      } else {
        /* the reference's TC programs parse untagged frames only
         * (nat44.c:573-581, antispoof.c:194-219): a tagged non-DHCP frame
         * reads as non-IP there and is allowed through */
        v = antispoof_process(p, len, P.bindings, P.bmask, P.acfg,
                              P.spoof_ring, P.spoof_hdr, P.now_ns, AF);
        if (v == BNG_FWD && ip_ok && !c.tagged) {
          bng_subctx* ctxe = subctx_lookup(P.subctx, P.subctx_mask, c.saddr);
          v = nat_egress_process(c, NT, NF, ctxe, true);
          if (v == BNG_FWD)
            v = qos_process_ctx(c, ctxe, P.now_ns, QF);
        }
      }
      P.verdict[pid] = (uint8_t)v;
      P.out_len[pid] = ol;
    }
    dhcp_commit_stats(DF, P.dhcp_stats);
    nat_commit_stats(NF, P.nat_stats);
    stat_inc(&P.as_stats[BNG_AS_ALLOWED], AF.allowed);
    stat_inc(&P.as_stats[BNG_AS_DROPPED], AF.dropped);
    stat_inc(&P.as_stats[BNG_AS_LOGGED], AF.logged);
    stat_inc(&P.as_stats[BNG_AS_V4_VIOLATIONS], AF.v4viol);
    stat_inc(&P.as_stats[BNG_AS_V6_VIOLATIONS], AF.v6viol);
    stat_inc(&P.qos_stats[BNG_QS_PKT_PASSED], QF.passed);
    stat_inc(&P.qos_stats[BNG_QS_PKT_DROPPED], QF.dropped);
    stat_add(&P.qos_stats[BNG_QS_BYTES_PASSED], QF.bytes, QF.passed);
    stat_add(&P.qos_stats[BNG_QS_BYTES_DROPPED], QF.bytes, QF.dropped);
  }
}

/* fused downlink pipeline: internet -> subscriber return path.
 * NAT44 DNAT (reverse map -> session -> rewrite) then QoS egress
 * (download shaping keyed by the post-DNAT destination = the
 * subscriber's IP) in one pass — the reference's TC-ingress nat44 +
 * TC-egress qos hook chain (nat44.c:805-948, qos_ratelimit.c:126-172). */
__global__ __launch_bounds__(256, 5)
void downlink_pipeline_kernel(bng_uplink_params P) {
  if (P.now_ptr) { P.now_ns = P.now_ptr[0]; P.now_sec = P.now_ptr[1]; }
  nat_tables NT{P.sessions, P.sess_mask, P.reverse, P.rev_mask, P.eim,
                P.eim_mask, P.subctx, P.subctx_mask, P.ncfg, P.hairpin_ips,
                P.n_hairpin, P.nat_stats, P.log_ring, P.log_hdr, P.now_ns};
  int tid = blockIdx.x * blockDim.x + threadIdx.x;
  int nthreads = gridDim.x * blockDim.x;
  for (int base = 0; base < P.n; base += nthreads) {
    int i = base + tid;
    nat_flags NF; NF.clear();
    qos_flags QF{false, false, 0};
    if (i < P.n) {
      int pid = P.order ? P.order[i] : i;
      uint8_t* p = P.data + (size_t)pid * P.stride;
      int len = P.in_len[pid];
      pktctx c;
      bool ip_ok = parse_pkt(c, p, len, false);
      int v = BNG_FWD;
      if (ip_ok) {
        v = nat_ingress_process(c, NT, NF);
        if (v == BNG_FWD && c.ip_off >= 0) {
          /* re-read the (possibly rewritten) destination for QoS */
          c.daddr = ld_u32be(p + c.ip_off + 16);
          v = qos_process(c, P.qos_eg, P.qos_eg_mask, /*egress=*/true,
                          P.now_ns, QF);
        }
      }
      P.verdict[pid] = (uint8_t)v;
      P.out_len[pid] = (uint16_t)len;
    }
    nat_commit_stats(NF, P.nat_stats);
    stat_inc(&P.qos_stats[BNG_QS_PKT_PASSED], QF.passed);
    stat_inc(&P.qos_stats[BNG_QS_PKT_DROPPED], QF.dropped);
    stat_add(&P.qos_stats[BNG_QS_BYTES_PASSED], QF.bytes, QF.passed);
    stat_add(&P.qos_stats[BNG_QS_BYTES_DROPPED], QF.bytes, QF.dropped);
  }
}

/* classify packets for type-sorting: 1 = DHCP (UDP dst 67), 0 = other */
__global__ void pkt_class_kernel(const uint8_t* __restrict__ data,
                                 const uint16_t* __restrict__ in_len,
                                 uint8_t* __restrict__ cls, int n,
                                 int stride) {
  int tid = blockIdx.x * blockDim.x + threadIdx.x;
  int nthreads = gridDim.x * blockDim.x;
  for (int pid = tid; pid < n; pid += nthreads) {
    pktctx c;
    bool ip_ok = parse_pkt(c, const_cast<uint8_t*>(data) +
                           (size_t)pid * stride, in_len[pid], true);
    cls[pid] = (ip_ok && c.ip_off >= 0 && c.proto == 17 && c.l4_ok &&
                c.dport == 67) ? 1 : 0;
  }
}

/* ======================================= host CRUD kernels (pkg/ebpf) */
/* Stream-ordered table upserts/deletes: the MI355X analog of BPF map
 * update syscalls (pkg/ebpf/loader.go:349-661).  Launched on the same HIP
 * stream as the dataplane kernels, so every batch sees a consistent table
 * snapshot (BPF map semantics for free). */

/* HA standby promotion: bulk-restore NAT sessions (+ reverse + EIM)
 * from the active's export records, recreating exactly what the egress
 * create path builds (round-1 VERDICT task 3; ref ha/sync.go session
 * replication keeps NAT bindings across failover). */
__global__ void sess_import_kernel(
    bng_nat_session* sess, uint32_t sess_mask,
    bng_nat_reverse* rev, uint32_t rev_mask,
    bng_eim_entry* eim, uint32_t eim_mask,
    const bng_sess_export* batch, int n, int* rc) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  bng_sess_export e = batch[i];
  uint64_t sig = bng_tuple_sig(e.src_ip, e.dst_ip, e.src_port, e.dst_port,
                               e.protocol);
  bool claimed, found;
  bng_nat_session* s = sig_find_or_claim(sess, sess_mask, sig,
                                         &claimed, &found);
  if (!s) { if (rc) rc[i] = 1; return; }
  /* import is authoritative (runs on a quiesced standby): overwrite on
   * found as well as on claim */
  s->key = bng_nat_tuple{e.src_ip, e.dst_ip, e.src_port, e.dst_port,
                         e.protocol, {0, 0, 0}};
  s->nat_ip = e.nat_ip; s->nat_port = e.nat_port;
  s->orig_port = e.src_port; s->orig_ip = e.src_ip;
  s->state = e.state; s->is_hairpin = e.is_hairpin;
  s->last_seen = e.last_seen; s->created = e.created;
  s->packets_out = 0; s->packets_in = 0;
  s->bytes_out = 0; s->bytes_in = 0;
  bng_publish_ready(&s->ready);

  uint64_t rsig = bng_tuple_sig(e.dst_ip, e.nat_ip, e.dst_port,
                                e.nat_port, e.protocol);
  bng_nat_reverse* r = sig_find_or_claim(rev, rev_mask, rsig,
                                         &claimed, &found);
  if (r) {
    r->key = bng_nat_tuple{e.dst_ip, e.nat_ip, e.dst_port, e.nat_port,
                           e.protocol, {0, 0, 0}};
    r->orig = s->key;
    bng_publish_ready(&r->ready);
  } else if (rc) {
    rc[i] = 2;
  }

  if (e.flags & 1) {
    uint64_t esig = bng_eim_sig(e.src_ip, e.src_port, e.protocol);
    bng_eim_entry* m = sig_find_or_claim(eim, eim_mask, esig,
                                         &claimed, &found);
    if (m) {
      m->internal_ip = e.src_ip; m->internal_port = e.src_port;
      m->protocol = e.protocol;
      m->external_ip = e.nat_ip; m->external_port = e.eim_port;
      m->created = e.created; m->last_used = e.last_seen;
      if (claimed) m->ref_count = 1;
      m->flags = 0;
      bng_publish_ready(&m->ready);
    }
  }
  if (rc && rc[i] == 0) rc[i] = 0;
}

__global__ void sub_upsert_kernel(bng_sub_entry* t, uint32_t mask,
                                  const bng_sub_entry* batch, int n,
                                  int* rc) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  bng_sub_entry e = batch[i];
  uint32_t slot = (uint32_t)bng_mix64(e.key) & mask;
  int first_tomb = -1;
  for (int k = 0; k < BNG_MAX_PROBE; ++k) {
    bng_sub_entry* s = &t[(slot + k) & mask];
    uint64_t cur = __hip_atomic_load(&s->key, __ATOMIC_RELAXED,
                                     __HIP_MEMORY_SCOPE_AGENT);
    if (cur == e.key) {   /* overwrite value */
      s->pool_id = e.pool_id; s->allocated_ip = e.allocated_ip;
      s->lease_expiry = e.lease_expiry; s->vlan_id = e.vlan_id;
      s->client_class = e.client_class; s->flags = e.flags;
      if (rc) rc[i] = 0;
      return;
    }
    if (cur == BNG_KEY_TOMBSTONE && first_tomb < 0)
      first_tomb = (int)((slot + k) & mask);
    if (cur == BNG_KEY_EMPTY) {
      int target = first_tomb >= 0 ? first_tomb : (int)((slot + k) & mask);
      bng_sub_entry* d = &t[target];
      uint64_t expect = first_tomb >= 0 ? BNG_KEY_TOMBSTONE : BNG_KEY_EMPTY;
      if (__hip_atomic_compare_exchange_strong(
              &d->key, &expect, e.key, __ATOMIC_RELAXED, __ATOMIC_RELAXED,
              __HIP_MEMORY_SCOPE_AGENT)) {
        d->pool_id = e.pool_id; d->allocated_ip = e.allocated_ip;
        d->lease_expiry = e.lease_expiry; d->vlan_id = e.vlan_id;
        d->client_class = e.client_class; d->flags = e.flags;
        if (rc) rc[i] = 0;
        return;
      }
      /* raced with another upsert in this batch; retry from here */
      first_tomb = -1;
      continue;
    }
  }
  if (rc) rc[i] = -1;   /* probe bound exceeded: table too full */
}

__global__ void sub_delete_kernel(bng_sub_entry* t, uint32_t mask,
                                  const uint64_t* keys, int n) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  uint64_t key = keys[i];
  uint32_t slot = (uint32_t)bng_mix64(key) & mask;
  for (int k = 0; k < BNG_MAX_PROBE; ++k) {
    bng_sub_entry* s = &t[(slot + k) & mask];
    uint64_t cur = s->key;
    if (cur == key) {
      __hip_atomic_store(&s->key, BNG_KEY_TOMBSTONE, __ATOMIC_RELAXED,
                         __HIP_MEMORY_SCOPE_AGENT);
      return;
    }
    if (cur == BNG_KEY_EMPTY) return;
  }
}

/* Merge-upsert into the subscriber context.  update_mask selects which
 * half changes (BNG_CTX_SET_NAT / SET_QOS / CLR_QOS / CLR_NAT) so the
 * NAT manager and the QoS manager can each own their fields without
 * clobbering the other's — the BPF-map analog of two maps sharing a key. */
__global__ void subctx_upsert_kernel(bng_subctx* t, uint32_t mask,
                                     const bng_subctx* batch, int n,
                                     uint32_t um, int* rc) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  bng_subctx e = batch[i];
  uint32_t slot = (uint32_t)bng_mix64(e.key_ip) & mask;
  for (int k = 0; k < BNG_MAX_PROBE; ++k) {
    bng_subctx* s = &t[(slot + k) & mask];
    uint32_t cur = __hip_atomic_load(&s->key_ip, __ATOMIC_RELAXED,
                                     __HIP_MEMORY_SCOPE_AGENT);
    if (cur == e.key_ip || cur == 0) {
      if (cur == 0) {
        if (!(um & (BNG_CTX_SET_NAT | BNG_CTX_SET_QOS))) {
          if (rc) rc[i] = 0;   /* clear of an absent entry: no-op */
          return;
        }
        uint32_t expect = 0;
        if (!__hip_atomic_compare_exchange_strong(
                &s->key_ip, &expect, e.key_ip, __ATOMIC_RELAXED,
                __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT))
          continue;
      }
      if (um & BNG_CTX_SET_NAT) {
        s->public_ip = e.public_ip;
        s->port_start = e.port_start; s->port_end = e.port_end;
        s->next_port = e.next_port; s->subscriber_id = e.subscriber_id;
        s->flags = e.flags; s->nat_valid = 1;
      }
      if (um & BNG_CTX_SET_QOS) {
        s->rate_bps = e.rate_bps; s->tokens = e.tokens;
        s->last_update = e.last_update; s->burst_bytes = e.burst_bytes;
        s->priority = e.priority; s->qos_valid = 1;
      }
      if (um & BNG_CTX_CLR_QOS) s->qos_valid = 0;
      if (um & BNG_CTX_CLR_NAT) s->nat_valid = 0;
      if (rc) rc[i] = 0;
      return;
    }
  }
  if (rc) rc[i] = -1;
}

__global__ void qos_upsert_kernel(bng_qos_bucket* t, uint32_t mask,
                                  const bng_qos_bucket* batch, int n,
                                  int* rc) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  bng_qos_bucket e = batch[i];
  uint32_t slot = (uint32_t)bng_mix64(e.key_ip) & mask;
  for (int k = 0; k < BNG_MAX_PROBE; ++k) {
    bng_qos_bucket* s = &t[(slot + k) & mask];
    uint32_t cur = __hip_atomic_load(&s->key_ip, __ATOMIC_RELAXED,
                                     __HIP_MEMORY_SCOPE_AGENT);
    if (cur == e.key_ip || cur == 0) {
      if (cur == 0) {
        uint32_t expect = 0;
        if (!__hip_atomic_compare_exchange_strong(
                &s->key_ip, &expect, e.key_ip, __ATOMIC_RELAXED,
                __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT))
          continue;
      }
      s->rate_bps = e.rate_bps; s->tokens = e.tokens;
      s->last_update = e.last_update; s->burst_bytes = e.burst_bytes;
      s->priority = e.priority;
      s->valid = e.valid;     /* valid=0 upsert == remove policy */
      if (rc) rc[i] = 0;
      return;
    }
  }
  if (rc) rc[i] = -1;
}

__global__ void binding_upsert_kernel(bng_binding_entry* t, uint32_t mask,
                                      const bng_binding_entry* batch, int n,
                                      int* rc) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  bng_binding_entry e = batch[i];
  uint32_t slot = (uint32_t)bng_mix64(e.key_mac) & mask;
  int first_tomb = -1;
  for (int k = 0; k < BNG_MAX_PROBE; ++k) {
    bng_binding_entry* s = &t[(slot + k) & mask];
    uint64_t cur = __hip_atomic_load(&s->key_mac, __ATOMIC_RELAXED,
                                     __HIP_MEMORY_SCOPE_AGENT);
    if (cur == e.key_mac) {
      s->ipv4_addr = e.ipv4_addr; s->ipv4_valid = e.ipv4_valid;
      s->ipv6_valid = e.ipv6_valid; s->mode = e.mode;
      #pragma unroll
      for (int j = 0; j < 16; ++j) s->ipv6_addr[j] = e.ipv6_addr[j];
      if (rc) rc[i] = 0;
      return;
    }
    if (cur == BNG_KEY_TOMBSTONE && first_tomb < 0)
      first_tomb = (int)((slot + k) & mask);
    if (cur == BNG_KEY_EMPTY) {
      int target = first_tomb >= 0 ? first_tomb : (int)((slot + k) & mask);
      bng_binding_entry* d = &t[target];
      uint64_t expect = first_tomb >= 0 ? BNG_KEY_TOMBSTONE : BNG_KEY_EMPTY;
      if (__hip_atomic_compare_exchange_strong(
              &d->key_mac, &expect, e.key_mac, __ATOMIC_RELAXED,
              __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT)) {
        d->ipv4_addr = e.ipv4_addr; d->ipv4_valid = e.ipv4_valid;
        d->ipv6_valid = e.ipv6_valid; d->mode = e.mode;
        #pragma unroll
        for (int j = 0; j < 16; ++j) d->ipv6_addr[j] = e.ipv6_addr[j];
        if (rc) rc[i] = 0;
        return;
      }
      first_tomb = -1;
      continue;
    }
  }
  if (rc) rc[i] = -1;
}

__global__ void binding_delete_kernel(bng_binding_entry* t, uint32_t mask,
                                      const uint64_t* keys, int n) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  uint64_t key = keys[i];
  uint32_t slot = (uint32_t)bng_mix64(key) & mask;
  for (int k = 0; k < BNG_MAX_PROBE; ++k) {
    bng_binding_entry* s = &t[(slot + k) & mask];
    uint64_t cur = s->key_mac;
    if (cur == key) {
      __hip_atomic_store(&s->key_mac, BNG_KEY_TOMBSTONE, __ATOMIC_RELAXED,
                         __HIP_MEMORY_SCOPE_AGENT);
      return;
    }
    if (cur == BNG_KEY_EMPTY) return;
  }
}

/* NAT session sweeper: reclaim timed-out sessions (the LRU-map analog,
 * host-triggered; ref relies on BPF LRU eviction).  Walks the table in
 * parallel; each thread owns a contiguous range of slots. */
__global__ void nat_sweep_kernel(bng_nat_session* sessions, uint32_t n_slots,
                                 bng_nat_reverse* reverse, uint32_t rev_mask,
                                 bng_subctx* ctx,
                                 uint32_t ctx_mask,
                                 bng_eim_entry* eim, uint32_t eim_slots,
                                 uint64_t eim_to,
                                 uint64_t now_ns, uint64_t udp_to,
                                 uint64_t tcp_est_to, uint64_t tcp_tr_to,
                                 uint64_t icmp_to,
                                 unsigned long long* stats) {
  int tid = blockIdx.x * blockDim.x + threadIdx.x;
  int nthreads = gridDim.x * blockDim.x;
  bool any = false;
  for (uint32_t i = tid; i < n_slots; i += nthreads) {
    bng_nat_session* s = &sessions[i];
    uint64_t sig = s->sig;
    if (sig == BNG_KEY_EMPTY || sig == BNG_KEY_TOMBSTONE || !s->ready)
      continue;
    /* fold the hot-path packed accounting deltas ({bytes:40,pkts:24} in
     * _pad2[0/1]) into the u64 counters */
    {
      unsigned long long po = atomicExch(
          (unsigned long long*)&s->_pad2[0], 0ull);
      unsigned long long pi = atomicExch(
          (unsigned long long*)&s->_pad2[1], 0ull);
      if (po) {
        s->packets_out += po & 0xFFFFFFull;
        s->bytes_out += po >> 24;
      }
      if (pi) {
        s->packets_in += pi & 0xFFFFFFull;
        s->bytes_in += pi >> 24;
      }
    }
    uint64_t to = udp_to;
    if (s->key.protocol == 6)
      to = (s->state == BNG_NAT_ESTABLISHED) ? tcp_est_to : tcp_tr_to;
    else if (s->key.protocol == 1)
      to = icmp_to;
    if (now_ns - s->last_seen < to && s->state != BNG_NAT_CLOSING) continue;
    if (s->state == BNG_NAT_CLOSING && now_ns - s->last_seen < tcp_tr_to)
      continue;
    /* expire: tombstone session + its reverse entry, decrement counters */
    uint64_t rsig = bng_tuple_sig(s->key.dst_ip, s->nat_ip, s->key.dst_port,
                                  s->nat_port, s->key.protocol);
    bng_nat_reverse* rev = sig_lookup(reverse, rev_mask, rsig);
    if (rev)
      __hip_atomic_store(&rev->sig, BNG_KEY_TOMBSTONE, __ATOMIC_RELAXED,
                         __HIP_MEMORY_SCOPE_AGENT);
    bng_subctx* blk = subctx_lookup(ctx, ctx_mask, s->key.src_ip);
    if (blk) atomicSub(&blk->sessions_active, 1u);
    s->ready = 0;
    __hip_atomic_store(&s->sig, BNG_KEY_TOMBSTONE, __ATOMIC_RELAXED,
                       __HIP_MEMORY_SCOPE_AGENT);
    any = true;
    stat_inc(&stats[BNG_NS_SESS_EXPIRED], true);
  }
  (void)any;
  /* EIM idle expiry — the LRU-eviction analog for the endpoint-
   * independent mapping table (the reference's eim_table is an LRU
   * map, nat44.c:231-238).  Safe to expire under live sessions: the
   * reverse map demuxes ingress by the full remote endpoint, so only
   * endpoint-independence for FUTURE flows is (correctly) lost. */
  if (eim != nullptr && eim_to != 0) {
    for (uint32_t i = tid; i < eim_slots; i += nthreads) {
      bng_eim_entry* e = &eim[i];
      uint64_t sig = e->sig;
      if (sig == BNG_KEY_EMPTY || sig == BNG_KEY_TOMBSTONE || !e->ready)
        continue;
      if (now_ns - e->last_used < eim_to)
        continue;
      e->ready = 0;
      __hip_atomic_store(&e->sig, BNG_KEY_TOMBSTONE, __ATOMIC_RELAXED,
                         __HIP_MEMORY_SCOPE_AGENT);
    }
  }
}

/* shard-owner computation for multi-GPU steering: owner by subscriber
 * identity — source IP for upstream IPv4, chaddr MAC for DHCP.  Must match
 * bng_amd/parallel/hashring.py owner_of_* bit-for-bit. */
__global__ void shard_owner_kernel(const uint8_t* __restrict__ data,
                                   const uint16_t* __restrict__ in_len,
                                   int32_t* __restrict__ owner,
                                   int n, int stride, int n_shards) {
  int tid = blockIdx.x * blockDim.x + threadIdx.x;
  int nthreads = gridDim.x * blockDim.x;
  for (int pid = tid; pid < n; pid += nthreads) {
    const uint8_t* p = data + (size_t)pid * stride;
    int len = in_len[pid];
    uint64_t key = 0;
    if (len >= 34 && ld_u16be(p + 12) == 0x0800) {
      if (p[14 + 9] == 17 && len >= 42 && ld_u16be(p + 14 + 20 + 2) == 67 &&
          len >= 42 + 240) {
        /* DHCP: key by chaddr MAC */
        const uint8_t* ch = p + 42 + 28;
        #pragma unroll
        for (int i = 0; i < 6; ++i) key = (key << 8) | ch[i];
      } else {
        key = ld_u32be(p + 14 + 12);   /* source IP */
      }
    } else if (len >= 14) {
      #pragma unroll
      for (int i = 0; i < 6; ++i) key = (key << 8) | p[6 + i];  /* src MAC */
    }
    owner[pid] = (int32_t)(bng_mix64(key) % (uint64_t)n_shards);
  }
}

/* ============================= extern "C" launchers for the extension */

static inline int pkt_grid(int n) {
  int blocks = (n + 255) / 256;
  /* memory-bound grid cap per CDNA4 guide G11: ~2048 blocks + grid-stride */
  return blocks < 4096 ? blocks : 4096;
}

extern "C" {

void bng_launch_dhcp(void* data, const void* in_len, void* out_len,
                     void* verdict, int n, int stride,
                     const void* subs, uint32_t sub_mask,
                     const void* pools, uint32_t n_pools, const void* cfg,
                     void* stats, uint64_t now_sec, const void* now_ptr,
                     hipStream_t s) {
  hipLaunchKernelGGL(dhcp_fastpath_kernel, dim3(pkt_grid(n)), dim3(256), 0, s,
      (uint8_t*)data, (const uint16_t*)in_len, (uint16_t*)out_len,
      (uint8_t*)verdict, n, stride, (const bng_sub_entry*)subs, sub_mask,
      (const bng_ip_pool*)pools, n_pools, (const bng_server_config*)cfg,
      (unsigned long long*)stats, now_sec, (const uint64_t*)now_ptr);
}

void bng_launch_dhcp_service(void* ctrl, void* req, const void* in_len,
                             void* out_len, void* verdict, void* scratch,
                             int n_slots, int n_blocks, void* ctrs,
                             const void* subs,
                             uint32_t sub_mask, const void* pools,
                             uint32_t n_pools, const void* cfg, void* stats,
                             hipStream_t s) {
  hipLaunchKernelGGL(dhcp_service_kernel, dim3(n_blocks), dim3(256), 0, s,
      (bng_svc_ctrl*)ctrl, (uint8_t*)req, (const uint16_t*)in_len,
      (uint16_t*)out_len, (uint8_t*)verdict, (uint8_t*)scratch, n_slots,
      (uint32_t*)ctrs,
      (const bng_sub_entry*)subs, sub_mask, (const bng_ip_pool*)pools,
      n_pools, (const bng_server_config*)cfg,
      (unsigned long long*)stats);
}

void bng_launch_nat44(void* data, const void* in_len, void* verdict, int n,
                      int stride, int is_egress,
                      void* sessions, uint32_t sess_mask,
                      void* reverse, uint32_t rev_mask,
                      void* eim, uint32_t eim_mask,
                      void* subnat, uint32_t subnat_mask,
                      const void* cfg, const void* hairpin,
                      uint32_t n_hairpin, void* stats,
                      void* log_ring, void* log_hdr, uint64_t now_ns,
                      hipStream_t s) {
  hipLaunchKernelGGL(nat44_kernel, dim3(pkt_grid(n)), dim3(256), 0, s,
      (uint8_t*)data, (const uint16_t*)in_len, (uint8_t*)verdict, n, stride,
      is_egress, (bng_nat_session*)sessions, sess_mask,
      (bng_nat_reverse*)reverse, rev_mask, (bng_eim_entry*)eim, eim_mask,
      (bng_subctx*)subnat, subnat_mask, (const bng_nat_config*)cfg,
      (const uint32_t*)hairpin, n_hairpin, (unsigned long long*)stats,
      (bng_nat_log_entry*)log_ring, (bng_ring_header*)log_hdr, now_ns);
}

void bng_launch_qos(void* data, const void* in_len, void* verdict, int n,
                    int stride, int is_egress, void* table, uint32_t mask,
                    void* stats, uint64_t now_ns, hipStream_t s) {
  hipLaunchKernelGGL(qos_kernel, dim3(pkt_grid(n)), dim3(256), 0, s,
      (uint8_t*)data, (const uint16_t*)in_len, (uint8_t*)verdict, n, stride,
      is_egress, table, mask, (unsigned long long*)stats,
      now_ns);
}

void bng_launch_antispoof(void* data, const void* in_len, void* verdict,
                          int n, int stride, const void* bindings,
                          uint32_t bmask, const void* cfg, void* stats,
                          void* ring, void* hdr, uint64_t now_ns,
                          hipStream_t s) {
  hipLaunchKernelGGL(antispoof_kernel, dim3(pkt_grid(n)), dim3(256), 0, s,
      (uint8_t*)data, (const uint16_t*)in_len, (uint8_t*)verdict, n, stride,
      (const bng_binding_entry*)bindings, bmask,
      (const bng_antispoof_config*)cfg, (unsigned long long*)stats,
      (bng_spoof_event*)ring, (bng_ring_header*)hdr, now_ns);
}

void bng_launch_uplink(bng_uplink_params* P, hipStream_t s) {
  hipLaunchKernelGGL(uplink_pipeline_kernel, dim3(pkt_grid(P->n)), dim3(256),
                     0, s, *P);
}

void bng_launch_downlink(bng_uplink_params* P, hipStream_t s) {
  hipLaunchKernelGGL(downlink_pipeline_kernel, dim3(pkt_grid(P->n)),
                     dim3(256), 0, s, *P);
}

void bng_launch_pkt_class(const void* data, const void* in_len, void* cls,
                          int n, int stride, hipStream_t s) {
  hipLaunchKernelGGL(pkt_class_kernel, dim3(pkt_grid(n)), dim3(256), 0, s,
      (const uint8_t*)data, (const uint16_t*)in_len, (uint8_t*)cls, n,
      stride);
}

void bng_launch_sub_upsert(void* t, uint32_t mask, const void* batch, int n,
                           void* rc, hipStream_t s) {
  hipLaunchKernelGGL(sub_upsert_kernel, dim3((n + 255) / 256), dim3(256), 0,
      s, (bng_sub_entry*)t, mask, (const bng_sub_entry*)batch, n, (int*)rc);
}
void bng_launch_sess_import(void* sess, uint32_t sess_mask, void* rev,
                            uint32_t rev_mask, void* eim, uint32_t eim_mask,
                            const void* batch, int n, void* rc,
                            hipStream_t s) {
  hipLaunchKernelGGL(sess_import_kernel, dim3((n + 255) / 256), dim3(256),
      0, s, (bng_nat_session*)sess, sess_mask, (bng_nat_reverse*)rev,
      rev_mask, (bng_eim_entry*)eim, eim_mask,
      (const bng_sess_export*)batch, n, (int*)rc);
}
void bng_launch_sub_delete(void* t, uint32_t mask, const void* keys, int n,
                           hipStream_t s) {
  hipLaunchKernelGGL(sub_delete_kernel, dim3((n + 255) / 256), dim3(256), 0,
      s, (bng_sub_entry*)t, mask, (const uint64_t*)keys, n);
}
void bng_launch_subctx_upsert(void* t, uint32_t mask, const void* batch,
                              int n, uint32_t um, void* rc, hipStream_t s) {
  hipLaunchKernelGGL(subctx_upsert_kernel, dim3((n + 255) / 256), dim3(256),
      0, s, (bng_subctx*)t, mask, (const bng_subctx*)batch, n, um,
      (int*)rc);
}
void bng_launch_qos_upsert(void* t, uint32_t mask, const void* batch, int n,
                           void* rc, hipStream_t s) {
  hipLaunchKernelGGL(qos_upsert_kernel, dim3((n + 255) / 256), dim3(256), 0,
      s, (bng_qos_bucket*)t, mask, (const bng_qos_bucket*)batch, n, (int*)rc);
}
void bng_launch_binding_upsert(void* t, uint32_t mask, const void* batch,
                               int n, void* rc, hipStream_t s) {
  hipLaunchKernelGGL(binding_upsert_kernel, dim3((n + 255) / 256), dim3(256),
      0, s, (bng_binding_entry*)t, mask, (const bng_binding_entry*)batch, n,
      (int*)rc);
}
void bng_launch_binding_delete(void* t, uint32_t mask, const void* keys,
                               int n, hipStream_t s) {
  hipLaunchKernelGGL(binding_delete_kernel, dim3((n + 255) / 256), dim3(256),
      0, s, (bng_binding_entry*)t, mask, (const uint64_t*)keys, n);
}

void bng_launch_nat_sweep(void* sessions, uint32_t n_slots, void* reverse,
                          uint32_t rev_mask, void* subnat,
                          uint32_t subnat_mask, void* eim,
                          uint32_t eim_slots, uint64_t eim_to,
                          uint64_t now_ns,
                          uint64_t udp_to, uint64_t tcp_est_to,
                          uint64_t tcp_tr_to, uint64_t icmp_to, void* stats,
                          hipStream_t s) {
  hipLaunchKernelGGL(nat_sweep_kernel, dim3(2048), dim3(256), 0, s,
      (bng_nat_session*)sessions, n_slots, (bng_nat_reverse*)reverse,
      rev_mask, (bng_subctx*)subnat, subnat_mask, (bng_eim_entry*)eim,
      eim_slots, eim_to, now_ns, udp_to,
      tcp_est_to, tcp_tr_to, icmp_to, (unsigned long long*)stats);
}

void bng_launch_shard_owner(const void* data, const void* in_len,
                            void* owner, int n, int stride, int n_shards,
                            hipStream_t s) {
  hipLaunchKernelGGL(shard_owner_kernel, dim3(pkt_grid(n)), dim3(256), 0, s,
      (const uint8_t*)data, (const uint16_t*)in_len, (int32_t*)owner, n,
      stride, n_shards);
}

} /* extern "C" */
