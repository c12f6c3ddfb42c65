/* bng_params.h — packed launch-parameter blocks.
 *
 * The fused pipeline takes ~35 arguments; passing them individually left
 * the compiler holding every pointer live in SGPRs (187 SGPR spills,
 * occupancy 4 waves/SIMD).  One by-value struct lets it s_load fields on
 * demand from the kernarg segment instead.
 */
#ifndef BNG_PARAMS_H
#define BNG_PARAMS_H

#include <stdint.h>
#include "bng_abi.h"

#ifdef __cplusplus
extern "C" {
#endif

typedef struct bng_uplink_params {
  /* batch */
  uint8_t* data;
  const uint16_t* in_len;
  uint16_t* out_len;
  uint8_t* verdict;
  const int32_t* order;       /* optional type-sorted indirection; NULL=id */
  int n;
  int stride;
  /* dhcp */
  const bng_sub_entry* subs; uint32_t sub_mask;
  const bng_ip_pool* pools;  uint32_t n_pools;
  const bng_server_config* scfg;
  unsigned long long* dhcp_stats;
  /* antispoof */
  const bng_binding_entry* bindings; uint32_t bmask;
  const bng_antispoof_config* acfg;
  unsigned long long* as_stats;
  bng_spoof_event* spoof_ring; bng_ring_header* spoof_hdr;
  /* nat */
  bng_nat_session* sessions; uint32_t sess_mask;
  bng_nat_reverse* reverse;  uint32_t rev_mask;
  bng_eim_entry* eim;        uint32_t eim_mask;
  bng_subctx* subctx;        uint32_t subctx_mask;
  const bng_nat_config* ncfg;
  const uint32_t* hairpin_ips; uint32_t n_hairpin;
  unsigned long long* nat_stats;
  bng_nat_log_entry* log_ring; bng_ring_header* log_hdr;
  /* qos: ingress lives in subctx; this is the downlink egress table */
  bng_qos_bucket* qos_eg; uint32_t qos_eg_mask;
  unsigned long long* qos_stats;
  /* time: now_ptr (device {now_ns, now_sec}) overrides the scalars when
   * non-NULL — required under hipGraph replay, where kernel args are
   * frozen at capture but batch time must advance */
  const uint64_t* now_ptr;
  uint64_t now_ns;
  uint64_t now_sec;
} bng_uplink_params;

#ifdef __cplusplus
}
#endif

#endif /* BNG_PARAMS_H */
