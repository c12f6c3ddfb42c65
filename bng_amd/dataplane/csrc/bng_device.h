/* bng_device.h — shared device-side helpers for the CDNA4 BNG kernels.
 *
 * Written for gfx950 only: wave64, device-scope atomics for cross-XCD
 * visibility, agent-scope release/acquire for intra-launch entry publishing
 * (per the CDNA4 guide's Guideline 16 — per-XCD L2s are not coherent and a
 * CU's L1 is never refreshed by another CU's stores).
 */
#ifndef BNG_DEVICE_H
#define BNG_DEVICE_H

#include <hip/hip_runtime.h>
#include "bng_abi.h"

#define BNG_DEV __device__ __forceinline__

/* ------------------------------------------------------------- hashing */
BNG_DEV uint64_t bng_mix64(uint64_t x) {
  /* splitmix64 finalizer; bit-for-bit identical to abi.mix64 */
  x += 0x9E3779B97F4A7C15ULL;
  uint64_t z = x;
  z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ULL;
  z = (z ^ (z >> 27)) * 0x94D049BB133111EBULL;
  return z ^ (z >> 31);
}

BNG_DEV uint64_t bng_fnv1a64(const uint8_t* p, int n) {
  uint64_t h = 0xCBF29CE484222325ULL;
  for (int i = 0; i < n; ++i) { h ^= p[i]; h *= 0x100000001B3ULL; }
  return h;
}

BNG_DEV uint64_t bng_tuple_sig(uint32_t src_ip, uint32_t dst_ip,
                               uint16_t src_port, uint16_t dst_port,
                               uint8_t proto) {
  uint64_t a = ((uint64_t)src_ip << 32) | dst_ip;
  uint64_t b = ((uint64_t)src_port << 24) | ((uint64_t)dst_port << 8) | proto;
  uint64_t s = bng_mix64(bng_mix64(a) ^ b);
  s |= 1ULL;
  if (s == BNG_KEY_TOMBSTONE) s -= 2;
  return s;
}

BNG_DEV uint64_t bng_eim_sig(uint32_t ip, uint16_t port, uint8_t proto) {
  uint64_t s = bng_mix64(((uint64_t)ip << 24) | ((uint64_t)port << 8) | proto);
  s |= 1ULL;
  if (s == BNG_KEY_TOMBSTONE) s -= 2;
  return s;
}

/* ------------------------------------------- unaligned big-endian loads */
BNG_DEV uint8_t  ld_u8 (const uint8_t* p)  { return *p; }
BNG_DEV uint16_t ld_u16be(const uint8_t* p) {
  return ((uint16_t)p[0] << 8) | p[1];
}
BNG_DEV uint32_t ld_u32be(const uint8_t* p) {
  return ((uint32_t)p[0] << 24) | ((uint32_t)p[1] << 16) |
         ((uint32_t)p[2] << 8) | p[3];
}
BNG_DEV void st_u16be(uint8_t* p, uint16_t v) {
  p[0] = (uint8_t)(v >> 8); p[1] = (uint8_t)v;
}
BNG_DEV void st_u32be(uint8_t* p, uint32_t v) {
  p[0] = (uint8_t)(v >> 24); p[1] = (uint8_t)(v >> 16);
  p[2] = (uint8_t)(v >> 8);  p[3] = (uint8_t)v;
}
BNG_DEV void bng_zero(uint8_t* p, int n) {
  /* byte head to 4-alignment, u32 body, byte tail */
  int i = 0;
  while (((uintptr_t)(p + i) & 3) && i < n) p[i++] = 0;
  for (; i + 16 <= n && !((uintptr_t)(p + i) & 15); i += 16)
    *(uint4*)(p + i) = make_uint4(0, 0, 0, 0);
  for (; i + 4 <= n; i += 4) *(uint32_t*)(p + i) = 0;
  while (i < n) p[i++] = 0;
}

/* ------------------------------------------------ wave-aggregated stats */
/* One atomicAdd per wave instead of per lane: a single contended counter
 * word saturates at ~88 atomics/us on MI355X (microarch 'dequeue' row) —
 * per-lane atomics would cap the chip at ~88 Mpps per counter. */
BNG_DEV void stat_inc(unsigned long long* counter, bool cond) {
  uint64_t mask = __ballot(cond);
  if (mask == 0) return;
  int leader = __ffsll((unsigned long long)mask) - 1;
  int lane = threadIdx.x & 63;
  if (lane == leader)
    atomicAdd(counter, (unsigned long long)__popcll(mask));
}

BNG_DEV void stat_add(unsigned long long* counter, uint64_t v, bool cond) {
  /* wave-reduce v (0 where !cond), one atomic per wave */
  uint64_t x = cond ? v : 0;
  for (int off = 32; off > 0; off >>= 1)
    x += __shfl_down((unsigned long long)x, off, 64);
  uint64_t any = __ballot(cond);
  int lane = threadIdx.x & 63;
  if (lane == 0 && x) atomicAdd(counter, (unsigned long long)x);
  (void)any;
}

/* --------------------------------------------- publish/consume protocol */
/* Device-side hash-entry creation inside one launch (session/EIM insert):
 * winner CASes the slot signature, writes the payload with plain stores,
 * issues an agent-scope release, then sets `ready` with a relaxed
 * agent-scope store.  Readers poll `ready` relaxed (bounded), then one
 * agent-scope acquire before plain reads.  Guide §6 G16. */
BNG_DEV void bng_publish_ready(uint8_t* ready_flag) {
  __builtin_amdgcn_fence(__ATOMIC_RELEASE, "agent");
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __hip_atomic_store(ready_flag, (uint8_t)1, __ATOMIC_RELAXED,
                     __HIP_MEMORY_SCOPE_AGENT);
}

BNG_DEV bool bng_wait_ready(const uint8_t* ready_flag, int max_spins) {
  for (int i = 0; i < max_spins; ++i) {
    if (__hip_atomic_load(ready_flag, __ATOMIC_RELAXED,
                          __HIP_MEMORY_SCOPE_AGENT)) {
      __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
      return true;
    }
    __builtin_amdgcn_s_sleep(1);
  }
  return false;
}

/* ----------------------------------------------------------- log rings */
BNG_DEV uint32_t ring_claim(bng_ring_header* hdr) {
  return atomicAdd(&hdr->widx, 1u);
}

#endif /* BNG_DEVICE_H */
