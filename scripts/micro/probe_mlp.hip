// Probe-MLP microbenchmark: random 16-byte table probes, P independent
// probes in flight per lane.  Decides whether a staged SoA pipeline
// (tiny per-probe state => many probes/lane) can beat the fused
// kernel's ~1 chain/lane, or whether the L2/TCC random service rate is
// the wall.  Build: hipcc --offload-arch=gfx950 -O3 -o probe_mlp
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdint>

__device__ __forceinline__ uint64_t mix64(uint64_t x) {
  x += 0x9E3779B97F4A7C15ull;
  uint64_t z = x;
  z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ull;
  z = (z ^ (z >> 27)) * 0x94D049BB133111EBull;
  return z ^ (z >> 31);
}

__device__ __forceinline__ uint64_t rlx64(const uint64_t* p) {
  return __hip_atomic_load(p, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
}

template <int P>
__global__ __launch_bounds__(256)
void probe_kernel(const uint64_t* __restrict__ table, uint32_t mask,
                  uint64_t seed, int iters, uint64_t* __restrict__ sink) {
  int tid = blockIdx.x * blockDim.x + threadIdx.x;
  uint64_t acc = 0;
  uint64_t key[P];
  #pragma unroll
  for (int j = 0; j < P; ++j) key[j] = seed + tid * P + j;
  for (int it = 0; it < iters; ++it) {
    uint64_t v[P];
    #pragma unroll
    for (int j = 0; j < P; ++j) {        // issue P independent probes
      uint32_t slot = (uint32_t)mix64(key[j]) & mask;
      v[j] = rlx64(&table[slot * 2]);    // 16B entry, first word
    }
    #pragma unroll
    for (int j = 0; j < P; ++j) {        // consume (new keys depend on
      acc ^= v[j];                       //  loaded values: true chain)
      key[j] = key[j] * 6364136223846793005ull + v[j] + 1;
    }
  }
  if (acc == 0xDEAD) sink[tid & 255] = acc;
}

template <int P>
float run(const uint64_t* tab, uint32_t mask, uint64_t* sink, int iters,
          int blocks) {
  hipEvent_t a, b;
  hipEventCreate(&a); hipEventCreate(&b);
  // warmup
  hipLaunchKernelGGL((probe_kernel<P>), dim3(blocks), dim3(256), 0, 0,
                     tab, mask, 1, iters / 10, sink);
  hipDeviceSynchronize();
  hipEventRecord(a);
  hipLaunchKernelGGL((probe_kernel<P>), dim3(blocks), dim3(256), 0, 0,
                     tab, mask, 2, iters, sink);
  hipEventRecord(b);
  hipEventSynchronize(b);
  float ms;
  hipEventElapsedTime(&ms, a, b);
  double probes = (double)blocks * 256 * P * iters;
  printf("P=%d  %8.2f ms  %8.2f Gprobe/s  (vgpr-limited waves vary)\n",
         P, ms, probes / ms / 1e6);
  return ms;
}

int main() {
  const uint32_t n = 1u << 23;           // 8M entries x 16B = 128 MB
  uint64_t* tab; uint64_t* sink;
  hipMalloc(&tab, (size_t)n * 16);
  hipMalloc(&sink, 256 * 8);
  hipMemset(tab, 0x5A, (size_t)n * 16);
  const int iters = 4000, blocks = 4096;
  run<1>(tab, n - 1, sink, iters, blocks);
  run<2>(tab, n - 1, sink, iters, blocks);
  run<4>(tab, n - 1, sink, iters, blocks);
  run<8>(tab, n - 1, sink, iters, blocks);
  run<16>(tab, n - 1, sink, iters, blocks);
  hipFree(tab); hipFree(sink);
  return 0;
}
