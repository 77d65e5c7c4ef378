// LDS vs global atomic-add throughput on gfx950 — sizing the histogram
// kernel's real constraint.  Variants:
//  lds_same    : all 64 lanes → one LDS address (worst case)
//  lds_rand    : random LDS addresses (histogram-like)
//  lds_spread  : lane-distinct addresses, bank = lane%32 (best case)
//  lds_u64     : random ds_add_u64 (packed-pair candidate)
//  glb_rand    : random global atomics into a 300 KB L2-resident buffer
// Build: hipcc --offload-arch=gfx950 -O3 -munsafe-fp-atomics tools/atomic_bench.hip -o tools/atomic_bench
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdlib>
#include <vector>

#define CHECK(x) do { hipError_t e=(x); if(e){printf("ERR %s\n", hipGetErrorString(e)); exit(1);} } while(0)

constexpr int LDS_WORDS = 12288;  // 48 KB
constexpr int ITERS = 2000;

template <int V>
__global__ void k(const unsigned* __restrict__ rnd, float* out,
                  unsigned long long* gbuf64, float* gbuf) {
  __shared__ float lds[LDS_WORDS];
  __shared__ unsigned long long lds64[LDS_WORDS / 2];
  const int tid = threadIdx.x;
  for (int i = tid; i < LDS_WORDS; i += blockDim.x) lds[i] = 0;
  for (int i = tid; i < LDS_WORDS / 2; i += blockDim.x) lds64[i] = 0;
  __syncthreads();
  unsigned seed = rnd[(blockIdx.x * blockDim.x + tid) & 65535];
#pragma unroll 4
  for (int it = 0; it < ITERS; ++it) {
    seed = seed * 1664525u + 1013904223u;
    unsigned addr;
    if (V == 0) addr = 0;
    else if (V == 2) addr = (tid & 63) + ((seed >> 8) % (LDS_WORDS / 64)) * 64;
    else addr = seed % LDS_WORDS;
    if (V == 3) {
      atomicAdd(&lds64[addr % (LDS_WORDS / 2)], 1ull);
    } else if (V == 4) {
      atomicAdd(&gbuf[(blockIdx.x % 2) * 65536 + (seed % 65536)], 1.0f);
    } else {
      atomicAdd(&lds[addr], 1.0f);
    }
  }
  __syncthreads();
  if (tid == 0) out[blockIdx.x] = lds[1] + (float)lds64[1];
}

template <int V>
float run(const unsigned* rnd, float* out, unsigned long long* g64, float* g,
          int blocks) {
  hipLaunchKernelGGL((k<V>), dim3(blocks), dim3(256), 0, 0, rnd, out, g64, g);
  CHECK(hipDeviceSynchronize());
  hipEvent_t a, b;
  (void)hipEventCreate(&a); (void)hipEventCreate(&b);
  (void)hipEventRecord(a);
  for (int i = 0; i < 3; ++i)
    hipLaunchKernelGGL((k<V>), dim3(blocks), dim3(256), 0, 0, rnd, out, g64, g);
  (void)hipEventRecord(b);
  CHECK(hipEventSynchronize(b));
  float ms;
  (void)hipEventElapsedTime(&ms, a, b);
  return ms / 3;
}

int main() {
  unsigned* rnd;
  float* out;
  float* gbuf;
  unsigned long long* g64;
  CHECK(hipMalloc(&rnd, 65536 * 4));
  CHECK(hipMalloc(&out, 65536 * 4));
  CHECK(hipMalloc(&gbuf, 2 * 65536 * 4));
  CHECK(hipMalloc(&g64, 65536 * 8));
  std::vector<unsigned> h(65536);
  srand(2);
  for (auto& x : h) x = rand();
  CHECK(hipMemcpy(rnd, h.data(), h.size() * 4, hipMemcpyHostToDevice));
  const int blocks = 2048;
  const double ops = (double)blocks * 256 * ITERS;
  auto rep = [&](const char* name, float ms) {
    printf("%-10s %8.3f ms  %7.1f G lane-atomics/s\n", name, ms,
           ops / ms / 1e6);
  };
  rep("lds_rand", run<1>(rnd, out, g64, gbuf, blocks));
  rep("lds_same", run<0>(rnd, out, g64, gbuf, blocks));
  rep("lds_spread", run<2>(rnd, out, g64, gbuf, blocks));
  rep("lds_u64", run<3>(rnd, out, g64, gbuf, blocks));
  rep("glb_rand", run<4>(rnd, out, g64, gbuf, blocks));
  return 0;
}
