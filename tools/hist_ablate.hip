// Standalone ablation probe for the GBDT histogram kernel (no torch).
// Variants isolate: memory loads vs LDS atomics vs ILP depth vs packed
// (g,h) atomics — following the "ablate before optimizing" rule.
// Build: hipcc --offload-arch=gfx950 -O3 -std=c++17 tools/hist_ablate.hip -o gpurun_out/hist_ablate
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdlib>
#include <vector>

#define CHECK(x) do { hipError_t e = (x); if (e) { printf("ERR %s @%d\n", hipGetErrorString(e), __LINE__); exit(1);} } while (0)

constexpr int NB = 255;

// V0: current structure (GPB groups, 3 scalar ds_add per feature)
// V1: loads only (atomics replaced by asm-kept accumulate)
// V2: atomics only (bins synthesized, no binned load)
// V3: 4-row ILP (batch 4 rows' loads before atomics)
// V4: 8-row ILP
template <int GPB, int VARIANT>
__global__ void hist_k(const uchar4* __restrict__ binned, long n_rows,
                       const int* __restrict__ rows, long m,
                       const float* __restrict__ grad,
                       const float* __restrict__ hess,
                       float* __restrict__ hist, int ngroups, long chunk) {
  constexpr int STRIDE = 3;
  constexpr int ILP = (VARIANT == 4) ? 8 : 4;
  extern __shared__ float lds[];
  const int tid = threadIdx.x;
  const int lds_elems = GPB * 4 * NB * STRIDE;
  for (int i = tid; i < lds_elems; i += blockDim.x) lds[i] = 0.0f;
  __syncthreads();
  const int gq0 = blockIdx.y * GPB;
  const long start = (long)blockIdx.x * chunk;
  const long end = min(start + chunk, m);

  if (VARIANT == 3 || VARIANT == 4) {
    long i = start + tid;
    for (; i + (ILP - 1) * blockDim.x < end; i += ILP * blockDim.x) {
      int r[ILP];
      float g[ILP], h[ILP];
#pragma unroll
      for (int u = 0; u < ILP; ++u) r[u] = rows[i + u * blockDim.x];
#pragma unroll
      for (int u = 0; u < ILP; ++u) { g[u] = grad[r[u]]; h[u] = hess[r[u]]; }
#pragma unroll
      for (int q = 0; q < GPB; ++q) {
        const int grp = gq0 + q;
        if (grp >= ngroups) break;
        uchar4 b4[ILP];
#pragma unroll
        for (int u = 0; u < ILP; ++u) b4[u] = binned[(size_t)grp * n_rows + r[u]];
#pragma unroll
        for (int u = 0; u < ILP; ++u) {
          const unsigned char bs[4] = {b4[u].x, b4[u].y, b4[u].z, b4[u].w};
#pragma unroll
          for (int j = 0; j < 4; ++j) {
            float* cell = &lds[((q * 4 + j) * NB + bs[j]) * STRIDE];
            atomicAdd(cell + 0, g[u]);
            atomicAdd(cell + 1, h[u]);
            atomicAdd(cell + 2, 1.0f);
          }
        }
      }
    }
    for (; i < end; i += blockDim.x) {  // tail
      const int r = rows[i];
      const float g = grad[r], h = hess[r];
      for (int q = 0; q < GPB; ++q) {
        const int grp = gq0 + q;
        if (grp >= ngroups) break;
        const uchar4 b4 = binned[(size_t)grp * n_rows + r];
        const unsigned char bs[4] = {b4.x, b4.y, b4.z, b4.w};
        for (int j = 0; j < 4; ++j) {
          float* cell = &lds[((q * 4 + j) * NB + bs[j]) * STRIDE];
          atomicAdd(cell + 0, g);
          atomicAdd(cell + 1, h);
          atomicAdd(cell + 2, 1.0f);
        }
      }
    }
  } else {
    for (long i = start + tid; i < end; i += blockDim.x) {
      const int r = rows[i];
      const float g = grad[r];
      const float h = hess[r];
      float keep = 0.0f;
#pragma unroll
      for (int q = 0; q < GPB; ++q) {
        const int grp = gq0 + q;
        if (grp >= ngroups) break;
        uchar4 b4;
        if (VARIANT == 2) {
          b4 = make_uchar4((r + q) & 255 % NB, (r * 7 + q) % NB,
                           (r * 13 + q) % NB, (r * 29 + q) % NB);
        } else {
          b4 = binned[(size_t)grp * n_rows + r];
        }
        const unsigned char bs[4] = {b4.x, b4.y, b4.z, b4.w};
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          if (VARIANT == 1) {
            keep += bs[j] * g + h;
          } else {
            float* cell = &lds[((q * 4 + j) * NB + bs[j]) * 3];
            atomicAdd(cell + 0, g);
            atomicAdd(cell + 1, h);
            atomicAdd(cell + 2, 1.0f);
          }
        }
      }
      if (VARIANT == 1) asm volatile("" ::"v"(keep));
    }
  }
  __syncthreads();
  const size_t base = (size_t)gq0 * 4 * NB * 3;
  const int valid_f = min(GPB * 4, (ngroups - gq0) * 4);
  for (int i = tid; i < valid_f * NB * 3; i += blockDim.x) {
    const float v = lds[i];
    if (v != 0.0f) atomicAdd(&hist[base + i], v);
  }
}

template <int GPB, int V>
float run(const uchar4* binned, long n, const int* rows, long m,
          const float* grad, const float* hess, float* hist, int ngroups,
          int iters) {
  const int n_fblocks = (ngroups + GPB - 1) / GPB;
  long chunks = (2048 + n_fblocks - 1) / n_fblocks;
  long chunk = (m + chunks - 1) / chunks;
  if (chunk < 1024) chunk = 1024;
  chunks = (m + chunk - 1) / chunk;
  dim3 grid((unsigned)chunks, (unsigned)n_fblocks);
  size_t lds_bytes = (size_t)GPB * 4 * NB * 3 * sizeof(float);
  // warmup
  hipLaunchKernelGGL((hist_k<GPB, V>), grid, dim3(256), lds_bytes, 0, binned,
                     n, rows, m, grad, hess, hist, ngroups, chunk);
  CHECK(hipDeviceSynchronize());
  hipEvent_t a, b;
  hipEventCreate(&a);
  hipEventCreate(&b);
  hipEventRecord(a);
  for (int it = 0; it < iters; ++it)
    hipLaunchKernelGGL((hist_k<GPB, V>), grid, dim3(256), lds_bytes, 0, binned,
                       n, rows, m, grad, hess, hist, ngroups, chunk);
  hipEventRecord(b);
  CHECK(hipEventSynchronize(b));
  float ms;
  hipEventElapsedTime(&ms, a, b);
  return ms / iters;
}

int main() {
  const long n = 10'000'000;
  const int ngroups = 25;
  uchar4* binned;
  int* rows;
  float *grad, *hess, *hist;
  CHECK(hipMalloc(&binned, (size_t)ngroups * n * 4));
  CHECK(hipMalloc(&rows, n * 4));
  CHECK(hipMalloc(&grad, n * 4));
  CHECK(hipMalloc(&hess, n * 4));
  CHECK(hipMalloc(&hist, (size_t)ngroups * 4 * NB * 3 * 4));
  // init on host
  {
    std::vector<unsigned char> hb((size_t)ngroups * n * 4);
    srand(1);
    for (size_t i = 0; i < hb.size(); ++i) hb[i] = rand() % NB;
    CHECK(hipMemcpy(binned, hb.data(), hb.size(), hipMemcpyHostToDevice));
    std::vector<int> hr(n);
    for (long i = 0; i < n; ++i) hr[i] = (int)i;
    CHECK(hipMemcpy(rows, hr.data(), n * 4, hipMemcpyHostToDevice));
    std::vector<float> hg(n, 0.5f);
    CHECK(hipMemcpy(grad, hg.data(), n * 4, hipMemcpyHostToDevice));
    CHECK(hipMemcpy(hess, hg.data(), n * 4, hipMemcpyHostToDevice));
  }
  printf("V0 full  GPB4: %.2f ms\n", run<4, 0>(binned, n, rows, n, grad, hess, hist, ngroups, 5));
  printf("V1 loads GPB4: %.2f ms\n", run<4, 1>(binned, n, rows, n, grad, hess, hist, ngroups, 5));
  printf("V2 atom  GPB4: %.2f ms\n", run<4, 2>(binned, n, rows, n, grad, hess, hist, ngroups, 5));
  printf("V3 ilp4  GPB4: %.2f ms\n", run<4, 3>(binned, n, rows, n, grad, hess, hist, ngroups, 5));
  printf("V4 ilp8  GPB4: %.2f ms\n", run<4, 4>(binned, n, rows, n, grad, hess, hist, ngroups, 5));
  printf("V0 full  GPB2: %.2f ms\n", run<2, 0>(binned, n, rows, n, grad, hess, hist, ngroups, 5));
  printf("V3 ilp4  GPB2: %.2f ms\n", run<2, 3>(binned, n, rows, n, grad, hess, hist, ngroups, 5));
  printf("V4 ilp8  GPB2: %.2f ms\n", run<2, 4>(binned, n, rows, n, grad, hess, hist, ngroups, 5));
  printf("V3 ilp4  GPB8: %.2f ms\n", run<8, 3>(binned, n, rows, n, grad, hess, hist, ngroups, 5));
  printf("V4 ilp8  GPB8: %.2f ms\n", run<8, 4>(binned, n, rows, n, grad, hess, hist, ngroups, 5));
  return 0;
}
