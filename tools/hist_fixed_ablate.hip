// Ablation v2: the fixed-point u64 histogram kernel — GPB (occupancy) × ILP
// (row batching) sweep. The production kernel is latency-bound
// (SQ_WAIT_ANY >> SQ_BUSY); µbench says the u64-atomic floor is ~1.1 ms per
// full pass, measured 5.2 ms.
// Build: hipcc --offload-arch=gfx950 -O3 -std=c++17 tools/hist_fixed_ablate.hip -o tools/hist_fixed_ablate
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdlib>
#include <vector>

#define CHECK(x) do { hipError_t e=(x); if(e){printf("ERR %s @%d\n", hipGetErrorString(e), __LINE__); exit(1);} } while(0)

constexpr int NB = 255;

template <int GPB, int ILP>
__global__ void histf_k(const uchar4* __restrict__ binned, long n_rows,
                        const int* __restrict__ rows, long m,
                        const float* __restrict__ grad,
                        const float* __restrict__ hess,
                        long long* __restrict__ hist, int ngroups, long chunk,
                        double sg, double sh) {
  extern __shared__ unsigned long long lds64[];
  const int tid = threadIdx.x;
  const int lds_elems = GPB * 4 * NB * 2;
  for (int i = tid; i < lds_elems; i += blockDim.x) lds64[i] = 0ull;
  __syncthreads();
  const int gq0 = blockIdx.y * GPB;
  const long start = (long)blockIdx.x * chunk;
  const long end = min(start + chunk, m);
  constexpr unsigned long long CNT_ONE = 1ull << 44;

  long i = start + tid;
  for (; i + (ILP - 1) * (long)blockDim.x < end; i += ILP * blockDim.x) {
    int r[ILP];
    long long gq[ILP];
    unsigned long long hq[ILP];
#pragma unroll
    for (int u = 0; u < ILP; ++u) r[u] = rows[i + u * blockDim.x];
#pragma unroll
    for (int u = 0; u < ILP; ++u) {
      gq[u] = (long long)llrint((double)grad[r[u]] * sg);
      hq[u] = CNT_ONE | (unsigned long long)llrint((double)hess[r[u]] * sh);
    }
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
          unsigned long long* cell =
              &lds64[((q * 4 + j) * NB + bs[j]) * 2];
          atomicAdd(cell + 0, (unsigned long long)gq[u]);
          atomicAdd(cell + 1, hq[u]);
        }
      }
    }
  }
  for (; i < end; i += blockDim.x) {
    const int r = rows[i];
    const long long gq = (long long)llrint((double)grad[r] * sg);
    const unsigned long long hq =
        CNT_ONE | (unsigned long long)llrint((double)hess[r] * sh);
    for (int q = 0; q < GPB; ++q) {
      const int grp = gq0 + q;
      if (grp >= ngroups) break;
      const uchar4 b4 = binned[(size_t)grp * n_rows + r];
      const unsigned char bs[4] = {b4.x, b4.y, b4.z, b4.w};
      for (int j = 0; j < 4; ++j) {
        unsigned long long* cell = &lds64[((q * 4 + j) * NB + bs[j]) * 2];
        atomicAdd(cell + 0, (unsigned long long)gq);
        atomicAdd(cell + 1, hq);
      }
    }
  }
  __syncthreads();
  const int valid_f = min(GPB * 4, (ngroups - gq0) * 4);
  for (int k = tid; k < valid_f * NB; k += blockDim.x) {
    const int f = k / NB, b = k % NB;
    const unsigned long long gsum = lds64[(f * NB + b) * 2 + 0];
    const unsigned long long hp = lds64[(f * NB + b) * 2 + 1];
    if (gsum == 0ull && hp == 0ull) continue;
    long long* out = hist + ((size_t)(gq0 * 4 + f) * NB + b) * 3;
    atomicAdd((unsigned long long*)(out + 0), gsum);
    atomicAdd((unsigned long long*)(out + 1), hp & ((1ull << 44) - 1));
    atomicAdd((unsigned long long*)(out + 2), hp >> 44);
  }
}

template <int GPB, int ILP, int BLK = 256>
float run(const uchar4* binned, long n, const int* rows, long m,
          const float* grad, const float* hess, long long* hist, int ngroups,
          int iters) {
  const int n_fblocks = (ngroups + GPB - 1) / GPB;
  long chunks = (2048 + n_fblocks - 1) / n_fblocks;
  long chunk = (m + chunks - 1) / chunks;
  if (chunk < 1024) chunk = 1024;
  if (chunk > (1l << 19)) chunk = 1l << 19;
  chunks = (m + chunk - 1) / chunk;
  dim3 grid((unsigned)chunks, (unsigned)n_fblocks);
  size_t lds = (size_t)GPB * 4 * NB * 2 * 8;
  hipLaunchKernelGGL((histf_k<GPB, ILP>), grid, dim3(BLK), lds, 0, binned, n,
                     rows, m, grad, hess, hist, ngroups, chunk, 1e9, 1e6);
  CHECK(hipDeviceSynchronize());
  hipEvent_t a, b;
  (void)hipEventCreate(&a); (void)hipEventCreate(&b);
  (void)hipEventRecord(a);
  for (int it = 0; it < iters; ++it)
    hipLaunchKernelGGL((histf_k<GPB, ILP>), grid, dim3(BLK), lds, 0, binned,
                       n, rows, m, grad, hess, hist, ngroups, chunk, 1e9, 1e6);
  (void)hipEventRecord(b);
  CHECK(hipEventSynchronize(b));
  float ms;
  (void)hipEventElapsedTime(&ms, a, b);
  return ms / iters;
}

int main() {
  const long n = 10'000'000;
  const int ngroups = 25;
  uchar4* binned; int* rows; float *grad, *hess; long long* hist;
  CHECK(hipMalloc(&binned, (size_t)ngroups * n * 4));
  CHECK(hipMalloc(&rows, n * 4));
  CHECK(hipMalloc(&grad, n * 4));
  CHECK(hipMalloc(&hess, n * 4));
  CHECK(hipMalloc(&hist, (size_t)ngroups * 4 * NB * 3 * 8));
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
  printf("GPB2 ILP1 (prod): %.2f ms\n", run<2, 1>(binned, n, rows, n, grad, hess, hist, ngroups, 5));
  printf("GPB2 ILP4       : %.2f ms\n", run<2, 4>(binned, n, rows, n, grad, hess, hist, ngroups, 5));
  printf("GPB2 ILP8       : %.2f ms\n", run<2, 8>(binned, n, rows, n, grad, hess, hist, ngroups, 5));
  printf("GPB1 ILP1       : %.2f ms\n", run<1, 1>(binned, n, rows, n, grad, hess, hist, ngroups, 5));
  printf("GPB1 ILP4       : %.2f ms\n", run<1, 4>(binned, n, rows, n, grad, hess, hist, ngroups, 5));
  printf("GPB1 ILP8       : %.2f ms\n", run<1, 8>(binned, n, rows, n, grad, hess, hist, ngroups, 5));
  printf("GPB4 ILP1       : %.2f ms\n", run<4, 1>(binned, n, rows, n, grad, hess, hist, ngroups, 5));
  printf("GPB4 ILP4       : %.2f ms\n", run<4, 4>(binned, n, rows, n, grad, hess, hist, ngroups, 5));
  printf("GPB2 ILP4 512thr: %.2f ms\n", (run<2, 4, 512>(binned, n, rows, n, grad, hess, hist, ngroups, 5)));
  printf("GPB1 ILP4 512thr: %.2f ms\n", (run<1, 4, 512>(binned, n, rows, n, grad, hess, hist, ngroups, 5)));
  // half-data subset (leaf-like): sorted stride-2 rows
  {
    std::vector<int> hr(n / 2);
    for (long i = 0; i < n / 2; ++i) hr[i] = (int)(2 * i);
    CHECK(hipMemcpy(rows, hr.data(), hr.size() * 4, hipMemcpyHostToDevice));
    printf("half GPB2 ILP1  : %.2f ms\n", run<2, 1>(binned, n, rows, n / 2, grad, hess, hist, ngroups, 5));
    printf("half GPB1 ILP4  : %.2f ms\n", run<1, 4>(binned, n, rows, n / 2, grad, hess, hist, ngroups, 5));
  }
  return 0;
}
