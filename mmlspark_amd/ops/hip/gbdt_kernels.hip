// CDNA4 (gfx950 / MI355X) kernels for the GBDT path — written HIP-first.
//
// These implement, natively on MI355X, the compute the reference delegates to
// LightGBM's CPU library behind LGBM_BoosterUpdateOneIter /
// LGBM_BoosterPredictForMat: per-leaf feature-histogram build,
// ensemble traversal scoring, and quantile binning.
//
// Design notes (cf. /opt/skills/guides/cdna_hip_programming.md):
//  * wave = 64; blocks are 256 threads (4 waves).
//  * hist_build: LDS-staged per-(feature,bin) accumulation — each workgroup
//    owns GPB*4 features × n_bins bins in LDS (~49 KB at GPB=4, 3 blocks/CU)
//    and a contiguous row chunk; flush with device atomics. Binned data is
//    feature-interleaved uchar4 so one 4-byte load yields 4 features' bins.
//  * grid is sized ≫256 workgroups (row-chunk × feature-block 2-D grid) to
//    fill 8 XCDs; row lists are kept sorted so grad/hess gathers are
//    near-coalesced.
#include <hip/hip_runtime.h>
#include <cstdint>
#include <cstdlib>

#define WAVE 64
#define DEV_INLINE __device__ __forceinline__

// ---------------------------------------------------------------- histogram
// binned: (ngroups, n_rows) uchar4; rows: (m,) i32 sorted; grad/hess: (n_rows,)
// hist:   (ngroups*4, n_bins, 3) f32, pre-zeroed.
template <int GPB>
__global__ void hist_build_k(const uchar4* __restrict__ binned, long n_rows,
                             const int* __restrict__ rows, long m,
                             const float* __restrict__ grad,
                             const float* __restrict__ hess,
                             float* __restrict__ hist, int n_bins,
                             int ngroups, long chunk) {
  extern __shared__ float lds[];  // [GPB*4][n_bins][3]
  const int tid = threadIdx.x;
  const int nfb = GPB * 4;
  const int lds_elems = nfb * n_bins * 3;
  for (int i = tid; i < lds_elems; i += blockDim.x) lds[i] = 0.0f;
  __syncthreads();

  const int gq0 = blockIdx.y * GPB;  // first feature-group of this block
  const long start = (long)blockIdx.x * chunk;
  const long end = min(start + chunk, m);

  for (long i = start + tid; i < end; i += blockDim.x) {
    const int r = rows[i];
    const float g = grad[r];
    const float h = hess[r];
#pragma unroll
    for (int q = 0; q < GPB; ++q) {
      const int grp = gq0 + q;
      if (grp >= ngroups) break;
      const uchar4 b4 = binned[(size_t)grp * n_rows + r];
      const unsigned char bs[4] = {b4.x, b4.y, b4.z, b4.w};
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        float* cell = &lds[((q * 4 + j) * n_bins + bs[j]) * 3];
        atomicAdd(cell + 0, g);
        atomicAdd(cell + 1, h);
        atomicAdd(cell + 2, 1.0f);
      }
    }
  }
  __syncthreads();

  // flush LDS histograms to global with device atomics (row chunks overlap)
  const size_t base = (size_t)gq0 * 4 * n_bins * 3;
  const int valid_f = min(nfb, (ngroups - gq0) * 4);
  const int flush_elems = valid_f * n_bins * 3;
  for (int i = tid; i < flush_elems; i += blockDim.x) {
    const float v = lds[i];
    if (v != 0.0f) atomicAdd(&hist[base + i], v);
  }
}

// ---------------------------------------------------- fixed-point histogram
// gfx950 measured (tools/atomic_bench): random-address ds_add_f32 runs at
// ~200 G lane-ops/s while integer ds_add_u64 runs at ~1.8 T — 9×.  So the
// production histogram accumulates FIXED-POINT integers: per (feature,bin)
// two u64 cells, [g_fixed] and [count:20|h_fixed:44] (hessians are ≥0 so the
// packed add never borrows; chunk ≤ 2^20 rows bounds the count field; scales
// are chosen per iteration from max|g|, max h so per-chunk sums fit).
// Bonus: integer histograms make multi-rank all_reduce and the sibling
// subtraction trick bit-exact.
template <int GPB>
__global__ void hist_build_fixed_k(const uchar4* __restrict__ binned,
                                   long n_rows, const int* __restrict__ rows,
                                   long m, const float* __restrict__ grad,
                                   const float* __restrict__ hess,
                                   long long* __restrict__ hist, int n_bins,
                                   int ngroups, long chunk, double scale_g,
                                   double scale_h,
                                   const int* __restrict__ nl_dev, int side) {
  // side >= 0: child-of-partition mode — m is the PARENT row count (an
  // upper bound for grid sizing) and the actual child range comes from the
  // device-side left count nl_dev[0]: left = rows[0, nl), right = rows[nl,
  // m).  Lets the distributed grower enqueue the smaller child's histogram
  // without a host readback of the local partition count.
  long base = 0, m_eff = m;
  if (side >= 0) {
    const long nl = nl_dev[0];
    m_eff = (side == 0) ? nl : m - nl;
    base = (side == 0) ? 0 : nl;
  }
  if ((long)blockIdx.x * chunk >= m_eff) return;  // surplus block: no work
  rows += base;
  extern __shared__ unsigned long long lds64[];  // [GPB*4][n_bins][2]
  const int tid = threadIdx.x;
  const int nfb = GPB * 4;
  const int lds_elems = nfb * n_bins * 2;
  for (int i = tid; i < lds_elems; i += blockDim.x) lds64[i] = 0ull;
  __syncthreads();

  const int gq0 = blockIdx.y * GPB;
  const long start = (long)blockIdx.x * chunk;
  const long end = min(start + chunk, m_eff);
  constexpr unsigned long long CNT_ONE = 1ull << 44;

  constexpr int ILP = 4;  // probe: +10% via deeper load batching
  long i = start + tid;
  for (; i + (ILP - 1) * (long)blockDim.x < end; i += ILP * blockDim.x) {
    int r[ILP];
    long long gq[ILP];
    unsigned long long hq[ILP];
#pragma unroll
    for (int u = 0; u < ILP; ++u) r[u] = rows[i + u * blockDim.x];
#pragma unroll
    for (int u = 0; u < ILP; ++u) {
      gq[u] = (long long)llrint((double)grad[r[u]] * scale_g);
      hq[u] = CNT_ONE
              | (unsigned long long)llrint((double)hess[r[u]] * scale_h);
    }
#pragma unroll
    for (int q = 0; q < GPB; ++q) {
      const int grp = gq0 + q;
      if (grp >= ngroups) break;
      uchar4 b4[ILP];
#pragma unroll
      for (int u = 0; u < ILP; ++u)
        b4[u] = binned[(size_t)grp * n_rows + r[u]];
#pragma unroll
      for (int u = 0; u < ILP; ++u) {
        const unsigned char bs[4] = {b4[u].x, b4[u].y, b4[u].z, b4[u].w};
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          unsigned long long* cell =
              &lds64[((q * 4 + j) * n_bins + bs[j]) * 2];
          atomicAdd(cell + 0, (unsigned long long)gq[u]);
          atomicAdd(cell + 1, hq[u]);
        }
      }
    }
  }
  for (; i < end; i += blockDim.x) {  // tail
    const int r = rows[i];
    const long long gq = (long long)llrint((double)grad[r] * scale_g);
    const unsigned long long hq =
        CNT_ONE | (unsigned long long)llrint((double)hess[r] * scale_h);
#pragma unroll
    for (int q = 0; q < GPB; ++q) {
      const int grp = gq0 + q;
      if (grp >= ngroups) break;
      const uchar4 b4 = binned[(size_t)grp * n_rows + r];
      const unsigned char bs[4] = {b4.x, b4.y, b4.z, b4.w};
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        unsigned long long* cell = &lds64[((q * 4 + j) * n_bins + bs[j]) * 2];
        atomicAdd(cell + 0, (unsigned long long)gq);
        atomicAdd(cell + 1, hq);
      }
    }
  }
  __syncthreads();

  // flush: unpack (count | h) and add to global int64 hist (nf, n_bins, 3)
  const int valid_f = min(nfb, (ngroups - gq0) * 4);
  for (int i = tid; i < valid_f * n_bins; i += blockDim.x) {
    const int f = i / n_bins;
    const int b = i % n_bins;
    const unsigned long long gsum = lds64[(f * n_bins + b) * 2 + 0];
    const unsigned long long hpacked = lds64[(f * n_bins + b) * 2 + 1];
    if (gsum == 0ull && hpacked == 0ull) continue;
    const unsigned long long cnt = hpacked >> 44;
    const unsigned long long hsum = hpacked & ((1ull << 44) - 1ull);
    long long* out = hist + ((size_t)(gq0 * 4 + f) * n_bins + b) * 3;
    atomicAdd((unsigned long long*)(out + 0), gsum);
    atomicAdd((unsigned long long*)(out + 1), hsum);
    atomicAdd((unsigned long long*)(out + 2), cnt);
  }
}

// --------------------------------------------- paired-plane fixed histogram
// binned_pair: (npairs, n_rows) u64, low 4B = group 2p's uchar4, high 4B =
// group 2p+1's.  A gathered row costs ONE cacheline per feature-chunk block
// instead of two (the two planes of a GPB=2 block are ~40 MB apart in the
// plane-major layout, so every sparse row paid two line fetches; measured
// 2.3x gather penalty on mid-size leaves, tools/hist_gather_probe.py).
// Trailing-group padding (odd ngroups) accumulates into pad features that
// the split scan masks out via nf_real.
// GHQ=true: grad/hess come pre-quantized as one 16B record per row
// ((gq:i64, CNT|hq:u64) in ghq) — each pair-block then gathers 2 cachelines
// per sparse row (binned + ghq) instead of 3 (binned + grad + hess), and
// the double->fixed conversion runs once per row instead of once per
// (row, pair-block).
template <bool GHQ>
__global__ void hist_build_fixed_pair_k(
    const unsigned long long* __restrict__ binned_pair, long n_rows,
    const int* __restrict__ rows, long m, const float* __restrict__ grad,
    const float* __restrict__ hess, const longlong2* __restrict__ ghq,
    long long* __restrict__ hist, int n_bins,
    int npairs, int tail_bytes, long chunk, double scale_g, double scale_h,
    const int* __restrict__ nl_dev, int side) {
  long base = 0, m_eff = m;
  if (side >= 0) {
    const long nl = nl_dev[0];
    m_eff = (side == 0) ? nl : m - nl;
    base = (side == 0) ? 0 : nl;
  }
  if ((long)blockIdx.x * chunk >= m_eff) return;
  rows += base;
  extern __shared__ unsigned long long lds64[];  // [8][n_bins][2]
  const int tid = threadIdx.x;
  const int lds_elems = 8 * n_bins * 2;
  for (int i = tid; i < lds_elems; i += blockDim.x) lds64[i] = 0ull;
  __syncthreads();

  const int pair = blockIdx.y;
  // zero-padded bytes of the last pair would funnel EVERY row into bin 0 of
  // the pad features — a same-address LDS-atomic hotspot — so skip them
  const int jmax = (pair == npairs - 1) ? tail_bytes : 8;
  const unsigned long long* plane = binned_pair + (size_t)pair * n_rows;
  const long start = (long)blockIdx.x * chunk;
  const long end = min(start + chunk, m_eff);
  constexpr unsigned long long CNT_ONE = 1ull << 44;

  constexpr int ILP = 4;
  long i = start + tid;
  for (; i + (ILP - 1) * (long)blockDim.x < end; i += ILP * blockDim.x) {
    int r[ILP];
    long long gq[ILP];
    unsigned long long hq[ILP], v[ILP];
#pragma unroll
    for (int u = 0; u < ILP; ++u) r[u] = rows[i + u * blockDim.x];
#pragma unroll
    for (int u = 0; u < ILP; ++u) v[u] = plane[r[u]];
#pragma unroll
    for (int u = 0; u < ILP; ++u) {
      if (GHQ) {
        const longlong2 q = ghq[r[u]];
        gq[u] = q.x;
        hq[u] = (unsigned long long)q.y;
      } else {
        gq[u] = (long long)llrint((double)grad[r[u]] * scale_g);
        hq[u] = CNT_ONE
                | (unsigned long long)llrint((double)hess[r[u]] * scale_h);
      }
    }
#pragma unroll
    for (int u = 0; u < ILP; ++u) {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        if (j >= jmax) break;
        const int b = (int)((v[u] >> (8 * j)) & 0xffull);
        unsigned long long* cell = &lds64[(j * n_bins + b) * 2];
        atomicAdd(cell + 0, (unsigned long long)gq[u]);
        atomicAdd(cell + 1, hq[u]);
      }
    }
  }
  for (; i < end; i += blockDim.x) {  // tail
    const int r = rows[i];
    const unsigned long long v = plane[r];
    long long gq;
    unsigned long long hq;
    if (GHQ) {
      const longlong2 q = ghq[r];
      gq = q.x;
      hq = (unsigned long long)q.y;
    } else {
      gq = (long long)llrint((double)grad[r] * scale_g);
      hq = CNT_ONE | (unsigned long long)llrint((double)hess[r] * scale_h);
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      if (j >= jmax) break;
      const int b = (int)((v >> (8 * j)) & 0xffull);
      unsigned long long* cell = &lds64[(j * n_bins + b) * 2];
      atomicAdd(cell + 0, (unsigned long long)gq);
      atomicAdd(cell + 1, hq);
    }
  }
  __syncthreads();

  // flush to (npairs*8, n_bins, 3) int64
  for (int i = tid; i < jmax * n_bins; i += blockDim.x) {
    const int f = i / n_bins;
    const int b = i % n_bins;
    const unsigned long long gsum = lds64[(f * n_bins + b) * 2 + 0];
    const unsigned long long hpacked = lds64[(f * n_bins + b) * 2 + 1];
    if (gsum == 0ull && hpacked == 0ull) continue;
    const unsigned long long cnt = hpacked >> 44;
    const unsigned long long hsum = hpacked & ((1ull << 44) - 1ull);
    long long* out = hist + ((size_t)(pair * 8 + f) * n_bins + b) * 3;
    atomicAdd((unsigned long long*)(out + 0), gsum);
    atomicAdd((unsigned long long*)(out + 1), hsum);
    atomicAdd((unsigned long long*)(out + 2), cnt);
  }
}

// one-shot per tree: quantize grad/hess to (gq, CNT|hq) 16B records
__global__ void quantize_gh_k(const float* __restrict__ grad,
                              const float* __restrict__ hess, long n,
                              double scale_g, double scale_h,
                              longlong2* __restrict__ out) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  longlong2 q;
  q.x = (long long)llrint((double)grad[i] * scale_g);
  q.y = (long long)((1ull << 44)
                    | (unsigned long long)llrint((double)hess[i] * scale_h));
  out[i] = q;
}

extern "C" void launch_quantize_gh(const float* grad, const float* hess,
                                   long n, double scale_g, double scale_h,
                                   void* out, hipStream_t stream) {
  if (n == 0) return;
  const long blocks = (n + 255) / 256;
  hipLaunchKernelGGL(quantize_gh_k, dim3((unsigned)blocks), dim3(256), 0,
                     stream, grad, hess, n, scale_g, scale_h,
                     (longlong2*)out);
}

extern "C" void launch_hist_build_fixed_pair(
    const void* binned_pair, long n_rows, const int* rows, long m,
    const float* grad, const float* hess, const void* ghq, long long* hist,
    int n_bins, int npairs, int tail_bytes, double scale_g, double scale_h,
    const int* nl_dev, int side, hipStream_t stream) {
  if (m == 0) return;
  long chunks = (2048 + npairs - 1) / npairs;
  long chunk = (m + chunks - 1) / chunks;
  if (chunk < 16384) {
    chunk = (m + 7) / 8;
    if (chunk < 2048) chunk = 2048;
    if (chunk > 16384) chunk = 16384;
  }
  if (chunk > (1l << 19)) chunk = 1l << 19;
  chunks = (m + chunk - 1) / chunk;
  dim3 grid((unsigned)chunks, (unsigned)npairs);
  const size_t lds_bytes = (size_t)8 * n_bins * 2 * sizeof(long long);
  if (ghq) {
    hipLaunchKernelGGL(hist_build_fixed_pair_k<true>, grid, dim3(256),
                       lds_bytes, stream,
                       (const unsigned long long*)binned_pair, n_rows, rows,
                       m, grad, hess, (const longlong2*)ghq, hist, n_bins,
                       npairs, tail_bytes, chunk, scale_g, scale_h, nl_dev,
                       side);
  } else {
    hipLaunchKernelGGL(hist_build_fixed_pair_k<false>, grid, dim3(256),
                       lds_bytes, stream,
                       (const unsigned long long*)binned_pair, n_rows, rows,
                       m, grad, hess, nullptr, hist, n_bins, npairs,
                       tail_bytes, chunk, scale_g, scale_h, nl_dev, side);
  }
}

extern "C" void launch_hist_build_fixed_child(
    const void* binned, long n_rows, const int* rows, long m,
    const float* grad, const float* hess, long long* hist, int n_bins,
    int ngroups, double scale_g, double scale_h, const int* nl_dev, int side,
    hipStream_t stream) {
  if (m == 0) return;
  constexpr int GPB = 2;
  const int n_fblocks = (ngroups + GPB - 1) / GPB;
  long chunks = (2048 + n_fblocks - 1) / n_fblocks;
  long chunk = (m + chunks - 1) / chunks;
  if (chunk < 16384) {
    chunk = (m + 7) / 8;
    if (chunk < 2048) chunk = 2048;
    if (chunk > 16384) chunk = 16384;
  }
  if (chunk > (1l << 19)) chunk = 1l << 19;
  chunks = (m + chunk - 1) / chunk;  // m = parent size: grid upper bound
  dim3 grid((unsigned)chunks, (unsigned)n_fblocks);
  const size_t lds_bytes = (size_t)GPB * 4 * n_bins * 2 * sizeof(long long);
  hipLaunchKernelGGL((hist_build_fixed_k<GPB>), grid, dim3(256), lds_bytes,
                     stream, (const uchar4*)binned, n_rows, rows, m, grad,
                     hess, hist, n_bins, ngroups, chunk, scale_g, scale_h,
                     nl_dev, side);
}

extern "C" void launch_hist_build_fixed(const void* binned, long n_rows,
                                        const int* rows, long m,
                                        const float* grad, const float* hess,
                                        long long* hist, int n_bins,
                                        int ngroups, double scale_g,
                                        double scale_h, hipStream_t stream) {
  if (m == 0) return;
  constexpr int GPB = 2;
  const int n_fblocks = (ngroups + GPB - 1) / GPB;
  long chunks = (2048 + n_fblocks - 1) / n_fblocks;
  long chunk = (m + chunks - 1) / chunks;
  // every block flushes its whole LDS histogram with GLOBAL atomics, so
  // large leaves must not shatter into hundreds of row-chunks (profiled:
  // flush contention made calls ~0.3 ms at ~157 chunks) — but a SMALL leaf
  // at one coarse chunk uses only n_fblocks workgroups of a 256-CU chip, so
  // give small leaves at least 8 row-chunks (8-way flush contention is
  // negligible; measured floor drops ~3x)
  if (chunk < 16384) {
    chunk = (m + 7) / 8;
    if (chunk < 2048) chunk = 2048;
    if (chunk > 16384) chunk = 16384;
  }
  if (chunk > (1l << 19)) chunk = 1l << 19;  // count-field bound (< 2^20)
  chunks = (m + chunk - 1) / chunk;
  dim3 grid((unsigned)chunks, (unsigned)n_fblocks);
  const size_t lds_bytes = (size_t)GPB * 4 * n_bins * 2 * sizeof(long long);
  hipLaunchKernelGGL((hist_build_fixed_k<GPB>), grid, dim3(256), lds_bytes,
                     stream, (const uchar4*)binned, n_rows, rows, m, grad,
                     hess, hist, n_bins, ngroups, chunk, scale_g, scale_h,
                     nullptr, -1);
}

static int hist_gpb_env() {
  const char* e = getenv("MMLSPARK_HIST_GPB");
  if (e) {
    int v = atoi(e);
    if (v == 1 || v == 2 || v == 4 || v == 8) return v;
  }
  return 4;
}

extern "C" void launch_hist_build(const void* binned, long n_rows,
                                  const int* rows, long m, const float* grad,
                                  const float* hess, float* hist, int n_bins,
                                  int ngroups, hipStream_t stream) {
  if (m == 0) return;
  const int GPB = hist_gpb_env();
  const int n_fblocks = (ngroups + GPB - 1) / GPB;
  // target ≥ 2048 workgroups total to fill 256 CUs × 8 XCDs
  long chunks = (2048 + n_fblocks - 1) / n_fblocks;
  long min_chunk = 1024;  // enough rows per block to amortize the LDS flush
  long chunk = (m + chunks - 1) / chunks;
  if (chunk < min_chunk) chunk = min_chunk;
  chunks = (m + chunk - 1) / chunk;
  dim3 grid((unsigned)chunks, (unsigned)n_fblocks);
  const size_t lds_bytes = (size_t)GPB * 4 * n_bins * 3 * sizeof(float);
#define LAUNCH_HB(G) \
    hipLaunchKernelGGL((hist_build_k<G>), grid, dim3(256), lds_bytes, stream, \
                       (const uchar4*)binned, n_rows, rows, m, grad, hess,    \
                       hist, n_bins, ngroups, chunk)
  switch (GPB) {
    case 1: LAUNCH_HB(1); break;
    case 2: LAUNCH_HB(2); break;
    case 8: LAUNCH_HB(8); break;
    default: LAUNCH_HB(4); break;
  }
#undef LAUNCH_HB
}

// -------------------------------------------------------- ordered partition
// Stable row partition (left = bin[feature] <= thr) in 3 async launches +
// one count readback — replaces torch's two masked_selects whose internal
// nonzero() forced TWO device→host syncs per split (profiled at ~13 ms/iter
// host time).  Output preserves ascending row order on both sides, keeping
// later grad/hess gathers coalesced.
__global__ void part_count_k(const uchar4* __restrict__ binned, long n_rows,
                             const int* __restrict__ rows, long m, int grp,
                             int j, int thr, long chunk,
                             int* __restrict__ block_counts) {
  const long start = (long)blockIdx.x * chunk;
  const long end = min(start + chunk, m);
  int cnt = 0;
  for (long i = start + threadIdx.x; i < end; i += blockDim.x) {
    const uchar4 b4 = binned[(size_t)grp * n_rows + rows[i]];
    const unsigned char b = j == 0 ? b4.x : j == 1 ? b4.y : j == 2 ? b4.z : b4.w;
    cnt += (b <= thr);
  }
  __shared__ int sh[256];
  sh[threadIdx.x] = cnt;
  __syncthreads();
  for (int d = 128; d > 0; d >>= 1) {
    if (threadIdx.x < d) sh[threadIdx.x] += sh[threadIdx.x + d];
    __syncthreads();
  }
  if (threadIdx.x == 0) block_counts[blockIdx.x] = sh[0];
}

__global__ void part_scan_k(int* __restrict__ block_counts, int n_blocks,
                            int* __restrict__ total_left) {
  // single block: exclusive scan of block_counts in place
  __shared__ int carry;
  __shared__ int sh[256];
  if (threadIdx.x == 0) carry = 0;
  __syncthreads();
  for (int base = 0; base < n_blocks; base += blockDim.x) {
    const int i = base + threadIdx.x;
    const int v = (i < n_blocks) ? block_counts[i] : 0;
    sh[threadIdx.x] = v;
    __syncthreads();
    for (int d = 1; d < 256; d <<= 1) {  // Hillis-Steele inclusive scan
      const int add = threadIdx.x >= d ? sh[threadIdx.x - d] : 0;
      __syncthreads();
      sh[threadIdx.x] += add;
      __syncthreads();
    }
    if (i < n_blocks) block_counts[i] = carry + sh[threadIdx.x] - v;
    __syncthreads();
    if (threadIdx.x == 0) carry += sh[255];
    __syncthreads();
  }
  if (threadIdx.x == 0) *total_left = carry;
}

__global__ void part_scatter_k(const uchar4* __restrict__ binned, long n_rows,
                               const int* __restrict__ rows, long m, int grp,
                               int j, int thr, long chunk,
                               const int* __restrict__ block_offsets,
                               const int* __restrict__ total_left,
                               int* __restrict__ out) {
  const long start = (long)blockIdx.x * chunk;
  const long end = min(start + chunk, m);
  const long nl_total = *total_left;
  __shared__ long base_l, base_r;
  __shared__ int wave_l[4], wave_r[4];
  if (threadIdx.x == 0) {
    base_l = block_offsets[blockIdx.x];
    base_r = nl_total + (start - block_offsets[blockIdx.x]);
  }
  __syncthreads();
  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  for (long i0 = start; i0 < end; i0 += blockDim.x) {
    const long i = i0 + threadIdx.x;
    int r = 0;
    bool valid = i < end, left = false;
    if (valid) {
      r = rows[i];
      const uchar4 b4 = binned[(size_t)grp * n_rows + r];
      const unsigned char b = j == 0 ? b4.x : j == 1 ? b4.y : j == 2 ? b4.z : b4.w;
      left = (b <= thr);
    }
    const unsigned long long mask_l = __ballot(valid && left);
    const unsigned long long mask_r = __ballot(valid && !left);
    const unsigned long long lt = (1ull << lane) - 1ull;
    if (threadIdx.x == (unsigned)(wid << 6)) {  // lane 0 of each wave
      wave_l[wid] = __popcll(mask_l);
      wave_r[wid] = __popcll(mask_r);
    }
    __syncthreads();
    long wl = base_l, wr = base_r;
    for (int w = 0; w < wid; ++w) {
      wl += wave_l[w];
      wr += wave_r[w];
    }
    if (valid) {
      if (left) out[wl + __popcll(mask_l & lt)] = r;
      else out[wr + __popcll(mask_r & lt)] = r;
    }
    __syncthreads();
    if (threadIdx.x == 0) {
      base_l += wave_l[0] + wave_l[1] + wave_l[2] + wave_l[3];
      base_r += wave_r[0] + wave_r[1] + wave_r[2] + wave_r[3];
    }
    __syncthreads();
  }
}

// ------------------------------------------------------------- arena mode
// Per-tree physical repartitioning: the tree's working set lives in a
// double-buffered ARENA of paired bin planes + pre-quantized (gq,hq)
// records + original row ids.  Every split SCATTERS the parent's segment
// into the other buffer (left|right, stable), so child histograms read
// CONTIGUOUS rows — the sparse row gather that bounded child hist builds
// (tools/hist_gather_probe.py: 2.3x) disappears entirely, at the cost of
// moving ~124 B/row/split of arena payload (HBM streaming, ~8 TB/s).

__global__ void arena_gather_k(const unsigned long long* __restrict__ pair_src,
                               long n_src, const float* __restrict__ grad,
                               const float* __restrict__ hess,
                               const int* __restrict__ rows, long m,
                               int npairs, double scale_g, double scale_h,
                               unsigned long long* __restrict__ pair_dst,
                               longlong2* __restrict__ ghq_dst,
                               int* __restrict__ rowid_dst) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= m) return;
  const int r = rows[i];
  rowid_dst[i] = r;
  longlong2 q;
  q.x = (long long)llrint((double)grad[r] * scale_g);
  q.y = (long long)((1ull << 44)
                    | (unsigned long long)llrint((double)hess[r] * scale_h));
  ghq_dst[i] = q;
  for (int p = 0; p < npairs; ++p)
    pair_dst[(size_t)p * m + i] = pair_src[(size_t)p * n_src + r];
}

extern "C" void launch_arena_gather(const void* pair_src, long n_src,
                                    const float* grad, const float* hess,
                                    const int* rows, long m, int npairs,
                                    double scale_g, double scale_h,
                                    void* pair_dst, void* ghq_dst,
                                    int* rowid_dst, hipStream_t stream) {
  if (m == 0) return;
  const long blocks = (m + 255) / 256;
  hipLaunchKernelGGL(arena_gather_k, dim3((unsigned)blocks), dim3(256), 0,
                     stream, (const unsigned long long*)pair_src, n_src, grad,
                     hess, rows, m, npairs, scale_g, scale_h,
                     (unsigned long long*)pair_dst, (longlong2*)ghq_dst,
                     rowid_dst);
}

__global__ void part_arena_count_k(const unsigned long long* __restrict__ pair,
                                   long n_arena, long lo, long m, int pf,
                                   int jbyte, int thr,
                                   const unsigned* __restrict__ cat_bits,
                                   long chunk,
                                   int* __restrict__ block_counts) {
  const long start = (long)blockIdx.x * chunk;
  const long end = min(start + chunk, m);
  const unsigned long long* plane = pair + (size_t)pf * n_arena + lo;
  int cnt = 0;
  for (long i = start + threadIdx.x; i < end; i += blockDim.x) {
    const unsigned bv = (unsigned)((plane[i] >> (8 * jbyte)) & 0xffull);
    cnt += (int)(cat_bits ? ((cat_bits[bv >> 5] >> (bv & 31)) & 1u)
                          : (unsigned)(bv <= (unsigned)thr));
  }
  __shared__ int sh[256];
  sh[threadIdx.x] = cnt;
  __syncthreads();
  for (int d = 128; d > 0; d >>= 1) {
    if (threadIdx.x < d) sh[threadIdx.x] += sh[threadIdx.x + d];
    __syncthreads();
  }
  if (threadIdx.x == 0) block_counts[blockIdx.x] = sh[0];
}

// pass 1: per-row destination index (stable left|right ranks) + move the
// small payloads (row id + quantized gh); dst_idx drives the plane copies
__global__ void part_arena_index_k(
    const unsigned long long* __restrict__ pair_src,
    const longlong2* __restrict__ ghq_src, const int* __restrict__ rowid_src,
    longlong2* __restrict__ ghq_dst, int* __restrict__ rowid_dst,
    int* __restrict__ dst_idx, long n_arena, long lo, long m, int pf,
    int jbyte, int thr, const unsigned* __restrict__ cat_bits, long chunk,
    const int* __restrict__ block_offsets,
    const int* __restrict__ total_left) {
  const long start = (long)blockIdx.x * chunk;
  const long end = min(start + chunk, m);
  const long nl_total = *total_left;
  __shared__ long base_l, base_r;
  __shared__ int wave_l[4], wave_r[4];
  if (threadIdx.x == 0) {
    base_l = block_offsets[blockIdx.x];
    base_r = nl_total + (start - block_offsets[blockIdx.x]);
  }
  __syncthreads();
  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const unsigned long long* split_plane = pair_src + (size_t)pf * n_arena + lo;
  for (long i0 = start; i0 < end; i0 += blockDim.x) {
    const long i = i0 + threadIdx.x;
    bool valid = i < end, left = false;
    if (valid) {
      const unsigned bv = (unsigned)((split_plane[i] >> (8 * jbyte)) & 0xffull);
      left = cat_bits ? (((cat_bits[bv >> 5] >> (bv & 31)) & 1u) != 0u)
                      : (bv <= (unsigned)thr);
    }
    const unsigned long long mask_l = __ballot(valid && left);
    const unsigned long long mask_r = __ballot(valid && !left);
    const unsigned long long lt = (1ull << lane) - 1ull;
    if (lane == 0) {
      wave_l[wid] = __popcll(mask_l);
      wave_r[wid] = __popcll(mask_r);
    }
    __syncthreads();
    long wl = base_l, wr = base_r;
    for (int w = 0; w < wid; ++w) {
      wl += wave_l[w];
      wr += wave_r[w];
    }
    if (valid) {
      const long d = left ? wl + __popcll(mask_l & lt)
                          : wr + __popcll(mask_r & lt);  // relative to lo
      dst_idx[i] = (int)d;
      rowid_dst[lo + d] = rowid_src[lo + i];
      ghq_dst[lo + d] = ghq_src[lo + i];
    }
    __syncthreads();
    if (threadIdx.x == 0) {
      long al = 0, ar = 0;
      for (int w = 0; w < 4; ++w) {
        al += wave_l[w];
        ar += wave_r[w];
      }
      base_l += al;
      base_r += ar;
    }
    __syncthreads();
  }
}

// pass 2: plane copies driven by dst_idx — 2-D grid (row-chunk, plane) so
// all npairs planes stream in parallel with coalesced reads
__global__ void part_arena_copy_k(
    const unsigned long long* __restrict__ pair_src,
    unsigned long long* __restrict__ pair_dst,
    const int* __restrict__ dst_idx, long n_arena, long lo, long m,
    long chunk) {
  const long start = (long)blockIdx.x * chunk;
  const long end = min(start + chunk, m);
  const unsigned long long* src = pair_src + (size_t)blockIdx.y * n_arena + lo;
  unsigned long long* dst = pair_dst + (size_t)blockIdx.y * n_arena + lo;
  for (long i = start + threadIdx.x; i < end; i += blockDim.x)
    dst[dst_idx[i]] = src[i];
}

extern "C" void launch_partition_arena(
    const void* pair_src, const void* ghq_src, const int* rowid_src,
    void* pair_dst, void* ghq_dst, int* rowid_dst, int* dst_idx,
    long n_arena, int npairs, long lo, long m, int feature, int thr,
    const unsigned* cat_bits, int* scratch, int* total_left,
    hipStream_t stream) {
  if (m == 0) return;
  long chunk = 4096;
  long blocks = (m + chunk - 1) / chunk;
  if (blocks > 4096) {
    chunk = (m + 4095) / 4096;
    blocks = (m + chunk - 1) / chunk;
  }
  const int pf = feature / 8, jbyte = feature % 8;
  hipLaunchKernelGGL(part_arena_count_k, dim3((unsigned)blocks), dim3(256),
                     0, stream, (const unsigned long long*)pair_src, n_arena,
                     lo, m, pf, jbyte, thr, cat_bits, chunk, scratch);
  hipLaunchKernelGGL(part_scan_k, dim3(1), dim3(256), 0, stream, scratch,
                     (int)blocks, total_left);
  hipLaunchKernelGGL(part_arena_index_k, dim3((unsigned)blocks), dim3(256),
                     0, stream, (const unsigned long long*)pair_src,
                     (const longlong2*)ghq_src, rowid_src,
                     (longlong2*)ghq_dst, rowid_dst, dst_idx, n_arena, lo, m,
                     pf, jbyte, thr, cat_bits, chunk, scratch, total_left);
  long cchunk = 2048;
  long cblocks = (m + cchunk - 1) / cchunk;
  if (cblocks > 4096) {
    cchunk = (m + 4095) / 4096;
    cblocks = (m + cchunk - 1) / cchunk;
  }
  hipLaunchKernelGGL(part_arena_copy_k, dim3((unsigned)cblocks,
                                             (unsigned)npairs), dim3(256), 0,
                     stream, (const unsigned long long*)pair_src,
                     (unsigned long long*)pair_dst, dst_idx, n_arena, lo, m,
                     cchunk);
}

// range-mode histogram over arena segments: rows are CONTIGUOUS [lo+base,
// lo+base+m_eff) of a buffer — no index gather at all.  side resolves the
// child range from the device left count exactly like the rows-list kernel.
__global__ void hist_pair_range_k(
    const unsigned long long* __restrict__ pair, const longlong2* __restrict__ ghq,
    long n_arena, long lo, long m, long long* __restrict__ hist, int n_bins,
    int npairs, int tail_bytes, long chunk,
    const int* __restrict__ nl_dev, int side) {
  long base = 0, m_eff = m;
  if (side >= 0) {
    const long nl = nl_dev[0];
    m_eff = (side == 0) ? nl : m - nl;
    base = (side == 0) ? 0 : nl;
  }
  if ((long)blockIdx.x * chunk >= m_eff) return;
  extern __shared__ unsigned long long lds64[];
  const int tid = threadIdx.x;
  const int lds_elems = 8 * n_bins * 2;
  for (int i = tid; i < lds_elems; i += blockDim.x) lds64[i] = 0ull;
  __syncthreads();

  const int pair_id = blockIdx.y;
  const int jmax = (pair_id == npairs - 1) ? tail_bytes : 8;
  const unsigned long long* plane =
      pair + (size_t)pair_id * n_arena + lo + base;
  const longlong2* gh = ghq + lo + base;
  const long start = (long)blockIdx.x * chunk;
  const long end = min(start + chunk, m_eff);

  constexpr int ILP = 4;
  long i = start + tid;
  for (; i + (ILP - 1) * (long)blockDim.x < end; i += ILP * blockDim.x) {
    unsigned long long v[ILP];
    longlong2 q[ILP];
#pragma unroll
    for (int u = 0; u < ILP; ++u) v[u] = plane[i + u * blockDim.x];
#pragma unroll
    for (int u = 0; u < ILP; ++u) q[u] = gh[i + u * blockDim.x];
#pragma unroll
    for (int u = 0; u < ILP; ++u) {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        if (j >= jmax) break;
        const int b = (int)((v[u] >> (8 * j)) & 0xffull);
        unsigned long long* cell = &lds64[(j * n_bins + b) * 2];
        atomicAdd(cell + 0, (unsigned long long)q[u].x);
        atomicAdd(cell + 1, (unsigned long long)q[u].y);
      }
    }
  }
  for (; i < end; i += blockDim.x) {
    const unsigned long long v = plane[i];
    const longlong2 q = gh[i];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      if (j >= jmax) break;
      const int b = (int)((v >> (8 * j)) & 0xffull);
      unsigned long long* cell = &lds64[(j * n_bins + b) * 2];
      atomicAdd(cell + 0, (unsigned long long)q.x);
      atomicAdd(cell + 1, (unsigned long long)q.y);
    }
  }
  __syncthreads();

  for (int i2 = tid; i2 < jmax * n_bins; i2 += blockDim.x) {
    const int f = i2 / n_bins;
    const int b = i2 % n_bins;
    const unsigned long long gsum = lds64[(f * n_bins + b) * 2 + 0];
    const unsigned long long hpacked = lds64[(f * n_bins + b) * 2 + 1];
    if (gsum == 0ull && hpacked == 0ull) continue;
    const unsigned long long cnt = hpacked >> 44;
    const unsigned long long hsum = hpacked & ((1ull << 44) - 1ull);
    long long* out = hist + ((size_t)(pair_id * 8 + f) * n_bins + b) * 3;
    atomicAdd((unsigned long long*)(out + 0), gsum);
    atomicAdd((unsigned long long*)(out + 1), hsum);
    atomicAdd((unsigned long long*)(out + 2), cnt);
  }
}

extern "C" void launch_hist_pair_range(
    const void* pair, const void* ghq, long n_arena, long lo, long m,
    long long* hist, int n_bins, int npairs, int tail_bytes,
    const int* nl_dev, int side, hipStream_t stream) {
  if (m == 0) return;
  long chunks = (2048 + npairs - 1) / npairs;
  long chunk = (m + chunks - 1) / chunks;
  if (chunk < 16384) {
    chunk = (m + 7) / 8;
    if (chunk < 2048) chunk = 2048;
    if (chunk > 16384) chunk = 16384;
  }
  if (chunk > (1l << 19)) chunk = 1l << 19;
  chunks = (m + chunk - 1) / chunk;
  dim3 grid((unsigned)chunks, (unsigned)npairs);
  const size_t lds_bytes = (size_t)8 * n_bins * 2 * sizeof(long long);
  hipLaunchKernelGGL(hist_pair_range_k, grid, dim3(256), lds_bytes, stream,
                     (const unsigned long long*)pair, (const longlong2*)ghq,
                     n_arena, lo, m, hist, n_bins, npairs, tail_bytes, chunk,
                     nl_dev, side);
}

extern "C" void launch_partition(const void* binned, long n_rows,
                                 const int* rows, long m, int feature,
                                 int thr, int* out, int* scratch,
                                 int* total_left, hipStream_t stream) {
  if (m == 0) return;
  long chunk = 4096;
  long blocks = (m + chunk - 1) / chunk;
  if (blocks > 4096) {
    chunk = (m + 4095) / 4096;
    blocks = (m + chunk - 1) / chunk;
  }
  const int grp = feature / 4, j = feature % 4;
  hipLaunchKernelGGL(part_count_k, dim3((unsigned)blocks), dim3(256), 0,
                     stream, (const uchar4*)binned, n_rows, rows, m, grp, j,
                     thr, chunk, scratch);
  hipLaunchKernelGGL(part_scan_k, dim3(1), dim3(256), 0, stream, scratch,
                     (int)blocks, total_left);
  hipLaunchKernelGGL(part_scatter_k, dim3((unsigned)blocks), dim3(256), 0,
                     stream, (const uchar4*)binned, n_rows, rows, m, grp, j,
                     thr, chunk, scratch, total_left, out);
}

// ------------------------------------------------------------ forest predict
// Flat node arrays across trees; per-thread row traversal.  Categorical
// nodes (cat_offset[idx] >= 0) test a 256-bit category bitset instead of the
// numeric threshold (LightGBM categorical-split semantics).
DEV_INLINE bool go_left_node(float xv, long idx, const float* __restrict__ thr,
                             const int* __restrict__ catoff,
                             const unsigned* __restrict__ catw) {
  if (catoff != nullptr) {
    const int off = catoff[idx];
    if (off >= 0) {
      if (isnan(xv) || xv < 0.0f || xv >= 256.0f) return true;
      const int b = (int)rintf(xv);
      return (catw[off * 8 + (b >> 5)] >> (b & 31)) & 1u;
    }
  }
  return xv <= thr[idx] || isnan(xv);
}

__global__ void predict_forest_k(const int* __restrict__ feat,
                                 const float* __restrict__ thr,
                                 const int* __restrict__ left,
                                 const int* __restrict__ right,
                                 const float* __restrict__ val,
                                 const long* __restrict__ offsets,
                                 const float* __restrict__ tw,
                                 const float* __restrict__ X, long n, int nf,
                                 float* __restrict__ out, int n_outputs,
                                 int t0, int t1,
                                 const int* __restrict__ catoff,
                                 const unsigned* __restrict__ catw) {
  const long row0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long row = row0; row < n; row += stride) {
    const float* x = X + row * nf;
    if (n_outputs == 1) {
      float acc = 0.0f;
      int t = t0;
      // traversal is a serial dependent-load chain per tree — walk 4
      // independent trees at once so their node loads overlap (ILP)
      for (; t + 3 < t1; t += 4) {
        long idx[4];
        long base[4];
        bool done[4] = {false, false, false, false};
#pragma unroll
        for (int u = 0; u < 4; ++u) {
          base[u] = offsets[t + u];
          idx[u] = base[u];
          done[u] = feat[idx[u]] < 0;
        }
        while (!(done[0] & done[1] & done[2] & done[3])) {
#pragma unroll
          for (int u = 0; u < 4; ++u) {
            if (!done[u]) {
              const int f = feat[idx[u]];
              const float xv = x[f];
              idx[u] = base[u]
                       + (go_left_node(xv, idx[u], thr, catoff, catw)
                              ? left[idx[u]] : right[idx[u]]);
              done[u] = feat[idx[u]] < 0;
            }
          }
        }
#pragma unroll
        for (int u = 0; u < 4; ++u) acc += tw[t + u] * val[idx[u]];
      }
      for (; t < t1; ++t) {
        long idx = offsets[t];
        const long base = idx;
        int f = feat[idx];
        while (f >= 0) {
          const float xv = x[f];
          idx = base + (go_left_node(xv, idx, thr, catoff, catw) ? left[idx]
                                                                 : right[idx]);
          f = feat[idx];
        }
        acc += tw[t] * val[idx];
      }
      out[row] += acc;
    } else {
      for (int t = t0; t < t1; ++t) {
        long idx = offsets[t];
        const long base = idx;
        int f = feat[idx];
        while (f >= 0) {
          const float xv = x[f];
          idx = base + (go_left_node(xv, idx, thr, catoff, catw) ? left[idx]
                                                                 : right[idx]);
          f = feat[idx];
        }
        out[row * n_outputs + (t % n_outputs)] += tw[t] * val[idx];
      }
    }
  }
}

extern "C" void launch_predict_forest(const int* feat, const float* thr,
                                      const int* left, const int* right,
                                      const float* val, const long* offsets,
                                      const float* tw, const float* X, long n,
                                      int nf, float* out, int n_outputs,
                                      int t0, int t1, const int* catoff,
                                      const unsigned* catw,
                                      hipStream_t stream) {
  if (n == 0 || t1 <= t0) return;
  long blocks = (n + 255) / 256;
  if (blocks > 8192) blocks = 8192;
  hipLaunchKernelGGL(predict_forest_k, dim3((unsigned)blocks), dim3(256), 0,
                     stream, feat, thr, left, right, val, offsets, tw, X, n,
                     nf, out, n_outputs, t0, t1, catoff, catw);
}

// ------------------------------------------------------------- leaf indices
__global__ void predict_leaf_k(const int* __restrict__ feat,
                               const float* __restrict__ thr,
                               const int* __restrict__ left,
                               const int* __restrict__ right,
                               const int* __restrict__ leaf_index,
                               const long* __restrict__ offsets,
                               const float* __restrict__ X, long n, int nf,
                               int* __restrict__ out, int n_trees,
                               const int* __restrict__ catoff,
                               const unsigned* __restrict__ catw) {
  const long row0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long row = row0; row < n; row += stride) {
    const float* x = X + row * nf;
    for (int t = 0; t < n_trees; ++t) {
      long idx = offsets[t];
      const long base = idx;
      int f = feat[idx];
      while (f >= 0) {
        const float xv = x[f];
        idx = base + (go_left_node(xv, idx, thr, catoff, catw) ? left[idx]
                                                               : right[idx]);
        f = feat[idx];
      }
      out[row * n_trees + t] = leaf_index[idx];
    }
  }
}

extern "C" void launch_predict_leaf(const int* feat, const float* thr,
                                    const int* left, const int* right,
                                    const int* leaf_index, const long* offsets,
                                    const float* X, long n, int nf, int* out,
                                    int n_trees, const int* catoff,
                                    const unsigned* catw, hipStream_t stream) {
  if (n == 0) return;
  long blocks = (n + 255) / 256;
  if (blocks > 8192) blocks = 8192;
  hipLaunchKernelGGL(predict_leaf_k, dim3((unsigned)blocks), dim3(256), 0,
                     stream, feat, thr, left, right, leaf_index, offsets, X, n,
                     nf, out, n_trees, catoff, catw);
}

// ----------------------------------------------------------------- binning
// X: (n, nf) f32 row-major; ub: (nf, n_bins-1) ascending; out: (ngroups, n) uchar4
__global__ void bin_matrix_k(const float* __restrict__ X,
                             const float* __restrict__ ub, long n, int nf,
                             int n_bins, int ngroups, uchar4* __restrict__ out) {
  const long r0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  const int nb = n_bins - 1;
  for (long r = r0; r < n; r += stride) {
    for (int g = 0; g < ngroups; ++g) {
      unsigned char b[4] = {0, 0, 0, 0};
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const int f = g * 4 + j;
        if (f >= nf) break;
        const float x = X[r * nf + f];
        if (isnan(x)) { b[j] = 0; continue; }
        const float* bounds = ub + (size_t)f * nb;
        // lower_bound: first idx with bounds[idx] >= x
        int lo = 0, hi = nb;
        while (lo < hi) {
          const int mid = (lo + hi) >> 1;
          if (bounds[mid] < x) lo = mid + 1; else hi = mid;
        }
        b[j] = (unsigned char)min(lo, n_bins - 1);
      }
      out[(size_t)g * n + r] = make_uchar4(b[0], b[1], b[2], b[3]);
    }
  }
}

extern "C" void launch_bin_matrix(const float* X, const float* ub, long n,
                                  int nf, int n_bins, int ngroups, void* out,
                                  hipStream_t stream) {
  if (n == 0) return;
  long blocks = (n + 255) / 256;
  if (blocks > 8192) blocks = 8192;
  hipLaunchKernelGGL(bin_matrix_k, dim3((unsigned)blocks), dim3(256), 0,
                     stream, X, ub, n, nf, n_bins, ngroups, (uchar4*)out);
}

// ------------------------------------------------------------- split scan
// Fused best-split search over one or two sibling histograms — replaces the
// ~24-launch torch cumsum/argmax soup per leaf with 2 launches + one 12-float
// readback (profiling showed the torch scan was host-launch-bound at
// ~50 ms/iter vs 8 ms of GPU time).
// hist: (n_hists, nf_pad, n_bins, 3); scratch: (n_hists, nf_pad, 6);
// out: (n_hists, 6) = {gain, feat, bin, GL, HL, CL}.
// FIXED=true reads int64 fixed-point histograms directly (scales applied
// in-kernel) — saves the int64→float conversion launches per split.
template <bool FIXED>
__global__ void split_scan_k(const void* __restrict__ hist_v, int n_bins,
                             long nf_pad, float l1, float l2, float min_data,
                             float min_hess, float min_gain, long nf_real,
                             const bool* __restrict__ feat_mask,
                             float* __restrict__ scratch,
                             double inv_g, double inv_h) {
  const int f = blockIdx.x;
  const int hi = blockIdx.y;
  const int tid = threadIdx.x;
  float* out = scratch + ((size_t)hi * nf_pad + f) * 6;
  if (f >= nf_real || (feat_mask && !feat_mask[f])) {
    if (tid == 0) {
      out[0] = -INFINITY; out[1] = 0; out[2] = 0;
      out[3] = 0; out[4] = 0; out[5] = 0;
    }
    return;
  }
  __shared__ float sg[256], sh[256], sc[256];
  const bool in = tid < n_bins;
  if (FIXED) {
    const long long* H =
        (const long long*)hist_v + ((size_t)hi * nf_pad + f) * n_bins * 3;
    sg[tid] = in ? (float)((double)H[tid * 3 + 0] * inv_g) : 0.0f;
    sh[tid] = in ? (float)((double)H[tid * 3 + 1] * inv_h) : 0.0f;
    sc[tid] = in ? (float)H[tid * 3 + 2] : 0.0f;
  } else {
    const float* H =
        (const float*)hist_v + ((size_t)hi * nf_pad + f) * n_bins * 3;
    sg[tid] = in ? H[tid * 3 + 0] : 0.0f;
    sh[tid] = in ? H[tid * 3 + 1] : 0.0f;
    sc[tid] = in ? H[tid * 3 + 2] : 0.0f;
  }
  __syncthreads();
  // degenerate feature: every row in ONE bin (e.g. a sparse feature absent
  // from this leaf — all mass in its zero bin) can never split validly;
  // skip the 3 prefix scans + gain pass.  Dominant on wide-sparse shapes
  // where most of 100k features are untouched per leaf.
  const int nz_bins = __syncthreads_count(in && sc[tid] > 0.0f);
  if (nz_bins <= 1) {
    if (tid == 0) {
      out[0] = -INFINITY; out[1] = (float)f; out[2] = 0;
      out[3] = 0; out[4] = 0; out[5] = 0;
    }
    return;
  }
  // Hillis-Steele inclusive scan over 256 slots
#pragma unroll
  for (int d = 1; d < 256; d <<= 1) {
    float g = sg[tid], h = sh[tid], c = sc[tid];
    float ga = 0, ha = 0, ca = 0;
    if (tid >= d) { ga = sg[tid - d]; ha = sh[tid - d]; ca = sc[tid - d]; }
    __syncthreads();
    sg[tid] = g + ga; sh[tid] = h + ha; sc[tid] = c + ca;
    __syncthreads();
  }
  const float G = sg[n_bins - 1], Ht = sh[n_bins - 1], C = sc[n_bins - 1];
  auto leaf_sc = [l1, l2](float Gs, float Hs) {
    float Ga = fabsf(Gs) - l1;
    Ga = Ga > 0 ? Ga : 0.0f;
    return Ga * Ga / (Hs + l2 + 1e-32f);
  };
  const float parent = leaf_sc(G, Ht);
  float gain = -INFINITY;
  float GL = 0, HL = 0, CL = 0;
  if (in && tid < n_bins - 1) {
    const float gl = sg[tid], hl = sh[tid], cl = sc[tid];
    const float gr = G - gl, hr = Ht - hl, cr = C - cl;
    if (cl >= min_data && cr >= min_data && hl >= min_hess && hr >= min_hess) {
      gain = leaf_sc(gl, hl) + leaf_sc(gr, hr) - parent;
      GL = gl; HL = hl; CL = cl;
    }
  }
  // block argmax (first-index tie-break to match torch argmax semantics)
  __shared__ float bg[256];
  __shared__ int bb[256];
  bg[tid] = gain;
  bb[tid] = tid;
  __syncthreads();
#pragma unroll
  for (int d = 128; d > 0; d >>= 1) {
    if (tid < d) {
      if (bg[tid + d] > bg[tid] ||
          (bg[tid + d] == bg[tid] && bb[tid + d] < bb[tid])) {
        bg[tid] = bg[tid + d];
        bb[tid] = bb[tid + d];
      }
    }
    __syncthreads();
  }
  __shared__ int best_bin_s;
  if (tid == 0) best_bin_s = bb[0];
  __syncthreads();
  if (tid == best_bin_s) {
    out[0] = bg[0] > -INFINITY ? bg[0] : -INFINITY;
    out[1] = (float)f;
    out[2] = (float)tid;
    out[3] = GL; out[4] = HL; out[5] = CL;
  }
}

__global__ void split_reduce_k(const float* __restrict__ scratch, long nf_pad,
                               float* __restrict__ out) {
  const int hi = blockIdx.x;
  const int tid = threadIdx.x;
  const float* S = scratch + (size_t)hi * nf_pad * 6;
  float best = -INFINITY;
  int best_f = -1;
  for (long f = tid; f < nf_pad; f += blockDim.x) {
    const float g = S[f * 6];
    if (g > best || (g == best && best_f >= 0 && f < best_f)) {
      best = g;
      best_f = (int)f;
    }
  }
  __shared__ float bg[256];
  __shared__ int bf[256];
  bg[tid] = best;
  bf[tid] = best_f;
  __syncthreads();
#pragma unroll
  for (int d = 128; d > 0; d >>= 1) {
    if (tid < d) {
      if (bg[tid + d] > bg[tid] ||
          (bg[tid + d] == bg[tid] && bf[tid + d] >= 0 && bf[tid + d] < bf[tid])) {
        bg[tid] = bg[tid + d];
        bf[tid] = bf[tid + d];
      }
    }
    __syncthreads();
  }
  if (tid == 0) {
    const int f = bf[0] >= 0 ? bf[0] : 0;
    const float* s = S + (size_t)f * 6;
#pragma unroll
    for (int j = 0; j < 6; ++j) out[hi * 6 + j] = s[j];
  }
}

extern "C" void launch_split_scan(const float* hist, int n_hists, long nf_pad,
                                  int n_bins, float l1, float l2,
                                  float min_data, float min_hess,
                                  float min_gain, long nf_real,
                                  const bool* feat_mask, float* scratch,
                                  float* out, hipStream_t stream) {
  dim3 grid1((unsigned)nf_pad, (unsigned)n_hists);
  hipLaunchKernelGGL((split_scan_k<false>), grid1, dim3(256), 0, stream, hist,
                     n_bins, nf_pad, l1, l2, min_data, min_hess, min_gain,
                     nf_real, feat_mask, scratch, 1.0, 1.0);
  hipLaunchKernelGGL(split_reduce_k, dim3((unsigned)n_hists), dim3(256), 0,
                     stream, scratch, nf_pad, out);
}

extern "C" void launch_split_scan_fixed(const long long* hist, int n_hists,
                                        long nf_pad, int n_bins, float l1,
                                        float l2, float min_data,
                                        float min_hess, float min_gain,
                                        long nf_real, const bool* feat_mask,
                                        float* scratch, float* out,
                                        double inv_g, double inv_h,
                                        hipStream_t stream) {
  dim3 grid1((unsigned)nf_pad, (unsigned)n_hists);
  hipLaunchKernelGGL((split_scan_k<true>), grid1, dim3(256), 0, stream, hist,
                     n_bins, nf_pad, l1, l2, min_data, min_hess, min_gain,
                     nf_real, feat_mask, scratch, inv_g, inv_h);
  hipLaunchKernelGGL(split_reduce_k, dim3((unsigned)n_hists), dim3(256), 0,
                     stream, scratch, nf_pad, out);
}

// ------------------------------------------------------------- tree SHAP
// Path-dependent TreeSHAP contributions on GPU (SURVEY hard-part #4: the
// featuresShap column at scale).  Thread = one (row, tree) pair doing a DFS
// that keeps only the CURRENT root→node path (the Shapley path weights are
// permutation-symmetric, so at each leaf we merge duplicate features and run
// ONE extend over the merged elements — identical results to the recursive
// formulation, verified against the CPU reference, with ~500 B of per-thread
// state instead of per-level path copies).
template <int TS_MAXD>
__global__ void tree_shap_k(const int* __restrict__ feat,
                            const float* __restrict__ thr,
                            const int* __restrict__ left,
                            const int* __restrict__ right,
                            const float* __restrict__ val,   // leaf values ×w
                            const float* __restrict__ cnt,
                            const long* __restrict__ offsets,
                            const int* __restrict__ catoff,
                            const unsigned* __restrict__ catw,
                            const float* __restrict__ X, long n, int nf,
                            int n_trees, int n_outputs,
                            float* __restrict__ out /*(n, n_outputs*(nf+1))*/) {
  const long pair0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  const long n_pairs = n * (long)n_trees;

  // current path (edge elements root→node) + per-leaf merge scratch
  int pd[TS_MAXD + 2];
  float pz[TS_MAXD + 2], po[TS_MAXD + 2];
  int md[TS_MAXD + 2];
  // deep paths need f64: the Shapley weights carry 1/C(m,k) factors that
  // underflow f32 beyond ~20 path elements (CPU reference is float64 too)
  double mz[TS_MAXD + 2], mo[TS_MAXD + 2], w[TS_MAXD + 2];
  int nstack[TS_MAXD + 2];
  signed char phase[TS_MAXD + 2];

  for (long pair = pair0; pair < n_pairs; pair += stride) {
    const long row = pair / n_trees;
    const int t = (int)(pair % n_trees);
    const long base = offsets[t];
    const float* x = X + row * nf;
    float* phi = out + (row * n_outputs + (t % n_outputs)) * (nf + 1);

    int sp = 0;             // stack level == path length
    nstack[0] = (int)base;
    phase[0] = 0;
    while (sp >= 0) {
      const int nd = nstack[sp];
      const int ft = feat[nd];
      if (phase[sp] == 0) {
        phase[sp] = 1;
        if (ft < 0) {  // ---- leaf: merge path, extend once, credit features
          int m = 0;
          for (int i = 0; i < sp; ++i) {
            int k = -1;
            for (int j = 0; j < m; ++j)
              if (md[j] == pd[i]) { k = j; break; }
            if (k >= 0) { mz[k] *= pz[i]; mo[k] *= po[i]; }
            else { md[m] = pd[i]; mz[m] = pz[i]; mo[m] = po[i]; ++m; }
          }
          // extend: element 0 is the root sentinel (z=1, o=1)
          w[0] = 1.0;
          for (int e = 0; e < m; ++e) {  // append element e (len = e+1 after)
            w[e + 1] = 0.0;
            for (int i = e; i >= 0; --i) {
              w[i + 1] += mo[e] * w[i] * (double)(i + 1) / (double)(e + 2);
              w[i] = mz[e] * w[i] * (double)(e + 1 - i) / (double)(e + 2);
            }
          }
          const double leaf_v = val[nd];
          for (int i = 0; i < m; ++i) {
            // unwound sum: remove element i from the extended set
            double total = 0.0;
            double nrun = w[m];
            if (mo[i] != 0.0) {
              for (int j = m - 1; j >= 0; --j) {
                const double tmp =
                    nrun * (double)(m + 1) / ((double)(j + 1) * mo[i]);
                total += tmp;
                nrun = w[j] - tmp * mz[i] * (double)(m - j) / (double)(m + 1);
              }
            } else {
              for (int j = m - 1; j >= 0; --j)
                total += w[j] * (double)(m + 1) / (mz[i] * (double)(m - j));
            }
            atomicAdd(&phi[md[i]],
                      (float)(total * (mo[i] - mz[i]) * leaf_v));
          }
          --sp;
          continue;
        }
        // interior: descend hot child first
        const float xv = x[ft];
        const bool gl = go_left_node(xv, nd, thr, catoff, catw);
        const int hot = (int)base + (gl ? left[nd] : right[nd]);
        if (sp < TS_MAXD) {
          pd[sp] = ft;
          pz[sp] = cnt[hot] / fmaxf(cnt[nd], 1e-12f);
          po[sp] = 1.0f;
          nstack[sp + 1] = hot;
          phase[sp + 1] = 0;
          ++sp;
        }
        continue;
      }
      if (phase[sp] == 1) {  // descend cold child
        phase[sp] = 2;
        const float xv = x[ft];
        const bool gl = go_left_node(xv, nd, thr, catoff, catw);
        const int cold = (int)base + (gl ? right[nd] : left[nd]);
        if (sp < TS_MAXD) {
          pd[sp] = ft;
          pz[sp] = cnt[cold] / fmaxf(cnt[nd], 1e-12f);
          po[sp] = 0.0f;
          nstack[sp + 1] = cold;
          phase[sp + 1] = 0;
          ++sp;
        }
        continue;
      }
      --sp;
    }
  }
}

extern "C" void launch_tree_shap(const int* feat, const float* thr,
                                 const int* left, const int* right,
                                 const float* val, const float* cnt,
                                 const long* offsets, const int* catoff,
                                 const unsigned* catw, const float* X, long n,
                                 int nf, int n_trees, int n_outputs,
                                 int max_depth, float* out,
                                 hipStream_t stream) {
  if (n == 0 || n_trees == 0) return;
  long pairs = n * (long)n_trees;
  long blocks = (pairs + 255) / 256;
  if (blocks > 8192) blocks = 8192;
#define TSLAUNCH(D)                                                      \
  hipLaunchKernelGGL((tree_shap_k<D>), dim3((unsigned)blocks), dim3(256), \
                     0, stream, feat, thr, left, right, val, cnt, offsets, \
                     catoff, catw, X, n, nf, n_trees, n_outputs, out)
  if (max_depth < 8) TSLAUNCH(8);
  else if (max_depth < 16) TSLAUNCH(16);
  else TSLAUNCH(32);
#undef TSLAUNCH
}

// -------------------------------------------------------------- sparse CSR
// Histogram over STORED entries only (the reference's CSR ingestion,
// DatasetAggregator.scala:442); implicit zeros are corrected host-side by
// subtraction from exact integer leaf totals (LightGBM zero-bin trick).
// Fixed-point int64 global atomics keep the multi-rank all_reduce bit-exact.
// Thread-per-row grid-stride; rows are sorted so indptr/grad reads coalesce.
__global__ void csr_hist_fixed_k(const long* __restrict__ indptr,
                                 const int* __restrict__ col,
                                 const unsigned char* __restrict__ binv,
                                 const long long* __restrict__ gq,
                                 const long long* __restrict__ hq,
                                 const int* __restrict__ rows, long m,
                                 long long* __restrict__ hist, int n_bins) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < m;
       i += (long)gridDim.x * blockDim.x) {
    const int r = rows[i];
    const unsigned long long g = (unsigned long long)gq[r];
    const unsigned long long h = (unsigned long long)hq[r];
    const long s = indptr[r], e = indptr[r + 1];
    for (long j = s; j < e; ++j) {
      unsigned long long* cell = (unsigned long long*)hist
          + ((size_t)col[j] * n_bins + binv[j]) * 3;
      atomicAdd(cell + 0, g);
      atomicAdd(cell + 1, h);
      atomicAdd(cell + 2, 1ull);
    }
  }
}

extern "C" void launch_csr_hist_fixed(const long* indptr, const int* col,
                                      const unsigned char* binv,
                                      const long long* gq,
                                      const long long* hq, const int* rows,
                                      long m, long long* hist, int n_bins,
                                      hipStream_t stream) {
  if (m == 0) return;
  const int threads = 256;
  const long want = (m + threads - 1) / threads;
  const int blocks = (int)(want < 8192 ? want : 8192);  // ≫256 WGs, 8 XCDs
  hipLaunchKernelGGL(csr_hist_fixed_k, dim3(blocks), dim3(threads), 0,
                     stream, indptr, col, binv, gq, hq, rows, m, hist,
                     n_bins);
}

// Bin of one feature per row: binary search the row's sorted column
// segment; missing -> the feature's zero bin. Used for leaf partition.
__global__ void csr_gather_bin_k(const long* __restrict__ indptr,
                                 const int* __restrict__ col,
                                 const unsigned char* __restrict__ binv,
                                 const int* __restrict__ rows, long m,
                                 int feature, int zero_bin,
                                 int* __restrict__ out) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < m;
       i += (long)gridDim.x * blockDim.x) {
    const int r = rows[i];
    long lo = indptr[r], hi = indptr[r + 1];
    int b = zero_bin;
    while (lo < hi) {
      const long mid = (lo + hi) >> 1;
      const int c = col[mid];
      if (c == feature) { b = binv[mid]; break; }
      if (c < feature) lo = mid + 1; else hi = mid;
    }
    out[i] = b;
  }
}

extern "C" void launch_csr_gather_bin(const long* indptr, const int* col,
                                      const unsigned char* binv,
                                      const int* rows, long m, int feature,
                                      int zero_bin, int* out,
                                      hipStream_t stream) {
  if (m == 0) return;
  const int threads = 256;
  const long want = (m + threads - 1) / threads;
  const int blocks = (int)(want < 8192 ? want : 8192);
  hipLaunchKernelGGL(csr_gather_bin_k, dim3(blocks), dim3(threads), 0,
                     stream, indptr, col, binv, rows, m, feature, zero_bin,
                     out);
}

// v2 sparse histogram: one WAVE cooperates on 64 consecutive rows of the
// (sorted) row list.  Row bounds + quantized (g,h) stage through LDS; an
// in-wave prefix over row lengths maps entry slots to rows, so col/binv
// reads are coalesced runs instead of 20-strided gathers, and gq/hq come
// from LDS instead of per-entry global gathers.  Leaf totals (sum gq, sum
// hq, count) accumulate in-kernel (one atomic per wave) so the host-side
// correction needs no extra passes over the row list.
__global__ void csr_hist_fixed_v2_k(const long* __restrict__ indptr,
                                    const int* __restrict__ col,
                                    const unsigned char* __restrict__ binv,
                                    const long long* __restrict__ gq,
                                    const long long* __restrict__ hq,
                                    const int* __restrict__ rows, long m,
                                    long long* __restrict__ hist, int n_bins,
                                    long long* __restrict__ tot) {
  __shared__ long s_start[4][64];
  __shared__ int s_len[4][65];      // inclusive prefix at +1
  __shared__ long long s_g[4][64], s_h[4][64];
  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  unsigned long long tg = 0, th = 0, tc = 0;

  // block-uniform outer loop (same trip count for all 4 waves — the
  // barriers below must not diverge); each wave owns a 64-row sub-chunk
  for (long bbase = (long)blockIdx.x * 256; bbase < m;
       bbase += (long)gridDim.x * 256) {
    const long base = bbase + (long)wid * 64;
    const int nrows = (int)max(0l, min((long)64, m - base));
    long my_len = 0;
    if (lane < nrows) {
      const int r = rows[base + lane];
      const long s = indptr[r];
      s_start[wid][lane] = s;
      my_len = indptr[r + 1] - s;
      const long long g = gq[r];
      const long long h = hq[r];
      s_g[wid][lane] = g;
      s_h[wid][lane] = h;
      tg += (unsigned long long)g;
      th += (unsigned long long)h;
      tc += 1;
    }
    // inclusive prefix of row lengths across the wave (LDS scan)
    s_len[wid][0] = 0;
    int v = (int)my_len;
#pragma unroll
    for (int d = 1; d < 64; d <<= 1) {
      const int up = __shfl_up((int)v, d);
      if (lane >= d) v += up;
    }
    s_len[wid][lane + 1] = v;
    __syncthreads();
    const int total = s_len[wid][nrows];
    const int* cum = s_len[wid];
    for (int e = lane; e < total; e += 64) {
      // binary search: largest k with cum[k] <= e
      int lo2 = 0, hi2 = nrows;
      while (lo2 + 1 < hi2) {
        const int mid = (lo2 + hi2) >> 1;
        if (cum[mid] <= e) lo2 = mid; else hi2 = mid;
      }
      const long j = s_start[wid][lo2] + (e - cum[lo2]);
      unsigned long long* cell = (unsigned long long*)hist
          + ((size_t)col[j] * n_bins + binv[j]) * 3;
      atomicAdd(cell + 0, (unsigned long long)s_g[wid][lo2]);
      atomicAdd(cell + 1, (unsigned long long)s_h[wid][lo2]);
      atomicAdd(cell + 2, 1ull);
    }
    __syncthreads();
  }
  // wave-reduce totals, one atomic triple per wave
#pragma unroll
  for (int d = 32; d > 0; d >>= 1) {
    tg += (unsigned long long)__shfl_down((long long)tg, d);
    th += (unsigned long long)__shfl_down((long long)th, d);
    tc += (unsigned long long)__shfl_down((long long)tc, d);
  }
  if (lane == 0 && tot) {
    atomicAdd((unsigned long long*)tot + 0, tg);
    atomicAdd((unsigned long long*)tot + 1, th);
    atomicAdd((unsigned long long*)tot + 2, tc);
  }
}

extern "C" void launch_csr_hist_fixed_v2(const long* indptr, const int* col,
                                         const unsigned char* binv,
                                         const long long* gq,
                                         const long long* hq,
                                         const int* rows, long m,
                                         long long* hist, int n_bins,
                                         long long* tot,
                                         hipStream_t stream) {
  if (m == 0) return;
  const int threads = 256;                  // 4 waves x 64 rows
  const long want = (m + 255) / 256;
  const int blocks = (int)(want < 4096 ? (want > 0 ? want : 1) : 4096);
  hipLaunchKernelGGL(csr_hist_fixed_v2_k, dim3(blocks), dim3(threads), 0,
                     stream, indptr, col, binv, gq, hq, rows, m, hist,
                     n_bins, tot);
}

// LDS-privatized sparse histogram for small/medium nf: grid dim y carves
// features into chunks of 8; each block stages an 8-feature × n_bins packed
// (g, count|h) histogram in LDS (the 9× integer-LDS-atomic win of the dense
// kernel) and streams its row chunk's entries COALESCED, filtering to its
// feature chunk.  Global atomics drop from 3-per-entry to one flush per
// LDS cell.  Row staging + entry→row mapping as in v2.  Row count per
// block must stay ≤ 2^19 so the 44-bit packed h-sum cannot overflow
// (launcher enforces via grid sizing).
template <int FW, bool ILP = false>  // FW: feature-chunk width; ILP: 4-entry load batching
__global__ void csr_hist_fixed_lds_k(const long* __restrict__ indptr,
                                     const int* __restrict__ col,
                                     const unsigned char* __restrict__ binv,
                                     const long long* __restrict__ gq,
                                     const long long* __restrict__ hq,
                                     const int* __restrict__ rows, long m,
                                     long long* __restrict__ hist,
                                     int n_bins, int nf,
                                     long long* __restrict__ tot) {
  extern __shared__ unsigned long long lds64[];  // [FW][n_bins][2]
  const int f0 = blockIdx.y * FW;
  const int tid = threadIdx.x;
  const int lds_elems = FW * n_bins * 2;
  for (int i = tid; i < lds_elems; i += blockDim.x) lds64[i] = 0ull;

  __shared__ long s_start[4][64];
  __shared__ int s_len[4][65];
  __shared__ long long s_g[4][64], s_h[4][64];
  const int wid = tid >> 6;
  const int lane = tid & 63;
  constexpr unsigned long long CNT_ONE = 1ull << 44;
  unsigned long long tg = 0, th = 0, tc = 0;
  __syncthreads();

  for (long bbase = (long)blockIdx.x * 256; bbase < m;
       bbase += (long)gridDim.x * 256) {
    const long base = bbase + (long)wid * 64;
    const int nrows = (int)max(0l, min((long)64, m - base));
    long my_len = 0;
    if (lane < nrows) {
      const int r = rows[base + lane];
      const long s = indptr[r];
      s_start[wid][lane] = s;
      my_len = indptr[r + 1] - s;
      const long long g = gq[r];
      const long long h = hq[r];
      s_g[wid][lane] = g;
      s_h[wid][lane] = h;
      if (blockIdx.y == 0) {  // totals once, not per feature chunk
        tg += (unsigned long long)g;
        th += (unsigned long long)h;
        tc += 1;
      }
    }
    s_len[wid][0] = 0;
    int v = (int)my_len;
#pragma unroll
    for (int d = 1; d < 64; d <<= 1) {
      const int up = __shfl_up((int)v, d);
      if (lane >= d) v += up;
    }
    s_len[wid][lane + 1] = v;
    __syncthreads();
    // contiguous per-lane entry ranges: ONE binary search per lane, then
    // an amortized forward row walk — the strided variant's per-entry
    // binary search made the kernel latency-bound (PMC: 21x wait/busy)
    const int total = s_len[wid][nrows];
    const int* cum = s_len[wid];
    const int per = (total + 63) >> 6;
    const int e0 = lane * per;
    const int e1 = min(total, e0 + per);
    if (e0 < e1) {
      int k = 0, hi2 = nrows;
      while (k + 1 < hi2) {
        const int mid = (k + hi2) >> 1;
        if (cum[mid] <= e0) k = mid; else hi2 = mid;
      }
      if (ILP) {
        // batch 4 entries: resolve rows + issue all col/bin loads before
        // any LDS atomic.  A/B-measured on MI355X @10M×100 nnz=20: 131M vs
        // 211M rows/s — 1.6× SLOWER (the ks/js/cs/bs staging spills and the
        // LDS-atomic chain, not load latency, is the bound).  Kept gated
        // OFF behind MMLSPARK_AMD_SPARSE_ILP as the recorded negative result
        for (int e = e0; e < e1;) {
          int ks[4]; long js[4]; int cnt = 0;
          for (; cnt < 4 && e + cnt < e1; ++cnt) {
            while (cum[k + 1] <= e + cnt) ++k;
            ks[cnt] = k;
            js[cnt] = s_start[wid][k] + (e + cnt - cum[k]);
          }
          int cs[4]; unsigned char bs[4];
          for (int t = 0; t < cnt; ++t) {
            cs[t] = col[js[t]];
            bs[t] = binv[js[t]];
          }
          for (int t = 0; t < cnt; ++t) {
            if (cs[t] >= f0 && cs[t] < f0 + FW) {
              unsigned long long* cell =
                  &lds64[(((cs[t] - f0) * n_bins) + bs[t]) * 2];
              atomicAdd(cell + 0, (unsigned long long)s_g[wid][ks[t]]);
              atomicAdd(cell + 1,
                        CNT_ONE + (unsigned long long)s_h[wid][ks[t]]);
            }
          }
          e += cnt;
        }
      } else {
        for (int e = e0; e < e1; ++e) {
          while (cum[k + 1] <= e) ++k;
          const long j = s_start[wid][k] + (e - cum[k]);
          const int c = col[j];
          if (c >= f0 && c < f0 + FW) {
          unsigned long long* cell =
              &lds64[(((c - f0) * n_bins) + binv[j]) * 2];
          atomicAdd(cell + 0, (unsigned long long)s_g[wid][k]);
          atomicAdd(cell + 1, CNT_ONE + (unsigned long long)s_h[wid][k]);
          }
        }
      }
    }
    __syncthreads();
  }

  // flush LDS chunk to the global (nf, n_bins, 3) histogram
  for (int i = tid; i < FW * n_bins; i += blockDim.x) {
    const int f = i / n_bins;
    if (f0 + f >= nf) break;
    const unsigned long long gsum = lds64[i * 2 + 0];
    const unsigned long long hpacked = lds64[i * 2 + 1];
    if (gsum == 0ull && hpacked == 0ull) continue;
    long long* out = hist + ((size_t)(f0 + f) * n_bins + (i % n_bins)) * 3;
    atomicAdd((unsigned long long*)(out + 0), gsum);
    atomicAdd((unsigned long long*)(out + 1),
              hpacked & ((1ull << 44) - 1ull));
    atomicAdd((unsigned long long*)(out + 2), hpacked >> 44);
  }
  if (blockIdx.y == 0) {
#pragma unroll
    for (int d = 32; d > 0; d >>= 1) {
      tg += (unsigned long long)__shfl_down((long long)tg, d);
      th += (unsigned long long)__shfl_down((long long)th, d);
      tc += (unsigned long long)__shfl_down((long long)tc, d);
    }
    if (lane == 0 && tot) {
      atomicAdd((unsigned long long*)tot + 0, tg);
      atomicAdd((unsigned long long*)tot + 1, th);
      atomicAdd((unsigned long long*)tot + 2, tc);
    }
  }
}

extern "C" void launch_csr_hist_fixed_lds(const long* indptr, const int* col,
                                          const unsigned char* binv,
                                          const long long* gq,
                                          const long long* hq,
                                          const int* rows, long m,
                                          long long* hist, int n_bins,
                                          int nf, long long* tot,
                                          hipStream_t stream) {
  if (m == 0) return;
  long bx = (m + 2047) / 2048;
  if (bx > 4096) bx = 4096;
  const long min_bx = (m + (1l << 19) - 1) >> 19;  // ≤2^19 rows per block
  if (bx < min_bx) bx = min_bx;
  if (bx < 1) bx = 1;
  // A/B-measured on MI355X at 10M×100 nnz=20: FW=16 (half the passes,
  // half the occupancy) ties FW=8; FW=4 (more occupancy, double passes)
  // is ~1.7× SLOWER — reads win over latency hiding below FW=8, atomics
  // cap above it.  FW=8 is the sweet spot and the default
  static const int fw_env = [] {
    const char* e = getenv("MMLSPARK_AMD_SPARSE_FW");
    return e ? atoi(e) : 8;
  }();
  static const bool ilp_env = [] {
    const char* e = getenv("MMLSPARK_AMD_SPARSE_ILP");
    return e && atoi(e) != 0;
  }();
  const int FW = (fw_env == 16) ? 16 : (fw_env == 4 ? 4 : 8);
  const int by = (nf + FW - 1) / FW;
  const size_t lds_bytes = (size_t)FW * n_bins * 2 * sizeof(long long);
  if (FW == 8 && ilp_env) {
    hipLaunchKernelGGL((csr_hist_fixed_lds_k<8, true>),
                       dim3((unsigned)bx, (unsigned)by), dim3(256),
                       lds_bytes, stream, indptr, col, binv, gq, hq, rows,
                       m, hist, n_bins, nf, tot);
    return;
  }
  if (FW == 4)
    hipLaunchKernelGGL(csr_hist_fixed_lds_k<4>,
                       dim3((unsigned)bx, (unsigned)by), dim3(256),
                       lds_bytes, stream, indptr, col, binv, gq, hq, rows,
                       m, hist, n_bins, nf, tot);
  else if (FW == 8)
    hipLaunchKernelGGL(csr_hist_fixed_lds_k<8>,
                       dim3((unsigned)bx, (unsigned)by), dim3(256),
                       lds_bytes, stream, indptr, col, binv, gq, hq, rows,
                       m, hist, n_bins, nf, tot);
  else
    hipLaunchKernelGGL(csr_hist_fixed_lds_k<16>,
                       dim3((unsigned)bx, (unsigned)by), dim3(256),
                       lds_bytes, stream, indptr, col, binv, gq, hq, rows,
                       m, hist, n_bins, nf, tot);
}

// Fused CSR leaf partition: predicate = binary-searched bin of the split
// feature (missing → zero bin) compared against thr / a category bitset —
// replaces csr_gather + torch boolean masks (two nonzero passes + gathers)
// with the same 3-kernel stable ordered partition the dense path uses.
DEV_INLINE int csr_bin_of(const long* indptr, const int* col,
                          const unsigned char* binv, int r, int feature,
                          int zero_bin) {
  long lo = indptr[r], hi = indptr[r + 1];
  while (lo < hi) {
    const long mid = (lo + hi) >> 1;
    const int c = col[mid];
    if (c == feature) return binv[mid];
    if (c < feature) lo = mid + 1; else hi = mid;
  }
  return zero_bin;
}

DEV_INLINE bool csr_goes_left(const long* indptr, const int* col,
                              const unsigned char* binv, int r, int feature,
                              int zero_bin, int thr,
                              const unsigned* cat_bits) {
  const int b = csr_bin_of(indptr, col, binv, r, feature, zero_bin);
  if (cat_bits) return ((cat_bits[b >> 5] >> (b & 31)) & 1u) != 0u;
  return b <= thr;
}

__global__ void csr_part_count_k(const long* __restrict__ indptr,
                                 const int* __restrict__ col,
                                 const unsigned char* __restrict__ binv,
                                 const int* __restrict__ rows, long m,
                                 int feature, int zero_bin, int thr,
                                 const unsigned* __restrict__ cat_bits,
                                 long chunk, int* __restrict__ block_counts,
                                 unsigned char* __restrict__ pred) {
  const long start = (long)blockIdx.x * chunk;
  const long end = min(start + chunk, m);
  int cnt = 0;
  for (long i = start + threadIdx.x; i < end; i += blockDim.x) {
    const bool left = csr_goes_left(indptr, col, binv, rows[i], feature,
                                    zero_bin, thr, cat_bits);
    pred[i] = (unsigned char)left;  // scatter pass reuses the predicate
    cnt += (int)left;
  }
  __shared__ int sh[256];
  sh[threadIdx.x] = cnt;
  __syncthreads();
  for (int d = 128; d > 0; d >>= 1) {
    if (threadIdx.x < d) sh[threadIdx.x] += sh[threadIdx.x + d];
    __syncthreads();
  }
  if (threadIdx.x == 0) block_counts[blockIdx.x] = sh[0];
}

__global__ void csr_part_scatter_k(const unsigned char* __restrict__ pred,
                                   const int* __restrict__ rows, long m,
                                   long chunk,
                                   const int* __restrict__ block_offsets,
                                   const int* __restrict__ total_left,
                                   int* __restrict__ out) {
  const long start = (long)blockIdx.x * chunk;
  const long end = min(start + chunk, m);
  const long nl_total = *total_left;
  __shared__ long base_l, base_r;
  __shared__ int wave_l[4], wave_r[4];
  if (threadIdx.x == 0) {
    base_l = block_offsets[blockIdx.x];
    base_r = nl_total + (start - block_offsets[blockIdx.x]);
  }
  __syncthreads();
  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  for (long i0 = start; i0 < end; i0 += blockDim.x) {
    const long i = i0 + threadIdx.x;
    int r = 0;
    bool valid = i < end, left = false;
    if (valid) {
      r = rows[i];
      left = pred[i] != 0;
    }
    const unsigned long long mask_l = __ballot(valid && left);
    const unsigned long long mask_r = __ballot(valid && !left);
    const unsigned long long lt = (1ull << lane) - 1ull;
    if (lane == 0) {
      wave_l[wid] = __popcll(mask_l);
      wave_r[wid] = __popcll(mask_r);
    }
    __syncthreads();
    long wl = base_l, wr = base_r;
    for (int w = 0; w < wid; ++w) {
      wl += wave_l[w];
      wr += wave_r[w];
    }
    if (valid) {
      if (left) out[wl + __popcll(mask_l & lt)] = r;
      else out[wr + __popcll(mask_r & lt)] = r;
    }
    __syncthreads();
    if (threadIdx.x == 0) {
      long al = 0, ar = 0;
      for (int w = 0; w < 4; ++w) {
        al += wave_l[w];
        ar += wave_r[w];
      }
      base_l += al;
      base_r += ar;
    }
    __syncthreads();
  }
}

extern "C" void launch_csr_partition(const long* indptr, const int* col,
                                     const unsigned char* binv,
                                     const int* rows, long m, int feature,
                                     int zero_bin, int thr,
                                     const unsigned* cat_bits,
                                     unsigned char* pred, int* out,
                                     int* scratch, int* total_left,
                                     hipStream_t stream) {
  if (m == 0) return;
  long chunk = 4096;
  long blocks = (m + chunk - 1) / chunk;
  if (blocks > 4096) {
    chunk = (m + 4095) / 4096;
    blocks = (m + chunk - 1) / chunk;
  }
  hipLaunchKernelGGL(csr_part_count_k, dim3((unsigned)blocks), dim3(256), 0,
                     stream, indptr, col, binv, rows, m, feature, zero_bin,
                     thr, cat_bits, chunk, scratch, pred);
  hipLaunchKernelGGL(part_scan_k, dim3(1), dim3(256), 0, stream, scratch,
                     (int)blocks, total_left);
  hipLaunchKernelGGL(csr_part_scatter_k, dim3((unsigned)blocks), dim3(256),
                     0, stream, pred, rows, m, chunk, scratch, total_left,
                     out);
}
