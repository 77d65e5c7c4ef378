// CDNA4 kernels for the VW-equivalent path: hashed sparse SGD over a 2^b
// weight table in HBM.  Replaces the compute inside the reference's
// VowpalWabbitNative example.learn loop (VowpalWabbitBase.scala:261-292):
// per-example sparse dot product + adaptive (AdaGrad-style) per-weight update.
// One 64-lane wave per example; lanes stride the example's features; the dot
// product reduces with __shfl_xor over the full wave.  Minibatch updates are
// hogwild (atomics) — end-of-pass weight sync is an RCCL all_reduce
// (spanning-tree AllReduce → RCCL over xGMI).
#include <hip/hip_runtime.h>
#include <cstdint>

#define WAVE 64

__device__ __forceinline__ float wave_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE);
  return v;
}

// loss: 0=squared 1=logistic(label ±1) 2=hinge(label ±1)
__device__ __forceinline__ float dloss(int loss, float pred, float y) {
  switch (loss) {
    case 0: return pred - y;
    case 1: { // d/dp log(1+exp(-y p)) = -y * sigmoid(-y p)
      const float z = -y * pred;
      const float s = 1.0f / (1.0f + __expf(-z));
      return -y * s;
    }
    case 2: return (y * pred < 1.0f) ? -y : 0.0f;
  }
  return 0.0f;
}

// Importance-weight-invariant prediction-space move (VW --invariant,
// Karampatziakis & Langford): exact integral of dp/dh = -eta*dloss(p),
// never overshoots the label/margin for any importance weight h.
__device__ __forceinline__ float invariant_dp(int loss, float pred, float y,
                                              float h_eta) {
  switch (loss) {
    case 0:  // squared: p(h) = y + (p0-y) e^{-h eta}
      return (y - pred) * (-expm1f(-h_eta));
    case 1: {  // logistic: q=y*p obeys e^q + q = e^{q0}+q0+h_eta
      const float q0 = y * pred;
      if (q0 > 30.0f) return y * (h_eta * __expf(-q0));
      const float A = __expf(q0);
      float d = log1pf(h_eta / (A + 1.0f));
#pragma unroll
      for (int it = 0; it < 8; ++it) {  // Newton on convex A*expm1(d)+d-h_eta
        const float g = A * expm1f(d) + d - h_eta;
        d = fmaxf(d - g / (A * __expf(d) + 1.0f), 0.0f);
      }
      return y * d;
    }
    case 2: {  // hinge: move to the margin, never past
      const float q0 = y * pred;
      return y * fminf(h_eta, fmaxf(1.0f - q0, 0.0f));
    }
  }
  return 0.0f;
}

// s_tbl (nullable): per-weight running max|x| — VW's --normalized scale.
// atomicMax on the float bit pattern is order-correct for non-negative floats.
__global__ void vw_sgd_k(const int* __restrict__ idx,
                         const float* __restrict__ val,
                         const long* __restrict__ off,
                         const float* __restrict__ label,
                         const float* __restrict__ ex_weight,
                         float* __restrict__ w_tbl, float* __restrict__ g_tbl,
                         float* __restrict__ s_tbl,
                         float lr, float l2, float power_t, int loss,
                         int invariant, long n_ex,
                         float* __restrict__ preds_out) {
  const long wid0 = ((long)blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const long n_waves = ((long)gridDim.x * blockDim.x) / WAVE;
  for (long ex = wid0; ex < n_ex; ex += n_waves) {
    const long s = off[ex], e = off[ex + 1];
    float dot = 0.0f;
    for (long k = s + lane; k < e; k += WAVE) dot += w_tbl[idx[k]] * val[k];
    const float pred = wave_sum(dot);
    if (preds_out && lane == 0) preds_out[ex] = pred;
    const float y = label[ex];
    const float h = ex_weight ? ex_weight[ex] : 1.0f;
    float gl = dloss(loss, pred, y) * h;
    if (invariant) {
      // pass 1: sensitivity x_norm = sum x_i^2 * scale_i with the same
      // per-coordinate scales the update will use (G + x^2 proxy)
      float xn = 0.0f;
      for (long k = s + lane; k < e; k += WAVE) {
        const int i = idx[k];
        const float x = val[k];
        float sc;
        const float Gp = g_tbl[i] + x * x;
        if (power_t == 0.5f) sc = __frsqrt_rn(Gp + 1e-10f);
        else sc = __powf(Gp + 1e-10f, -power_t);
        if (s_tbl) {
          const float ax = fabsf(x);
          atomicMax((int*)&s_tbl[i], __float_as_int(ax));
          const float sn = fmaxf(__int_as_float(((const int*)s_tbl)[i]), ax);
          if (sn > 0.0f) sc /= sn;
        }
        xn += x * x * sc;
      }
      const float x_norm = wave_sum(xn);
      if (x_norm <= 0.0f) continue;
      const float dp = invariant_dp(loss, pred, y, h * lr * x_norm);
      const float kk = dp / x_norm;
      for (long k = s + lane; k < e; k += WAVE) {
        const int i = idx[k];
        const float x = val[k];
        float sc;
        const float Gp = g_tbl[i] + x * x;
        if (power_t == 0.5f) sc = __frsqrt_rn(Gp + 1e-10f);
        else sc = __powf(Gp + 1e-10f, -power_t);
        if (s_tbl) {
          const float sn = fmaxf(__int_as_float(((const int*)s_tbl)[i]),
                                 fabsf(x));
          if (sn > 0.0f) sc /= sn;
        }
        const float g = gl * x + l2 * w_tbl[i];
        atomicAdd(&w_tbl[i], (kk * x - lr * l2 * w_tbl[i]) * sc);
        atomicAdd(&g_tbl[i], g * g);
      }
      continue;
    }
    if (gl == 0.0f) continue;
    for (long k = s + lane; k < e; k += WAVE) {
      const int i = idx[k];
      const float x = val[k];
      float g = gl * x + l2 * w_tbl[i];
      const float Gold = atomicAdd(&g_tbl[i], g * g);
      const float G = Gold + g * g;
      // adaptive per-weight rate: lr * G^(-power_t); power_t=0.5 → rsqrt
      float scale;
      if (power_t == 0.5f) scale = __frsqrt_rn(G + 1e-10f);
      else scale = __powf(G + 1e-10f, -power_t);
      if (s_tbl) {  // --normalized: divide by running max|x| per weight
        const float ax = fabsf(x);
        atomicMax((int*)&s_tbl[i], __float_as_int(ax));
        const float sn = fmaxf(__int_as_float(((const int*)s_tbl)[i]), ax);
        if (sn > 0.0f) scale /= sn;
      }
      atomicAdd(&w_tbl[i], -lr * g * scale);
    }
  }
}

extern "C" void launch_vw_sgd(const int* idx, const float* val,
                              const long* off, const float* label,
                              const float* ex_weight,
                              float* w_tbl, float* g_tbl, float* s_tbl,
                              float lr, float l2,
                              float power_t, int loss, int invariant,
                              long n_ex, float* preds_out,
                              hipStream_t stream) {
  if (n_ex == 0) return;
  long waves = n_ex;
  long blocks = (waves * WAVE + 255) / 256;
  if (blocks > 4096) blocks = 4096;
  hipLaunchKernelGGL(vw_sgd_k, dim3((unsigned)blocks), dim3(256), 0, stream,
                     idx, val, off, label, ex_weight, w_tbl, g_tbl, s_tbl,
                     lr, l2, power_t, loss, invariant, n_ex, preds_out);
}

__global__ void vw_predict_k(const int* __restrict__ idx,
                             const float* __restrict__ val,
                             const long* __restrict__ off,
                             const float* __restrict__ w_tbl, long n_ex,
                             float* __restrict__ out) {
  const long wid0 = ((long)blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const long n_waves = ((long)gridDim.x * blockDim.x) / WAVE;
  for (long ex = wid0; ex < n_ex; ex += n_waves) {
    const long s = off[ex], e = off[ex + 1];
    float dot = 0.0f;
    for (long k = s + lane; k < e; k += WAVE) dot += w_tbl[idx[k]] * val[k];
    const float pred = wave_sum(dot);
    if (lane == 0) out[ex] = pred;
  }
}

extern "C" void launch_vw_predict(const int* idx, const float* val,
                                  const long* off, const float* w_tbl,
                                  long n_ex, float* out, hipStream_t stream) {
  if (n_ex == 0) return;
  long blocks = (n_ex * WAVE + 255) / 256;
  if (blocks > 4096) blocks = 4096;
  hipLaunchKernelGGL(vw_predict_k, dim3((unsigned)blocks), dim3(256), 0,
                     stream, idx, val, off, w_tbl, n_ex, out);
}
