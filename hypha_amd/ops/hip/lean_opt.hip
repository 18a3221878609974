// "Lean" memory-mode kernels (BASELINE configs 4/5): train models whose
// fp32 optimizer state cannot fit 288 GB (Mixtral-8x7B = 46.7B params,
// Llama-3-70B) on one GPU per worker:
//   * adamw8_lean_: AdamW with NO fp32 master — the bf16 parameters are the
//     only weight copy, updated with STOCHASTIC ROUNDING (unbiased: the
//     expected stored value equals the fp32 result, so small updates
//     accumulate instead of vanishing under round-to-nearest), and m/v in
//     blockwise-uint8 (sqrt-domain v) as in adamw8.hip.
//   * extract_delta_bf16 / nesterov_bf16_: the outer DiLoCo step over bf16
//     theta0/momentum chunks streamed from host memory (theta0 and the
//     outer momentum live host-side; the outer sync is chunked H2D->
//     compute->D2H, amortized over H inner steps).

#include <torch/extension.h>

#include "hip_common.h"

namespace {
constexpr int QBLOCK = 2048;

// xorshift-style per-element hash for stochastic rounding
__device__ __forceinline__ unsigned rnd_hash(unsigned long long idx, unsigned seed) {
  unsigned x = (unsigned)(idx ^ (idx >> 31)) * 0x9E3779B9u + seed;
  x ^= x >> 16;
  x *= 0x85EBCA6Bu;
  x ^= x >> 13;
  return x;
}

__device__ __forceinline__ short f2bf_sr(float f, unsigned r) {
  union {
    float f;
    unsigned i;
  } c;
  c.f = f;
  if ((c.i & 0x7fffffffu) > 0x7f800000u) return (short)0x7fc0;
  // add uniform noise below the bf16 mantissa, then truncate: E[result] = f
  return (short)((c.i + (r & 0xFFFFu)) >> 16);
}

__global__ void adamw8_lean_kernel(short* __restrict__ param,
                                   const short* __restrict__ grad,
                                   unsigned char* __restrict__ m8,
                                   unsigned char* __restrict__ v8,
                                   float* __restrict__ m_scale,
                                   float* __restrict__ v_scale, long long n, float lr,
                                   float beta1, float beta2, float eps, float wd,
                                   float inv_bc1, float inv_bc2, unsigned seed) {
  __shared__ float red[8];
  const long long nblocks = (n + QBLOCK - 1) / QBLOCK;
  for (long long blk = blockIdx.x; blk < nblocks; blk += gridDim.x) {
    const long long base = blk * QBLOCK;
    const int count = (int)((n - base) < QBLOCK ? (n - base) : QBLOCK);
    const float ms = m_scale[blk];
    const float vs = v_scale[blk];
    float mv[8], vv[8];
    float local_am = 0.f, local_av = 0.f;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      int i = threadIdx.x * 8 + j;
      if (i < count) {
        float g = bf2f(grad[base + i]);
        float m = ((float)m8[base + i] - 127.f) * ms;
        float sv = (float)v8[base + i] * vs;
        float v = sv * sv;
        m = beta1 * m + (1.f - beta1) * g;
        v = beta2 * v + (1.f - beta2) * g * g;
        float denom = sqrtf(v * inv_bc2) + eps;
        float w = bf2f(param[base + i]);
        w = w * (1.f - lr * wd) - lr * inv_bc1 * m / denom;
        param[base + i] = f2bf_sr(w, rnd_hash(base + i, seed));
        mv[j] = m;
        vv[j] = sqrtf(v);
        local_am = fmaxf(local_am, fabsf(m));
        local_av = fmaxf(local_av, vv[j]);
      } else {
        mv[j] = 0.f;
        vv[j] = 0.f;
      }
    }
    float am = wave_reduce_max(local_am);
    float av = wave_reduce_max(local_av);
    int wid = threadIdx.x / 64;
    if ((threadIdx.x & 63) == 0) {
      red[wid] = am;
      red[4 + wid] = av;
    }
    __syncthreads();
    am = fmaxf(fmaxf(red[0], red[1]), fmaxf(red[2], red[3]));
    av = fmaxf(fmaxf(red[4], red[5]), fmaxf(red[6], red[7]));
    __syncthreads();
    const float new_ms = am > 0.f ? am / 127.f : 1e-12f;
    const float new_vs = av > 0.f ? av / 255.f : 1e-12f;
    if (threadIdx.x == 0) {
      m_scale[blk] = new_ms;
      v_scale[blk] = new_vs;
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      int i = threadIdx.x * 8 + j;
      if (i < count) {
        int q = (int)rintf(mv[j] / new_ms) + 127;
        m8[base + i] = (unsigned char)(q < 0 ? 0 : (q > 254 ? 254 : q));
        int qv = (int)rintf(vv[j] / new_vs);
        v8[base + i] = (unsigned char)(qv < 0 ? 0 : (qv > 255 ? 255 : qv));
      }
    }
    __syncthreads();
  }
}

__global__ void extract_delta_bf16_kernel(const short* __restrict__ theta_t,
                                          const short* __restrict__ theta0,
                                          short* __restrict__ out, long long n) {
  const long long stride = (long long)gridDim.x * blockDim.x * 8;
  for (long long base = ((long long)blockIdx.x * blockDim.x + threadIdx.x) * 8; base < n;
       base += stride) {
    if (base + 8 <= n) {
      s16x8 a = *reinterpret_cast<const s16x8*>(theta_t + base);
      s16x8 b = *reinterpret_cast<const s16x8*>(theta0 + base);
      s16x8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j) o[j] = f2bf(bf2f(a[j]) - bf2f(b[j]));
      *reinterpret_cast<s16x8*>(out + base) = o;
    } else {
      for (long long i = base; i < n; ++i)
        out[i] = f2bf(bf2f(theta_t[i]) - bf2f(theta0[i]));
    }
  }
}

// bf16 outer Nesterov chunk: m <- mu*m + d ; theta <- theta + lr*(mu*m + d)
__global__ void nesterov_bf16_kernel(short* __restrict__ theta,
                                     const short* __restrict__ delta,
                                     short* __restrict__ mom, long long n, float lr,
                                     float mu, unsigned seed) {
  const long long stride = (long long)gridDim.x * blockDim.x * 8;
  for (long long base = ((long long)blockIdx.x * blockDim.x + threadIdx.x) * 8; base < n;
       base += stride) {
    long long end = base + 8 <= n ? base + 8 : n;
    for (long long i = base; i < end; ++i) {
      float d = bf2f(delta[i]);
      float m = mu * bf2f(mom[i]) + d;
      float t = bf2f(theta[i]) + lr * (mu * m + d);
      mom[i] = f2bf(m);
      theta[i] = f2bf_sr(t, rnd_hash(i, seed));
    }
  }
}

}  // namespace

void adamw8_lean_(torch::Tensor param, torch::Tensor grad, torch::Tensor m8,
                  torch::Tensor v8, torch::Tensor m_scale, torch::Tensor v_scale,
                  double lr, double beta1, double beta2, double eps, double wd,
                  long step, long seed) {
  TORCH_CHECK(param.dtype() == torch::kBFloat16 && m8.dtype() == torch::kUInt8);
  long long n = param.numel();
  long long nblocks = (n + QBLOCK - 1) / QBLOCK;
  TORCH_CHECK(m_scale.numel() >= nblocks && v_scale.numel() >= nblocks);
  float inv_bc1 = 1.f / (1.f - powf((float)beta1, (float)step));
  float inv_bc2 = 1.f / (1.f - powf((float)beta2, (float)step));
  int grid = (int)(nblocks < 2048 ? nblocks : 2048);
  hipLaunchKernelGGL(adamw8_lean_kernel, dim3(grid), dim3(256), 0, hypha_stream(),
                     (short*)param.data_ptr(), (const short*)grad.data_ptr(),
                     m8.data_ptr<unsigned char>(), v8.data_ptr<unsigned char>(),
                     m_scale.data_ptr<float>(), v_scale.data_ptr<float>(), n, (float)lr,
                     (float)beta1, (float)beta2, (float)eps, (float)wd, inv_bc1, inv_bc2,
                     (unsigned)seed);
}

void extract_delta_bf16(torch::Tensor theta_t, torch::Tensor theta0, torch::Tensor out) {
  long long n = theta_t.numel();
  TORCH_CHECK(theta0.numel() == n && out.numel() == n);
  hipLaunchKernelGGL(extract_delta_bf16_kernel, dim3(elementwise_grid((n + 7) / 8)),
                     dim3(256), 0, hypha_stream(), (const short*)theta_t.data_ptr(),
                     (const short*)theta0.data_ptr(), (short*)out.data_ptr(), n);
}

void nesterov_bf16_(torch::Tensor theta, torch::Tensor delta, torch::Tensor mom,
                    double lr, double mu, long seed) {
  long long n = theta.numel();
  TORCH_CHECK(delta.numel() == n && mom.numel() == n);
  hipLaunchKernelGGL(nesterov_bf16_kernel, dim3(elementwise_grid((n + 7) / 8)), dim3(256),
                     0, hypha_stream(), (short*)theta.data_ptr(),
                     (const short*)delta.data_ptr(), (short*)mom.data_ptr(), n, (float)lr,
                     (float)mu, (unsigned)seed);
}
