// LayerNorm (weight+bias) and tanh-GELU forward/backward — the GPT-2 block's
// normalization/activation (K1 in SURVEY.md §2.10; the reference runs GPT-2
// through HF transformers, executors/accelerate/.../model.py). Same design
// as rmsnorm.hip: one block per row, short8-vectorized bf16 IO, f32
// accumulation, LDS block reductions; dw/db via striped column reduction
// with one atomic per column per stripe.

#include <torch/extension.h>

#include "hip_common.h"

namespace {

__device__ __forceinline__ float ln_block_reduce(float v, float* scratch) {
  v = wave_reduce_sum(v);
  int wid = threadIdx.x / 64;
  if ((threadIdx.x & 63) == 0) scratch[wid] = v;
  __syncthreads();
  float total = scratch[0] + scratch[1] + scratch[2] + scratch[3];
  __syncthreads();
  return total;
}

__global__ void layernorm_fwd_kernel(const short* __restrict__ x,
                                     const short* __restrict__ w,
                                     const short* __restrict__ b,
                                     short* __restrict__ y,
                                     float* __restrict__ mean_out,
                                     float* __restrict__ rstd_out, int D, float eps) {
  __shared__ float scratch[4];
  const long long row = blockIdx.x;
  const short* xr = x + row * D;
  short* yr = y + row * D;

  float s = 0.f, ss = 0.f;
  for (int i = threadIdx.x * 8; i < D; i += blockDim.x * 8) {
    s16x8 v8 = *reinterpret_cast<const s16x8*>(xr + i);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = bf2f(v8[j]);
      s += f;
      ss += f * f;
    }
  }
  s = ln_block_reduce(s, scratch);
  ss = ln_block_reduce(ss, scratch);
  float mu = s / D;
  float var = ss / D - mu * mu;
  float rstd = rsqrtf(var > 0.f ? var + eps : eps);
  if (threadIdx.x == 0) {
    mean_out[row] = mu;
    rstd_out[row] = rstd;
  }

  for (int i = threadIdx.x * 8; i < D; i += blockDim.x * 8) {
    s16x8 v8 = *reinterpret_cast<const s16x8*>(xr + i);
    s16x8 w8 = *reinterpret_cast<const s16x8*>(w + i);
    s16x8 b8 = *reinterpret_cast<const s16x8*>(b + i);
    s16x8 o8;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      o8[j] = f2bf((bf2f(v8[j]) - mu) * rstd * bf2f(w8[j]) + bf2f(b8[j]));
    *reinterpret_cast<s16x8*>(yr + i) = o8;
  }
}

// dx = rstd * (g - mean(g) - xhat * mean(g * xhat)), g = dy*w
__global__ void layernorm_bwd_dx_kernel(const short* __restrict__ dy,
                                        const short* __restrict__ x,
                                        const short* __restrict__ w,
                                        const float* __restrict__ mean,
                                        const float* __restrict__ rstd,
                                        short* __restrict__ dx, int D) {
  __shared__ float scratch[4];
  const long long row = blockIdx.x;
  const short* dyr = dy + row * D;
  const short* xr = x + row * D;
  short* dxr = dx + row * D;
  const float mu = mean[row];
  const float rs = rstd[row];

  float gsum = 0.f, gxsum = 0.f;
  for (int i = threadIdx.x * 8; i < D; i += blockDim.x * 8) {
    s16x8 d8 = *reinterpret_cast<const s16x8*>(dyr + i);
    s16x8 x8 = *reinterpret_cast<const s16x8*>(xr + i);
    s16x8 w8 = *reinterpret_cast<const s16x8*>(w + i);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float g = bf2f(d8[j]) * bf2f(w8[j]);
      float xhat = (bf2f(x8[j]) - mu) * rs;
      gsum += g;
      gxsum += g * xhat;
    }
  }
  gsum = ln_block_reduce(gsum, scratch) / D;
  gxsum = ln_block_reduce(gxsum, scratch) / D;

  for (int i = threadIdx.x * 8; i < D; i += blockDim.x * 8) {
    s16x8 d8 = *reinterpret_cast<const s16x8*>(dyr + i);
    s16x8 x8 = *reinterpret_cast<const s16x8*>(xr + i);
    s16x8 w8 = *reinterpret_cast<const s16x8*>(w + i);
    s16x8 o8;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float g = bf2f(d8[j]) * bf2f(w8[j]);
      float xhat = (bf2f(x8[j]) - mu) * rs;
      o8[j] = f2bf(rs * (g - gsum - xhat * gxsum));
    }
    *reinterpret_cast<s16x8*>(dxr + i) = o8;
  }
}

// dw[j] = sum_rows dy*xhat ; db[j] = sum_rows dy
__global__ void layernorm_bwd_dwdb_kernel(const short* __restrict__ dy,
                                          const short* __restrict__ x,
                                          const float* __restrict__ mean,
                                          const float* __restrict__ rstd,
                                          float* __restrict__ dw,
                                          float* __restrict__ db, long long N, int D) {
  const int col = blockIdx.x * blockDim.x + threadIdx.x;
  if (col >= D) return;
  const long long rows_per_stripe = (N + gridDim.y - 1) / gridDim.y;
  const long long r0 = blockIdx.y * rows_per_stripe;
  const long long r1 = (r0 + rows_per_stripe < N) ? r0 + rows_per_stripe : N;
  float aw = 0.f, ab = 0.f;
  for (long long i = r0; i < r1; ++i) {
    float d = bf2f(dy[i * D + col]);
    aw += d * (bf2f(x[i * D + col]) - mean[i]) * rstd[i];
    ab += d;
  }
  atomicAdd(dw + col, aw);
  atomicAdd(db + col, ab);
}

// tanh-GELU (the GPT-2 activation): y = 0.5 x (1 + tanh(c (x + a x^3)))
constexpr float GELU_C = 0.7978845608028654f;  // sqrt(2/pi)
constexpr float GELU_A = 0.044715f;

__device__ __forceinline__ float gelu_f(float x) {
  float u = GELU_C * (x + GELU_A * x * x * x);
  return 0.5f * x * (1.f + tanhf(u));
}

__device__ __forceinline__ float gelu_grad_f(float x) {
  float x2 = x * x;
  float u = GELU_C * (x + GELU_A * x * x2);
  float t = tanhf(u);
  float du = GELU_C * (1.f + 3.f * GELU_A * x2);
  return 0.5f * (1.f + t) + 0.5f * x * (1.f - t * t) * du;
}

__global__ void gelu_fwd_kernel(const short* __restrict__ x, short* __restrict__ y,
                                long long n) {
  const long long stride = (long long)gridDim.x * blockDim.x * 8;
  for (long long base = ((long long)blockIdx.x * blockDim.x + threadIdx.x) * 8; base < n;
       base += stride) {
    if (base + 8 <= n) {
      s16x8 v8 = *reinterpret_cast<const s16x8*>(x + base);
      s16x8 o8;
#pragma unroll
      for (int j = 0; j < 8; ++j) o8[j] = f2bf(gelu_f(bf2f(v8[j])));
      *reinterpret_cast<s16x8*>(y + base) = o8;
    } else {
      for (long long i = base; i < n; ++i) y[i] = f2bf(gelu_f(bf2f(x[i])));
    }
  }
}

__global__ void gelu_bwd_kernel(const short* __restrict__ dy, const short* __restrict__ x,
                                short* __restrict__ dx, long long n) {
  const long long stride = (long long)gridDim.x * blockDim.x * 8;
  for (long long base = ((long long)blockIdx.x * blockDim.x + threadIdx.x) * 8; base < n;
       base += stride) {
    if (base + 8 <= n) {
      s16x8 d8 = *reinterpret_cast<const s16x8*>(dy + base);
      s16x8 v8 = *reinterpret_cast<const s16x8*>(x + base);
      s16x8 o8;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        o8[j] = f2bf(bf2f(d8[j]) * gelu_grad_f(bf2f(v8[j])));
      *reinterpret_cast<s16x8*>(dx + base) = o8;
    } else {
      for (long long i = base; i < n; ++i)
        dx[i] = f2bf(bf2f(dy[i]) * gelu_grad_f(bf2f(x[i])));
    }
  }
}

}  // namespace

std::vector<torch::Tensor> layernorm_fwd(torch::Tensor x, torch::Tensor w,
                                         torch::Tensor b, double eps) {
  TORCH_CHECK(x.dim() == 2 && x.dtype() == torch::kBFloat16 && x.is_contiguous());
  TORCH_CHECK(x.size(1) % 8 == 0, "D must be a multiple of 8");
  TORCH_CHECK(w.dtype() == torch::kBFloat16 && b.dtype() == torch::kBFloat16);
  long long N = x.size(0);
  int D = (int)x.size(1);
  auto y = torch::empty_like(x);
  auto mean = torch::empty({N}, x.options().dtype(torch::kFloat32));
  auto rstd = torch::empty({N}, x.options().dtype(torch::kFloat32));
  hipLaunchKernelGGL(layernorm_fwd_kernel, dim3((unsigned)N), dim3(256), 0,
                     hypha_stream(), (const short*)x.data_ptr(),
                     (const short*)w.data_ptr(), (const short*)b.data_ptr(),
                     (short*)y.data_ptr(), mean.data_ptr<float>(),
                     rstd.data_ptr<float>(), D, (float)eps);
  return {y, mean, rstd};
}

std::vector<torch::Tensor> layernorm_bwd(torch::Tensor dy, torch::Tensor x,
                                         torch::Tensor w, torch::Tensor mean,
                                         torch::Tensor rstd) {
  long long N = x.size(0);
  int D = (int)x.size(1);
  auto dx = torch::empty_like(x);
  auto dw = torch::zeros({D}, x.options().dtype(torch::kFloat32));
  auto db = torch::zeros({D}, x.options().dtype(torch::kFloat32));
  hipStream_t stream = hypha_stream();
  hipLaunchKernelGGL(layernorm_bwd_dx_kernel, dim3((unsigned)N), dim3(256), 0, stream,
                     (const short*)dy.data_ptr(), (const short*)x.data_ptr(),
                     (const short*)w.data_ptr(), mean.data_ptr<float>(),
                     rstd.data_ptr<float>(), (short*)dx.data_ptr(), D);
  int threads = 256;
  int col_blocks = (D + threads - 1) / threads;
  long long stripes = N / 16;
  if (stripes < 1) stripes = 1;
  if (stripes > 64) stripes = 64;
  hipLaunchKernelGGL(layernorm_bwd_dwdb_kernel, dim3(col_blocks, (unsigned)stripes),
                     dim3(threads), 0, stream, (const short*)dy.data_ptr(),
                     (const short*)x.data_ptr(), mean.data_ptr<float>(),
                     rstd.data_ptr<float>(), dw.data_ptr<float>(),
                     db.data_ptr<float>(), N, D);
  return {dx, dw, db};
}

torch::Tensor gelu_fwd(torch::Tensor x) {
  TORCH_CHECK(x.dtype() == torch::kBFloat16 && x.is_contiguous());
  long long n = x.numel();
  auto y = torch::empty_like(x);
  hipLaunchKernelGGL(gelu_fwd_kernel, dim3(elementwise_grid((n + 7) / 8)), dim3(256), 0,
                     hypha_stream(), (const short*)x.data_ptr(), (short*)y.data_ptr(), n);
  return y;
}

torch::Tensor gelu_bwd(torch::Tensor dy, torch::Tensor x) {
  long long n = x.numel();
  auto dx = torch::empty_like(x);
  hipLaunchKernelGGL(gelu_bwd_kernel, dim3(elementwise_grid((n + 7) / 8)), dim3(256), 0,
                     hypha_stream(), (const short*)dy.data_ptr(),
                     (const short*)x.data_ptr(), (short*)dx.data_ptr(), n);
  return dx;
}
