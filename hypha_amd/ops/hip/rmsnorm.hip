// RMSNorm forward/backward (K1 normalization piece).
// Memory-bound: one block per row (fwd / bwd-dx), bf16 loads vectorized as
// short8 (guide G13), f32 accumulation, block-level reduction in LDS.
// dw uses a separate column-reduction kernel (no atomic contention).

#include <torch/extension.h>

#include "hip_common.h"

// block reduce over 256 threads = 4 waves
__device__ __forceinline__ float block_reduce_sum(float v, float* scratch) {
  v = wave_reduce_sum(v);
  int wid = threadIdx.x / 64;
  if ((threadIdx.x & 63) == 0) scratch[wid] = v;
  __syncthreads();
  float total = scratch[0] + scratch[1] + scratch[2] + scratch[3];
  __syncthreads();
  return total;
}

__global__ void rmsnorm_fwd_kernel(const short* __restrict__ x, const short* __restrict__ w,
                                   short* __restrict__ y, float* __restrict__ rstd_out,
                                   int D, float eps) {
  __shared__ float scratch[4];
  const long long row = blockIdx.x;
  const short* xr = x + row * D;
  short* yr = y + row * D;

  float ss = 0.f;
  for (int i = threadIdx.x * 8; i < D; i += blockDim.x * 8) {
    s16x8 v8 = *reinterpret_cast<const s16x8*>(xr + i);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = bf2f(v8[j]);
      ss += f * f;
    }
  }
  ss = block_reduce_sum(ss, scratch);
  float rstd = rsqrtf(ss / D + eps);
  if (threadIdx.x == 0) rstd_out[row] = rstd;

  for (int i = threadIdx.x * 8; i < D; i += blockDim.x * 8) {
    s16x8 v8 = *reinterpret_cast<const s16x8*>(xr + i);
    s16x8 w8 = *reinterpret_cast<const s16x8*>(w + i);
    s16x8 o8;
#pragma unroll
    for (int j = 0; j < 8; ++j) o8[j] = f2bf(bf2f(v8[j]) * rstd * bf2f(w8[j]));
    *reinterpret_cast<s16x8*>(yr + i) = o8;
  }
}

// dx = rstd * (g - xhat * mean(g * xhat)), g = dy*w, xhat = x*rstd
__global__ void rmsnorm_bwd_dx_kernel(const short* __restrict__ dy,
                                      const short* __restrict__ x,
                                      const short* __restrict__ w,
                                      const float* __restrict__ rstd,
                                      short* __restrict__ dx, int D) {
  __shared__ float scratch[4];
  const long long row = blockIdx.x;
  const short* dyr = dy + row * D;
  const short* xr = x + row * D;
  short* dxr = dx + row * D;
  const float rs = rstd[row];

  float dot = 0.f;
  for (int i = threadIdx.x * 8; i < D; i += blockDim.x * 8) {
    s16x8 d8 = *reinterpret_cast<const s16x8*>(dyr + i);
    s16x8 x8 = *reinterpret_cast<const s16x8*>(xr + i);
    s16x8 w8 = *reinterpret_cast<const s16x8*>(w + i);
#pragma unroll
    for (int j = 0; j < 8; ++j)
      dot += bf2f(d8[j]) * bf2f(w8[j]) * bf2f(x8[j]) * rs;
  }
  dot = block_reduce_sum(dot, scratch) / D;

  for (int i = threadIdx.x * 8; i < D; i += blockDim.x * 8) {
    s16x8 d8 = *reinterpret_cast<const s16x8*>(dyr + i);
    s16x8 x8 = *reinterpret_cast<const s16x8*>(xr + i);
    s16x8 w8 = *reinterpret_cast<const s16x8*>(w + i);
    s16x8 o8;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float g = bf2f(d8[j]) * bf2f(w8[j]);
      float xhat = bf2f(x8[j]) * rs;
      o8[j] = f2bf(rs * (g - xhat * dot));
    }
    *reinterpret_cast<s16x8*>(dxr + i) = o8;
  }
}

// dw[j] = sum_rows dy[i,j] * x[i,j] * rstd[i] — 2D grid: blockIdx.x owns a
// 256-column slice, blockIdx.y a row stripe; one f32 atomicAdd per column
// per stripe (G12: partial reduction first, few atomics).
__global__ void rmsnorm_bwd_dw_kernel(const short* __restrict__ dy,
                                      const short* __restrict__ x,
                                      const float* __restrict__ rstd,
                                      float* __restrict__ dw, long long N, int D) {
  const int col = blockIdx.x * blockDim.x + threadIdx.x;
  if (col >= D) return;
  const long long rows_per_stripe = (N + gridDim.y - 1) / gridDim.y;
  const long long r0 = blockIdx.y * rows_per_stripe;
  const long long r1 = (r0 + rows_per_stripe < N) ? r0 + rows_per_stripe : N;
  float acc = 0.f;
  for (long long i = r0; i < r1; ++i)
    acc += bf2f(dy[i * D + col]) * bf2f(x[i * D + col]) * rstd[i];
  atomicAdd(dw + col, acc);
}

std::vector<torch::Tensor> rmsnorm_fwd(torch::Tensor x, torch::Tensor w, double eps) {
  TORCH_CHECK(x.dim() == 2 && x.dtype() == torch::kBFloat16 && x.is_contiguous());
  TORCH_CHECK(x.size(1) % 8 == 0, "D must be a multiple of 8");
  long long N = x.size(0);
  int D = x.size(1);
  auto y = torch::empty_like(x);
  auto rstd = torch::empty({N}, x.options().dtype(torch::kFloat32));
  hipStream_t stream = hypha_stream();
  hipLaunchKernelGGL(rmsnorm_fwd_kernel, dim3((unsigned)N), dim3(256), 0, stream,
                     (const short*)x.data_ptr(), (const short*)w.data_ptr(),
                     (short*)y.data_ptr(), rstd.data_ptr<float>(), D, (float)eps);
  return {y, rstd};
}

std::vector<torch::Tensor> rmsnorm_bwd(torch::Tensor dy, torch::Tensor x, torch::Tensor w,
                                       torch::Tensor rstd) {
  long long N = x.size(0);
  int D = x.size(1);
  auto dx = torch::empty_like(x);
  auto dw = torch::zeros({D}, x.options().dtype(torch::kFloat32));
  hipStream_t stream = hypha_stream();
  hipLaunchKernelGGL(rmsnorm_bwd_dx_kernel, dim3((unsigned)N), dim3(256), 0, stream,
                     (const short*)dy.data_ptr(), (const short*)x.data_ptr(),
                     (const short*)w.data_ptr(), rstd.data_ptr<float>(),
                     (short*)dx.data_ptr(), D);
  int threads = 256;
  int col_blocks = (D + threads - 1) / threads;
  long long stripes_ll = N / 16; if (stripes_ll < 1) stripes_ll = 1; if (stripes_ll > 64) stripes_ll = 64;
  int stripes = (int)stripes_ll;
  hipLaunchKernelGGL(rmsnorm_bwd_dw_kernel, dim3(col_blocks, stripes), dim3(threads), 0, stream,
                     (const short*)dy.data_ptr(), (const short*)x.data_ptr(),
                     rstd.data_ptr<float>(), dw.data_ptr<float>(), N, D);
  return {dx, dw};
}

// ---------------------------------------------------------------------------
// Fused residual-add + RMSNorm (round-2 step-time sweep): h = a + b is
// computed once, kept in registers for the normalization, and written as the
// residual stream — removing the separate elementwise add pass (and its
// extra read of h) per transformer sub-block. Backward fuses the incoming
// residual gradient into the dx pass the same way.
// ---------------------------------------------------------------------------

__global__ void add_rmsnorm_fwd_kernel(const short* __restrict__ a,
                                       const short* __restrict__ b,
                                       const short* __restrict__ w,
                                       short* __restrict__ h, short* __restrict__ y,
                                       float* __restrict__ rstd_out, int D,
                                       float eps) {
  __shared__ float scratch[4];
  constexpr int MAXIT = 8;  // D <= 16384 at blockDim 256 x 8-vectors
  const long long row = blockIdx.x;
  const short* ar = a + row * D;
  const short* br = b + row * D;
  short* hr = h + row * D;
  short* yr = y + row * D;

  float hv[MAXIT][8];
  float ss = 0.f;
  int it = 0;
  for (int i = threadIdx.x * 8; i < D; i += blockDim.x * 8, ++it) {
    s16x8 a8 = *reinterpret_cast<const s16x8*>(ar + i);
    s16x8 b8 = *reinterpret_cast<const s16x8*>(br + i);
    s16x8 o8;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = bf2f(a8[j]) + bf2f(b8[j]);
      hv[it][j] = f;
      o8[j] = f2bf(f);
      ss += f * f;
    }
    *reinterpret_cast<s16x8*>(hr + i) = o8;
  }
  ss = block_reduce_sum(ss, scratch);
  float rstd = rsqrtf(ss / D + eps);
  if (threadIdx.x == 0) rstd_out[row] = rstd;

  it = 0;
  for (int i = threadIdx.x * 8; i < D; i += blockDim.x * 8, ++it) {
    s16x8 w8 = *reinterpret_cast<const s16x8*>(w + i);
    s16x8 o8;
#pragma unroll
    for (int j = 0; j < 8; ++j) o8[j] = f2bf(hv[it][j] * rstd * bf2f(w8[j]));
    *reinterpret_cast<s16x8*>(yr + i) = o8;
  }
}

// dx = rstd * (g - xhat * mean(g*xhat)) + dh  (residual grad fused in)
__global__ void add_rmsnorm_bwd_dx_kernel(const short* __restrict__ dy,
                                          const short* __restrict__ dh,
                                          const short* __restrict__ x,
                                          const short* __restrict__ w,
                                          const float* __restrict__ rstd,
                                          short* __restrict__ dx, int D) {
  __shared__ float scratch[4];
  const long long row = blockIdx.x;
  const short* dyr = dy + row * D;
  const short* dhr = dh + row * D;
  const short* xr = x + row * D;
  short* dxr = dx + row * D;
  const float rs = rstd[row];

  float dot = 0.f;
  for (int i = threadIdx.x * 8; i < D; i += blockDim.x * 8) {
    s16x8 d8 = *reinterpret_cast<const s16x8*>(dyr + i);
    s16x8 x8 = *reinterpret_cast<const s16x8*>(xr + i);
    s16x8 w8 = *reinterpret_cast<const s16x8*>(w + i);
#pragma unroll
    for (int j = 0; j < 8; ++j)
      dot += bf2f(d8[j]) * bf2f(w8[j]) * bf2f(x8[j]) * rs;
  }
  dot = block_reduce_sum(dot, scratch) / D;

  for (int i = threadIdx.x * 8; i < D; i += blockDim.x * 8) {
    s16x8 d8 = *reinterpret_cast<const s16x8*>(dyr + i);
    s16x8 e8 = *reinterpret_cast<const s16x8*>(dhr + i);
    s16x8 x8 = *reinterpret_cast<const s16x8*>(xr + i);
    s16x8 w8 = *reinterpret_cast<const s16x8*>(w + i);
    s16x8 o8;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float g = bf2f(d8[j]) * bf2f(w8[j]);
      float xhat = bf2f(x8[j]) * rs;
      o8[j] = f2bf(rs * (g - xhat * dot) + bf2f(e8[j]));
    }
    *reinterpret_cast<s16x8*>(dxr + i) = o8;
  }
}

std::vector<torch::Tensor> add_rmsnorm_fwd(torch::Tensor a, torch::Tensor b,
                                           torch::Tensor w, double eps) {
  TORCH_CHECK(a.dim() == 2 && a.dtype() == torch::kBFloat16 && a.is_contiguous());
  TORCH_CHECK(b.sizes() == a.sizes() && b.is_contiguous());
  TORCH_CHECK(a.size(1) % 8 == 0 && a.size(1) <= 16384);
  long long N = a.size(0);
  int D = a.size(1);
  auto h = torch::empty_like(a);
  auto y = torch::empty_like(a);
  auto rstd = torch::empty({N}, a.options().dtype(torch::kFloat32));
  hipStream_t stream = hypha_stream();
  hipLaunchKernelGGL(add_rmsnorm_fwd_kernel, dim3((unsigned)N), dim3(256), 0, stream,
                     (const short*)a.data_ptr(), (const short*)b.data_ptr(),
                     (const short*)w.data_ptr(), (short*)h.data_ptr(),
                     (short*)y.data_ptr(), rstd.data_ptr<float>(), D, (float)eps);
  return {h, y, rstd};
}

std::vector<torch::Tensor> add_rmsnorm_bwd(torch::Tensor dy, torch::Tensor dh,
                                           torch::Tensor h, torch::Tensor w,
                                           torch::Tensor rstd) {
  long long N = h.size(0);
  int D = h.size(1);
  auto dx = torch::empty_like(h);
  auto dw = torch::zeros({D}, h.options().dtype(torch::kFloat32));
  hipStream_t stream = hypha_stream();
  hipLaunchKernelGGL(add_rmsnorm_bwd_dx_kernel, dim3((unsigned)N), dim3(256), 0, stream,
                     (const short*)dy.data_ptr(), (const short*)dh.data_ptr(),
                     (const short*)h.data_ptr(), (const short*)w.data_ptr(),
                     rstd.data_ptr<float>(), (short*)dx.data_ptr(), D);
  int threads = 256;
  int col_blocks = (D + threads - 1) / threads;
  long long stripes_ll = N / 16; if (stripes_ll < 1) stripes_ll = 1; if (stripes_ll > 64) stripes_ll = 64;
  hipLaunchKernelGGL(rmsnorm_bwd_dw_kernel, dim3(col_blocks, (int)stripes_ll),
                     dim3(threads), 0, stream, (const short*)dy.data_ptr(),
                     (const short*)h.data_ptr(), rstd.data_ptr<float>(),
                     dw.data_ptr<float>(), N, D);
  return {dx, dw};
}
