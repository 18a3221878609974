// Fused optimizer + elementwise kernels (memory-bound; bf16 vectorized per
// guide G13: 8 bf16 per lane via short8 reinterpret, fp32 state via float4).
//
// K2/K4: fused inner-AdamW over flat buffers — replaces the reference's
//   torch AdamW step + extract_gradients file pass (training.py:113,
//   utils.py:118-123) with one single-pass kernel.
// K7/K8: fused outer-Nesterov — replaces parameter_server.rs:386-446's
//   two-pass mmapped-safetensors pipeline with one kernel on device memory.

#include <torch/extension.h>

#include "hip_common.h"

// ---------------------------------------------------------------------------
// AdamW: per 8-element group: g (bf16x8), m/v/master (2x float4 each),
// param out (bf16x8). ~26 B/elem traffic.
// ---------------------------------------------------------------------------

__global__ void adamw_kernel(float* __restrict__ master, short* __restrict__ param,
                             const short* __restrict__ grad, float* __restrict__ m,
                             float* __restrict__ v, long long n, float lr, float beta1,
                             float beta2, float eps, float wd, float inv_bc1,
                             float inv_bc2) {
  const long long stride = (long long)gridDim.x * blockDim.x * 8;
  for (long long base = ((long long)blockIdx.x * blockDim.x + threadIdx.x) * 8; base < n;
       base += stride) {
    if (base + 8 <= n) {
      s16x8 g8 = *reinterpret_cast<const s16x8*>(grad + base);
      f32x4 m0 = *reinterpret_cast<f32x4*>(m + base);
      f32x4 m1 = *reinterpret_cast<f32x4*>(m + base + 4);
      f32x4 v0 = *reinterpret_cast<f32x4*>(v + base);
      f32x4 v1 = *reinterpret_cast<f32x4*>(v + base + 4);
      f32x4 w0 = *reinterpret_cast<f32x4*>(master + base);
      f32x4 w1 = *reinterpret_cast<f32x4*>(master + base + 4);
      s16x8 p8;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float g = bf2f(g8[j]);
        float mm = j < 4 ? m0[j] : m1[j - 4];
        float vv = j < 4 ? v0[j] : v1[j - 4];
        float ww = j < 4 ? w0[j] : w1[j - 4];
        mm = beta1 * mm + (1.f - beta1) * g;
        vv = beta2 * vv + (1.f - beta2) * g * g;
        float denom = sqrtf(vv * inv_bc2) + eps;
        ww = ww * (1.f - lr * wd) - lr * inv_bc1 * mm / denom;
        if (j < 4) { m0[j] = mm; v0[j] = vv; w0[j] = ww; }
        else { m1[j - 4] = mm; v1[j - 4] = vv; w1[j - 4] = ww; }
        p8[j] = f2bf(ww);
      }
      *reinterpret_cast<f32x4*>(m + base) = m0;
      *reinterpret_cast<f32x4*>(m + base + 4) = m1;
      *reinterpret_cast<f32x4*>(v + base) = v0;
      *reinterpret_cast<f32x4*>(v + base + 4) = v1;
      *reinterpret_cast<f32x4*>(master + base) = w0;
      *reinterpret_cast<f32x4*>(master + base + 4) = w1;
      *reinterpret_cast<s16x8*>(param + base) = p8;
    } else {
      for (long long i = base; i < n; ++i) {
        float g = bf2f(grad[i]);
        float mm = beta1 * m[i] + (1.f - beta1) * g;
        float vv = beta2 * v[i] + (1.f - beta2) * g * g;
        float denom = sqrtf(vv * inv_bc2) + eps;
        float ww = master[i] * (1.f - lr * wd) - lr * inv_bc1 * mm / denom;
        m[i] = mm;
        v[i] = vv;
        master[i] = ww;
        param[i] = f2bf(ww);
      }
    }
  }
}

void adamw_step_(torch::Tensor master, torch::Tensor param, torch::Tensor grad,
                 torch::Tensor m, torch::Tensor v, double lr, double beta1, double beta2,
                 double eps, double wd, long step) {
  TORCH_CHECK(master.is_cuda() && master.dtype() == torch::kFloat32);
  TORCH_CHECK(param.dtype() == torch::kBFloat16 && grad.dtype() == torch::kBFloat16);
  long long n = master.numel();
  TORCH_CHECK(param.numel() == n && grad.numel() == n && m.numel() == n && v.numel() == n);
  float inv_bc1 = 1.f / (1.f - powf((float)beta1, (float)step));
  float inv_bc2 = 1.f / (1.f - powf((float)beta2, (float)step));
  hipStream_t stream = hypha_stream();
  hipLaunchKernelGGL(adamw_kernel, dim3(elementwise_grid((n + 7) / 8)), dim3(256), 0,
                     stream, master.data_ptr<float>(), (short*)param.data_ptr(),
                     (const short*)grad.data_ptr(), m.data_ptr<float>(),
                     v.data_ptr<float>(), n, (float)lr, (float)beta1, (float)beta2,
                     (float)eps, (float)wd, inv_bc1, inv_bc2);
}

// ---------------------------------------------------------------------------
// Outer Nesterov: m <- mu*m + d ; theta <- theta + lr*(mu*m + d)
// delta arrives in the bf16 comm buffer (post all-reduce).
// ---------------------------------------------------------------------------

__global__ void nesterov_kernel(float* __restrict__ theta, const short* __restrict__ delta,
                                float* __restrict__ mom, long long n, float lr, float mu) {
  const long long stride = (long long)gridDim.x * blockDim.x * 8;
  for (long long base = ((long long)blockIdx.x * blockDim.x + threadIdx.x) * 8; base < n;
       base += stride) {
    if (base + 8 <= n) {
      s16x8 d8 = *reinterpret_cast<const s16x8*>(delta + base);
      f32x4 mo0 = *reinterpret_cast<f32x4*>(mom + base);
      f32x4 mo1 = *reinterpret_cast<f32x4*>(mom + base + 4);
      f32x4 t0 = *reinterpret_cast<f32x4*>(theta + base);
      f32x4 t1 = *reinterpret_cast<f32x4*>(theta + base + 4);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float d = bf2f(d8[j]);
        float mm = j < 4 ? mo0[j] : mo1[j - 4];
        float tt = j < 4 ? t0[j] : t1[j - 4];
        mm = mu * mm + d;
        tt += lr * (mu * mm + d);
        if (j < 4) { mo0[j] = mm; t0[j] = tt; }
        else { mo1[j - 4] = mm; t1[j - 4] = tt; }
      }
      *reinterpret_cast<f32x4*>(mom + base) = mo0;
      *reinterpret_cast<f32x4*>(mom + base + 4) = mo1;
      *reinterpret_cast<f32x4*>(theta + base) = t0;
      *reinterpret_cast<f32x4*>(theta + base + 4) = t1;
    } else {
      for (long long i = base; i < n; ++i) {
        float mm = mu * mom[i] + bf2f(delta[i]);
        mom[i] = mm;
        theta[i] += lr * (mu * mm + bf2f(delta[i]));
      }
    }
  }
}

void nesterov_step_(torch::Tensor theta, torch::Tensor delta, torch::Tensor mom, double lr,
                    double mu) {
  TORCH_CHECK(theta.dtype() == torch::kFloat32 && mom.dtype() == torch::kFloat32);
  TORCH_CHECK(delta.dtype() == torch::kBFloat16);
  long long n = theta.numel();
  TORCH_CHECK(delta.numel() == n && mom.numel() == n);
  hipStream_t stream = hypha_stream();
  hipLaunchKernelGGL(nesterov_kernel, dim3(elementwise_grid((n + 7) / 8)), dim3(256), 0,
                     stream, theta.data_ptr<float>(), (const short*)delta.data_ptr(),
                     mom.data_ptr<float>(), n, (float)lr, (float)mu);
}

// ---------------------------------------------------------------------------
// extract_delta: out_bf16 = master - theta0 (fused sub + downcast for comm)
// ---------------------------------------------------------------------------

__global__ void extract_delta_kernel(const float* __restrict__ master,
                                     const float* __restrict__ theta0,
                                     short* __restrict__ out, long long n) {
  const long long stride = (long long)gridDim.x * blockDim.x * 8;
  for (long long base = ((long long)blockIdx.x * blockDim.x + threadIdx.x) * 8; base < n;
       base += stride) {
    if (base + 8 <= n) {
      f32x4 a0 = *reinterpret_cast<const f32x4*>(master + base);
      f32x4 a1 = *reinterpret_cast<const f32x4*>(master + base + 4);
      f32x4 b0 = *reinterpret_cast<const f32x4*>(theta0 + base);
      f32x4 b1 = *reinterpret_cast<const f32x4*>(theta0 + base + 4);
      s16x8 o8;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        o8[j] = f2bf((j < 4 ? a0[j] - b0[j] : a1[j - 4] - b1[j - 4]));
      *reinterpret_cast<s16x8*>(out + base) = o8;
    } else {
      for (long long i = base; i < n; ++i) out[i] = f2bf(master[i] - theta0[i]);
    }
  }
}

void extract_delta(torch::Tensor master, torch::Tensor theta0, torch::Tensor out) {
  TORCH_CHECK(out.dtype() == torch::kBFloat16);
  long long n = master.numel();
  TORCH_CHECK(theta0.numel() == n && out.numel() == n);
  hipStream_t stream = hypha_stream();
  hipLaunchKernelGGL(extract_delta_kernel, dim3(elementwise_grid((n + 7) / 8)), dim3(256),
                     0, stream, master.data_ptr<float>(), theta0.data_ptr<float>(),
                     (short*)out.data_ptr(), n);
}

// ---------------------------------------------------------------------------
// SwiGLU: out = silu(gate) * up  (fwd);  bwd: dgate, dup.
// ---------------------------------------------------------------------------

__global__ void swiglu_fwd_kernel(const short* __restrict__ gate,
                                  const short* __restrict__ up, short* __restrict__ out,
                                  long long n) {
  const long long stride = (long long)gridDim.x * blockDim.x * 8;
  for (long long base = ((long long)blockIdx.x * blockDim.x + threadIdx.x) * 8; base < n;
       base += stride) {
    if (base + 8 <= n) {
      s16x8 g8 = *reinterpret_cast<const s16x8*>(gate + base);
      s16x8 u8 = *reinterpret_cast<const s16x8*>(up + base);
      s16x8 o8;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float g = bf2f(g8[j]);
        float s = 1.f / (1.f + __expf(-g));
        o8[j] = f2bf(g * s * bf2f(u8[j]));
      }
      *reinterpret_cast<s16x8*>(out + base) = o8;
    } else {
      for (long long i = base; i < n; ++i) {
        float g = bf2f(gate[i]);
        float s = 1.f / (1.f + __expf(-g));
        out[i] = f2bf(g * s * bf2f(up[i]));
      }
    }
  }
}

__global__ void swiglu_bwd_kernel(const short* __restrict__ dout,
                                  const short* __restrict__ gate,
                                  const short* __restrict__ up, short* __restrict__ dgate,
                                  short* __restrict__ dup, long long n) {
  const long long stride = (long long)gridDim.x * blockDim.x * 8;
  for (long long base = ((long long)blockIdx.x * blockDim.x + threadIdx.x) * 8; base < n;
       base += stride) {
    long long end = base + 8 <= n ? base + 8 : n;
    if (base + 8 <= n) {
      s16x8 d8 = *reinterpret_cast<const s16x8*>(dout + base);
      s16x8 g8 = *reinterpret_cast<const s16x8*>(gate + base);
      s16x8 u8 = *reinterpret_cast<const s16x8*>(up + base);
      s16x8 dg8, du8;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float d = bf2f(d8[j]), g = bf2f(g8[j]), u = bf2f(u8[j]);
        float s = 1.f / (1.f + __expf(-g));
        float silu = g * s;
        dg8[j] = f2bf(d * u * (s * (1.f + g * (1.f - s))));
        du8[j] = f2bf(d * silu);
      }
      *reinterpret_cast<s16x8*>(dgate + base) = dg8;
      *reinterpret_cast<s16x8*>(dup + base) = du8;
    } else {
      for (long long i = base; i < end; ++i) {
        float d = bf2f(dout[i]), g = bf2f(gate[i]), u = bf2f(up[i]);
        float s = 1.f / (1.f + __expf(-g));
        dgate[i] = f2bf(d * u * (s * (1.f + g * (1.f - s))));
        dup[i] = f2bf(d * g * s);
      }
    }
  }
}

torch::Tensor swiglu_fwd(torch::Tensor gate, torch::Tensor up) {
  TORCH_CHECK(gate.dtype() == torch::kBFloat16 && gate.is_contiguous());
  auto out = torch::empty_like(gate);
  long long n = gate.numel();
  hipStream_t stream = hypha_stream();
  hipLaunchKernelGGL(swiglu_fwd_kernel, dim3(elementwise_grid((n + 7) / 8)), dim3(256), 0,
                     stream, (const short*)gate.data_ptr(), (const short*)up.data_ptr(),
                     (short*)out.data_ptr(), n);
  return out;
}

std::vector<torch::Tensor> swiglu_bwd(torch::Tensor dout, torch::Tensor gate,
                                      torch::Tensor up) {
  auto dgate = torch::empty_like(gate);
  auto dup = torch::empty_like(up);
  long long n = gate.numel();
  hipStream_t stream = hypha_stream();
  hipLaunchKernelGGL(swiglu_bwd_kernel, dim3(elementwise_grid((n + 7) / 8)), dim3(256), 0,
                     stream, (const short*)dout.data_ptr(), (const short*)gate.data_ptr(),
                     (const short*)up.data_ptr(), (short*)dgate.data_ptr(),
                     (short*)dup.data_ptr(), n);
  return {dgate, dup};
}

// ---------------------------------------------------------------------------
// grad_norm_sq: sum of squares of a flat bf16 tensor (fp32 accumulation,
// wave reduce + one atomic per block) — the clip-norm reduce without
// torch's generic reduction overhead (measured 2.2 TB/s -> HBM-bound).
// ---------------------------------------------------------------------------

__global__ void norm_sq_kernel(const short* __restrict__ x, float* __restrict__ out,
                               long long n) {
  const long long stride = (long long)gridDim.x * blockDim.x * 8;
  float acc = 0.f;
  for (long long base = ((long long)blockIdx.x * blockDim.x + threadIdx.x) * 8; base < n;
       base += stride) {
    if (base + 8 <= n) {
      s16x8 v8 = *reinterpret_cast<const s16x8*>(x + base);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float f = bf2f(v8[j]);
        acc += f * f;
      }
    } else {
      for (long long i = base; i < n; ++i) {
        float f = bf2f(x[i]);
        acc += f * f;
      }
    }
  }
  acc = wave_reduce_sum(acc);
  __shared__ float sc[4];
  int wid = threadIdx.x / 64;
  if ((threadIdx.x & 63) == 0) sc[wid] = acc;
  __syncthreads();
  if (threadIdx.x == 0) atomicAdd(out, sc[0] + sc[1] + sc[2] + sc[3]);
}

torch::Tensor grad_norm_sq(torch::Tensor x) {
  TORCH_CHECK(x.dtype() == torch::kBFloat16);
  auto out = torch::zeros({}, x.options().dtype(torch::kFloat32));
  long long n = x.numel();
  hipLaunchKernelGGL(norm_sq_kernel, dim3(elementwise_grid((n + 7) / 8)), dim3(256), 0,
                     hypha_stream(), (const short*)x.data_ptr(), out.data_ptr<float>(),
                     n);
  return out;
}
