// Fused softmax-cross-entropy over the vocab (K1 loss piece).
// fwd: one block per row; online max+sum in a single pass over V (bf16x8
// vector loads), block-combine (m, s) pairs; saves lse[N] f32.
// bwd: dlogits = scale * (exp(l - lse) - onehot), written IN PLACE over the
// logits buffer — avoids materialising a second [N, V] tensor (the logits
// tensor of Llama-3-8B at B4xS2048 is ~4 GB).

#include <torch/extension.h>

#include "hip_common.h"

__global__ void ce_fwd_kernel(const short* __restrict__ logits,
                              const long* __restrict__ targets,
                              float* __restrict__ lse_out, float* __restrict__ loss_sum,
                              int* __restrict__ n_valid, long long N, int V) {
  __shared__ float sm[4], ss[4];
  const long long row = blockIdx.x;
  if (row >= N) return;
  const short* lr = logits + row * V;
  const long tgt = targets[row];

  // online (max, sumexp) per thread
  float m = -1e30f, s = 0.f;
  for (int i = threadIdx.x * 8; i < V; i += blockDim.x * 8) {
    if (i + 8 <= V) {
      s16x8 v8 = *reinterpret_cast<const s16x8*>(lr + i);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float x = bf2f(v8[j]);
        if (x > m) {
          s *= __expf(m - x);
          m = x;
        }
        s += __expf(x - m);
      }
    } else {
      for (int k = i; k < V; ++k) {
        float x = bf2f(lr[k]);
        if (x > m) {
          s *= __expf(m - x);
          m = x;
        }
        s += __expf(x - m);
      }
    }
  }
  // wave combine
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    float mo = __shfl_xor(m, off, 64);
    float so = __shfl_xor(s, off, 64);
    float mn = fmaxf(m, mo);
    s = s * __expf(m - mn) + so * __expf(mo - mn);
    m = mn;
  }
  int wid = threadIdx.x / 64;
  if ((threadIdx.x & 63) == 0) {
    sm[wid] = m;
    ss[wid] = s;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    float M = fmaxf(fmaxf(sm[0], sm[1]), fmaxf(sm[2], sm[3]));
    float S = 0.f;
#pragma unroll
    for (int w = 0; w < 4; ++w) S += ss[w] * __expf(sm[w] - M);
    float lse = M + __logf(S);
    lse_out[row] = lse;
    if (tgt >= 0) {
      atomicAdd(loss_sum, lse - bf2f(lr[tgt]));
      atomicAdd(n_valid, 1);
    }
  }
}

__global__ void ce_bwd_kernel(short* __restrict__ logits, const long* __restrict__ targets,
                              const float* __restrict__ lse, float scale, long long N,
                              int V) {
  const long long row = blockIdx.x;
  if (row >= N) return;
  short* lr = logits + row * V;
  const long tgt = targets[row];
  const float l = lse[row];
  if (tgt < 0) {  // ignored row: zero gradient
    for (int i = threadIdx.x * 8; i < V; i += blockDim.x * 8) {
      if (i + 8 <= V) {
        s16x8 z = {};
        *reinterpret_cast<s16x8*>(lr + i) = z;
      } else {
        for (int k = i; k < V; ++k) lr[k] = 0;
      }
    }
    return;
  }
  for (int i = threadIdx.x * 8; i < V; i += blockDim.x * 8) {
    if (i + 8 <= V) {
      s16x8 v8 = *reinterpret_cast<const s16x8*>(lr + i);
      s16x8 o8;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float p = __expf(bf2f(v8[j]) - l);
        if (i + j == tgt) p -= 1.f;
        o8[j] = f2bf(p * scale);
      }
      *reinterpret_cast<s16x8*>(lr + i) = o8;
    } else {
      for (int k = i; k < V; ++k) {
        float p = __expf(bf2f(lr[k]) - l);
        if (k == tgt) p -= 1.f;
        lr[k] = f2bf(p * scale);
      }
    }
  }
}

std::vector<torch::Tensor> ce_fwd(torch::Tensor logits, torch::Tensor targets) {
  TORCH_CHECK(logits.dim() == 2 && logits.dtype() == torch::kBFloat16 &&
              logits.is_contiguous());
  TORCH_CHECK(targets.dtype() == torch::kInt64);
  long long N = logits.size(0);
  int V = logits.size(1);
  auto lse = torch::empty({N}, logits.options().dtype(torch::kFloat32));
  auto loss = torch::zeros({}, logits.options().dtype(torch::kFloat32));
  auto nv = torch::zeros({}, logits.options().dtype(torch::kInt32));
  hipStream_t stream = hypha_stream();
  hipLaunchKernelGGL(ce_fwd_kernel, dim3((unsigned)N), dim3(256), 0, stream,
                     (const short*)logits.data_ptr(), targets.data_ptr<long>(),
                     lse.data_ptr<float>(), loss.data_ptr<float>(), nv.data_ptr<int>(), N,
                     V);
  return {loss, lse, nv};
}

torch::Tensor ce_bwd_(torch::Tensor logits, torch::Tensor targets, torch::Tensor lse,
                      double scale) {
  long long N = logits.size(0);
  int V = logits.size(1);
  hipStream_t stream = hypha_stream();
  hipLaunchKernelGGL(ce_bwd_kernel, dim3((unsigned)N), dim3(256), 0, stream,
                     (short*)logits.data_ptr(), targets.data_ptr<long>(),
                     lse.data_ptr<float>(), (float)scale, N, V);
  return logits;
}
