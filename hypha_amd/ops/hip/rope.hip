// RoPE forward (and inverse = backward) for q and k in one call.
// Llama "rotate-half" convention: pairs (d, d + D/2). Host-precomputed
// fp32 cos/sin tables (guide Appendix B: never on-device trig per element).
// x layout [B, H, S, D] contiguous; one thread handles 4 pairs (2x 8B loads).

#include <torch/extension.h>

#include "hip_common.h"

__global__ void rope_kernel(const short* __restrict__ x, short* __restrict__ out,
                            const float* __restrict__ cos_t, const float* __restrict__ sin_t,
                            long long n_tokens,  // B*H*S
                            int S, int D, float sign,
                            int s_div) {  // position = (token / s_div) % S
  const int half = D / 2;
  const long long stride = (long long)gridDim.x * blockDim.x;
  const long long total = n_tokens * (half / 4);  // work items: 4 pairs each
  for (long long it = (long long)blockIdx.x * blockDim.x + threadIdx.x; it < total;
       it += stride) {
    const long long tok = it / (half / 4);
    const int d0 = (int)(it % (half / 4)) * 4;
    const int s = (int)((tok / s_div) % S);
    const short* xp = x + tok * D;
    short* op = out + tok * D;
    const float* cp = cos_t + (long long)s * half + d0;
    const float* sp = sin_t + (long long)s * half + d0;
    s16x4 x1 = *reinterpret_cast<const s16x4*>(xp + d0);
    s16x4 x2 = *reinterpret_cast<const s16x4*>(xp + d0 + half);
    f32x4 c = *reinterpret_cast<const f32x4*>(cp);
    f32x4 sn = *reinterpret_cast<const f32x4*>(sp);
    s16x4 o1, o2;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float a = bf2f(x1[j]);
      float b = bf2f(x2[j]);
      float sj = sign * sn[j];
      o1[j] = f2bf(a * c[j] - b * sj);
      o2[j] = f2bf(b * c[j] + a * sj);
    }
    *reinterpret_cast<s16x4*>(op + d0) = o1;
    *reinterpret_cast<s16x4*>(op + d0 + half) = o2;
  }
}

// layout "bhsd": [B,H,S,D]; "bshd": [B,S,H,D] (no transpose copies).
std::vector<torch::Tensor> rope_fwd_ex(torch::Tensor q, torch::Tensor k,
                                       torch::Tensor cos_t, torch::Tensor sin_t,
                                       bool inverse, const std::string& layout) {
  TORCH_CHECK(q.dim() == 4 && q.dtype() == torch::kBFloat16 && q.is_contiguous());
  TORCH_CHECK(cos_t.dtype() == torch::kFloat32 && cos_t.is_contiguous());
  const bool bshd = layout == "bshd";
  int D = q.size(3);
  int S = bshd ? q.size(1) : q.size(2);
  TORCH_CHECK(D % 8 == 0);
  TORCH_CHECK(cos_t.size(0) >= S && cos_t.size(1) == D / 2, "rope table too small");
  auto qo = torch::empty_like(q);
  auto ko = torch::empty_like(k);
  float sign = inverse ? -1.f : 1.f;
  hipStream_t stream = hypha_stream();
  for (auto& pair : {std::make_pair(&q, &qo), std::make_pair(&k, &ko)}) {
    auto& t = *pair.first;
    auto& o = *pair.second;
    long long n_tokens = t.size(0) * t.size(1) * t.size(2);
    int s_div = bshd ? (int)t.size(2) : 1;  // [B,S,H,D]: position changes every H rows
    long long items = n_tokens * (D / 8);
    hipLaunchKernelGGL(rope_kernel, dim3(elementwise_grid(items)), dim3(256), 0, stream,
                       (const short*)t.data_ptr(), (short*)o.data_ptr(),
                       cos_t.data_ptr<float>(), sin_t.data_ptr<float>(), n_tokens, S, D,
                       sign, s_div);
  }
  return {qo, ko};
}

std::vector<torch::Tensor> rope_fwd(torch::Tensor q, torch::Tensor k, torch::Tensor cos_t,
                                    torch::Tensor sin_t, bool inverse) {
  return rope_fwd_ex(q, k, cos_t, sin_t, inverse, "bhsd");
}
