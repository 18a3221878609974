// MFMA fragment-layout probes: tiny single-wave GEMMs used by GPU tests to
// verify the lane->element maps this codebase assumes (guide §3 + G9:
// asymmetric-B transpose-detecting checks).
//
// Assumed maps (gfx950):
//   v_mfma_f32_32x32x16_bf16: A[m][k]: m=l&31, k=8*(l>>5)+j (j=0..7)
//                             B[k][n]: n=l&31, k=8*(l>>5)+j
//                             D[m][n]: n=l&31, m=(r&3)+8*(r>>2)+4*(l>>5), r=0..15
//   v_mfma_f32_16x16x32_bf16: A[m][k]: m=l&15, k=8*(l>>4)+j
//                             B[k][n]: n=l&15, k=8*(l>>4)+j
//                             D[m][n]: n=l&15, m=(l>>4)*4+r, r=0..3

#include <torch/extension.h>

#include "hip_common.h"

__global__ void probe32_kernel(const short* __restrict__ a, const short* __restrict__ b,
                               float* __restrict__ d) {
  const int l = threadIdx.x;  // one wave
  s16x8 af, bf;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    int k = 8 * (l >> 5) + j;
    af[j] = a[(l & 31) * 16 + k];
    bf[j] = b[k * 32 + (l & 31)];
  }
  f32x16 acc = {};
  acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(af, bf, acc, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    int m = (r & 3) + 8 * (r >> 2) + 4 * (l >> 5);
    d[m * 32 + (l & 31)] = acc[r];
  }
}

__global__ void probe16_kernel(const short* __restrict__ a, const short* __restrict__ b,
                               float* __restrict__ d) {
  const int l = threadIdx.x;
  s16x8 af, bf;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    int k = 8 * (l >> 4) + j;
    af[j] = a[(l & 15) * 32 + k];
    bf[j] = b[k * 16 + (l & 15)];
  }
  f32x4 acc = {};
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bf, acc, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    int m = (l >> 4) * 4 + r;
    d[m * 16 + (l & 15)] = acc[r];
  }
}

torch::Tensor mfma_probe_32x32x16(torch::Tensor a, torch::Tensor b) {
  TORCH_CHECK(a.sizes() == torch::IntArrayRef({32, 16}) &&
              b.sizes() == torch::IntArrayRef({16, 32}));
  TORCH_CHECK(a.dtype() == torch::kBFloat16 && a.is_contiguous() && b.is_contiguous());
  auto d = torch::empty({32, 32}, a.options().dtype(torch::kFloat32));
  hipLaunchKernelGGL(probe32_kernel, dim3(1), dim3(64), 0, hypha_stream(),
                     (const short*)a.data_ptr(), (const short*)b.data_ptr(),
                     d.data_ptr<float>());
  return d;
}

torch::Tensor mfma_probe_16x16x32(torch::Tensor a, torch::Tensor b) {
  TORCH_CHECK(a.sizes() == torch::IntArrayRef({16, 32}) &&
              b.sizes() == torch::IntArrayRef({32, 16}));
  auto d = torch::empty({16, 16}, a.options().dtype(torch::kFloat32));
  hipLaunchKernelGGL(probe16_kernel, dim3(1), dim3(64), 0, hypha_stream(),
                     (const short*)a.data_ptr(), (const short*)b.data_ptr(),
                     d.data_ptr<float>());
  return d;
}
