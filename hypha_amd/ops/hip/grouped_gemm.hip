// Grouped GEMM (bf16, MFMA) for the MoE expert computation: tokens sorted by
// expert run as one kernel over ragged per-expert row groups —
// out[r] = x[r] @ W[g]^T for r in group g (BASELINE config 4's
// "MoE grouped-GEMM in inner step"; the training backward currently runs
// per-expert hipBLASLt GEMMs, this kernel serves the forward/inference path
// and small-M regimes where per-expert launches dominate).
//
// Structure: one workgroup per (group, 128-row, 128-col) tile descriptor;
// 4 waves x (64x64) outputs via 32x32x16 MFMAs; x and W tiles staged through
// LDS with pitch-40 rows (conflict-free ds_read_b128) and T14 register
// prefetch (global loads for step k+1 issue before the MFMAs of step k).

#include <torch/extension.h>

#include "hip_common.h"

namespace {

constexpr int BM = 128, BN = 128, BK = 32;
constexpr int LP = BK + 8;  // LDS row pitch in elements (80 B: conflict-free)

__global__ __launch_bounds__(256, 2) void grouped_gemm_kernel(
    const short* __restrict__ xg,  // [T, K]
    const short* __restrict__ wg,  // [E, N, K]
    short* __restrict__ outg,      // [T, N]
    const int* __restrict__ desc,  // [ntiles, 4]: group, m0, mlen, n0
    int K, int N) {
  __shared__ __attribute__((aligned(16))) short a_lds[BM * LP];
  __shared__ __attribute__((aligned(16))) short b_lds[BN * LP];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int hi = lane >> 5;
  const int ln = lane & 31;
  const int wr = wid >> 1, wc = wid & 1;  // wave's 64x64 quadrant

  const int g = desc[blockIdx.x * 4 + 0];
  const int m0 = desc[blockIdx.x * 4 + 1];
  const int mlen = desc[blockIdx.x * 4 + 2];
  const int n0 = desc[blockIdx.x * 4 + 3];
  const long long wbase = (long long)g * N * K;

  // staging map: thread t covers chunks 2t, 2t+1 of the 512 8-elem chunks
  // (row = chunk / 4, e0 = (chunk % 4) * 8)
  s16x8 areg[2], breg[2];
  auto load_step = [&](int k0) {
#pragma unroll
    for (int c = 0; c < 2; ++c) {
      int chunk = 2 * tid + c;
      int row = chunk >> 2, e0 = (chunk & 3) * 8;
      int arow = row < mlen ? row : 0;  // clamp ragged rows (masked at store)
      areg[c] = *reinterpret_cast<const s16x8*>(xg + (long long)(m0 + arow) * K + k0 + e0);
      breg[c] = *reinterpret_cast<const s16x8*>(wg + wbase + (long long)(n0 + row) * K + k0 + e0);
    }
  };
  auto write_step = [&]() {
#pragma unroll
    for (int c = 0; c < 2; ++c) {
      int chunk = 2 * tid + c;
      int row = chunk >> 2, e0 = (chunk & 3) * 8;
      *reinterpret_cast<s16x8*>(a_lds + row * LP + e0) = areg[c];
      *reinterpret_cast<s16x8*>(b_lds + row * LP + e0) = breg[c];
    }
  };

  f32x16 acc[2][2] = {};
  load_step(0);
  const int ksteps = K / BK;
  for (int ks = 0; ks < ksteps; ++ks) {
    __syncthreads();
    write_step();
    if (ks + 1 < ksteps) load_step((ks + 1) * BK);
    __syncthreads();
#pragma unroll
    for (int kc = 0; kc < 2; ++kc) {
#pragma unroll
      for (int mi = 0; mi < 2; ++mi) {
        s16x8 af = *reinterpret_cast<const s16x8*>(
            a_lds + (64 * wr + 32 * mi + ln) * LP + 16 * kc + 8 * hi);
#pragma unroll
        for (int ni = 0; ni < 2; ++ni) {
          s16x8 bf = *reinterpret_cast<const s16x8*>(
              b_lds + (64 * wc + 32 * ni + ln) * LP + 16 * kc + 8 * hi);
          acc[mi][ni] =
              __builtin_amdgcn_mfma_f32_32x32x16_bf16(af, bf, acc[mi][ni], 0, 0, 0);
        }
      }
    }
  }

  // epilogue: D[m][n] lane col n = ln, rows via reg map; ragged-M masked
#pragma unroll
  for (int mi = 0; mi < 2; ++mi)
#pragma unroll
    for (int ni = 0; ni < 2; ++ni)
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        int mrel = 64 * wr + 32 * mi + (r & 3) + 8 * (r >> 2) + 4 * hi;
        if (mrel < mlen) {
          int col = n0 + 64 * wc + 32 * ni + ln;
          outg[(long long)(m0 + mrel) * N + col] = f2bf(acc[mi][ni][r]);
        }
      }
}

}  // namespace

// x [T, K] bf16 (rows grouped by expert), w [E, N, K] bf16,
// group_offsets [E+1] int32 (row ranges) -> out [T, N] bf16.
torch::Tensor grouped_gemm(torch::Tensor x, torch::Tensor w,
                           torch::Tensor group_offsets) {
  TORCH_CHECK(x.dim() == 2 && w.dim() == 3 && x.dtype() == torch::kBFloat16);
  TORCH_CHECK(x.is_contiguous() && w.is_contiguous());
  const int T = x.size(0), K = x.size(1);
  const int E = w.size(0), N = w.size(1);
  TORCH_CHECK(w.size(2) == K && K % BK == 0 && N % BN == 0,
              "grouped_gemm needs K%32==0 and N%128==0");
  TORCH_CHECK(group_offsets.numel() == E + 1);
  auto off = group_offsets.to(torch::kCPU, torch::kInt32).contiguous();
  const int* offp = off.data_ptr<int>();
  std::vector<int> desc;
  for (int g = 0; g < E; ++g) {
    for (int m0 = offp[g]; m0 < offp[g + 1]; m0 += BM) {
      int mlen = std::min(BM, offp[g + 1] - m0);
      for (int n0 = 0; n0 < N; n0 += BN) {
        desc.push_back(g);
        desc.push_back(m0);
        desc.push_back(mlen);
        desc.push_back(n0);
      }
    }
  }
  auto out = torch::empty({T, N}, x.options());
  if (desc.empty()) return out;
  auto desc_t = torch::from_blob(desc.data(), {(long)desc.size() / 4, 4},
                                 torch::TensorOptions().dtype(torch::kInt32))
                    .to(x.device());
  hipLaunchKernelGGL(grouped_gemm_kernel, dim3((unsigned)(desc.size() / 4)), dim3(256),
                     0, hypha_stream(), (const short*)x.data_ptr(),
                     (const short*)w.data_ptr(), (short*)out.data_ptr(),
                     desc_t.data_ptr<int>(), K, N);
  return out;
}
