// Skinny-M GEMM for single-token decode: out[M, N] = x[M, K] @ W[N, K]^T
// with M <= 8 (batched decode GEMV). Decode GEMMs are weight-bandwidth
// bound, but hipBLASLt at M=8 N=4096 launches only ~16 workgroups (6% of
// the chip) and lands at ~1.8 TB/s. This kernel's shape: one LANE owns one
// W row over a k-chunk (no cross-lane reduction — a first version
// wave-reduced 8 accumulators per row, 48 dependent ds_bpermutes, and
// measured 0.4 TB/s), the x slab is a broadcast LDS read, and split-K is
// chosen host-side so the grid fills all 256 CUs even at N=1024. fp32
// atomic accumulation across k-chunks (distinct addresses), bf16 cast at
// the end.

#include <torch/extension.h>

#include "hip_common.h"

namespace {

constexpr int MMAX = 8;
constexpr int KCMAX = 1024;  // max k-chunk (LDS slab 8 x 1024 bf16 = 16 KB)

__global__ __launch_bounds__(256) void skinny_gemm_kernel(
    const short* __restrict__ xg, const short* __restrict__ wg,
    float* __restrict__ out_f32, int M, int N, int K, int kc) {
  __shared__ __attribute__((aligned(16))) short xs[MMAX * (KCMAX + 8)];
  constexpr int XP = KCMAX + 8;

  const int tid = threadIdx.x;
  const int k0 = blockIdx.y * kc;
  const int ks = min(kc, K - k0);
  const int n = blockIdx.x * 256 + tid;  // this lane's W row

  // ---- stage x[:, k0:k0+ks] (tiny) ----
  for (int c = tid * 8; c < M * kc; c += 256 * 8) {
    int m = c / kc, k = c % kc;
    s16x8 v{};
    if (k + 8 <= ks) {
      v = *reinterpret_cast<const s16x8*>(xg + (long long)m * K + k0 + k);
    } else {
      for (int j = 0; j < 8; ++j)
        v[j] = (k + j < ks) ? xg[(long long)m * K + k0 + k + j] : (short)0;
    }
    *reinterpret_cast<s16x8*>(xs + m * XP + k) = v;
  }
  __syncthreads();
  if (n >= N) return;

  const short* wr = wg + (long long)n * K + k0;
  float acc[MMAX];
#pragma unroll
  for (int m = 0; m < MMAX; ++m) acc[m] = 0.f;

  int k = 0;
  for (; k + 8 <= ks; k += 8) {
    s16x8 w8 = *reinterpret_cast<const s16x8*>(wr + k);
    float wv[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) wv[j] = bf2f(w8[j]);
    for (int m = 0; m < M; ++m) {
      s16x8 x8 = *reinterpret_cast<const s16x8*>(xs + m * XP + k);
      float a = acc[m];
#pragma unroll
      for (int j = 0; j < 8; ++j) a += wv[j] * bf2f(x8[j]);
      acc[m] = a;
    }
  }
  for (; k < ks; ++k) {  // K tail
    float wvk = bf2f(wr[k]);
    for (int m = 0; m < M; ++m) acc[m] += wvk * bf2f(xs[m * XP + k]);
  }

  if (gridDim.y == 1) {
    for (int m = 0; m < M; ++m) out_f32[(long long)m * N + n] = acc[m];
  } else {
    for (int m = 0; m < M; ++m)
      atomicAdd(out_f32 + (long long)m * N + n, acc[m]);
  }
}

__global__ void f32_to_bf16_kernel(const float* __restrict__ in,
                                   short* __restrict__ out, long long n) {
  long long i = ((long long)blockIdx.x * blockDim.x + threadIdx.x) * 8;
  if (i + 8 <= n) {
    s16x8 o8;
#pragma unroll
    for (int j = 0; j < 8; ++j) o8[j] = f2bf(in[i + j]);
    *reinterpret_cast<s16x8*>(out + i) = o8;
  } else {
    for (long long kk = i; kk < n; ++kk) out[kk] = f2bf(in[kk]);
  }
}

}  // namespace

// x [M, K] bf16 (M <= 8), w [N, K] bf16 row-major -> out [M, N] bf16.
torch::Tensor skinny_gemm(torch::Tensor x, torch::Tensor w) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 2 && x.dtype() == torch::kBFloat16 &&
              x.is_contiguous());
  TORCH_CHECK(w.dim() == 2 && w.is_contiguous() && w.size(1) == x.size(1));
  const int M = x.size(0), K = x.size(1), N = w.size(0);
  TORCH_CHECK(M <= MMAX, "skinny_gemm: M must be <= 8");
  auto out32 = torch::empty({M, N}, x.options().dtype(torch::kFloat32));
  auto out = torch::empty({M, N}, x.options());
  hipStream_t stream = hypha_stream();
  const int nblk = (N + 255) / 256;
  // split K until the grid covers the chip (>= 512 blocks), k-chunk in
  // [128, KCMAX] multiples of 8
  int kc = KCMAX;
  while (kc > 128 && (long long)nblk * ((K + kc - 1) / kc) < 512 && kc / 2 >= 128)
    kc /= 2;
  const int ksplit = (K + kc - 1) / kc;
  if (ksplit > 1) out32.zero_();
  dim3 grid(nblk, ksplit);
  hipLaunchKernelGGL(skinny_gemm_kernel, grid, dim3(256), 0, stream,
                     (const short*)x.data_ptr(), (const short*)w.data_ptr(),
                     out32.data_ptr<float>(), M, N, K, kc);
  long long n = (long long)M * N;
  hipLaunchKernelGGL(f32_to_bf16_kernel, dim3((unsigned)((n + 2047) / 2048)),
                     dim3(256), 0, stream, out32.data_ptr<float>(),
                     (short*)out.data_ptr(), n);
  return out;
}
