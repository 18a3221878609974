// 8-bit optimizer-state AdamW (BASELINE config 5: "fp8 weights + 8-bit
// optimizer" for Llama-3-70B in 288 GB HBM).
//
// m and v are stored as uint8 with one fp32 absmax scale per 2048-element
// block (bitsandbytes-style dynamic blockwise quantization). m is symmetric
// linear; v is stored in the SQRT domain (u = sqrt(v)/scale) so the 255
// levels quantize the denominator directly — linear-quantized v collapses
// small entries to 0 and blows up m/(sqrt(v)+eps). One workgroup owns one
// block per iteration: dequantize, AdamW update, block-reduce the new
// absmax, requantize — state traffic is 2 B/elem instead of 8 B/elem.

#include <torch/extension.h>

#include "hip_common.h"

namespace {
constexpr int QBLOCK = 2048;  // elements per quant block (= 256 threads x 8)

__global__ void adamw8_kernel(float* __restrict__ master, short* __restrict__ param,
                              const short* __restrict__ grad,
                              unsigned char* __restrict__ m8,
                              unsigned char* __restrict__ v8,
                              float* __restrict__ m_scale, float* __restrict__ v_scale,
                              long long n, float lr, float beta1, float beta2, float eps,
                              float wd, float inv_bc1, float inv_bc2) {
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
        // m: symmetric int8 around 0; v stored as sqrt(v) (unsigned)
        float m = ((float)m8[base + i] - 127.f) * ms;
        float sv = (float)v8[base + i] * vs;
        float v = sv * sv;
        m = beta1 * m + (1.f - beta1) * g;
        v = beta2 * v + (1.f - beta2) * g * g;
        float denom = sqrtf(v * inv_bc2) + eps;
        float w = master[base + i] * (1.f - lr * wd) - lr * inv_bc1 * m / denom;
        master[base + i] = w;
        param[base + i] = f2bf(w);
        mv[j] = m;
        vv[j] = sqrtf(v);
        local_am = fmaxf(local_am, fabsf(m));
        local_av = fmaxf(local_av, vv[j]);
      } else {
        mv[j] = 0.f;
        vv[j] = 0.f;
      }
    }
    // block absmax reduce (new scales)
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

}  // namespace

void adamw8_step_(torch::Tensor master, torch::Tensor param, torch::Tensor grad,
                  torch::Tensor m8, torch::Tensor v8, torch::Tensor m_scale,
                  torch::Tensor v_scale, double lr, double beta1, double beta2,
                  double eps, double wd, long step) {
  TORCH_CHECK(master.dtype() == torch::kFloat32 && m8.dtype() == torch::kUInt8);
  long long n = master.numel();
  long long nblocks = (n + QBLOCK - 1) / QBLOCK;
  TORCH_CHECK(m_scale.numel() == nblocks && v_scale.numel() == nblocks);
  float inv_bc1 = 1.f / (1.f - powf((float)beta1, (float)step));
  float inv_bc2 = 1.f / (1.f - powf((float)beta2, (float)step));
  int grid = (int)(nblocks < 2048 ? nblocks : 2048);
  hipLaunchKernelGGL(adamw8_kernel, dim3(grid), dim3(256), 0, hypha_stream(),
                     master.data_ptr<float>(), (short*)param.data_ptr(),
                     (const short*)grad.data_ptr(), m8.data_ptr<unsigned char>(),
                     v8.data_ptr<unsigned char>(), m_scale.data_ptr<float>(),
                     v_scale.data_ptr<float>(), n, (float)lr, (float)beta1,
                     (float)beta2, (float)eps, (float)wd, inv_bc1, inv_bc2);
}
