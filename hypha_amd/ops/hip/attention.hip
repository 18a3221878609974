// Fused causal GQA flash attention, forward + backward, for CDNA4 (gfx950).
// Covers K1's attention piece natively — the reference
// delegates this to HF/PyTorch SDPA inside its Python executor.
//
// Forward structure (one workgroup = 4 waves = 128 q rows; KV tile = 64):
//  * swapped QK^T: S^T = mfma(A=K, B=Q) so each lane owns ONE q column and
//    the online-softmax state (m, l) is lane-local — no cross-lane reduce
//    beyond the in-reg accumulator sweep + one __shfl_xor(32) half combine.
//  * O is accumulated TRANSPOSED: O^T = mfma(A=V^T, B=P^T), keeping the
//    per-q rescale factor lane-local as well; the epilogue transposes O
//    back through LDS once per q block for coalesced stores.
//  * P^T (f32 accum regs) is packed to bf16 in-register (pack pairs +
//    __builtin_amdgcn_permlane32_swap half exchange) and feeds the PV MFMA
//    B operand directly — no LDS round trip for P.
//  * K tile lives in LDS row-major with an XOR-16 swizzle (bank-conflict
//    free ds_read_b128); V is transposed into LDS at staging time (pitch
//    padded to 72 elems, conflict-free b128 rows).
//
// Backward (one workgroup = 4 waves = one 32-row KV tile, q-tiles split
// round-robin across waves): recomputes P^T from q/k/lse, stages P^T, dS
// tiles and transposed operand images through LDS, accumulates dK/dV in
// f32 MFMA accumulators (cross-wave combine through LDS at the end), and
// accumulates dQ with global f32 atomics (FA2-style).

#include <torch/extension.h>

#include "hip_common.h"

namespace {

typedef __attribute__((ext_vector_type(2))) __bf16 bf16x2;

constexpr int KVB = 64;   // kv tile (fwd)
constexpr int QB = 32;    // q rows per wave
constexpr int NWAVE = 4;  // waves per workgroup
constexpr int WGQ = QB * NWAVE;
constexpr int PADV = 8;   // V^T pitch pad (72 elems -> conflict-free b128)
constexpr int KVT_COMB = 32;  // bwd combine-buffer kv rows

__device__ __forceinline__ unsigned pack_bf16x2(float lo, float hi) {
  return (unsigned)(unsigned short)f2bf(lo) | ((unsigned)(unsigned short)f2bf(hi) << 16);
}

// Swizzled byte offset inside the K tile: row-major [KVB][HD] bf16 with
// byte ^= (row & 15) << 4 (guide G4 XOR swizzle: conflict-free on
// ds_read_b128 when the 16-lane group's rows are distinct mod 16 — ours are).
template <int HD>
__device__ __forceinline__ int k_lds_off(int row, int elem) {
  return (row * HD + elem) * 2 ^ ((row & 15) << 4);
}

template <int HD>
__global__ __launch_bounds__(256, 1) void attn_fwd_kernel(
    const short* __restrict__ qg, const short* __restrict__ kg,
    const short* __restrict__ vg, short* __restrict__ og, float* __restrict__ lseg,
    int B, int Hq, int Hkv, int S, float scale, bool causal,
    long long q_sb, long long q_sh, long long q_ss,  // q/o strides (elements)
    long long kv_sb, long long kv_sh, long long kv_ss) {
  constexpr int KC = HD / 16;        // 16-wide k chunks over head dim
  constexpr int DBLK = HD / 32;      // 32-row d blocks
  constexpr int VT_PITCH = KVB + PADV;
  __shared__ __attribute__((aligned(16))) short k_lds[KVB * HD];
  __shared__ __attribute__((aligned(16))) short vt_lds[HD * VT_PITCH];
  __shared__ __attribute__((aligned(16))) short ot_lds[NWAVE][QB * (HD + 8)];  // epilogue transpose

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int hi = lane >> 5;  // half-wave
  const int ln = lane & 31;

  const int bh = blockIdx.y;
  const int b = bh / Hq;
  const int hq = bh % Hq;
  const int hkv = hq / (Hq / Hkv);
  const long long qbase = (long long)b * q_sb + (long long)hq * q_sh;
  const long long kvbase = (long long)b * kv_sb + (long long)hkv * kv_sh;

  const int q0wg = blockIdx.x * WGQ;
  const int q0 = q0wg + wid * QB;   // this wave's q rows
  const int qrow = q0 + ln;         // this lane's q row (its S^T column)

  // ---- Q fragments: B operand of S^T = mfma(K, Q); lane ln = q column ----
  s16x8 qfrag[KC];
#pragma unroll
  for (int kc = 0; kc < KC; ++kc)
    qfrag[kc] = *reinterpret_cast<const s16x8*>(qg + qbase + (long long)qrow * q_ss +
                                                16 * kc + 8 * hi);

  f32x16 ot[DBLK] = {};
  float m_run = -INFINITY, l_run = 0.f;

  // ---- register-staged K/V prefetch (guide T14: issue the next tile's
  // global loads before computing the current one; HBM latency hides under
  // the MFMA phase instead of stalling every wave at the staging barrier) --
  constexpr int KCH = KVB * HD / 8 / 256;  // 16B K chunks per thread
  s16x8 kreg[KCH];
  s16x8 vreg[HD / 32];
  auto load_tile = [&](int kv0) {
#pragma unroll
    for (int c = 0; c < KCH; ++c) {
      int cc = c * 256 + tid;
      int row = cc / (HD / 8), e0 = (cc % (HD / 8)) * 8;
      kreg[c] = *reinterpret_cast<const s16x8*>(kg + kvbase + (long long)(kv0 + row) * kv_ss + e0);
    }
#pragma unroll
    for (int i = 0; i < HD / 32; ++i) {
      int kvr = tid & 63;
      int e0 = (i * 4 + (tid >> 6)) * 8;
      vreg[i] = *reinterpret_cast<const s16x8*>(vg + kvbase + (long long)(kv0 + kvr) * kv_ss + e0);
    }
  };
  auto write_tile = [&]() {
#pragma unroll
    for (int c = 0; c < KCH; ++c) {
      int cc = c * 256 + tid;
      int row = cc / (HD / 8), e0 = (cc % (HD / 8)) * 8;
      *reinterpret_cast<s16x8*>((char*)k_lds + k_lds_off<HD>(row, e0)) = kreg[c];
    }
    // V^T transpose: lane-per-kv stores walk the contiguous image row
    // (conflict-free; the d-major pattern was a 16-way bank conflict)
#pragma unroll
    for (int i = 0; i < HD / 32; ++i) {
      int kvr = tid & 63;
      int e0 = (i * 4 + (tid >> 6)) * 8;
#pragma unroll
      for (int j = 0; j < 8; ++j) vt_lds[(e0 + j) * VT_PITCH + kvr] = vreg[i][j];
    }
  };

  const int kv_end = causal ? (q0wg + WGQ) : S;  // exclusive upper bound
  load_tile(0);
  for (int kv0 = 0; kv0 < kv_end; kv0 += KVB) {
    __syncthreads();  // previous tile's LDS consumers done
    write_tile();
    if (kv0 + KVB < kv_end) load_tile(kv0 + KVB);  // prefetch next tile
    __syncthreads();  // this tile's LDS image ready

    if (causal && kv0 > q0 + QB - 1) continue;  // wave fully above the diagonal

    // ---- S^T = K Q^T : two 32-kv accumulators ----
    f32x16 st[2];
#pragma unroll
    for (int mb = 0; mb < 2; ++mb) {
      f32x16 acc = {};
#pragma unroll
      for (int kc = 0; kc < KC; ++kc) {
        s16x8 kf = *reinterpret_cast<const s16x8*>(
            (char*)k_lds + k_lds_off<HD>(32 * mb + ln, 16 * kc + 8 * hi));
        acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qfrag[kc], acc, 0, 0, 0);
      }
      st[mb] = acc;
    }

    // ---- online softmax over the 32 regs (all kv of this tile, own q) ----
    float tmax = -INFINITY;
#pragma unroll
    for (int mb = 0; mb < 2; ++mb)
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        int kv = kv0 + 32 * mb + (r & 3) + 8 * (r >> 2) + 4 * hi;
        float s = st[mb][r] * scale;
        if (causal && kv > qrow) s = -INFINITY;
        st[mb][r] = s;
        tmax = fmaxf(tmax, s);
      }
    tmax = fmaxf(tmax, __shfl_xor(tmax, 32, 64));
    const float m_new = fmaxf(m_run, tmax);
    const float alpha = __expf(m_run - m_new);
    float psum = 0.f;
#pragma unroll
    for (int mb = 0; mb < 2; ++mb)
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        float p = __expf(st[mb][r] - m_new);
        st[mb][r] = p;
        psum += p;
      }
    psum += __shfl_xor(psum, 32, 64);
    l_run = l_run * alpha + psum;
    m_run = m_new;
#pragma unroll
    for (int db = 0; db < DBLK; ++db)
#pragma unroll
      for (int r = 0; r < 16; ++r) ot[db][r] *= alpha;

    // ---- pack P^T to bf16 B-fragments (pair-pack + half swap) ----
    s16x8 pfrag[4];
#pragma unroll
    for (int mb = 0; mb < 2; ++mb) {
#pragma unroll
      for (int half = 0; half < 2; ++half) {
        unsigned c0 = pack_bf16x2(st[mb][8 * half + 0], st[mb][8 * half + 1]);
        unsigned c1 = pack_bf16x2(st[mb][8 * half + 2], st[mb][8 * half + 3]);
        unsigned c2 = pack_bf16x2(st[mb][8 * half + 4], st[mb][8 * half + 5]);
        unsigned c3 = pack_bf16x2(st[mb][8 * half + 6], st[mb][8 * half + 7]);
        auto r02 = __builtin_amdgcn_permlane32_swap(c0, c2, false, false);
        auto r13 = __builtin_amdgcn_permlane32_swap(c1, c3, false, false);
        unsigned frag[4] = {(unsigned)r02[0], (unsigned)r13[0], (unsigned)r02[1],
                            (unsigned)r13[1]};
        pfrag[2 * mb + half] = *reinterpret_cast<s16x8*>(frag);
      }
    }

    // ---- O^T += V^T P^T ----
#pragma unroll
    for (int db = 0; db < DBLK; ++db) {
#pragma unroll
      for (int kc = 0; kc < 4; ++kc) {
        s16x8 vf = *reinterpret_cast<const s16x8*>(
            vt_lds + (32 * db + ln) * VT_PITCH + 16 * kc + 8 * hi);
        ot[db] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vf, pfrag[kc], ot[db], 0, 0, 0);
      }
    }
  }

  // ---- epilogue: normalize, transpose through LDS, coalesced store ----
  const float inv_l = l_run > 0.f ? 1.f / l_run : 0.f;
  if (lane < 32) lseg[(long long)bh * S + qrow] = m_run + __logf(fmaxf(l_run, 1e-30f));

  constexpr int OPITCH = HD + 8;
#pragma unroll
  for (int db = 0; db < DBLK; ++db)
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      int d = 32 * db + (r & 3) + 8 * (r >> 2) + 4 * hi;
      ot_lds[wid][ln * OPITCH + d] = f2bf(ot[db][r] * inv_l);
    }
  __builtin_amdgcn_s_waitcnt(0);  // lgkmcnt(0): own-wave LDS writes visible
#pragma unroll
  for (int it = 0; it < QB * HD / 8 / 64; ++it) {
    int c = it * 64 + lane;
    int row = c / (HD / 8), e0 = (c % (HD / 8)) * 8;
    s16x8 o8 = *reinterpret_cast<const s16x8*>(ot_lds[wid] + row * OPITCH + e0);
    *reinterpret_cast<s16x8*>(og + qbase + (long long)(q0 + row) * q_ss + e0) = o8;
  }
}

}  // namespace

// layout: "bhsd" = [B,H,S,D] contiguous; "bshd" = [B,S,H,D] contiguous
// (the model's natural projection layout — no transpose copies needed).
std::vector<torch::Tensor> attn_fwd_ex(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                                       bool causal, const std::string& layout) {
  TORCH_CHECK(q.dim() == 4 && q.dtype() == torch::kBFloat16 && q.is_contiguous());
  const bool bshd = layout == "bshd";
  const int B = q.size(0);
  const int Hq = bshd ? q.size(2) : q.size(1);
  const int S = bshd ? q.size(1) : q.size(2);
  const int HD = q.size(3);
  const int Hkv = bshd ? k.size(2) : k.size(1);
  TORCH_CHECK(Hq % Hkv == 0 && (HD == 64 || HD == 128));
  TORCH_CHECK(S % WGQ == 0, "seq len must be a multiple of 128");
  auto o = torch::empty_like(q);
  auto lse = torch::empty({B, Hq, S}, q.options().dtype(torch::kFloat32));
  const float scale = 1.0f / sqrtf((float)HD);
  dim3 grid(S / WGQ, B * Hq);
  hipStream_t stream = hypha_stream();
  long long q_sb, q_sh, q_ss, kv_sb, kv_sh, kv_ss;
  if (bshd) {
    q_sh = HD; q_ss = (long long)Hq * HD; q_sb = (long long)S * Hq * HD;
    kv_sh = HD; kv_ss = (long long)Hkv * HD; kv_sb = (long long)S * Hkv * HD;
  } else {
    q_ss = HD; q_sh = (long long)S * HD; q_sb = (long long)Hq * S * HD;
    kv_ss = HD; kv_sh = (long long)S * HD; kv_sb = (long long)Hkv * S * HD;
  }
#define LAUNCH_FWD(HDV)                                                                  hipLaunchKernelGGL(attn_fwd_kernel<HDV>, grid, dim3(256), 0, stream,                                      (const short*)q.data_ptr(), (const short*)k.data_ptr(),                                (const short*)v.data_ptr(), (short*)o.data_ptr(),                                      lse.data_ptr<float>(), B, Hq, Hkv, S, scale, causal, q_sb, q_sh,                       q_ss, kv_sb, kv_sh, kv_ss)
  if (HD == 128)
    LAUNCH_FWD(128);
  else
    LAUNCH_FWD(64);
#undef LAUNCH_FWD
  return {o, lse};
}

std::vector<torch::Tensor> attn_fwd(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                                    bool causal) {
  return attn_fwd_ex(q, k, v, causal, "bhsd");
}

// ===========================================================================
// Backward
// ===========================================================================

namespace {

// Di = rowsum(dO * O) per (b,h,q) row — FA2 preprocess. One wave per row.
template <int HD>
__global__ void attn_bwd_di_kernel(const short* __restrict__ dog,
                                   const short* __restrict__ og,
                                   float* __restrict__ dig, long long nrows, int Hq,
                                   int S, long long sb, long long sh, long long ss) {
  const long long row = (long long)blockIdx.x * 4 + (threadIdx.x >> 6);
  if (row >= nrows) return;
  const long long b = row / ((long long)Hq * S);
  const long long h = (row / S) % Hq;
  const long long sq = row % S;
  const long long addr = b * sb + h * sh + sq * ss;
  const int lane = threadIdx.x & 63;
  float acc = 0.f;
#pragma unroll
  for (int i = 0; i < HD / 64; ++i) {
    int d = lane + 64 * i;
    acc += bf2f(dog[addr + d]) * bf2f(og[addr + d]);
  }
  acc = wave_reduce_sum(acc);
  if (lane == 0) dig[row] = acc;
}

__global__ void cast_f32_to_bf16_kernel(const float* __restrict__ in,
                                        short* __restrict__ out, long long n) {
  const long long stride = (long long)gridDim.x * blockDim.x * 4;
  for (long long base = ((long long)blockIdx.x * blockDim.x + threadIdx.x) * 4; base < n;
       base += stride) {
    if (base + 4 <= n) {
      f32x4 v = *reinterpret_cast<const f32x4*>(in + base);
      s16x4 o;
#pragma unroll
      for (int j = 0; j < 4; ++j) o[j] = f2bf(v[j]);
      *reinterpret_cast<s16x4*>(out + base) = o;
    } else {
      for (long long i = base; i < n; ++i) out[i] = f2bf(in[i]);
    }
  }
}

// Main backward: one workgroup owns TWO 32-row kv tiles of one (b, hkv);
// q-tiles (and the GQA q-head group) are walked by all 4 waves round-robin.
// Per staged q-tile the wave processes BOTH kv tiles (staging, lse/Di loads
// and dQ atomics amortize over 2x the MFMA work; dQ accumulates across both
// tiles in registers so the global f32 atomics halve). dK/dV accumulate in
// per-wave, per-tile MFMA accumulators (AGPR-backed at 1 wave/SIMD) and are
// combined through LDS at the end. P/dS are recomputed from q/k/lse
// (FA2-style).
template <int HD>
__global__ __launch_bounds__(256, 1) void attn_bwd_kernel(
    const short* __restrict__ qg, const short* __restrict__ kg,
    const short* __restrict__ vg, const short* __restrict__ dog,
    const float* __restrict__ lseg, const float* __restrict__ dig,
    float* __restrict__ dqg, short* __restrict__ dkg, short* __restrict__ dvg,
    int B, int Hq, int Hkv, int S, float scale, bool causal,
    long long q_sb, long long q_sh, long long q_ss,
    long long kv_sb, long long kv_sh, long long kv_ss) {
  constexpr int KVT = 32;        // kv rows per tile
  constexpr int NT = 2;          // kv tiles per workgroup
  constexpr int QT = 32;         // q rows per tile
  constexpr int KC = HD / 16;    // chunks over head dim
  constexpr int DBLK = HD / 32;  // d blocks
  constexpr int TP = 40;         // ptds/dst pitch (16B-aligned rows)
  constexpr int DSP = NT * KVT + 8;  // ds image pitch (72: 16B-aligned)
  constexpr int KTP = NT * KVT + 8;  // kt image pitch
  // q-image: [HD][32] bf16 with an 8-elem group swizzle (conflict-free b128)
  __shared__ __attribute__((aligned(16))) short k_img[NT * KVT * HD];   // [kv][d] swizzled
  __shared__ __attribute__((aligned(16))) short v_img[NT * KVT * HD];   // [kv][d] swizzled
  __shared__ __attribute__((aligned(16))) short kt_img[HD * KTP];       // [d][kv]
  __shared__ __attribute__((aligned(16))) short qt_img[NWAVE][HD * 32];   // [d][q] swizzled
  __shared__ __attribute__((aligned(16))) short dot_img[NWAVE][HD * 32];  // [d][q] swizzled
  __shared__ __attribute__((aligned(16))) short ptds_img[NWAVE][KVT * TP];  // ptT per tile
  __shared__ __attribute__((aligned(16))) short dst_img[NWAVE][KVT * TP];   // dsT per tile
  __shared__ __attribute__((aligned(16))) short ds_img[NWAVE][QT * DSP];    // dS [q][kv 0..63]

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int hi = lane >> 5;
  const int ln = lane & 31;

  const int bhkv = blockIdx.y;
  const int b = bhkv / Hkv;
  const int hkv = bhkv % Hkv;
  const int G = Hq / Hkv;
  const int kv_base = blockIdx.x * (NT * KVT);
  const long long kvbase = (long long)b * kv_sb + (long long)hkv * kv_sh;

  // swizzled q-image offset: 8-elem groups XORed by (d>>2)&3
  auto qimg_off = [](int d, int qe) {
    return d * 32 + ((((qe >> 3) ^ ((d >> 2) & 3)) << 3) | (qe & 7));
  };

  // ---- stage K, V (swizzled row-major, both tiles) and K^T ----
  {
    constexpr int CH = NT * KVT * HD / 8;
    for (int c = tid; c < CH; c += 256) {
      int row = c / (HD / 8), e0 = (c % (HD / 8)) * 8;
      s16x8 k8 = *reinterpret_cast<const s16x8*>(kg + kvbase + (long long)(kv_base + row) * kv_ss + e0);
      *reinterpret_cast<s16x8*>((char*)k_img + k_lds_off<HD>(row, e0)) = k8;
      s16x8 v8 = *reinterpret_cast<const s16x8*>(vg + kvbase + (long long)(kv_base + row) * kv_ss + e0);
      *reinterpret_cast<s16x8*>((char*)v_img + k_lds_off<HD>(row, e0)) = v8;
    }
    // K^T: lane-per-kv transpose staging (conflict-free scalar stores)
    for (int i = 0; i < HD / 32; ++i) {
      int kvr = tid & 63;
      int e0 = (i * 4 + (tid >> 6)) * 8;
      s16x8 k8 = *reinterpret_cast<const s16x8*>(kg + kvbase + (long long)(kv_base + kvr) * kv_ss + e0);
#pragma unroll
      for (int j = 0; j < 8; ++j) kt_img[(e0 + j) * KTP + kvr] = k8[j];
    }
  }
  __syncthreads();

  f32x16 dk_acc[NT][DBLK] = {};  // D[m=kv][n=d] per tile
  f32x16 dv_acc[NT][DBLK] = {};  // D[m=d][n=kv] per tile

  short* qt = qt_img[wid];
  short* dot = dot_img[wid];
  short* ptds = ptds_img[wid];
  short* dst = dst_img[wid];
  short* dsw = ds_img[wid];

  const int t0 = causal ? kv_base / QT : 0;
  const int Tq = S / QT;

  for (int hq = hkv * G; hq < (hkv + 1) * G; ++hq) {
    const long long qbase = (long long)b * q_sb + (long long)hq * q_sh;
    const long long lsebase = (long long)(b * Hq + hq) * S;
    for (int t = t0 + wid; t < Tq; t += NWAVE) {
      const int q0 = t * QT;
      const int qrow = q0 + ln;  // this lane's q (for B-operand frags)

      // ---- per-wave staging: swizzled Q^T / dO^T images (lane-per-q) ----
#pragma unroll
      for (int it = 0; it < HD / 16; ++it) {
        int d0 = 8 * hi + 16 * it;
        s16x8 q8 = *reinterpret_cast<const s16x8*>(qg + qbase + (long long)(q0 + ln) * q_ss + d0);
        s16x8 d8 = *reinterpret_cast<const s16x8*>(dog + qbase + (long long)(q0 + ln) * q_ss + d0);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          qt[qimg_off(d0 + j, ln)] = q8[j];
          dot[qimg_off(d0 + j, ln)] = d8[j];
        }
      }
      const float lse = lseg[lsebase + qrow];
      const float di = dig[lsebase + qrow];

#pragma unroll
      for (int tile = 0; tile < NT; ++tile) {
        const int kv0 = kv_base + tile * KVT;
        if (causal && kv0 > q0 + QT - 1) continue;  // tile above the diagonal

        // ---- S^T = K Q^T (Q frags straight from global) ----
        f32x16 st = {};
#pragma unroll
        for (int kc = 0; kc < KC; ++kc) {
          s16x8 qf = *reinterpret_cast<const s16x8*>(qg + qbase + (long long)qrow * q_ss +
                                                     16 * kc + 8 * hi);
          s16x8 kf = *reinterpret_cast<const s16x8*>(
              (char*)k_img + k_lds_off<HD>(tile * KVT + ln, 16 * kc + 8 * hi));
          st = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qf, st, 0, 0, 0);
        }

        // ---- P^T = exp(scale*S^T - lse) with causal mask; write ptT ----
        f32x16 pt;
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          int kv = kv0 + (r & 3) + 8 * (r >> 2) + 4 * hi;
          float pv = __expf(st[r] * scale - lse);
          if (causal && kv > qrow) pv = 0.f;
          pt[r] = pv;
        }
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          int kv = (r & 3) + 8 * (r >> 2) + 4 * hi;
          ptds[kv * TP + ln] = f2bf(pt[r]);
        }

        // ---- dV^T += dO^T P  (A = dO^T image, B = ptT image) ----
#pragma unroll
        for (int db = 0; db < DBLK; ++db) {
#pragma unroll
          for (int kcq = 0; kcq < QT / 16; ++kcq) {
            s16x8 af = *reinterpret_cast<const s16x8*>(
                dot + qimg_off(32 * db + ln, 16 * kcq + 8 * hi));
            s16x8 bf = *reinterpret_cast<const s16x8*>(ptds + ln * TP + 16 * kcq + 8 * hi);
            dv_acc[tile][db] =
                __builtin_amdgcn_mfma_f32_32x32x16_bf16(af, bf, dv_acc[tile][db], 0, 0, 0);
          }
        }

        // ---- dP^T = V dO^T (A = V image, B = dO frags from global) ----
        f32x16 dpt = {};
#pragma unroll
        for (int kc = 0; kc < KC; ++kc) {
          s16x8 df = *reinterpret_cast<const s16x8*>(dog + qbase + (long long)qrow * q_ss +
                                                     16 * kc + 8 * hi);
          s16x8 vf = *reinterpret_cast<const s16x8*>(
              (char*)v_img + k_lds_off<HD>(tile * KVT + ln, 16 * kc + 8 * hi));
          dpt = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vf, df, dpt, 0, 0, 0);
        }

        // ---- dS^T = P^T * (dP^T - Di) * scale; write dsT + ds columns ----
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          int kv = (r & 3) + 8 * (r >> 2) + 4 * hi;
          float v = pt[r] * (dpt[r] - di) * scale;
          short vb = f2bf(v);
          dst[kv * TP + ln] = vb;
          dsw[ln * DSP + tile * KVT + kv] = vb;
        }

        // ---- dK += dS^T Q (A = dsT image, B = swizzled Q^T image) ----
#pragma unroll
        for (int db = 0; db < DBLK; ++db) {
#pragma unroll
          for (int kcq = 0; kcq < QT / 16; ++kcq) {
            s16x8 af = *reinterpret_cast<const s16x8*>(dst + ln * TP + 16 * kcq + 8 * hi);
            s16x8 bf = *reinterpret_cast<const s16x8*>(
                qt + qimg_off(32 * db + ln, 16 * kcq + 8 * hi));
            dk_acc[tile][db] =
                __builtin_amdgcn_mfma_f32_32x32x16_bf16(af, bf, dk_acc[tile][db], 0, 0, 0);
          }
        }
      }

      // ---- dQ(tile-pair) = dS K over kv 0..63; one atomic pass ----
      const int kck_hi = (causal && kv_base + KVT > q0 + QT - 1) ? NT : 2 * NT;
#pragma unroll
      for (int db = 0; db < DBLK; ++db) {
        f32x16 dq = {};
        for (int kck = 0; kck < kck_hi && kck < 2 * NT; ++kck) {
          s16x8 af = *reinterpret_cast<const s16x8*>(dsw + ln * DSP + 16 * kck + 8 * hi);
          s16x8 bf = *reinterpret_cast<const s16x8*>(kt_img + (32 * db + ln) * KTP +
                                                     16 * kck + 8 * hi);
          dq = __builtin_amdgcn_mfma_f32_32x32x16_bf16(af, bf, dq, 0, 0, 0);
        }
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          int qi = q0 + (r & 3) + 8 * (r >> 2) + 4 * hi;
          atomicAdd(dqg + (qbase + (long long)qi * q_ss + 32 * db + ln), dq[r]);
        }
      }
    }
  }

  // ---- combine dK/dV across waves (phased, no atomics), store bf16 ----
  // the combine buffer reuses the (now dead) qt image area: 16 KB f32 within
  // the 64 KB qt arena; every use is bracketed by __syncthreads()
  float* comb2 = reinterpret_cast<float*>(&qt_img[0][0]);
  __syncthreads();
  auto combine_store = [&](int tile, const f32x16* acc, short* outg, bool transposed) {
    constexpr int KVT = 32;
    for (int c = tid; c < KVT * HD; c += 256) comb2[c] = 0.f;
    __syncthreads();
    for (int w = 0; w < NWAVE; ++w) {
      if (w == wid) {
#pragma unroll
        for (int db = 0; db < DBLK; ++db)
#pragma unroll
          for (int r = 0; r < 16; ++r) {
            int m = (r & 3) + 8 * (r >> 2) + 4 * hi;
            int kv = transposed ? ln : m;
            int d = transposed ? (32 * db + m) : (32 * db + ln);
            comb2[kv * HD + d] += acc[db][r];
          }
      }
      __syncthreads();
    }
    const int kv0 = kv_base + tile * KVT;
    for (int c = tid; c < KVT * HD / 8; c += 256) {
      int row = c / (HD / 8), e0 = (c % (HD / 8)) * 8;
      s16x8 o8;
#pragma unroll
      for (int j = 0; j < 8; ++j) o8[j] = f2bf(comb2[row * HD + e0 + j]);
      *reinterpret_cast<s16x8*>(outg + kvbase + (long long)(kv0 + row) * kv_ss + e0) = o8;
    }
    __syncthreads();
  };
#pragma unroll
  for (int tile = 0; tile < NT; ++tile) {
    combine_store(tile, dk_acc[tile], dkg, false);
    combine_store(tile, dv_acc[tile], dvg, true);
  }
}

// Backward v3 (experimental, HYPHA_ATTN_BWD_V3=1): occupancy-first variant.
// All 4 waves co-process ONE q-tile per iteration (v2 gives each wave its own
// q-tile), splitting work by (kv-tile = wid&1, d-block subset = wid>>1):
//   * qt/dO^T staging is cooperative (4x cheaper per wave);
//   * each (tile, db) accumulator is owned by exactly one wave, so dK/dV
//     store directly with NO cross-wave combine phase;
//   * the dS [q][kv] image for dQ reuses the ptds buffers (per tile, [32][40]
//     halves split at the kv-32 boundary) behind an explicit barrier;
//   * LDS = 76 KB -> TWO blocks/CU (v2: 152 KB -> one), 8 waves in flight.
// Cost: S^T and dP^T are computed twice (once per wave pair sharing a tile),
// +40% MFMA issue — the bet is that the kernel is latency- not MFMA-bound
// (PMC: SQ_WAIT_ANY ~3x SQ_BUSY on v2). A/B via tools/attn_bench.py.
template <int HD>
__global__ __launch_bounds__(256, 2) void attn_bwd_v3_kernel(
    const short* __restrict__ qg, const short* __restrict__ kg,
    const short* __restrict__ vg, const short* __restrict__ dog,
    const float* __restrict__ lseg, const float* __restrict__ dig,
    float* __restrict__ dqg, short* __restrict__ dkg, short* __restrict__ dvg,
    int B, int Hq, int Hkv, int S, float scale, bool causal,
    long long q_sb, long long q_sh, long long q_ss,
    long long kv_sb, long long kv_sh, long long kv_ss) {
  constexpr int KVT = 32;
  constexpr int NT = 2;
  constexpr int QT = 32;
  constexpr int KC = HD / 16;
  constexpr int DBLK = HD / 32;
  constexpr int TP = 40;
  constexpr int KTP = NT * KVT + 8;
  constexpr int NACC = DBLK >= 2 ? DBLK / 2 : 1;
  __shared__ __attribute__((aligned(16))) short k_img[NT * KVT * HD];
  __shared__ __attribute__((aligned(16))) short v_img[NT * KVT * HD];
  __shared__ __attribute__((aligned(16))) short kt_img[HD * KTP];
  __shared__ __attribute__((aligned(16))) short qt_img[HD * 32];       // shared, 1 copy
  __shared__ __attribute__((aligned(16))) short dot_img[HD * 32];      // shared, 1 copy
  __shared__ __attribute__((aligned(16))) short pd_img[NT][KVT * TP];  // ptT, then dS [q][kv]
  __shared__ __attribute__((aligned(16))) short dst_img[NT][KVT * TP];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int hi = lane >> 5;
  const int ln = lane & 31;
  const int tile = wid & 1;   // this wave's kv tile
  const int dbb = wid >> 1;   // this wave's d-block base (stride 2)

  const int bhkv = blockIdx.y;
  const int b = bhkv / Hkv;
  const int hkv = bhkv % Hkv;
  const int G = Hq / Hkv;
  const int kv_base = blockIdx.x * (NT * KVT);
  const long long kvbase = (long long)b * kv_sb + (long long)hkv * kv_sh;

  auto qimg_off = [](int d, int qe) {
    return d * 32 + ((((qe >> 3) ^ ((d >> 2) & 3)) << 3) | (qe & 7));
  };

  {  // stage K, V (swizzled) and K^T — identical to v2
    constexpr int CH = NT * KVT * HD / 8;
    for (int c = tid; c < CH; c += 256) {
      int row = c / (HD / 8), e0 = (c % (HD / 8)) * 8;
      s16x8 k8 = *reinterpret_cast<const s16x8*>(kg + kvbase + (long long)(kv_base + row) * kv_ss + e0);
      *reinterpret_cast<s16x8*>((char*)k_img + k_lds_off<HD>(row, e0)) = k8;
      s16x8 v8 = *reinterpret_cast<const s16x8*>(vg + kvbase + (long long)(kv_base + row) * kv_ss + e0);
      *reinterpret_cast<s16x8*>((char*)v_img + k_lds_off<HD>(row, e0)) = v8;
    }
    for (int i = 0; i < HD / 32; ++i) {
      int kvr = tid & 63;
      int e0 = (i * 4 + (tid >> 6)) * 8;
      s16x8 k8 = *reinterpret_cast<const s16x8*>(kg + kvbase + (long long)(kv_base + kvr) * kv_ss + e0);
#pragma unroll
      for (int j = 0; j < 8; ++j) kt_img[(e0 + j) * KTP + kvr] = k8[j];
    }
  }
  __syncthreads();

  f32x16 dk_acc[NACC] = {};
  f32x16 dv_acc[NACC] = {};

  const int t0 = causal ? kv_base / QT : 0;
  const int Tq = S / QT;
  const int kv0 = kv_base + tile * KVT;

  for (int hq = hkv * G; hq < (hkv + 1) * G; ++hq) {
    const long long qbase = (long long)b * q_sb + (long long)hq * q_sh;
    const long long lsebase = (long long)(b * Hq + hq) * S;
    for (int t = t0; t < Tq; ++t) {
      const int q0 = t * QT;
      const int qrow = q0 + ln;

      // cooperative Q^T / dO^T staging: the HD/16 column-chunks split 4 ways
      for (int it = wid; it < HD / 16; it += NWAVE) {
        int d0 = 8 * hi + 16 * it;
        s16x8 q8 = *reinterpret_cast<const s16x8*>(qg + qbase + (long long)(q0 + ln) * q_ss + d0);
        s16x8 d8 = *reinterpret_cast<const s16x8*>(dog + qbase + (long long)(q0 + ln) * q_ss + d0);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          qt_img[qimg_off(d0 + j, ln)] = q8[j];
          dot_img[qimg_off(d0 + j, ln)] = d8[j];
        }
      }
      const float lse = lseg[lsebase + qrow];
      const float di = dig[lsebase + qrow];
      __syncthreads();  // staged images ready

      const bool active = !(causal && kv0 > q0 + QT - 1);
      f32x16 pt;
      f32x16 dpt = {};
      if (active) {
        // S^T = K Q^T
        f32x16 st = {};
#pragma unroll
        for (int kc = 0; kc < KC; ++kc) {
          s16x8 qf = *reinterpret_cast<const s16x8*>(qg + qbase + (long long)qrow * q_ss +
                                                     16 * kc + 8 * hi);
          s16x8 kf = *reinterpret_cast<const s16x8*>(
              (char*)k_img + k_lds_off<HD>(tile * KVT + ln, 16 * kc + 8 * hi));
          st = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qf, st, 0, 0, 0);
        }
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          int kv = kv0 + (r & 3) + 8 * (r >> 2) + 4 * hi;
          float pv = __expf(st[r] * scale - lse);
          if (causal && kv > qrow) pv = 0.f;
          pt[r] = pv;
        }
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          int kv = (r & 3) + 8 * (r >> 2) + 4 * hi;
          pd_img[tile][kv * TP + ln] = f2bf(pt[r]);
        }
        // dV^T += dO^T P for this wave's d-blocks
#pragma unroll
        for (int i = 0; i < NACC; ++i) {
          int db = dbb + 2 * i;
#pragma unroll
          for (int kcq = 0; kcq < QT / 16; ++kcq) {
            s16x8 af = *reinterpret_cast<const s16x8*>(
                dot_img + qimg_off(32 * db + ln, 16 * kcq + 8 * hi));
            s16x8 bf = *reinterpret_cast<const s16x8*>(pd_img[tile] + ln * TP +
                                                       16 * kcq + 8 * hi);
            dv_acc[i] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(af, bf, dv_acc[i], 0, 0, 0);
          }
        }
        // dP^T = V dO^T
#pragma unroll
        for (int kc = 0; kc < KC; ++kc) {
          s16x8 df = *reinterpret_cast<const s16x8*>(dog + qbase + (long long)qrow * q_ss +
                                                     16 * kc + 8 * hi);
          s16x8 vf = *reinterpret_cast<const s16x8*>(
              (char*)v_img + k_lds_off<HD>(tile * KVT + ln, 16 * kc + 8 * hi));
          dpt = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vf, df, dpt, 0, 0, 0);
        }
      }
      __syncthreads();  // both tile-waves done reading ptds (dS reuses it)

      if (active) {
        // dS^T; write dsT (for dK) and the dS [q][kv] half (for dQ, aliased)
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          int kv = (r & 3) + 8 * (r >> 2) + 4 * hi;
          float v = pt[r] * (dpt[r] - di) * scale;
          short vb = f2bf(v);
          dst_img[tile][kv * TP + ln] = vb;
          pd_img[tile][ln * TP + kv] = vb;
        }
        // dK += dS^T Q for this wave's d-blocks
#pragma unroll
        for (int i = 0; i < NACC; ++i) {
          int db = dbb + 2 * i;
#pragma unroll
          for (int kcq = 0; kcq < QT / 16; ++kcq) {
            s16x8 af = *reinterpret_cast<const s16x8*>(dst_img[tile] + ln * TP +
                                                       16 * kcq + 8 * hi);
            s16x8 bf = *reinterpret_cast<const s16x8*>(
                qt_img + qimg_off(32 * db + ln, 16 * kcq + 8 * hi));
            dk_acc[i] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(af, bf, dk_acc[i], 0, 0, 0);
          }
        }
      }
      __syncthreads();  // dS halves visible to every wave

      // dQ = dS K over kv 0..63; d-blocks split across waves
      const int kck_hi = (causal && kv_base + KVT > q0 + QT - 1) ? NT : 2 * NT;
      for (int db = wid; db < DBLK; db += NWAVE) {
        f32x16 dq = {};
        for (int kck = 0; kck < kck_hi; ++kck) {
          s16x8 af = *reinterpret_cast<const s16x8*>(pd_img[kck >> 1] + ln * TP +
                                                     16 * (kck & 1) + 8 * hi);
          s16x8 bf = *reinterpret_cast<const s16x8*>(kt_img + (32 * db + ln) * KTP +
                                                     16 * kck + 8 * hi);
          dq = __builtin_amdgcn_mfma_f32_32x32x16_bf16(af, bf, dq, 0, 0, 0);
        }
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          int qi = q0 + (r & 3) + 8 * (r >> 2) + 4 * hi;
          atomicAdd(dqg + (qbase + (long long)qi * q_ss + 32 * db + ln), dq[r]);
        }
      }
      __syncthreads();  // before the next q-tile restages qt/dot/pd
    }
  }

  // Direct bf16 stores: each (tile, db) accumulator has exactly one owner
#pragma unroll
  for (int i = 0; i < NACC; ++i) {
    int db = dbb + 2 * i;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      int m = (r & 3) + 8 * (r >> 2) + 4 * hi;
      // dK: D[m=kv][n=d]
      dkg[kvbase + (long long)(kv0 + m) * kv_ss + 32 * db + ln] = f2bf(dk_acc[i][r]);
      // dV: transposed D[m=d][n=kv]
      dvg[kvbase + (long long)(kv0 + ln) * kv_ss + 32 * db + m] = f2bf(dv_acc[i][r]);
    }
  }
}

// Backward v4 (default for S%128==0): kv-split cooperative structure.
// One workgroup = 4 waves; each wave OWNS one 32-row kv tile (128 kv rows
// per workgroup, 2x v2), and ALL waves walk ALL q-tiles together:
//   * Q^T/dO^T staging is cooperative (1/4 of v2's per-wave cost) and
//     register-prefetched one q-tile ahead (T14 split: HBM latency hides
//     under the previous tile's MFMA phases);
//   * each wave's dK/dV accumulators see every q-tile, so they are COMPLETE
//     for its 32 kv rows -> direct bf16 stores, no cross-wave combine;
//   * dQ covers kv 0..127 in ONE atomic pass (1/4 of v2's atomic traffic
//     per flop) with d-blocks split across waves;
//   * P^T/dS^T images stay per-wave (no barrier); the shared dS [q][kv]
//     image and the staged q-images need 2 barriers per q-tile.
// Rationale (r1 profile): v2 at 1 wave/SIMD spent ~95% of cycles outside
// MFMA issue on per-wave staging, 4x the dQ atomics, and exposed HBM
// latency; this structure attacks all three without duplicating any MFMA
// (v3's mistake, measured -10%).
// `ablate` is a perf-diagnosis bitmask (HYPHA_ATTN_ABLATE, default 0=full):
// 1 = skip the dQ atomics, 2 = skip the whole dQ phase, 4 = skip the exp.
template <int HD>
__global__ __launch_bounds__(256, 1) void attn_bwd_v4_kernel(
    const short* __restrict__ qg, const short* __restrict__ kg,
    const short* __restrict__ vg, const short* __restrict__ dog,
    const float* __restrict__ lseg, const float* __restrict__ dig,
    float* __restrict__ dqg, short* __restrict__ dkg, short* __restrict__ dvg,
    int B, int Hq, int Hkv, int S, float scale, bool causal,
    long long q_sb, long long q_sh, long long q_ss,
    long long kv_sb, long long kv_sh, long long kv_ss, int ablate,
    int dq_bf16) {
  constexpr int KVT = 32;        // kv rows per wave
  constexpr int NT = 4;          // kv tiles (= waves) per workgroup
  constexpr int KVW = NT * KVT;  // 128 kv rows per workgroup
  constexpr int QT = 32;         // q rows per tile
  constexpr int KC = HD / 16;
  constexpr int DBLK = HD / 32;
  constexpr int TP = 40;           // per-wave ptds/dst pitch
  constexpr int DSP = KVW + 8;     // shared dS [q][kv] pitch (136: 16B rows)
  constexpr int KTP = KVW + 8;     // kt image pitch
  __shared__ __attribute__((aligned(16))) short k_img[KVW * HD];      // [kv][d] swizzled
  __shared__ __attribute__((aligned(16))) short v_img[KVW * HD];      // [kv][d] swizzled
  __shared__ __attribute__((aligned(16))) short kt_img[HD * KTP];     // [d][kv]
  __shared__ __attribute__((aligned(16))) short qt_img[HD * 32];      // [d][q] swizzled, shared
  __shared__ __attribute__((aligned(16))) short dot_img[HD * 32];     // [d][q] swizzled, shared
  __shared__ __attribute__((aligned(16))) short ptds_img[NWAVE][KVT * TP];  // per-wave ptT
  __shared__ __attribute__((aligned(16))) short dst_img[NWAVE][KVT * TP];   // per-wave dsT
  // double-buffered: tile t's dQ pass is DEFERRED into tile t+1's compute
  // phase (its MFMAs and atomics interleave with the next tile's work, and
  // causal-idle waves get dQ work) while tile t+1 writes the other buffer
  __shared__ __attribute__((aligned(16))) short ds_img[2][QT * DSP];  // dS [q][kv 0..127]

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int hi = lane >> 5;
  const int ln = lane & 31;

  const int bhkv = blockIdx.y;
  const int b = bhkv / Hkv;
  const int hkv = bhkv % Hkv;
  const int G = Hq / Hkv;
  const int kv_base = blockIdx.x * KVW;
  const long long kvbase = (long long)b * kv_sb + (long long)hkv * kv_sh;

  auto qimg_off = [](int d, int qe) {
    return d * 32 + ((((qe >> 3) ^ ((d >> 2) & 3)) << 3) | (qe & 7));
  };

  // ---- stage K, V (swizzled row-major) and K^T for all 128 kv rows ----
  {
    constexpr int CH = KVW * HD / 8;
    for (int c = tid; c < CH; c += 256) {
      int row = c / (HD / 8), e0 = (c % (HD / 8)) * 8;
      s16x8 k8 = *reinterpret_cast<const s16x8*>(kg + kvbase + (long long)(kv_base + row) * kv_ss + e0);
      *reinterpret_cast<s16x8*>((char*)k_img + k_lds_off<HD>(row, e0)) = k8;
      s16x8 v8 = *reinterpret_cast<const s16x8*>(vg + kvbase + (long long)(kv_base + row) * kv_ss + e0);
      *reinterpret_cast<s16x8*>((char*)v_img + k_lds_off<HD>(row, e0)) = v8;
    }
#pragma unroll
    for (int half = 0; half < KVW / 64; ++half) {
      for (int i = 0; i < HD / 32; ++i) {
        int kvr = (tid & 63) + 64 * half;
        int e0 = (i * 4 + (tid >> 6)) * 8;
        s16x8 k8 = *reinterpret_cast<const s16x8*>(kg + kvbase + (long long)(kv_base + kvr) * kv_ss + e0);
#pragma unroll
        for (int j = 0; j < 8; ++j) kt_img[(e0 + j) * KTP + kvr] = k8[j];
      }
    }
  }
  __syncthreads();

  f32x16 dk_acc[DBLK] = {};  // D[m=kv][n=d], this wave's 32 kv rows
  f32x16 dv_acc[DBLK] = {};  // D[m=d][n=kv]

  short* ptds = ptds_img[wid];
  short* dst = dst_img[wid];
  const int kv0 = kv_base + wid * KVT;  // this wave's kv tile

  const int t0 = causal ? kv_base / QT : 0;
  const int Tq = S / QT;
  // cooperative q-image staging: wave w writes d-chunks {w, w+4} of 8
  constexpr int NSTG = HD / 16 / NWAVE;  // chunks per wave (2 for HD=128)

  // deferred-dQ pipeline state: the previous q-tile whose dS image is
  // complete but whose dQ pass has not run yet (crosses hq boundaries:
  // kt_img is per-hkv, shared by every head in the GQA group)
  int prev_q0 = -1, prev_na = 0, pbuf = 0;
  long long prev_qbase = 0;

  auto run_dq = [&](int buf) {
    if (wid >= DBLK || prev_q0 < 0) return;
    const int db = wid;
    const int kck_hi = 2 * prev_na;
    f32x16 dq = {};
    for (int kck = 0; kck < kck_hi; ++kck) {
      s16x8 af = *reinterpret_cast<const s16x8*>(ds_img[buf] + ln * DSP + 16 * kck + 8 * hi);
      s16x8 bf = *reinterpret_cast<const s16x8*>(kt_img + (32 * db + ln) * KTP +
                                                 16 * kck + 8 * hi);
      dq = __builtin_amdgcn_mfma_f32_32x32x16_bf16(af, bf, dq, 0, 0, 0);
    }
    if (!(ablate & 1)) {
      if (dq_bf16) {
        // HYPHA_ATTN_DQ_BF16: halve the dQ RMW traffic with packed-bf16
        // atomics (adjacent d columns live in lane pairs: one shuffle
        // pairs them, even lanes issue one v2bf16 add for both)
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          int qi = prev_q0 + (r & 3) + 8 * (r >> 2) + 4 * hi;
          float other = __shfl_xor(dq[r], 1, 64);
          if ((ln & 1) == 0) {
            bf16x2 pkt;
            pkt.x = (__bf16)dq[r];
            pkt.y = (__bf16)other;
            __builtin_amdgcn_global_atomic_fadd_v2bf16(
                reinterpret_cast<bf16x2*>(
                    reinterpret_cast<short*>(dqg) +
                    (prev_qbase + (long long)qi * q_ss + 32 * db + ln)),
                pkt);
          }
        }
      } else {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          int qi = prev_q0 + (r & 3) + 8 * (r >> 2) + 4 * hi;
          atomicAdd(dqg + (prev_qbase + (long long)qi * q_ss + 32 * db + ln), dq[r]);
        }
      }
    } else {
      asm volatile("" ::"v"(dq[0]));  // keep the MFMA chain live (perf probe)
    }
  };

  for (int hq = hkv * G; hq < (hkv + 1) * G; ++hq) {
    const long long qbase = (long long)b * q_sb + (long long)hq * q_sh;
    const long long lsebase = (long long)(b * Hq + hq) * S;

    // prefetch the first q-tile of this head into registers
    s16x8 qreg[NSTG], doreg[NSTG];
    auto issue_prefetch = [&](int t) {
#pragma unroll
      for (int i = 0; i < NSTG; ++i) {
        int it = wid + NWAVE * i;
        int d0 = 8 * hi + 16 * it;
        qreg[i] = *reinterpret_cast<const s16x8*>(qg + qbase + (long long)(t * QT + ln) * q_ss + d0);
        doreg[i] = *reinterpret_cast<const s16x8*>(dog + qbase + (long long)(t * QT + ln) * q_ss + d0);
      }
    };
    issue_prefetch(t0);
    float lse = lseg[lsebase + t0 * QT + ln];
    float di = dig[lsebase + t0 * QT + ln];

    for (int t = t0; t < Tq; ++t) {
      const int q0 = t * QT;
      const int qrow = q0 + ln;

      // ---- (a) write the prefetched q-tile into the shared images ----
#pragma unroll
      for (int i = 0; i < NSTG; ++i) {
        int it = wid + NWAVE * i;
        int d0 = 8 * hi + 16 * it;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          qt_img[qimg_off(d0 + j, ln)] = qreg[i][j];
          dot_img[qimg_off(d0 + j, ln)] = doreg[i][j];
        }
      }
      __syncthreads();  // (b) staged images ready

      // ---- (c) issue next tile's loads; they land under the MFMA work ----
      const float lse_cur = lse, di_cur = di;
      if (t + 1 < Tq) {
        issue_prefetch(t + 1);
        lse = lseg[lsebase + (t + 1) * QT + ln];
        di = dig[lsebase + (t + 1) * QT + ln];
      }

      // active kv tiles for this q-tile (wave tiles above the diagonal idle)
      const int na = causal ? min(NT, (q0 + QT - 1 - kv_base) / KVT + 1) : NT;
      const int cbuf = pbuf ^ 1;

      // deferred dQ of the PREVIOUS tile: its dS buffer is complete and
      // nobody writes it this phase — the dQ MFMAs and atomics overlap the
      // current tile's compute instead of sitting in their own phase
      if (!(ablate & 2)) run_dq(pbuf);

      if (wid < na) {
        // ---- S^T = K Q^T (Q frags from global: L1-hot after staging) ----
        f32x16 st = {};
#pragma unroll
        for (int kc = 0; kc < KC; ++kc) {
          s16x8 qf = *reinterpret_cast<const s16x8*>(qg + qbase + (long long)qrow * q_ss +
                                                     16 * kc + 8 * hi);
          s16x8 kf = *reinterpret_cast<const s16x8*>(
              (char*)k_img + k_lds_off<HD>(wid * KVT + ln, 16 * kc + 8 * hi));
          st = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qf, st, 0, 0, 0);
        }
        // ---- P^T = exp(scale*S^T - lse), causal mask; per-wave ptT ----
        f32x16 pt;
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          int kv = kv0 + (r & 3) + 8 * (r >> 2) + 4 * hi;
          float pv = (ablate & 4) ? st[r] * scale : __expf(st[r] * scale - lse_cur);
          if (causal && kv > qrow) pv = 0.f;
          pt[r] = pv;
        }
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          int kv = (r & 3) + 8 * (r >> 2) + 4 * hi;
          ptds[kv * TP + ln] = f2bf(pt[r]);
        }
        // ---- dV^T += dO^T P (A from shared dO^T image, B from own ptT) ----
#pragma unroll
        for (int db = 0; db < DBLK; ++db) {
#pragma unroll
          for (int kcq = 0; kcq < QT / 16; ++kcq) {
            s16x8 af = *reinterpret_cast<const s16x8*>(
                dot_img + qimg_off(32 * db + ln, 16 * kcq + 8 * hi));
            s16x8 bf = *reinterpret_cast<const s16x8*>(ptds + ln * TP + 16 * kcq + 8 * hi);
            dv_acc[db] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(af, bf, dv_acc[db], 0, 0, 0);
          }
        }
        // ---- dP^T = V dO^T ----
        f32x16 dpt = {};
#pragma unroll
        for (int kc = 0; kc < KC; ++kc) {
          s16x8 df = *reinterpret_cast<const s16x8*>(dog + qbase + (long long)qrow * q_ss +
                                                     16 * kc + 8 * hi);
          s16x8 vf = *reinterpret_cast<const s16x8*>(
              (char*)v_img + k_lds_off<HD>(wid * KVT + ln, 16 * kc + 8 * hi));
          dpt = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vf, df, dpt, 0, 0, 0);
        }
        // ---- dS^T = P^T (dP^T - Di) scale; own dsT + shared dS columns ----
#pragma unroll
        for (int r = 0; r < 16; r += 2) {
          int kv = (r & 3) + 8 * (r >> 2) + 4 * hi;
          float v0 = pt[r] * (dpt[r] - di_cur) * scale;
          float v1 = pt[r + 1] * (dpt[r + 1] - di_cur) * scale;
          dst[kv * TP + ln] = f2bf(v0);
          dst[(kv + 1) * TP + ln] = f2bf(v1);
          // adjacent r -> adjacent kv: one packed b32 store into [q][kv]
          *reinterpret_cast<unsigned*>(ds_img[cbuf] + ln * DSP + wid * KVT + kv) =
              pack_bf16x2(v0, v1);
        }
        // ---- dK += dS^T Q (B from shared Q^T image) ----
#pragma unroll
        for (int db = 0; db < DBLK; ++db) {
#pragma unroll
          for (int kcq = 0; kcq < QT / 16; ++kcq) {
            s16x8 af = *reinterpret_cast<const s16x8*>(dst + ln * TP + 16 * kcq + 8 * hi);
            s16x8 bf = *reinterpret_cast<const s16x8*>(
                qt_img + qimg_off(32 * db + ln, 16 * kcq + 8 * hi));
            dk_acc[db] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(af, bf, dk_acc[db], 0, 0, 0);
          }
        }
      }
      __syncthreads();  // (e) shared dS complete; qt/dot free to rewrite

      prev_q0 = q0;
      prev_na = na;
      prev_qbase = qbase;
      pbuf = cbuf;
    }
  }

  // drain: the final q-tile's dQ pass (its dS buffer is complete after (e))
  if (!(ablate & 2)) run_dq(pbuf);

  // ---- direct bf16 stores: this wave owns kv rows [kv0, kv0+32) fully ----
#pragma unroll
  for (int db = 0; db < DBLK; ++db) {
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      int m = (r & 3) + 8 * (r >> 2) + 4 * hi;
      dkg[kvbase + (long long)(kv0 + m) * kv_ss + 32 * db + ln] = f2bf(dk_acc[db][r]);
      dvg[kvbase + (long long)(kv0 + ln) * kv_ss + 32 * db + m] = f2bf(dv_acc[db][r]);
    }
  }
}

}  // namespace

std::vector<torch::Tensor> attn_bwd_ex(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                                       torch::Tensor o, torch::Tensor dout,
                                       torch::Tensor lse, bool causal,
                                       const std::string& layout) {
  const bool bshd = layout == "bshd";
  const int B = q.size(0);
  const int Hq = bshd ? q.size(2) : q.size(1);
  const int S = bshd ? q.size(1) : q.size(2);
  const int HD = q.size(3);
  const int Hkv = bshd ? k.size(2) : k.size(1);
  TORCH_CHECK(S % 64 == 0 && (HD == 64 || HD == 128));
  auto dq32 = torch::empty({0}, q.options());  // allocated after variant choice
  auto dk = torch::empty_like(k);
  auto dv = torch::empty_like(v);
  auto di = torch::empty({B, Hq, S}, q.options().dtype(torch::kFloat32));
  const float scale = 1.0f / sqrtf((float)HD);
  hipStream_t stream = hypha_stream();
  long long nrows = (long long)B * Hq * S;
  long long q_sb, q_sh, q_ss, kv_sb, kv_sh, kv_ss;
  if (bshd) {
    q_sh = HD; q_ss = (long long)Hq * HD; q_sb = (long long)S * Hq * HD;
    kv_sh = HD; kv_ss = (long long)Hkv * HD; kv_sb = (long long)S * Hkv * HD;
  } else {
    q_ss = HD; q_sh = (long long)S * HD; q_sb = (long long)Hq * S * HD;
    kv_ss = HD; kv_sh = (long long)S * HD; kv_sb = (long long)Hkv * S * HD;
  }

  // kernel selection: v4 (kv-split cooperative, S%128) is the default;
  // HYPHA_ATTN_BWD=v2|v3|v4 overrides for A/B runs.
  static const int bwd_variant = [] {
    const char* e = getenv("HYPHA_ATTN_BWD");
    if (e && e[0] == 'v' && e[1]) return e[1] - '0';
    if (e && e[0] >= '2' && e[0] <= '4') return e[0] - '0';
    const char* e3 = getenv("HYPHA_ATTN_BWD_V3");  // legacy toggle
    return (e3 && e3[0] == '1') ? 3 : 4;
  }();
  const int variant = (bwd_variant == 4 && S % 128 != 0) ? 2 : bwd_variant;
  static const int bwd_ablate = [] {
    const char* e = getenv("HYPHA_ATTN_ABLATE");
    return e ? atoi(e) : 0;
  }();
  // packed-bf16 dQ accumulation (v4 only): halves the dominant RMW traffic;
  // costs ~3 bits of dq mantissa over 16 accumulations — opt-in
  static const bool dq_bf16_env = [] {
    const char* e = getenv("HYPHA_ATTN_DQ_BF16");
    return e && e[0] == '1';
  }();
  const bool dq_bf16 = dq_bf16_env && variant == 4;
  dq32 = dq_bf16 ? torch::zeros(q.sizes(), q.options())
                 : torch::zeros(q.sizes(), q.options().dtype(torch::kFloat32));

#define LAUNCH_BWD(KERN, NQ, HDV)                                                       \
  hipLaunchKernelGGL(KERN<HDV>, dim3(S / NQ, B * Hkv), dim3(256), 0, stream,            \
                     (const short*)q.data_ptr(), (const short*)k.data_ptr(),            \
                     (const short*)v.data_ptr(), (const short*)dout.data_ptr(),         \
                     lse.data_ptr<float>(), di.data_ptr<float>(),                       \
                     dq32.data_ptr<float>(), (short*)dk.data_ptr(),                     \
                     (short*)dv.data_ptr(), B, Hq, Hkv, S, scale, causal, q_sb,         \
                     q_sh, q_ss, kv_sb, kv_sh, kv_ss)

#define DISPATCH(HDV)                                                                   \
  do {                                                                                  \
    hipLaunchKernelGGL(attn_bwd_di_kernel<HDV>, dim3((unsigned)((nrows + 3) / 4)),      \
                       dim3(256), 0, stream, (const short*)dout.data_ptr(),             \
                       (const short*)o.data_ptr(), di.data_ptr<float>(), nrows, Hq, S,  \
                       q_sb, q_sh, q_ss);                                               \
    if (variant == 4)                                                                   \
      hipLaunchKernelGGL(attn_bwd_v4_kernel<HDV>, dim3(S / 128, B * Hkv), dim3(256),    \
                         0, stream, (const short*)q.data_ptr(),                         \
                         (const short*)k.data_ptr(), (const short*)v.data_ptr(),        \
                         (const short*)dout.data_ptr(), lse.data_ptr<float>(),          \
                         di.data_ptr<float>(), (float*)dq32.data_ptr(),                 \
                         (short*)dk.data_ptr(), (short*)dv.data_ptr(), B, Hq, Hkv, S,   \
                         scale, causal, q_sb, q_sh, q_ss, kv_sb, kv_sh, kv_ss,          \
                         bwd_ablate, dq_bf16 ? 1 : 0);                                  \
    else if (variant == 3)                                                              \
      LAUNCH_BWD(attn_bwd_v3_kernel, 64, HDV);                                          \
    else                                                                                \
      LAUNCH_BWD(attn_bwd_kernel, 64, HDV);                                             \
  } while (0)

  if (HD == 128)
    DISPATCH(128);
  else
    DISPATCH(64);
#undef DISPATCH

  if (dq_bf16) return {dq32, dk, dv};  // accumulated directly in bf16
  auto dq = torch::empty_like(q);
  long long n = dq32.numel();
  hipLaunchKernelGGL(cast_f32_to_bf16_kernel, dim3(elementwise_grid((n + 3) / 4)),
                     dim3(256), 0, stream, dq32.data_ptr<float>(), (short*)dq.data_ptr(),
                     n);
  return {dq, dk, dv};
}

std::vector<torch::Tensor> attn_bwd(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                                    torch::Tensor o, torch::Tensor dout,
                                    torch::Tensor lse, bool causal) {
  return attn_bwd_ex(q, k, v, o, dout, lse, causal, "bhsd");
}
