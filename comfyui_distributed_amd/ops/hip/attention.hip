// Flash-attention forward for diffusion UNet/DiT blocks (MI355X/gfx950).
//
// Hand-written CDNA4 kernel, v3. Structure per 64-key tile:
//   * 8-wave workgroup owns 128 q rows (wave w rows [16w, 16w+16)), so the
//     shared V^T stage is amortized over 8 waves and each SIMD carries 2+
//     waves of this kernel for latency overlap;
//   * K fragments are double-buffered in registers: tile t+1's eight
//     16-byte K loads are issued before tile t's PV phase (async-stage
//     split — HBM latency hides under MFMA work);
//   * online softmax with defer-max (skip the O-rescale and the m update
//     while the tile max stays within DEFER_THR of the running max; the
//     exp inputs stay bounded by e^DEFER_THR which fp32 accumulation
//     tolerates);
//   * epilogue divides replaced by one reciprocal per row.
//
// Contract (enforced by ops/attention.py): q [BH, Nq_pad, D_PAD] bf16 with
// Nq_pad % 128 == 0; k/v [BHk, Nk_pad, D_PAD] with Nk_pad % 64 == 0,
// zero-padded; D_PAD in {64, 96, 128, 160}; real Nk masks padded keys;
// grid.x = bh so one head's K/V stays on one XCD's L2 (dispatcher places
// block b on XCD b%8).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

typedef __attribute__((ext_vector_type(8))) short short8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

#define QROWS_PER_WAVE 16
#define NWAVES 8
#define QROWS_PER_BLOCK (QROWS_PER_WAVE * NWAVES)  // 128
#define KT 64
#define KFRAG (KT / 16)
#define VT_PITCH (KT + 8)
#define PT_PITCH (KT + 8)
#define DEFER_THR 8.0f

template <int D_PAD>
__global__ __launch_bounds__(NWAVES * 64) void attn_fwd_kernel(
    const uint16_t* __restrict__ q, const uint16_t* __restrict__ k,
    const uint16_t* __restrict__ v, uint16_t* __restrict__ o, int Nq_pad,
    int Nk_pad, int Nk, int H, int Hkv, float scale) {
  constexpr int DK = D_PAD / 32;
  constexpr int DN = D_PAD / 16;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int bh = blockIdx.x;
  const int h = bh % H;
  const int bhk = (bh / H) * Hkv + h / (H / Hkv);
  const int q_row0 = blockIdx.y * QROWS_PER_BLOCK + wave * QROWS_PER_WAVE;

  const uint16_t* qbase = q + ((long long)bh * Nq_pad + q_row0) * D_PAD;
  const uint16_t* kbase = k + (long long)bhk * Nk_pad * D_PAD;
  const uint16_t* vbase = v + (long long)bhk * Nk_pad * D_PAD;

  __shared__ __align__(16) uint16_t v_t[D_PAD][VT_PITCH];
  __shared__ __align__(16) uint16_t p_lds[NWAVES][QROWS_PER_WAVE][PT_PITCH];

  // ---- Q fragments resident ---------------------------------------------
  short8 qfrag[DK];
  {
    const int row = lane & 15;
    const int d0 = (lane >> 4) * 8;
#pragma unroll
    for (int kk = 0; kk < DK; ++kk)
      qfrag[kk] = *reinterpret_cast<const short8*>(
          qbase + (long long)row * D_PAD + kk * 32 + d0);
  }

  float m_run[4], l_run[4];
  f32x4 oacc[DN];
#pragma unroll
  for (int r = 0; r < 4; ++r) { m_run[r] = -1e30f; l_run[r] = 0.f; }
#pragma unroll
  for (int n = 0; n < DN; ++n) oacc[n] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int ntiles = Nk_pad / KT;
  const int kd0 = (lane >> 4) * 8;

  // K fragment double buffer: kf[kk][f] for the CURRENT tile.
  short8 kf[DK][KFRAG];
  auto load_kfrags = [&](int key0, short8 dst[DK][KFRAG]) {
#pragma unroll
    for (int kk = 0; kk < DK; ++kk)
#pragma unroll
      for (int f = 0; f < KFRAG; ++f) {
        const int key = key0 + f * 16 + (lane & 15);
        dst[kk][f] = *reinterpret_cast<const short8*>(
            kbase + (long long)key * D_PAD + kk * 32 + kd0);
      }
  };
  load_kfrags(0, kf);

  for (int t = 0; t < ntiles; ++t) {
    const int key0 = t * KT;

    // ---- stage V^T cooperatively ----------------------------------------
    __syncthreads();
    for (int c = threadIdx.x; c < KT * (D_PAD / 8); c += NWAVES * 64) {
      const int key = c & (KT - 1);
      const int d0 = (c / KT) * 8;
      short8 vv = *reinterpret_cast<const short8*>(
          vbase + (long long)(key0 + key) * D_PAD + d0);
#pragma unroll
      for (int j = 0; j < 8; ++j) v_t[d0 + j][key] = (uint16_t)vv[j];
    }
    __syncthreads();

    // ---- scores from the prefetched K fragments --------------------------
    f32x4 s[KFRAG];
#pragma unroll
    for (int f = 0; f < KFRAG; ++f) s[f] = f32x4{0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int kk = 0; kk < DK; ++kk)
#pragma unroll
      for (int f = 0; f < KFRAG; ++f)
        s[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qfrag[kk], kf[kk][f],
                                                       s[f], 0, 0, 0);

    // ---- prefetch next tile's K while softmax runs -----------------------
    if (t + 1 < ntiles) load_kfrags(key0 + KT, kf);

#pragma unroll
    for (int f = 0; f < KFRAG; ++f) {
      const int key = key0 + f * 16 + (lane & 15);
#pragma unroll
      for (int r = 0; r < 4; ++r)
        s[f][r] = (key < Nk) ? s[f][r] * scale : -1e30f;
    }

    // ---- online softmax with defer-max -----------------------------------
    float p[KFRAG][4];
    float mt[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float m = s[0][r];
#pragma unroll
      for (int f = 1; f < KFRAG; ++f) m = fmaxf(m, s[f][r]);
#pragma unroll
      for (int off = 8; off > 0; off >>= 1)
        m = fmaxf(m, __shfl_xor(m, off, 64));
      mt[r] = m;
    }
    // wave-uniform defer decision: skip rescale while every row's tile max
    // stays within DEFER_THR of its running max (T13; P bounded by e^THR)
    bool need = false;
#pragma unroll
    for (int r = 0; r < 4; ++r) need |= (mt[r] - m_run[r]) > DEFER_THR;
    if (__ballot(need) != 0ull) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const float m_new = fmaxf(m_run[r], mt[r]);
        const float a = __expf(m_run[r] - m_new);
        m_run[r] = m_new;
        l_run[r] *= a;
#pragma unroll
        for (int n = 0; n < DN; ++n) oacc[n][r] *= a;
      }
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float rowsum = 0.f;
#pragma unroll
      for (int f = 0; f < KFRAG; ++f) {
        p[f][r] = __expf(s[f][r] - m_run[r]);
        rowsum += p[f][r];
      }
#pragma unroll
      for (int off = 8; off > 0; off >>= 1)
        rowsum += __shfl_xor(rowsum, off, 64);
      l_run[r] += rowsum;
    }

    // ---- P -> LDS (wave-private) -----------------------------------------
    {
      const int col = lane & 15;
      const int rg = lane >> 4;
#pragma unroll
      for (int f = 0; f < KFRAG; ++f)
#pragma unroll
        for (int r = 0; r < 4; ++r)
          p_lds[wave][rg * 4 + r][f * 16 + col] = f32_to_bf16_bits(p[f][r]);
    }

    // ---- PV ----------------------------------------------------------------
    {
      const int prow = lane & 15;
      const int pk0 = (lane >> 4) * 8;
#pragma unroll
      for (int kc = 0; kc < KT / 32; ++kc) {
        short8 pfrag = *reinterpret_cast<const short8*>(
            &p_lds[wave][prow][kc * 32 + pk0]);
#pragma unroll
        for (int n = 0; n < DN; ++n) {
          short8 vfrag = *reinterpret_cast<const short8*>(
              &v_t[n * 16 + (lane & 15)][kc * 32 + pk0]);
          oacc[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pfrag, vfrag,
                                                            oacc[n], 0, 0, 0);
        }
      }
    }
  }

  // ---- epilogue: one reciprocal per row ---------------------------------
  {
    const int col = lane & 15;
    const int rg = lane >> 4;
    float rl[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) rl[r] = __builtin_amdgcn_rcpf(l_run[r]);
#pragma unroll
    for (int n = 0; n < DN; ++n) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = rg * 4 + r;
        o[((long long)bh * Nq_pad + q_row0 + row) * D_PAD + n * 16 + col] =
            f32_to_bf16_bits(oacc[n][r] * rl[r]);
      }
    }
  }
}

torch::Tensor attn_fwd(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                       int64_t heads, int64_t kv_heads, int64_t nk_real,
                       double scale) {
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == at::kBFloat16);
  TORCH_CHECK(q.dim() == 3 && k.dim() == 3 && v.dim() == 3,
              "q/k/v must be [B*H, N, D_PAD]");
  TORCH_CHECK(q.is_contiguous() && k.is_contiguous() && v.is_contiguous());
  const int D = q.size(2);
  TORCH_CHECK(D == 64 || D == 96 || D == 128 || D == 160,
              "D_PAD must be one of 64/96/128/160, got ", D);
  const int Nq_pad = q.size(1), Nk_pad = k.size(1);
  TORCH_CHECK(Nq_pad % QROWS_PER_BLOCK == 0 && Nk_pad % KT == 0,
              "pad Nq to ", QROWS_PER_BLOCK, " and Nk to ", KT);
  TORCH_CHECK(k.size(2) == D && v.size(2) == D);
  const int BH = q.size(0);
  TORCH_CHECK(BH % heads == 0, "BH must divide heads");
  TORCH_CHECK(heads % kv_heads == 0, "GQA ratio must be integral");

  auto o = torch::empty_like(q);
  dim3 grid(BH, Nq_pad / QROWS_PER_BLOCK);
  dim3 block(NWAVES * 64);
  auto stream = at::hip::getCurrentHIPStream();
#define LAUNCH_D(DP)                                                          \
  hipLaunchKernelGGL((attn_fwd_kernel<DP>), grid, block, 0, stream,           \
                     (const uint16_t*)q.data_ptr(),                           \
                     (const uint16_t*)k.data_ptr(),                           \
                     (const uint16_t*)v.data_ptr(), (uint16_t*)o.data_ptr(),  \
                     Nq_pad, Nk_pad, (int)nk_real, (int)heads, (int)kv_heads, \
                     (float)scale)
  switch (D) {
    case 64: LAUNCH_D(64); break;
    case 96: LAUNCH_D(96); break;
    case 128: LAUNCH_D(128); break;
    case 160: LAUNCH_D(160); break;
  }
#undef LAUNCH_D
  HIP_CHECK_LAUNCH();
  return o;
}
