// Flash-attention forward for diffusion UNet/DiT blocks (MI355X/gfx950).
//
// Hand-written CDNA4 kernel: MFMA 16x16x32 bf16 tiles, online softmax with
// in-register row state, V staged transposed through LDS (conflict-free
// padded layout), K fragments streamed from global (the K tile is
// L2-resident: the grid is ordered so every workgroup of one (batch, head)
// lands on the same XCD — blockIdx.x = bh and the dispatcher places block b
// on XCD b%8, so with BH % 8 == 0 a head's K/V stays in one XCD's L2).
//
// Contract (enforced by the Python wrapper ops/attention.py):
//   q      [BH,  Nq_pad, D_PAD]  bf16, Nq_pad % 64 == 0
//   k, v   [BHk, Nk_pad, D_PAD]  bf16, Nk_pad % 64 == 0, zero-padded
//   o      [BH,  Nq_pad, D_PAD]  bf16 (written)
//   D_PAD in {64, 96, 128, 160}; real D zero-padded up; real Nk passed for
//   the softmax mask. scale applied to scores. gqa = H / H_kv.
//
// Geometry: 4-wave workgroup owns 64 q rows (wave w rows [16w,16w+16));
// K-tile = 64 keys = four 16x16 score fragments per wave; two barriers per
// 64-key tile.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

typedef __attribute__((ext_vector_type(8))) short short8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

#define QROWS_PER_WAVE 16
#define QROWS_PER_BLOCK 64
#define KT 64            // keys per tile
#define KFRAG (KT / 16)  // score fragments per wave per tile
#define VT_PITCH (KT + 8)  // V_T row pitch: 144 B rows -> conflict-free b128
#define PT_PITCH (KT + 8)

template <int D_PAD>
__global__ __launch_bounds__(256) void attn_fwd_kernel(
    const uint16_t* __restrict__ q, const uint16_t* __restrict__ k,
    const uint16_t* __restrict__ v, uint16_t* __restrict__ o, int Nq_pad,
    int Nk_pad, int Nk, int H, int Hkv, float scale) {
  constexpr int DK = D_PAD / 32;   // QK^T k-steps
  constexpr int DN = D_PAD / 16;   // O column fragments
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int bh = blockIdx.x;        // b * H + h  (XCD affinity: bh % 8)
  const int h = bh % H;
  const int bhk = (bh / H) * Hkv + h / (H / Hkv);
  const int q_row0 = blockIdx.y * QROWS_PER_BLOCK + wave * QROWS_PER_WAVE;

  const uint16_t* qbase = q + ((long long)bh * Nq_pad + q_row0) * D_PAD;
  const uint16_t* kbase = k + (long long)bhk * Nk_pad * D_PAD;
  const uint16_t* vbase = v + (long long)bhk * Nk_pad * D_PAD;

  __shared__ __align__(16) uint16_t v_t[D_PAD][VT_PITCH];
  __shared__ __align__(16) uint16_t p_lds[4][QROWS_PER_WAVE][PT_PITCH];

  // ---- Q fragments resident in registers --------------------------------
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
  for (int t = 0; t < ntiles; ++t) {
    const int key0 = t * KT;

    // ---- stage V^T cooperatively: v_t[d][key] = V[key0+key][d] ----------
    __syncthreads();
    for (int c = threadIdx.x; c < KT * (D_PAD / 8); c += 256) {
      const int key = c & (KT - 1);
      const int d0 = (c / KT) * 8;
      short8 vv = *reinterpret_cast<const short8*>(
          vbase + (long long)(key0 + key) * D_PAD + d0);
#pragma unroll
      for (int j = 0; j < 8; ++j) v_t[d0 + j][key] = (uint16_t)vv[j];
    }
    __syncthreads();

    // ---- scores: KFRAG fragments of 16 keys -----------------------------
    f32x4 s[KFRAG];
#pragma unroll
    for (int f = 0; f < KFRAG; ++f) s[f] = f32x4{0.f, 0.f, 0.f, 0.f};
    {
      const int d0 = (lane >> 4) * 8;
#pragma unroll
      for (int kk = 0; kk < DK; ++kk) {
#pragma unroll
        for (int f = 0; f < KFRAG; ++f) {
          const int key = key0 + f * 16 + (lane & 15);
          short8 kf = *reinterpret_cast<const short8*>(
              kbase + (long long)key * D_PAD + kk * 32 + d0);
          s[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qfrag[kk], kf, s[f], 0, 0, 0);
        }
      }
    }
#pragma unroll
    for (int f = 0; f < KFRAG; ++f) {
      const int key = key0 + f * 16 + (lane & 15);
#pragma unroll
      for (int r = 0; r < 4; ++r)
        s[f][r] = (key < Nk) ? s[f][r] * scale : -1e30f;
    }

    // ---- online softmax --------------------------------------------------
    float p[KFRAG][4], alpha[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float mt = s[0][r];
#pragma unroll
      for (int f = 1; f < KFRAG; ++f) mt = fmaxf(mt, s[f][r]);
#pragma unroll
      for (int off = 8; off > 0; off >>= 1)
        mt = fmaxf(mt, __shfl_xor(mt, off, 64));
      const float m_new = fmaxf(m_run[r], mt);
      alpha[r] = __expf(m_run[r] - m_new);
      m_run[r] = m_new;
      float rowsum = 0.f;
#pragma unroll
      for (int f = 0; f < KFRAG; ++f) {
        p[f][r] = __expf(s[f][r] - m_new);
        rowsum += p[f][r];
      }
#pragma unroll
      for (int off = 8; off > 0; off >>= 1)
        rowsum += __shfl_xor(rowsum, off, 64);
      l_run[r] = l_run[r] * alpha[r] + rowsum;
    }
#pragma unroll
    for (int n = 0; n < DN; ++n)
#pragma unroll
      for (int r = 0; r < 4; ++r) oacc[n][r] *= alpha[r];

    // ---- P -> LDS (wave-private; compiler orders ds_write->ds_read) -----
    {
      const int col = lane & 15;
      const int rg = lane >> 4;
#pragma unroll
      for (int f = 0; f < KFRAG; ++f)
#pragma unroll
        for (int r = 0; r < 4; ++r)
          p_lds[wave][rg * 4 + r][f * 16 + col] = f32_to_bf16_bits(p[f][r]);
    }

    // ---- PV: O += P[16 x KT] @ V[KT x D_PAD] ----------------------------
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

  // ---- epilogue ---------------------------------------------------------
  {
    const int col = lane & 15;
    const int rg = lane >> 4;
#pragma unroll
    for (int n = 0; n < DN; ++n) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = rg * 4 + r;
        const float val = oacc[n][r] / l_run[r];
        o[((long long)bh * Nq_pad + q_row0 + row) * D_PAD + n * 16 + col] =
            f32_to_bf16_bits(val);
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
  TORCH_CHECK(Nq_pad % 64 == 0 && Nk_pad % KT == 0,
              "pad Nq to 64 and Nk to ", KT);
  TORCH_CHECK(k.size(2) == D && v.size(2) == D);
  const int BH = q.size(0);
  TORCH_CHECK(BH % heads == 0, "BH must divide heads");
  TORCH_CHECK(heads % kv_heads == 0, "GQA ratio must be integral");

  auto o = torch::empty_like(q);
  // blockIdx.x = bh -> XCD affinity per head (dispatcher: XCD = block % 8)
  dim3 grid(BH, Nq_pad / QROWS_PER_BLOCK);
  dim3 block(256);
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
