#include "hip/hip_runtime.h"
// Flash-attention forward for diffusion UNet/DiT blocks (MI355X/gfx950).
//
// The per-tile sampling hot path the reference delegates to ComfyUI's torch
// stack (self/cross attention inside common_ksampler) is
// implemented here as a hand-written CDNA4 kernel: MFMA 16x16x32 bf16 tiles,
// online softmax with in-register row state, V staged transposed through LDS
// (conflict-free padded layout), K/B fragments streamed straight from
// global (the K tile is L2-resident across the many workgroups of one head;
// LDS-staging it is pure overhead at these sizes — see the CDNA4 guide's
// "LDS-staging data that L2-fits" note).
//
// Contract (enforced by the Python wrapper ops/attention.py):
//   q      [BH,  Nq_pad, D_PAD]  bf16, Nq_pad % 64 == 0
//   k, v   [BHk, Nk_pad, D_PAD]  bf16, Nk_pad % 32 == 0, zero-padded
//   o      [BH,  Nq_pad, D_PAD]  bf16 (written)
//   D_PAD in {64, 96, 128, 160}; real D zero-padded up; real Nk passed for
//   the softmax mask. scale applied to scores. gqa = H / H_kv.
//
// Geometry: one 4-wave workgroup owns 64 q rows; wave w owns rows
// [16w, 16w+16). Per 32-key tile: two 16x16 score fragments per wave.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

typedef __attribute__((ext_vector_type(8))) short short8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

#define QROWS_PER_WAVE 16
#define QROWS_PER_BLOCK 64
#define KT 32  // keys per tile
#define VT_PITCH 40  // V_T row pitch in elements (32 keys + 8 pad: 80 B rows,
                     // stride 20 dwords -> conflict-free b128 lane groups)
#define PT_PITCH 40  // P row pitch (same reasoning)

template <int D_PAD>
__global__ __launch_bounds__(256) void attn_fwd_kernel(
    const uint16_t* __restrict__ q, const uint16_t* __restrict__ k,
    const uint16_t* __restrict__ v, uint16_t* __restrict__ o, int Nq_pad,
    int Nk_pad, int Nk, int H, int Hkv, float scale) {
  constexpr int DK = D_PAD / 32;   // QK^T k-steps per score fragment
  constexpr int DN = D_PAD / 16;   // O column fragments
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int bh = blockIdx.y;        // b * H + h
  const int h = bh % H;
  const int bhk = (bh / H) * Hkv + h / (H / Hkv);
  const int q_row0 = blockIdx.x * QROWS_PER_BLOCK + wave * QROWS_PER_WAVE;

  const uint16_t* qbase = q + ((long long)bh * Nq_pad + q_row0) * D_PAD;
  const uint16_t* kbase = k + (long long)bhk * Nk_pad * D_PAD;
  const uint16_t* vbase = v + (long long)bhk * Nk_pad * D_PAD;

  // LDS: V^T tile (shared) + per-wave P scratch. 16-B aligned: both are
  // read with ds_read_b128 (misalignment = 64-cycle replays, guide G17).
  __shared__ __align__(16) uint16_t v_t[D_PAD][VT_PITCH];
  __shared__ __align__(16) uint16_t p_lds[4][QROWS_PER_WAVE][PT_PITCH];

  // ---- Q fragments: resident in registers for the whole row block ----
  // A-fragment for mfma_f32_16x16x32: lane holds row (lane&15),
  // k-elements (lane>>4)*8 .. +8.
  short8 qfrag[DK];
  {
    const int row = lane & 15;
    const int d0 = (lane >> 4) * 8;
#pragma unroll
    for (int kk = 0; kk < DK; ++kk)
      qfrag[kk] = *reinterpret_cast<const short8*>(
          qbase + (long long)row * D_PAD + kk * 32 + d0);
  }

  // ---- online-softmax state (replicated across each 16-lane row group) --
  float m_run[4], l_run[4];
  f32x4 oacc[DN];
#pragma unroll
  for (int r = 0; r < 4; ++r) { m_run[r] = -1e30f; l_run[r] = 0.f; }
#pragma unroll
  for (int n = 0; n < DN; ++n) oacc[n] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int ntiles = Nk_pad / KT;
  for (int t = 0; t < ntiles; ++t) {
    const int key0 = t * KT;

    // ---- stage V^T tile cooperatively: v_t[d][key] = V[key0+key][d] ----
    __syncthreads();  // previous tile's reads done before overwrite
    for (int c = threadIdx.x; c < KT * (D_PAD / 8); c += 256) {
      const int key = c & (KT - 1);
      const int d0 = (c / KT) * 8;
      short8 vv = *reinterpret_cast<const short8*>(
          vbase + (long long)(key0 + key) * D_PAD + d0);
#pragma unroll
      for (int j = 0; j < 8; ++j) v_t[d0 + j][key] = (uint16_t)vv[j];
    }
    __syncthreads();

    // ---- scores: two 16x16 fragments (keys 0-15 / 16-31 of the tile) ----
    f32x4 s0 = {0.f, 0.f, 0.f, 0.f}, s1 = {0.f, 0.f, 0.f, 0.f};
    {
      const int key_a = key0 + (lane & 15);
      const int key_b = key_a + 16;
      const int d0 = (lane >> 4) * 8;
#pragma unroll
      for (int kk = 0; kk < DK; ++kk) {
        // B fragment: lane holds key column (lane&15), k-elems contiguous.
        short8 kf_a = *reinterpret_cast<const short8*>(
            kbase + (long long)key_a * D_PAD + kk * 32 + d0);
        short8 kf_b = *reinterpret_cast<const short8*>(
            kbase + (long long)key_b * D_PAD + kk * 32 + d0);
        s0 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qfrag[kk], kf_a, s0, 0, 0, 0);
        s1 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qfrag[kk], kf_b, s1, 0, 0, 0);
      }
    }
    // scale + mask keys beyond the real Nk
    const int key_a = key0 + (lane & 15);
    const int key_b = key_a + 16;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      s0[r] = (key_a < Nk) ? s0[r] * scale : -1e30f;
      s1[r] = (key_b < Nk) ? s1[r] * scale : -1e30f;
    }

    // ---- online softmax (row r lives on the 16 lanes of this row group) --
    float p0[4], p1[4], alpha[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float mt = fmaxf(s0[r], s1[r]);
#pragma unroll
      for (int off = 8; off > 0; off >>= 1)
        mt = fmaxf(mt, __shfl_xor(mt, off, 64));
      const float m_new = fmaxf(m_run[r], mt);
      alpha[r] = __expf(m_run[r] - m_new);
      m_run[r] = m_new;
      p0[r] = __expf(s0[r] - m_new);
      p1[r] = __expf(s1[r] - m_new);
      float rowsum = p0[r] + p1[r];
#pragma unroll
      for (int off = 8; off > 0; off >>= 1)
        rowsum += __shfl_xor(rowsum, off, 64);
      l_run[r] = l_run[r] * alpha[r] + rowsum;
    }

    // ---- rescale O ----
#pragma unroll
    for (int n = 0; n < DN; ++n)
#pragma unroll
      for (int r = 0; r < 4; ++r) oacc[n][r] *= alpha[r];

    // ---- P -> LDS (bf16) in A-fragment-friendly row-major [16][PT_PITCH] --
    {
      const int col = lane & 15;
      const int rg = lane >> 4;  // row group: rows rg*4 .. rg*4+3
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        p_lds[wave][rg * 4 + r][col] = f32_to_bf16_bits(p0[r]);
        p_lds[wave][rg * 4 + r][col + 16] = f32_to_bf16_bits(p1[r]);
      }
    }
    // wave-private LDS: ds_write -> ds_read ordering within the wave is
    // guaranteed by the compiler's lgkmcnt bookkeeping; no barrier needed.

    // ---- PV: O[16 x D_PAD] += P[16 x 32] @ V[32 x D_PAD] ----
    {
      const int prow = lane & 15;
      const int pk0 = (lane >> 4) * 8;
      short8 pfrag = *reinterpret_cast<const short8*>(&p_lds[wave][prow][pk0]);
#pragma unroll
      for (int n = 0; n < DN; ++n) {
        // B fragment: V[k][d]: lane holds d-col (lane&15), k contiguous
        //  -> row (n*16 + lane&15) of v_t, 8 elems from pk0.
        short8 vfrag = *reinterpret_cast<const short8*>(
            &v_t[n * 16 + (lane & 15)][pk0]);
        oacc[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pfrag, vfrag, oacc[n], 0, 0, 0);
      }
    }
  }

  // ---- epilogue: O /= l, store bf16 ----
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
  TORCH_CHECK(Nq_pad % 64 == 0 && Nk_pad % 32 == 0, "pad Nq to 64, Nk to 32");
  TORCH_CHECK(k.size(2) == D && v.size(2) == D);
  const int BH = q.size(0);
  TORCH_CHECK(BH % heads == 0, "BH must divide heads");
  TORCH_CHECK(heads % kv_heads == 0, "GQA ratio must be integral");

  auto o = torch::empty_like(q);
  dim3 grid(Nq_pad / QROWS_PER_BLOCK, BH);
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
