// Flash-attention forward for diffusion UNet/DiT blocks (MI355X/gfx950).
//
// Hand-written CDNA4 kernel, v4: generalized strided addressing so the
// kernel consumes the QKV projections' natural packed layout
// [B, N, H*D] directly — no pad, no permute, no copies on the host side.
// The head dim D only needs to be a multiple of 8: fragment loads are
// masked at 8-element granularity against the padded compute width D_PAD.
//
// Structure per 64-key tile (see profiles/ for measurements):
//   * 8-wave workgroup owns 128 q rows; shared V^T stage amortized over
//     all 8 waves; grid.x = b*H + h so one head's K/V stays on one XCD's
//     L2 (dispatcher places block b on XCD b%8);
//   * K fragments double-buffered in registers (tile t+1's loads issue
//     before tile t's PV phase);
//   * online softmax with defer-max (T13) and a reciprocal epilogue.

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

#define ZERO8 short8{0, 0, 0, 0, 0, 0, 0, 0}

template <int D_PAD>  // multiple of 16; QK^T K-depth rounds up to 32s
__global__ __launch_bounds__(NWAVES * 64) void attn_fwd_kernel(
    const uint16_t* __restrict__ q, const uint16_t* __restrict__ k,
    const uint16_t* __restrict__ v, uint16_t* __restrict__ o,
    const uint16_t* __restrict__ zp,  // >=16B zeros: OOB loads redirect
                                      // here so every load is UNconditional
                                      // (a per-lane branchy load makes
                                      // hipcc emit vmcnt(0) per cluster —
                                      // guide §5 trap (c) — serializing
                                      // the whole prefetch pipeline)
    int Nq, int Nk,
    int D, int H, int Hkv, float scale, long long q_bstride,
    long long q_hstride, long long q_rstride, long long k_bstride,
    long long k_hstride, long long k_rstride, long long o_bstride,
    long long o_hstride, long long o_rstride) {
  constexpr int DK = (D_PAD + 31) / 32;
  constexpr int DN = D_PAD / 16;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int bh = blockIdx.x;
  const int b = bh / H;
  const int h = bh % H;
  const int hk = h / (H / Hkv);
  const int q_row0 = blockIdx.y * QROWS_PER_BLOCK + wave * QROWS_PER_WAVE;

  const uint16_t* qbase = q + b * q_bstride + h * q_hstride;
  const uint16_t* kbase = k + b * k_bstride + hk * k_hstride;
  const uint16_t* vbase = v + b * k_bstride + hk * k_hstride;
  uint16_t* obase = o + b * o_bstride + h * o_hstride;

  __shared__ __align__(16) uint16_t v_t[D_PAD][VT_PITCH];
  __shared__ __align__(16) uint16_t p_lds[NWAVES][QROWS_PER_WAVE][PT_PITCH];

  const int kd0 = (lane >> 4) * 8;

  // T5 static priority (microarch guide, two-waves-per-SIMD item 4): the
  // second-dispatched wave half loses issue arbitration on every segment;
  // ONE setprio for that half — and no per-segment flips — removes its
  // start-of-segment penalty
  if (wave >= NWAVES / 2) __builtin_amdgcn_s_setprio(1);

  // ---- Q fragments resident (OOB rows/d -> zero page) -------------------
  short8 qfrag[DK];
  {
    const int row = q_row0 + (lane & 15);
    const bool row_ok = row < Nq;
#pragma unroll
    for (int kk = 0; kk < DK; ++kk) {
      const int d = kk * 32 + kd0;
      const uint16_t* src = (row_ok && d + 8 <= D)
          ? qbase + (long long)row * q_rstride + d : zp;
      qfrag[kk] = *reinterpret_cast<const short8*>(src);
    }
  }

  // per-lane softmax row state: this lane's q row is (lane & 15)
  float m_run = -1e30f, l_run = 0.f;
  f32x4 oacc[DN];
#pragma unroll
  for (int n = 0; n < DN; ++n) oacc[n] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int ntiles = (Nk + KT - 1) / KT;

  short8 kf[DK][KFRAG];
  auto load_kfrags = [&](int key0, short8 dst[DK][KFRAG]) {
#pragma unroll
    for (int kk = 0; kk < DK; ++kk) {
      const int d = kk * 32 + kd0;
      const bool d_ok = d + 8 <= D;
#pragma unroll
      for (int f = 0; f < KFRAG; ++f) {
        const int key = key0 + f * 16 + (lane & 15);
        const uint16_t* src = (d_ok && key < Nk)
            ? kbase + (long long)key * k_rstride + d : zp;
        dst[kk][f] = *reinterpret_cast<const short8*>(src);
      }
    }
  };
  load_kfrags(0, kf);

  // ---- V register double-buffer: tile t+1's V rows are loaded during
  // tile t's compute and written to LDS at tile start — the old
  // load-then-__syncthreads stage exposed the full L2/HBM latency at
  // every tile's barrier (vmcnt(0) drain). One short8 slot per thread
  // covers KT*D_PAD/8 elements; big D needs 2-3 slots.
  constexpr int VSLOTS = (KT * (D_PAD / 8) + NWAVES * 64 - 1) / (NWAVES * 64);
  short8 vreg[VSLOTS];
  auto load_v = [&](int key0) {
#pragma unroll
    for (int s = 0; s < VSLOTS; ++s) {
      const int c = (int)threadIdx.x + s * NWAVES * 64;
      const int key = c & (KT - 1);
      const int d0 = (c / KT) * 8;
      // unconditional load (zero page covers c overflow, key tail and
      // d beyond D) — a lane-divergent branch would serialize the cluster
      const bool ok = c < KT * (D_PAD / 8) && key0 + key < Nk && d0 + 8 <= D;
      const uint16_t* src =
          ok ? vbase + (long long)(key0 + key) * k_rstride + d0 : zp;
      vreg[s] = *reinterpret_cast<const short8*>(src);
    }
  };
  load_v(0);

  for (int t = 0; t < ntiles; ++t) {
    const int key0 = t * KT;

    // ---- commit the prefetched V^T rows (barrier-bracketed, no global
    // wait: the loads were issued one full tile ago) ----------------------
    __syncthreads();
#pragma unroll
    for (int s = 0; s < VSLOTS; ++s) {
      const int c = (int)threadIdx.x + s * NWAVES * 64;
      if (c < KT * (D_PAD / 8)) {
        const int key = c & (KT - 1);
        const int d0 = (c / KT) * 8;
#pragma unroll
        for (int j = 0; j < 8; ++j) v_t[d0 + j][key] = (uint16_t)vreg[s][j];
      }
    }
    __syncthreads();

    // ---- scores, SWAPPED: S^T = mfma(K, Q) so each lane's 16 score
    // values all belong to ONE q row (col = lane&15) — softmax becomes
    // in-register with only 2 cross-lane steps (guide's swapped-QK^T
    // idiom).  s[f][r] = score(key = key0 + f*16 + (lane>>4)*4 + r,
    //                          qrow = lane&15)
    f32x4 s[KFRAG];
#pragma unroll
    for (int f = 0; f < KFRAG; ++f) s[f] = f32x4{0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int kk = 0; kk < DK; ++kk)
#pragma unroll
      for (int f = 0; f < KFRAG; ++f)
        s[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(kf[kk][f], qfrag[kk],
                                                       s[f], 0, 0, 0);

    // t+1 prefetches issue HERE — after the QK^T MFMAs' kf-wait, so the
    // compiler's vmcnt(0) before the MFMAs never drains the fresh loads
    if (t + 1 < ntiles) {
      load_kfrags(key0 + KT, kf);
      load_v(key0 + KT);
    }

#pragma unroll
    for (int f = 0; f < KFRAG; ++f) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int key = key0 + f * 16 + (lane >> 4) * 4 + r;
        s[f][r] = (key < Nk) ? s[f][r] * scale : -1e30f;
      }
    }

    // ---- online softmax, per-lane row state ------------------------------
    float mt = s[0][0];
#pragma unroll
    for (int f = 0; f < KFRAG; ++f)
#pragma unroll
      for (int r = 0; r < 4; ++r) mt = fmaxf(mt, s[f][r]);
    mt = fmaxf(mt, __shfl_xor(mt, 16, 64));
    mt = fmaxf(mt, __shfl_xor(mt, 32, 64));

    const bool need = (mt - m_run) > DEFER_THR;
    if (__ballot(need) != 0ull) {
      const float m_new = fmaxf(m_run, mt);
      const float a = __expf(m_run - m_new);
      m_run = m_new;
      l_run *= a;
      // O fragment rows are (lane>>4)*4+r: fetch each row's alpha from a
      // lane holding that row's state (same 16-lane group, col == row)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int src = (lane & 48) | (((lane >> 4) * 4 + r) & 15);
        const float ar = __shfl(a, src, 64);
#pragma unroll
        for (int n = 0; n < DN; ++n) oacc[n][r] *= ar;
      }
    }
    float p[KFRAG][4];
    float rowsum = 0.f;
#pragma unroll
    for (int f = 0; f < KFRAG; ++f)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        p[f][r] = __expf(s[f][r] - m_run);
        rowsum += p[f][r];
      }
    rowsum += __shfl_xor(rowsum, 16, 64);
    rowsum += __shfl_xor(rowsum, 32, 64);
    l_run += rowsum;

    // ---- P -> LDS in the PV A-fragment layout ----------------------------
    {
      const int qrow = lane & 15;
      const int g = lane >> 4;
#pragma unroll
      for (int f = 0; f < KFRAG; ++f)
#pragma unroll
        for (int r = 0; r < 4; ++r)
          p_lds[wave][qrow][f * 16 + g * 4 + r] = f32_to_bf16_bits(p[f][r]);
    }

    // ---- PV --------------------------------------------------------------
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
    const float rl_own = __builtin_amdgcn_rcpf(l_run);
    float rl[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int src = (lane & 48) | ((rg * 4 + r) & 15);
      rl[r] = __shfl(rl_own, src, 64);
    }
#pragma unroll
    for (int n = 0; n < DN; ++n) {
      const int d = n * 16 + col;
      if (d >= D) continue;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = q_row0 + rg * 4 + r;
        if (row >= Nq) continue;
        obase[(long long)row * o_rstride + d] =
            f32_to_bf16_bits(oacc[n][r] * rl[r]);
      }
    }
  }
}

// intentionally leaked (see gemm.hip zero_page_ptr): exit-time tensor
// destruction would outlive the HIP context
static torch::Tensor* g_attn_zero_page = nullptr;

static const uint16_t* attn_zero_page(const torch::Tensor& like) {
  if (!g_attn_zero_page || g_attn_zero_page->device() != like.device())
    g_attn_zero_page = new torch::Tensor(
        torch::zeros({64}, like.options().dtype(at::kBFloat16)));
  return (const uint16_t*)g_attn_zero_page->data_ptr();
}

static torch::Tensor launch_attn_raw(const uint16_t* qp, const uint16_t* kp,
                                     const uint16_t* vp, torch::Tensor o,
                                     int H, int Hkv, int Nq, int Nk, int D,
                                     float scale, long long qb, long long qh,
                                     long long qr, long long kb, long long kh,
                                     long long kr, int batch) {
  const uint16_t* zp = attn_zero_page(o);
  // D=40 (SD1.5's hot dim) pads to 48, not 64: PV runs 3 d-fragments
  // instead of 4 and the V^T stage shrinks 25% on the dominant kernel
  const int dpad = D <= 48 ? 48
                 : (D <= 64 ? 64 : (D <= 96 ? 96 : (D <= 128 ? 128 : 160)));
  TORCH_CHECK(D % 8 == 0 && D <= 160, "head dim must be %8 and <=160, got ", D);
  // output is always a fresh packed [B, Nq, H*D] tensor
  const long long ob = (long long)Nq * H * D, oh = D, orr = (long long)H * D;
  dim3 grid(batch * H, (Nq + QROWS_PER_BLOCK - 1) / QROWS_PER_BLOCK);
  dim3 block(NWAVES * 64);
  auto stream = at::hip::getCurrentHIPStream();
#define LAUNCH_D(DP)                                                          \
  hipLaunchKernelGGL((attn_fwd_kernel<DP>), grid, block, 0, stream, qp, kp,   \
                     vp, (uint16_t*)o.data_ptr(), zp, Nq, Nk, D, H, Hkv,      \
                     scale, qb, qh, qr, kb, kh, kr, ob, oh, orr)
  switch (dpad) {
    case 48: LAUNCH_D(48); break;
    case 64: LAUNCH_D(64); break;
    case 96: LAUNCH_D(96); break;
    case 128: LAUNCH_D(128); break;
    case 160: LAUNCH_D(160); break;
  }
#undef LAUNCH_D
  HIP_CHECK_LAUNCH();
  return o;
}

// Legacy layout: q [B*H, N, D] contiguous (one head per batch row).
torch::Tensor attn_fwd(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                       int64_t heads, int64_t kv_heads, int64_t nk_real,
                       double scale) {
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == at::kBFloat16);
  TORCH_CHECK(q.dim() == 3 && q.is_contiguous() && k.is_contiguous()
              && v.is_contiguous());
  const int D = q.size(2);
  const int Nq = q.size(1), Nk = (int)nk_real;
  const int BH = q.size(0);
  TORCH_CHECK(BH % heads == 0);
  const int batch = BH / (int)heads;
  const long long qr = D, qh = (long long)Nq * D,
                  qb = (long long)heads * Nq * D;
  const long long kr = D, kh = (long long)k.size(1) * D,
                  kb = (long long)kv_heads * k.size(1) * D;
  auto o = torch::empty({batch, Nq, heads * D}, q.options());
  launch_attn_raw((const uint16_t*)q.data_ptr(), (const uint16_t*)k.data_ptr(),
                  (const uint16_t*)v.data_ptr(), o, (int)heads, (int)kv_heads,
                  Nq, Nk, D, (float)scale, qb, qh, qr, kb, kh, kr, batch);
  // packed output [B, Nq, H*D] -> legacy [B*H, Nq, D] view requires a
  // permute copy; callers of the legacy API accept it (tests, WAN path)
  return o.reshape({batch, Nq, (int)heads, D})
      .permute({0, 2, 1, 3})
      .reshape({BH, Nq, D})
      .contiguous();
}

// Packed layout: q [B, Nq, H*D], k/v [B, Nk, Hkv*D] — the natural output of
// fused QKV projections; zero host-side reshapes.
torch::Tensor attn_fwd_packed(torch::Tensor q, torch::Tensor k,
                              torch::Tensor v, int64_t heads,
                              int64_t kv_heads, double scale) {
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == at::kBFloat16);
  TORCH_CHECK(q.dim() == 3 && q.is_contiguous() && k.is_contiguous()
              && v.is_contiguous());
  const int B = q.size(0), Nq = q.size(1), Nk = k.size(1);
  const int D = q.size(2) / (int)heads;
  TORCH_CHECK((int)(heads * D) == q.size(2), "q last dim must be H*D");
  TORCH_CHECK((int)(kv_heads * D) == k.size(2), "k last dim must be Hkv*D");
  const long long qr = (long long)heads * D, qh = D,
                  qb = (long long)Nq * heads * D;
  const long long kr = (long long)kv_heads * D, kh = D,
                  kb = (long long)Nk * kv_heads * D;
  auto o = torch::empty_like(q);
  launch_attn_raw((const uint16_t*)q.data_ptr(), (const uint16_t*)k.data_ptr(),
                  (const uint16_t*)v.data_ptr(), o, (int)heads, (int)kv_heads,
                  Nq, Nk, D, (float)scale, qb, qh, qr, kb, kh, kr, B);
  return o;
}

// Fused self-attention: qkv [B, N, 3*H*D] from one projection GEMM; the
// kernel reads q/k/v through strides — zero splits, zero copies.
torch::Tensor attn_fwd_qkv(torch::Tensor qkv, int64_t heads, double scale) {
  TORCH_CHECK(qkv.is_cuda() && qkv.scalar_type() == at::kBFloat16);
  TORCH_CHECK(qkv.dim() == 3 && qkv.is_contiguous());
  const int B = qkv.size(0), N = qkv.size(1);
  const int HD = qkv.size(2) / 3;
  const int D = HD / (int)heads;
  TORCH_CHECK(3 * HD == qkv.size(2) && (int)heads * D == HD);
  const long long rstride = 3LL * HD, hstride = D,
                  bstride = (long long)N * 3 * HD;
  auto o = torch::empty({B, N, HD}, qkv.options());
  const uint16_t* base = (const uint16_t*)qkv.data_ptr();
  launch_attn_raw(base, base + HD, base + 2 * HD, o, (int)heads, (int)heads,
                  N, N, D, (float)scale, bstride, hstride, rstride, bstride,
                  hstride, rstride, B);
  return o;
}

// Fused cross-attention: q [B, Nq, H*D], kv [B, Nk, 2*H*D].
torch::Tensor attn_fwd_q_kv(torch::Tensor q, torch::Tensor kv, int64_t heads,
                            double scale) {
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == at::kBFloat16);
  TORCH_CHECK(q.is_contiguous() && kv.is_contiguous());
  const int B = q.size(0), Nq = q.size(1), Nk = kv.size(1);
  const int HD = q.size(2);
  const int D = HD / (int)heads;
  TORCH_CHECK(2 * HD == kv.size(2));
  const long long qr = HD, qh = D, qb = (long long)Nq * HD;
  const long long kr = 2LL * HD, kh = D, kb = (long long)Nk * 2 * HD;
  auto o = torch::empty_like(q);
  const uint16_t* kvp = (const uint16_t*)kv.data_ptr();
  launch_attn_raw((const uint16_t*)q.data_ptr(), kvp, kvp + HD, o, (int)heads,
                  (int)heads, Nq, Nk, D, (float)scale, qb, qh, qr, kb, kh, kr,
                  B);
  return o;
}
