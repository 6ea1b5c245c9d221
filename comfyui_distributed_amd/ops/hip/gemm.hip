// 256x256x64 MFMA GEMM / implicit-GEMM conv for the UNet trunk.
//
// Structure per the CDNA4 guide's "canonical GEMM" ladder (§5): 8 waves
// (2M x 4N), BM=BN=256, BK=64, both operands staged HBM->LDS with
// global_load_lds (glds) into double-buffered, st_16x32 XOR-swizzled
// images; one barrier pair per K-tile; MFMA 16x16x32 bf16 in quadrant
// clusters under s_setprio(1). This is the guide's "glds + 2 LDS buffers +
// BK=64" tier, which ties the best register pipeline (~1100-1200 TF on
// random data) at 32 fewer VGPRs; the further 8-phase fine interleave
// (1320-1470 TF) layers on top of this same skeleton.
//
// Two A-operand address modes share the kernel:
//   GEMM:  A[m][k]        (tokens x features; torch Linear with W[N][K])
//   CONV:  A = NHWC activations addressed per (pixel, tap) — the implicit
//          im2col: k = tap*C + c, tap shifts the source pixel, out-of-
//          bounds taps are redirected to a zero page (glds cannot
//          conditionally zero, but its per-lane SOURCE address is free).
//
// Replaces: hipBLASLt Linears (~9% of round-1 kernel time) and MIOpen
// igemm convs (~33%) — profiles/r01_prof_final_summary.csv. Reference
// counterpart: the ComfyUI conv/linear substrate of upscale/tile_ops.py
// (SURVEY.md §2.8 K6).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

typedef __attribute__((ext_vector_type(8))) short short8_g;
typedef __attribute__((ext_vector_type(4))) float f32x4_g;

#define GBM 256
#define GBN 256
#define GBK 64
// LDS: A[2][256][64] + B[2][256][64] bf16 = 128 KiB, one __shared__ object
// (a second __shared__ de-pipelines glds — guide §5 trap (a))
#define G_ABUF 0
#define G_BBUF 65536
#define G_TILE 32768  // one 256x64 bf16 image
#define G_HALF 16384  // one 128x64 half image

// st_16x32 swizzle within each 1 KiB subtile: spread ds_read_b128 lane
// groups across four 32 B slots instead of two (guide §5 "LDS swizzle is
// essential"; applied on the glds SOURCE address, LDS stays lane-linear)
__device__ __forceinline__ unsigned swz(unsigned byte_off) {
  return byte_off ^ (((byte_off >> 9) & 1u) << 5);
}

// one 16 B/lane HBM->LDS DMA; LDS destination is wave-uniform base +
// lane*16 (guide §5: the intrinsic's LDS side is lane-linear, the GLOBAL
// side is per-lane — swizzles live on the source address)
__device__ __forceinline__ void glds16(const void* gsrc, void* lds_dst) {
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) uint32_t*)gsrc,
      (__attribute__((address_space(3))) uint32_t*)lds_dst, 16, 0, 0);
}

struct ConvGeo {
  int H, W, C;     // INPUT spatial + channels
  int Ho, Wo;      // output spatial (Ho = H/stride for 3x3 pad1)
  int stride;      // 1 or 2 (the UNet/VAE Downsample convs)
  int up2;         // 1: conv consumes a VIRTUAL nearest-2x upsample of x
                   // (Ho = 2H) — the upsampled tensor never exists, the
                   // tap address just halves: the VAE decoder's biggest
                   // intermediates (F.interpolate output + its re-read)
                   // disappear entirely
  int RS;          // 9 for 3x3 (pad 1), 1 for 1x1
  int tile2d;      // 1: blocks cover 16x16 pixel tiles (H,W % 16 == 0) —
                   // a 3x3 tap re-reads an 18x18 halo (1.27x) instead of
                   // the scanline tile's 9x; the VAE-decode C=128/256
                   // shapes are HBM-bound on exactly that re-read
  int xcd_swz;     // 1: bijective blockIdx.x -> XCD-major remap (T1)
};

// bijective XCD-aware remap (guide §5 "XCD swizzle must be bijective"):
// consecutive original ids spread across the 8 XCDs' L2s
__device__ __forceinline__ unsigned xcd_remap(unsigned id, unsigned nwg) {
  const unsigned q = nwg / 8u, r = nwg % 8u;
  const unsigned xcd = id % 8u, pos = id / 8u;
  return (xcd < r ? xcd * (q + 1u) : r * (q + 1u) + (xcd - r) * q) + pos;
}

// Per-thread precomputed source descriptors for the 8 glds chunk slots
// (2 instrs x 2 halves x {A,B}).
struct AChunk {
  const uint16_t* row_base;  // pixel base (b,y,x)*C for conv, &A[m*K] for gemm
  int px_y, px_x;            // conv only
  bool ok;                   // m < M
};

template <bool IS_CONV, bool FUSE_SILU>
__global__ __launch_bounds__(512, 1) void gemm256_kernel(
    const uint16_t* __restrict__ A,   // gemm: [M,K]; conv: NHWC activations
    const uint16_t* __restrict__ Bw,  // [N, K] (conv: repacked [K_out][RS*C])
    const float* __restrict__ bias,   // [N] or nullptr
    const uint16_t* __restrict__ zero_page,  // >=16B of zeros
    const uint16_t* __restrict__ residual,   // [M, N] bf16 or nullptr:
                                             // fused skip-connection add
    uint16_t* __restrict__ Y,         // [M, N] bf16
    long long M, int N, int Kdim, ConvGeo geo) {
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wm = wave >> 2;  // 0..1: M half (128 rows)
  const int wn = wave & 3;   // 0..3: N strip (64 cols)

  // 128 KiB total (A images then B images), byte offsets per the macros
  __shared__ __align__(16) uint16_t lds[65536];

  unsigned bx = blockIdx.x;
  if (IS_CONV && geo.xcd_swz) bx = xcd_remap(bx, gridDim.x);
  long long m_blk = 0;
  int t_b = 0, t_py0 = 0, t_px0 = 0;
  if (IS_CONV && geo.tile2d) {
    const int tx_n = geo.Wo >> 4;
    const int per_img = tx_n * (geo.Ho >> 4);
    t_b = (int)(bx / per_img);
    const int rem = (int)(bx % per_img);
    t_py0 = (rem / tx_n) << 4;
    t_px0 = (rem % tx_n) << 4;
  } else {
    m_blk = (long long)bx * GBM;
  }
  const int n_blk = blockIdx.y * GBN;

  // ---- per-thread glds chunk descriptors (K-invariant) -----------------
  // Staging is 4 pieces of 16 KiB per K-tile, 2 glds each, issued in the
  // order [B0, B1, A0, A1] so counted vmcnt waits drain exactly what the
  // next quadrant phase reads (B fully + A's mq=0 rows at phase 0; A's
  // mq=1 rows only by phase 2). The A image is row-permuted so piece A0
  // holds the mq=0 rows of BOTH wave halves:
  //   image quarter qq = mq*2 + wm  ->  m_local = (qq&1)*128+(qq>>1)*64+r.
  // chunk slot (piece, instr i): chunk = tid + i*512; logical byte within
  // the 32 KiB image = swz(piece_off + chunk*16).
  AChunk a_desc[2][2];   // [a_piece][instr]
  const uint16_t* b_src[2][2];
  int c_local[2][2];     // [a_piece][instr] channel offset
  int cb_local[2][2];    // [b_piece][instr]
#pragma unroll
  for (int i = 0; i < 2; ++i) {
    const unsigned chunk16 = ((unsigned)tid + (unsigned)i * 512u) * 16u;
#pragma unroll
    for (int pc = 0; pc < 2; ++pc) {
      // ---- A piece pc: image rows pc*128..+128 (quarter-permuted) ----
      {
        const unsigned y = swz((unsigned)pc * 16384u + chunk16);
        const int img_row = (int)(y >> 7);
        const int qq = img_row >> 6;
        const int m_local = ((qq & 1) << 7) | ((qq >> 1) << 6) |
                            (img_row & 63);
        c_local[pc][i] = (int)((y & 127u) >> 1);
        AChunk d;
        if (IS_CONV && geo.tile2d) {
          d.ok = true;
          d.px_y = t_py0 + (m_local >> 4);
          d.px_x = t_px0 + (m_local & 15);
          d.row_base = A + (long long)t_b * geo.H * geo.W * geo.C;
        } else if (IS_CONV) {
          const long long m = m_blk + m_local;
          d.ok = m < M;
          const long long HWo = (long long)geo.Ho * geo.Wo;
          const long long mm = d.ok ? m : 0;
          const int pb = (int)(mm / HWo);
          const int rem = (int)(mm - (long long)pb * HWo);
          d.px_y = rem / geo.Wo;   // OUTPUT pixel coords; stride applied
          d.px_x = rem % geo.Wo;   // at the tap address
          d.row_base = A + (long long)pb * geo.H * geo.W * geo.C;
        } else {
          const long long m = m_blk + m_local;
          d.ok = m < M;
          d.row_base = A + (d.ok ? m * (long long)Kdim : 0);
          d.px_y = d.px_x = 0;
        }
        a_desc[pc][i] = d;
      }
      // ---- B piece pc: image rows pc*128..+128 (linear) --------------
      {
        const unsigned y = swz((unsigned)pc * 16384u + chunk16);
        const int row = (int)(y >> 7);
        cb_local[pc][i] = (int)((y & 127u) >> 1);
        const int n = n_blk + row;
        b_src[pc][i] = (n < N) ? Bw + (long long)n * Kdim : nullptr;
      }
    }
  }

  // ---- accumulators -----------------------------------------------------
  f32x4_g acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = f32x4_g{0.f, 0.f, 0.f, 0.f};

  const int fr = lane & 15;
  const int kgrp = lane >> 4;

  const int n_ktiles = Kdim / GBK;

  // ---- glds stage of one 16 KiB piece (2 instrs) into buffer p ---------
  // piece 0 = B rows 0..127, 1 = B rows 128..255, 2 = A quarters 0-1
  // (mq=0 rows of both wave halves), 3 = A quarters 2-3 (mq=1 rows)
  auto stage_piece = [&](int kt, int p, int piece) {
    const int kbase = kt * GBK;
    int tap_dy = 0, tap_dx = 0, cbase = kbase;
    if (IS_CONV && geo.RS == 9) {
      const int tap = kbase / geo.C;
      cbase = kbase - tap * geo.C;
      tap_dy = tap / 3 - 1;
      tap_dx = tap % 3 - 1;
    }
    const bool is_b = piece < 2;
    const int pc = piece & 1;
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      const uint16_t* src;
      unsigned base;
      if (is_b) {
        src = b_src[pc][i] ? b_src[pc][i] + kbase + cb_local[pc][i]
                           : zero_page;
        base = (unsigned)G_BBUF + p * G_TILE + pc * 16384u;
      } else {
        const AChunk& d = a_desc[pc][i];
        if (IS_CONV) {
          int sy, sx;
          bool ok;
          if (geo.up2) {
            // pad is applied on the VIRTUAL upsampled canvas [2H, 2W]
            const int uy = d.px_y + tap_dy;
            const int ux = d.px_x + tap_dx;
            ok = d.ok && uy >= 0 && uy < 2 * geo.H && ux >= 0 &&
                 ux < 2 * geo.W;
            sy = uy >> 1;
            sx = ux >> 1;
          } else {
            sy = d.px_y * geo.stride + tap_dy;
            sx = d.px_x * geo.stride + tap_dx;
            ok = d.ok && sy >= 0 && sy < geo.H && sx >= 0 && sx < geo.W;
          }
          src = ok ? d.row_base + ((long long)sy * geo.W + sx) * geo.C +
                         cbase + c_local[pc][i]
                   : zero_page;
        } else {
          src = d.ok ? d.row_base + kbase + c_local[pc][i] : zero_page;
        }
        base = (unsigned)G_ABUF + p * G_TILE + pc * 16384u;
      }
      // wave-uniform LDS base; hardware adds lane*16
      glds16(src, reinterpret_cast<uint8_t*>(lds) + base +
                      (wave * 64u + i * 512u) * 16u);
    }
  };

  // ---- one quadrant (16 MFMA over the K-tile's 64-deep K) --------------
  auto compute_quadrant = [&](int p, int mq, int nq) {
    const unsigned a_base = (unsigned)G_ABUF + p * G_TILE;
    const unsigned b_base = (unsigned)G_BBUF + p * G_TILE;
    const int qq = mq * 2 + wm;  // A image quarter for this wave
    short8_g af[4][2], bfr[2][2];
#pragma unroll
    for (int mf = 0; mf < 4; ++mf)
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        const unsigned lo =
            (qq * 64 + mf * 16 + fr) * 128u + ks * 64u + kgrp * 16u;
        af[mf][ks] = *reinterpret_cast<const short8_g*>(
            lds + ((a_base + swz(lo)) >> 1));
      }
#pragma unroll
    for (int nf = 0; nf < 2; ++nf)
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        const int brow = wn * 64 + (nq * 2 + nf) * 16 + fr;
        const unsigned lo = brow * 128u + ks * 64u + kgrp * 16u;
        bfr[nf][ks] = *reinterpret_cast<const short8_g*>(
            lds + ((b_base + swz(lo)) >> 1));
      }
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int mf = 0; mf < 4; ++mf)
#pragma unroll
      for (int nf = 0; nf < 2; ++nf)
#pragma unroll
        for (int ks = 0; ks < 2; ++ks)
          acc[mq * 4 + mf][nq * 2 + nf] =
              __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                  af[mf][ks], bfr[nf][ks], acc[mq * 4 + mf][nq * 2 + nf],
                  0, 0, 0);
    __builtin_amdgcn_s_setprio(0);
  };

  // ---- main loop ---------------------------------------------------------
  // Raw s_barrier + counted vmcnt (guide §5: __syncthreads with glds in
  // flight drains vmcnt(0) — the ~20% structural stall of the simple
  // variant). One barrier per K-tile; the next tile's pieces are issued
  // between quadrant clusters; vmcnt(2) at phase 0 drains this tile's
  // B0,B1,A0 (A1 may still fly), vmcnt(4) before the mq=1 quadrants
  // drains A1.
#pragma unroll
  for (int pc = 0; pc < 4; ++pc) stage_piece(0, 0, pc);
  for (int kt = 0; kt < n_ktiles; ++kt) {
    const int p = kt & 1;
    const bool more = kt + 1 < n_ktiles;
    // Full drain at tile start: the tile's 8 glds were issued a whole
    // K-tile ago, so vmcnt(0) is cheap here — and the barrier AFTER the
    // drain makes every wave's staging collectively visible (a counted
    // per-wave vmcnt alone cannot order OTHER waves' DMA against this
    // wave's ds_reads; the earlier staggered vmcnt(2)/(4) scheme needed a
    // second barrier for that and measured slower)
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    if (more) stage_piece(kt + 1, 1 - p, 0);
    compute_quadrant(p, 0, 0);
    if (more) stage_piece(kt + 1, 1 - p, 1);
    compute_quadrant(p, 0, 1);
    if (more) stage_piece(kt + 1, 1 - p, 2);
    compute_quadrant(p, 1, 0);
    if (more) stage_piece(kt + 1, 1 - p, 3);
    compute_quadrant(p, 1, 1);
  }

  // ---- epilogue: bias + optional SiLU, bf16 store ----------------------
  const long long m_wave = m_blk + wm * 128;
  const int n_wave = n_blk + wn * 64;
#pragma unroll
  for (int mf = 0; mf < 8; ++mf) {
#pragma unroll
    for (int nf = 0; nf < 4; ++nf) {
      const int n = n_wave + nf * 16 + fr;
      if (n >= N) continue;
      const float bval = bias ? bias[n] : 0.f;
#pragma unroll
      for (int rr = 0; rr < 4; ++rr) {
        long long m;
        if (IS_CONV && geo.tile2d) {
          const int idx = wm * 128 + mf * 16 + kgrp * 4 + rr;
          m = ((long long)t_b * geo.Ho + t_py0 + (idx >> 4)) * geo.Wo +
              t_px0 + (idx & 15);
        } else {
          m = m_wave + mf * 16 + kgrp * 4 + rr;
          if (m >= M) continue;
        }
        float v = acc[mf][nf][rr] + bval;
        if (FUSE_SILU) v = silu_f(v);
        if (residual) v += bf16_bits_to_f32(residual[m * N + n]);
        Y[m * N + n] = f32_to_bf16_bits(v);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// launchers
// ---------------------------------------------------------------------------

// intentionally leaked: a static torch::Tensor's destructor would run at
// process exit AFTER the HIP context is torn down
static torch::Tensor* g_zero_page = nullptr;

static const uint16_t* zero_page_ptr(const torch::Tensor& like) {
  if (!g_zero_page || g_zero_page->device() != like.device())
    g_zero_page =
        new torch::Tensor(torch::zeros({64}, like.options().dtype(at::kBFloat16)));
  return (const uint16_t*)g_zero_page->data_ptr();
}

torch::Tensor gemm256_bf16(torch::Tensor x, torch::Tensor w,
                           torch::Tensor bias, bool fuse_silu) {
  // (plain-GEMM entry has no residual fusion)
  // x [M, K] bf16, w [N, K] bf16 (torch Linear layout) -> y [M, N] bf16
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kBFloat16 &&
              x.is_contiguous());
  TORCH_CHECK(w.is_cuda() && w.scalar_type() == at::kBFloat16 &&
              w.is_contiguous());
  const long long M = x.size(0);
  const int K = (int)x.size(1), N = (int)w.size(0);
  TORCH_CHECK(w.size(1) == K && K % GBK == 0, "K must be a multiple of 64");
  auto y = torch::empty({M, (long long)N}, x.options());
  const float* bptr = nullptr;
  torch::Tensor bf32;
  if (bias.defined() && bias.numel() > 0) {
    bf32 = bias.contiguous().to(at::kFloat);
    bptr = bf32.data_ptr<float>();
  }
  dim3 grid((unsigned)((M + GBM - 1) / GBM), (unsigned)((N + GBN - 1) / GBN));
  auto stream = at::hip::getCurrentHIPStream();
  ConvGeo geo{0, 0, 0, 0, 0, 1, 0, 0, 0, 0};
  if (fuse_silu)
    hipLaunchKernelGGL((gemm256_kernel<false, true>), grid, dim3(512), 0,
                       stream, (const uint16_t*)x.data_ptr(),
                       (const uint16_t*)w.data_ptr(), bptr,
                       zero_page_ptr(x), (const uint16_t*)nullptr,
                       (uint16_t*)y.data_ptr(), M, N, K, geo);
  else
    hipLaunchKernelGGL((gemm256_kernel<false, false>), grid, dim3(512), 0,
                       stream, (const uint16_t*)x.data_ptr(),
                       (const uint16_t*)w.data_ptr(), bptr,
                       zero_page_ptr(x), (const uint16_t*)nullptr,
                       (uint16_t*)y.data_ptr(), M, N, K, geo);
  HIP_CHECK_LAUNCH();
  return y;
}

torch::Tensor conv256_nhwc(torch::Tensor x, torch::Tensor wt,
                           torch::Tensor bias, torch::Tensor residual,
                           int64_t B, int64_t H,
                           int64_t W, int64_t C, int64_t K, int64_t rs,
                           int64_t stride, bool up2, bool fuse_silu) {
  // x [B,H,W,C] bf16 NHWC, wt [K_out, rs*C] repacked -> y [B,Ho,Wo,K]
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kBFloat16 &&
              x.is_contiguous());
  TORCH_CHECK(wt.is_cuda() && wt.is_contiguous());
  TORCH_CHECK(rs == 9 || rs == 1);
  TORCH_CHECK(stride == 1 || (stride == 2 && rs == 9));
  TORCH_CHECK(!up2 || (stride == 1 && rs == 9));
  TORCH_CHECK(C % GBK == 0, "C must be a multiple of 64");
  // 3x3 always pad 1: Ho = ceil(H/stride) (stride 2 needs even dims for
  // the torch formula (H+2-3)/2+1 = H/2 when H even); up2 doubles
  const int64_t Ho = up2 ? 2 * H : (stride == 1 ? H : (H + 2 - 3) / 2 + 1);
  const int64_t Wo = up2 ? 2 * W : (stride == 1 ? W : (W + 2 - 3) / 2 + 1);
  const long long M = B * Ho * Wo;
  const int Kdim = (int)(rs * C);
  auto y = torch::empty({B, Ho, Wo, K}, x.options());
  const float* bptr = nullptr;
  torch::Tensor bf32;
  if (bias.defined() && bias.numel() > 0) {
    bf32 = bias.contiguous().to(at::kFloat);
    bptr = bf32.data_ptr<float>();
  }
  static const int xcd_swz = [] {
    const char* e = getenv("DISTGPU_CONV_XCDSWZ");
    return (!e || e[0] == '1') ? 1 : 0;
  }();
  const int tile2d = (Ho % 16 == 0 && Wo % 16 == 0) ? 1 : 0;
  const unsigned gx = tile2d
      ? (unsigned)(B * (Ho / 16) * (Wo / 16))
      : (unsigned)((M + GBM - 1) / GBM);
  dim3 grid(gx, (unsigned)((K + GBN - 1) / GBN));
  auto stream = at::hip::getCurrentHIPStream();
  ConvGeo geo{(int)H, (int)W, (int)C, (int)Ho, (int)Wo, (int)stride,
              up2 ? 1 : 0, (int)rs, tile2d, xcd_swz};
  const uint16_t* resptr = nullptr;
  if (residual.defined() && residual.numel() > 0) {
    TORCH_CHECK(residual.is_cuda() && residual.is_contiguous() &&
                residual.scalar_type() == at::kBFloat16 &&
                residual.numel() == M * K,
                "residual must be a contiguous bf16 [B,Ho,Wo,K] tensor");
    resptr = (const uint16_t*)residual.data_ptr();
  }
  if (fuse_silu)
    hipLaunchKernelGGL((gemm256_kernel<true, true>), grid, dim3(512), 0,
                       stream, (const uint16_t*)x.data_ptr(),
                       (const uint16_t*)wt.data_ptr(), bptr,
                       zero_page_ptr(x), resptr,
                       (uint16_t*)y.data_ptr(), M, (int)K, Kdim, geo);
  else
    hipLaunchKernelGGL((gemm256_kernel<true, false>), grid, dim3(512), 0,
                       stream, (const uint16_t*)x.data_ptr(),
                       (const uint16_t*)wt.data_ptr(), bptr,
                       zero_page_ptr(x), resptr,
                       (uint16_t*)y.data_ptr(), M, (int)K, Kdim, geo);
  HIP_CHECK_LAUNCH();
  return y;
}
