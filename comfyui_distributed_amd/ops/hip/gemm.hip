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
  int H, W, C;     // input spatial + channels
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
    const int tx_n = geo.W >> 4;
    const int per_img = tx_n * (geo.H >> 4);
    t_b = (int)(bx / per_img);
    const int rem = (int)(bx % per_img);
    t_py0 = (rem / tx_n) << 4;
    t_px0 = (rem % tx_n) << 4;
  } else {
    m_blk = (long long)bx * GBM;
  }
  const int n_blk = blockIdx.y * GBN;

  // ---- per-thread glds chunk descriptors (K-invariant) -----------------
  // chunk slot (instr i, half h): chunk = tid + i*512 within half h's
  // [128][64] image; logical byte = swz(chunk*16); row = logical>>7,
  // k-byte = logical&127 -> channel offset c_local = (logical&127)/2.
  AChunk a_desc[2][2];   // [half][instr]
  const uint16_t* b_src[2][2];
  int c_local[2];        // per instr (same for both halves)
  int a_cloc[2];
#pragma unroll
  for (int i = 0; i < 2; ++i) {
    const unsigned chunk = (unsigned)tid + (unsigned)i * 512u;
    const unsigned lo = swz(chunk * 16u);
    const int row = (int)(lo >> 7);
    a_cloc[i] = c_local[i] = (int)((lo & 127u) >> 1);
#pragma unroll
    for (int h = 0; h < 2; ++h) {
      // A row
      AChunk d;
      if (IS_CONV && geo.tile2d) {
        const int idx = h * 128 + row;  // 0..255 within the 16x16 tile
        d.ok = true;
        d.px_y = t_py0 + (idx >> 4);
        d.px_x = t_px0 + (idx & 15);
        d.row_base = A + (long long)t_b * geo.H * geo.W * geo.C;
      } else if (IS_CONV) {
        const long long m = m_blk + (long long)h * 128 + row;
        d.ok = m < M;
        const long long HW = (long long)geo.H * geo.W;
        const long long mm = d.ok ? m : 0;
        const int pb = (int)(mm / HW);
        const int rem = (int)(mm - (long long)pb * HW);
        d.px_y = rem / geo.W;
        d.px_x = rem % geo.W;
        d.row_base = A + ((long long)pb * HW) * geo.C;
      } else {
        const long long m = m_blk + (long long)h * 128 + row;
        d.ok = m < M;
        d.row_base = A + (d.ok ? m * (long long)Kdim : 0);
        d.px_y = d.px_x = 0;
      }
      a_desc[h][i] = d;
      // B row
      const int n = n_blk + h * 128 + row;
      b_src[h][i] = (n < N) ? Bw + (long long)n * Kdim : nullptr;
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

  // ---- glds stage of one K-tile into buffer p --------------------------
  auto stage = [&](int kt, int p) {
    const int kbase = kt * GBK;
    int tap_dy = 0, tap_dx = 0, cbase = kbase;
    if (IS_CONV && geo.RS == 9) {
      const int tap = kbase / geo.C;
      cbase = kbase - tap * geo.C;
      tap_dy = tap / 3 - 1;
      tap_dx = tap % 3 - 1;
    } else if (IS_CONV) {
      cbase = kbase;  // 1x1: k == c
    }
#pragma unroll
    for (int h = 0; h < 2; ++h) {
#pragma unroll
      for (int i = 0; i < 2; ++i) {
        // A half h, instr i
        const AChunk& d = a_desc[h][i];
        const uint16_t* src;
        if (IS_CONV) {
          const int sy = d.px_y + tap_dy;
          const int sx = d.px_x + tap_dx;
          const bool ok = d.ok && sy >= 0 && sy < geo.H && sx >= 0 &&
                          sx < geo.W;
          src = ok ? d.row_base + ((long long)sy * geo.W + sx) * geo.C +
                         cbase + a_cloc[i]
                   : zero_page;
        } else {
          src = d.ok ? d.row_base + kbase + a_cloc[i] : zero_page;
        }
        // wave-uniform LDS base; hardware adds lane*16
        const unsigned lds_off = (unsigned)G_ABUF + p * G_TILE + h * G_HALF +
                                 (wave * 64u + i * 512u) * 16u;
        glds16(src, reinterpret_cast<uint8_t*>(lds) + lds_off);
        // B half h, instr i
        const uint16_t* bsrc =
            b_src[h][i] ? b_src[h][i] + kbase + c_local[i] : zero_page;
        const unsigned lds_off_b = (unsigned)G_BBUF + p * G_TILE +
                                   h * G_HALF + (wave * 64u + i * 512u) * 16u;
        glds16(bsrc, reinterpret_cast<uint8_t*>(lds) + lds_off_b);
      }
    }
  };

  // ---- compute one K-tile from buffer p --------------------------------
  auto compute = [&](int p) {
    const unsigned a_base = (unsigned)G_ABUF + p * G_TILE + wm * G_HALF;
    // quadrants: (mq, nq) -> m-frags mq*4..+4, n-frags nq*2..+2
#pragma unroll
    for (int mq = 0; mq < 2; ++mq) {
#pragma unroll
      for (int nq = 0; nq < 2; ++nq) {
        short8_g af[4][2], bf[2][2];
#pragma unroll
        for (int mf = 0; mf < 4; ++mf)
#pragma unroll
          for (int ks = 0; ks < 2; ++ks) {
            const unsigned lo =
                ((mq * 4 + mf) * 16 + fr) * 128u + ks * 64u + kgrp * 16u;
            af[mf][ks] = *reinterpret_cast<const short8_g*>(
                lds + ((a_base + swz(lo)) >> 1));
          }
#pragma unroll
        for (int nf = 0; nf < 2; ++nf)
#pragma unroll
          for (int ks = 0; ks < 2; ++ks) {
            const int brow = wn * 64 + (nq * 2 + nf) * 16 + fr;
            const unsigned b_base =
                (unsigned)G_BBUF + p * G_TILE + (brow >> 7) * G_HALF;
            const unsigned lo = (brow & 127) * 128u + ks * 64u + kgrp * 16u;
            bf[nf][ks] = *reinterpret_cast<const short8_g*>(
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
                      af[mf][ks], bf[nf][ks], acc[mq * 4 + mf][nq * 2 + nf],
                      0, 0, 0);
        __builtin_amdgcn_s_setprio(0);
      }
    }
  };

  // ---- main loop: double-buffered glds, one barrier pair per K-tile ----
  stage(0, 0);
  for (int kt = 0; kt < n_ktiles; ++kt) {
    const int p = kt & 1;
    // __syncthreads() with glds in flight emits vmcnt(0): all of buffer
    // p's DMA (issued last iteration) has landed for every wave after the
    // barrier (guide §5 "glds, 2 LDS buffers, BK=64" row)
    __syncthreads();
    if (kt + 1 < n_ktiles) stage(kt + 1, 1 - p);
    compute(p);
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
          m = ((long long)t_b * geo.H + t_py0 + (idx >> 4)) * geo.W +
              t_px0 + (idx & 15);
        } else {
          m = m_wave + mf * 16 + kgrp * 4 + rr;
          if (m >= M) continue;
        }
        float v = acc[mf][nf][rr] + bval;
        if (FUSE_SILU) v = silu_f(v);
        Y[m * N + n] = f32_to_bf16_bits(v);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// launchers
// ---------------------------------------------------------------------------

static torch::Tensor g_zero_page;

static const uint16_t* zero_page_ptr(const torch::Tensor& like) {
  if (!g_zero_page.defined() || g_zero_page.device() != like.device())
    g_zero_page = torch::zeros({64}, like.options().dtype(at::kBFloat16));
  return (const uint16_t*)g_zero_page.data_ptr();
}

torch::Tensor gemm256_bf16(torch::Tensor x, torch::Tensor w,
                           torch::Tensor bias, bool fuse_silu) {
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
  ConvGeo geo{0, 0, 0, 0, 0, 0};
  if (fuse_silu)
    hipLaunchKernelGGL((gemm256_kernel<false, true>), grid, dim3(512), 0,
                       stream, (const uint16_t*)x.data_ptr(),
                       (const uint16_t*)w.data_ptr(), bptr,
                       zero_page_ptr(x), (uint16_t*)y.data_ptr(), M, N, K,
                       geo);
  else
    hipLaunchKernelGGL((gemm256_kernel<false, false>), grid, dim3(512), 0,
                       stream, (const uint16_t*)x.data_ptr(),
                       (const uint16_t*)w.data_ptr(), bptr,
                       zero_page_ptr(x), (uint16_t*)y.data_ptr(), M, N, K,
                       geo);
  HIP_CHECK_LAUNCH();
  return y;
}

torch::Tensor conv256_nhwc(torch::Tensor x, torch::Tensor wt,
                           torch::Tensor bias, int64_t B, int64_t H,
                           int64_t W, int64_t C, int64_t K, int64_t rs,
                           bool fuse_silu) {
  // x [B,H,W,C] bf16 NHWC, wt [K_out, rs*C] repacked -> y [B,H,W,K]
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kBFloat16 &&
              x.is_contiguous());
  TORCH_CHECK(wt.is_cuda() && wt.is_contiguous());
  TORCH_CHECK(rs == 9 || rs == 1);
  TORCH_CHECK(C % GBK == 0, "C must be a multiple of 64");
  const long long M = B * H * W;
  const int Kdim = (int)(rs * C);
  auto y = torch::empty({B, H, W, K}, x.options());
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
  const int tile2d = (H % 16 == 0 && W % 16 == 0) ? 1 : 0;
  const unsigned gx = tile2d
      ? (unsigned)(B * (H / 16) * (W / 16))
      : (unsigned)((M + GBM - 1) / GBM);
  dim3 grid(gx, (unsigned)((K + GBN - 1) / GBN));
  auto stream = at::hip::getCurrentHIPStream();
  ConvGeo geo{(int)H, (int)W, (int)C, (int)rs, tile2d, xcd_swz};
  if (fuse_silu)
    hipLaunchKernelGGL((gemm256_kernel<true, true>), grid, dim3(512), 0,
                       stream, (const uint16_t*)x.data_ptr(),
                       (const uint16_t*)wt.data_ptr(), bptr,
                       zero_page_ptr(x), (uint16_t*)y.data_ptr(), M, (int)K,
                       Kdim, geo);
  else
    hipLaunchKernelGGL((gemm256_kernel<true, false>), grid, dim3(512), 0,
                       stream, (const uint16_t*)x.data_ptr(),
                       (const uint16_t*)wt.data_ptr(), bptr,
                       zero_page_ptr(x), (uint16_t*)y.data_ptr(), M, (int)K,
                       Kdim, geo);
  HIP_CHECK_LAUNCH();
  return y;
}
