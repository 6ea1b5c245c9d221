// Hand-written CDNA4 convolution kernels for the VAE decode path.
//
// Implicit-GEMM formulation on MFMA 16x16x32 bf16: output pixels are GEMM
// rows (M = B*H*W), output channels are GEMM columns (N = K), the reduction
// runs over (r, s, c) with the channel dim innermost (NHWC activations, so
// every A fragment is a contiguous 16-byte load; no im2col materialization,
// no NCHW<->NHWC transposes). Weights are host-repacked once per module to
// W_t[k_out][(r*3+s)*C + c] so B fragments are contiguous too.
//
// Covers: 3x3 stride-1 pad-1 and 1x1 convs with C % 32 == 0, K % 16 == 0 —
// the whole VAE decoder trunk (512/256/128 channels). Odd-channel edges
// (conv_in from 4 latents, conv_out to RGB) stay on MIOpen.
//
// Reference counterpart: the VAEDecode ComfyUI op the reference calls per
// tile.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

typedef __attribute__((ext_vector_type(8))) short short8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

#define ZERO8_C short8{0, 0, 0, 0, 0, 0, 0, 0}

// Tile geometry: BM=64 pixels x BN=64 channels per workgroup, 4 waves in a
// 2x2 grid (each wave owns a 32x32 sub-tile = 2x2 MFMA fragments).
#define CBM 64
#define CBN 64

template <int RS>  // 9 for 3x3, 1 for 1x1
__global__ __launch_bounds__(256) void conv_nhwc_kernel(
    const uint16_t* __restrict__ x,   // [B, H, W, C]
    const uint16_t* __restrict__ wt,  // [K, RS*C] repacked
    const float* __restrict__ bias,   // [K] or nullptr
    uint16_t* __restrict__ y,         // [B, H, W, K]
    int B, int H, int W, int C, int K, int fuse_silu) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wm = wave >> 1;          // wave row (0..1)
  const int wn = wave & 1;           // wave col (0..1)
  const long long HW = (long long)H * W;
  const long long M = (long long)B * HW;

  const long long m_base = (long long)blockIdx.x * CBM + wm * 32;
  const int n_base = blockIdx.y * CBN + wn * 32;

  // accumulators: 2x2 fragments of 16x16
  f32x4 acc[2][2];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  // per-lane A row (pixel) and B row (output channel)
  const int fr = lane & 15;          // fragment row/col index
  const int kgrp = lane >> 4;        // k-group: elements kgrp*8..+8

  // decompose the two pixel rows this lane loads for A
  long long m_row[2];
  int px_y[2], px_x[2], px_b[2];
  bool m_ok[2];
#pragma unroll
  for (int i = 0; i < 2; ++i) {
    m_row[i] = m_base + i * 16 + fr;
    m_ok[i] = m_row[i] < M;
    long long mm = m_ok[i] ? m_row[i] : 0;
    px_b[i] = (int)(mm / HW);
    int rem = (int)(mm - (long long)px_b[i] * HW);
    px_y[i] = rem / W;
    px_x[i] = rem % W;
  }
  const int n_col[2] = {n_base + fr, n_base + 16 + fr};

  const int csteps = C / 32;
  for (int rs = 0; rs < RS; ++rs) {
    const int r = RS == 1 ? 0 : rs / 3;
    const int s = RS == 1 ? 0 : rs % 3;
    // source pixel offsets for this tap (pad 1 for 3x3)
    int sy[2], sx[2];
    bool tap_ok[2];
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      sy[i] = px_y[i] + (RS == 1 ? 0 : r - 1);
      sx[i] = px_x[i] + (RS == 1 ? 0 : s - 1);
      tap_ok[i] = m_ok[i] && sy[i] >= 0 && sy[i] < H && sx[i] >= 0 && sx[i] < W;
    }
    for (int cs = 0; cs < csteps; ++cs) {
      const int c0 = cs * 32 + kgrp * 8;
      // A fragments (2 pixel rows x 8 contiguous channels)
      short8 a[2];
#pragma unroll
      for (int i = 0; i < 2; ++i) {
        if (tap_ok[i]) {
          const uint16_t* p = x + (((long long)px_b[i] * H + sy[i]) * W + sx[i]) * C + c0;
          a[i] = *reinterpret_cast<const short8*>(p);
        } else {
          a[i] = short8{0, 0, 0, 0, 0, 0, 0, 0};
        }
      }
      // B fragments (2 output channels x 8 contiguous k-elements)
      const int kdim = rs * C + cs * 32 + kgrp * 8;
      short8 bfr[2];
#pragma unroll
      for (int j = 0; j < 2; ++j) {
        const int n = n_col[j];
        bfr[j] = (n < K)
            ? *reinterpret_cast<const short8*>(wt + (long long)n * (RS * C) + kdim)
            : short8{0, 0, 0, 0, 0, 0, 0, 0};
      }
#pragma unroll
      for (int i = 0; i < 2; ++i)
#pragma unroll
        for (int j = 0; j < 2; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a[i], bfr[j],
                                                              acc[i][j], 0, 0, 0);
    }
  }

  // epilogue: bias + optional SiLU, store NHWC
  const int out_col = lane & 15;
  const int rgrp = lane >> 4;
#pragma unroll
  for (int i = 0; i < 2; ++i) {
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      const int n = n_base + j * 16 + out_col;
      if (n >= K) continue;
      const float bval = bias ? bias[n] : 0.f;
#pragma unroll
      for (int rr = 0; rr < 4; ++rr) {
        const long long m = m_base + i * 16 + rgrp * 4 + rr;
        if (m >= M) continue;
        float v = acc[i][j][rr] + bval;
        if (fuse_silu) v = silu_f(v);
        y[m * K + n] = f32_to_bf16_bits(v);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// v2: 128x128 tile, 8 waves (2M x 4N wave grid, each wave 64x32 output),
// weights LDS-staged per 32-deep k-step in a conflict-free 80-byte-pitch
// image shared by all waves; A fragments stream from global (L1/L2-cached:
// consecutive k-steps walk a pixel row sequentially).
// ---------------------------------------------------------------------------

#define C2BM 128
#define C2BN 128
#define W_PITCH 40  // elements per LDS weight row (32 + 8 pad = 80 B)

template <int RS>
__global__ __launch_bounds__(512) void conv_nhwc2_kernel(
    const uint16_t* __restrict__ x, const uint16_t* __restrict__ wt,
    const float* __restrict__ bias, uint16_t* __restrict__ y, int B, int H,
    int W, int C, int K, int fuse_silu) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wm = wave >> 2;  // 0..1  (M strip of 64)
  const int wn = wave & 3;   // 0..3  (N strip of 32)
  const long long HW = (long long)H * W;
  const long long M = (long long)B * HW;

  const long long m_base = (long long)blockIdx.x * C2BM + wm * 64;
  const int n_base = blockIdx.y * C2BN + wn * 32;

  __shared__ __align__(16) uint16_t w_lds[C2BN][W_PITCH];

  f32x4 acc[4][2];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int fr = lane & 15;
  const int kgrp = lane >> 4;

  long long m_row[4];
  int px_y[4], px_x[4], px_b[4];
  bool m_ok[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    m_row[i] = m_base + i * 16 + fr;
    m_ok[i] = m_row[i] < M;
    long long mm = m_ok[i] ? m_row[i] : 0;
    px_b[i] = (int)(mm / HW);
    int rem = (int)(mm - (long long)px_b[i] * HW);
    px_y[i] = rem / W;
    px_x[i] = rem % W;
  }

  const int csteps = C / 32;
  for (int rs = 0; rs < RS; ++rs) {
    const int r = RS == 1 ? 0 : rs / 3;
    const int s = RS == 1 ? 0 : rs % 3;
    int sy[4], sx[4];
    bool tap_ok[4];
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      sy[i] = px_y[i] + (RS == 1 ? 0 : r - 1);
      sx[i] = px_x[i] + (RS == 1 ? 0 : s - 1);
      tap_ok[i] = m_ok[i] && sy[i] >= 0 && sy[i] < H && sx[i] >= 0 && sx[i] < W;
    }
    for (int cs = 0; cs < csteps; ++cs) {
      const int kdim = rs * C + cs * 32;
      // ---- stage the 128x32 weight chunk (one 16B piece per thread) ----
      __syncthreads();
      {
        const int n = blockIdx.y * C2BN + (threadIdx.x >> 2);      // 128 rows
        const int koff = (threadIdx.x & 3) * 8;                     // 4 pieces
        short8 wv = (n < K)
            ? *reinterpret_cast<const short8*>(wt + (long long)n * (RS * C) + kdim + koff)
            : short8{0, 0, 0, 0, 0, 0, 0, 0};
        *reinterpret_cast<short8*>(&w_lds[threadIdx.x >> 2][koff]) = wv;
      }
      __syncthreads();

      const int c0 = cs * 32 + kgrp * 8;
      short8 a[4];
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        if (tap_ok[i]) {
          const uint16_t* p =
              x + (((long long)px_b[i] * H + sy[i]) * W + sx[i]) * C + c0;
          a[i] = *reinterpret_cast<const short8*>(p);
        } else {
          a[i] = short8{0, 0, 0, 0, 0, 0, 0, 0};
        }
      }
      short8 bfr[2];
#pragma unroll
      for (int j = 0; j < 2; ++j)
        bfr[j] = *reinterpret_cast<const short8*>(
            &w_lds[wn * 32 + j * 16 + fr][kgrp * 8]);
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 2; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a[i], bfr[j],
                                                              acc[i][j], 0, 0, 0);
    }
  }

  const int out_col = lane & 15;
  const int rgrp = lane >> 4;
#pragma unroll
  for (int i = 0; i < 4; ++i) {
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      const int n = n_base + j * 16 + out_col;
      if (n >= K) continue;
      const float bval = bias ? bias[n] : 0.f;
#pragma unroll
      for (int rr = 0; rr < 4; ++rr) {
        const long long m = m_base + i * 16 + rgrp * 4 + rr;
        if (m >= M) continue;
        float v = acc[i][j][rr] + bval;
        if (fuse_silu) v = silu_f(v);
        y[m * K + n] = f32_to_bf16_bits(v);
      }
    }
  }
}

template <int RS>
__global__ void conv_nhwc3_kernel(const uint16_t*, const uint16_t*,
                                  const float*, uint16_t*, int, int, int,
                                  int, int, int);

torch::Tensor conv_nhwc(torch::Tensor x, torch::Tensor wt, torch::Tensor bias,
                        int64_t B, int64_t H, int64_t W, int64_t C, int64_t K,
                        int64_t rs, bool fuse_silu) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kBFloat16 && x.is_contiguous());
  TORCH_CHECK(wt.is_cuda() && wt.scalar_type() == at::kBFloat16 && wt.is_contiguous());
  TORCH_CHECK(C % 32 == 0, "C must be a multiple of 32");
  TORCH_CHECK(rs == 9 || rs == 1, "only 3x3 and 1x1");
  auto y = torch::empty({B, H, W, K}, x.options());
  const long long M = B * H * W;
  auto stream = at::hip::getCurrentHIPStream();
  const float* bptr = nullptr;
  torch::Tensor bf;
  if (bias.defined() && bias.numel() > 0) {
    bf = bias.contiguous().to(at::kFloat);
    bptr = bf.data_ptr<float>();
  }
  // v3 (64-deep k-steps) is opt-in until GPU-validated; v2 (LDS-staged
  // weights, 128x128 tile) when the N dim fills it; v1 otherwise. M tail
  // handled by in-kernel bounds in all variants.
  static const bool v3_enabled = [] {
    const char* e = getenv("DISTGPU_CONV_V3");
    return e && e[0] == '1';
  }();
  if (v3_enabled && (K % C2BN == 0) && (C % 64 == 0) && (M >= C2BM)) {
    dim3 grid((unsigned)((M + C2BM - 1) / C2BM), (unsigned)(K / C2BN));
    dim3 block(512);
    if (rs == 9)
      hipLaunchKernelGGL((conv_nhwc3_kernel<9>), grid, block, 0, stream,
                         (const uint16_t*)x.data_ptr(), (const uint16_t*)wt.data_ptr(),
                         bptr, (uint16_t*)y.data_ptr(), (int)B, (int)H, (int)W,
                         (int)C, (int)K, fuse_silu ? 1 : 0);
    else
      hipLaunchKernelGGL((conv_nhwc3_kernel<1>), grid, block, 0, stream,
                         (const uint16_t*)x.data_ptr(), (const uint16_t*)wt.data_ptr(),
                         bptr, (uint16_t*)y.data_ptr(), (int)B, (int)H, (int)W,
                         (int)C, (int)K, fuse_silu ? 1 : 0);
    HIP_CHECK_LAUNCH();
    return y;
  }
  const bool use_v2 = (K % C2BN == 0) && (M >= C2BM);
  if (use_v2) {
    dim3 grid((unsigned)((M + C2BM - 1) / C2BM), (unsigned)(K / C2BN));
    dim3 block(512);
    if (rs == 9)
      hipLaunchKernelGGL((conv_nhwc2_kernel<9>), grid, block, 0, stream,
                         (const uint16_t*)x.data_ptr(), (const uint16_t*)wt.data_ptr(),
                         bptr, (uint16_t*)y.data_ptr(), (int)B, (int)H, (int)W,
                         (int)C, (int)K, fuse_silu ? 1 : 0);
    else
      hipLaunchKernelGGL((conv_nhwc2_kernel<1>), grid, block, 0, stream,
                         (const uint16_t*)x.data_ptr(), (const uint16_t*)wt.data_ptr(),
                         bptr, (uint16_t*)y.data_ptr(), (int)B, (int)H, (int)W,
                         (int)C, (int)K, fuse_silu ? 1 : 0);
    HIP_CHECK_LAUNCH();
    return y;
  }
  dim3 grid((unsigned)((M + CBM - 1) / CBM), (unsigned)((K + CBN - 1) / CBN));
  dim3 block(256);
  if (rs == 9)
    hipLaunchKernelGGL((conv_nhwc_kernel<9>), grid, block, 0, stream,
                       (const uint16_t*)x.data_ptr(), (const uint16_t*)wt.data_ptr(),
                       bptr, (uint16_t*)y.data_ptr(), (int)B, (int)H, (int)W,
                       (int)C, (int)K, fuse_silu ? 1 : 0);
  else
    hipLaunchKernelGGL((conv_nhwc_kernel<1>), grid, block, 0, stream,
                       (const uint16_t*)x.data_ptr(), (const uint16_t*)wt.data_ptr(),
                       bptr, (uint16_t*)y.data_ptr(), (int)B, (int)H, (int)W,
                       (int)C, (int)K, fuse_silu ? 1 : 0);
  HIP_CHECK_LAUNCH();
  return y;
}

// ---------------------------------------------------------------------------
// v3 (experimental, env-gated DISTGPU_CONV_V3=1 until GPU-validated):
// like v2 but with 64-deep k-steps — one barrier pair covers TWO 32-deep
// MFMA sub-steps (the CDNA4 guide measures BK 32->64 at +16% on the
// equivalent GEMM structure: fewer barrier drains per MFMA).
// ---------------------------------------------------------------------------

#define W3_PITCH 72  // 64 elements + 8 pad = 144 B rows (16B-aligned,
                     // 36-dword stride -> conflict-free b128 lane groups)

template <int RS>
__global__ __launch_bounds__(512) void conv_nhwc3_kernel(
    const uint16_t* __restrict__ x, const uint16_t* __restrict__ wt,
    const float* __restrict__ bias, uint16_t* __restrict__ y, int B, int H,
    int W, int C, int K, int fuse_silu) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wm = wave >> 2;
  const int wn = wave & 3;
  const long long HW = (long long)H * W;
  const long long M = (long long)B * HW;

  const long long m_base = (long long)blockIdx.x * C2BM + wm * 64;
  const int n_base = blockIdx.y * C2BN + wn * 32;

  __shared__ __align__(16) uint16_t w_lds[C2BN][W3_PITCH];

  f32x4 acc[4][2];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int fr = lane & 15;
  const int kgrp = lane >> 4;

  long long m_row[4];
  int px_y[4], px_x[4], px_b[4];
  bool m_ok[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    m_row[i] = m_base + i * 16 + fr;
    m_ok[i] = m_row[i] < M;
    long long mm = m_ok[i] ? m_row[i] : 0;
    px_b[i] = (int)(mm / HW);
    int rem = (int)(mm - (long long)px_b[i] * HW);
    px_y[i] = rem / W;
    px_x[i] = rem % W;
  }

  const int csteps2 = C / 64;  // 64-deep k-steps per tap
  for (int rs = 0; rs < RS; ++rs) {
    const int r = RS == 1 ? 0 : rs / 3;
    const int s = RS == 1 ? 0 : rs % 3;
    int sy[4], sx[4];
    bool tap_ok[4];
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      sy[i] = px_y[i] + (RS == 1 ? 0 : r - 1);
      sx[i] = px_x[i] + (RS == 1 ? 0 : s - 1);
      tap_ok[i] = m_ok[i] && sy[i] >= 0 && sy[i] < H && sx[i] >= 0 && sx[i] < W;
    }
    for (int cs = 0; cs < csteps2; ++cs) {
      const int kdim = rs * C + cs * 64;
      // ---- stage the 128x64 weight chunk (two 16B pieces per thread) ----
      __syncthreads();
      {
        const int row = threadIdx.x >> 2;           // 128 rows
        const int piece = threadIdx.x & 3;          // 4 x 16B = 64B half
        const int n = blockIdx.y * C2BN + row;
#pragma unroll
        for (int h2 = 0; h2 < 2; ++h2) {
          const int koff = h2 * 32 + piece * 8;
          short8 wv = (n < K)
              ? *reinterpret_cast<const short8*>(
                    wt + (long long)n * (RS * C) + kdim + koff)
              : ZERO8_C;
          *reinterpret_cast<short8*>(&w_lds[row][koff]) = wv;
        }
      }
      __syncthreads();

      // ---- two 32-deep MFMA sub-steps under one barrier pair ----
#pragma unroll
      for (int half = 0; half < 2; ++half) {
        const int c0 = cs * 64 + half * 32 + kgrp * 8;
        short8 a[4];
#pragma unroll
        for (int i = 0; i < 4; ++i) {
          if (tap_ok[i]) {
            const uint16_t* p =
                x + (((long long)px_b[i] * H + sy[i]) * W + sx[i]) * C + c0;
            a[i] = *reinterpret_cast<const short8*>(p);
          } else {
            a[i] = ZERO8_C;
          }
        }
        short8 bfr[2];
#pragma unroll
        for (int j = 0; j < 2; ++j)
          bfr[j] = *reinterpret_cast<const short8*>(
              &w_lds[wn * 32 + j * 16 + fr][half * 32 + kgrp * 8]);
#pragma unroll
        for (int i = 0; i < 4; ++i)
#pragma unroll
          for (int j = 0; j < 2; ++j)
            acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                a[i], bfr[j], acc[i][j], 0, 0, 0);
      }
    }
  }

  const int out_col = lane & 15;
  const int rgrp = lane >> 4;
#pragma unroll
  for (int i = 0; i < 4; ++i) {
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      const int n = n_base + j * 16 + out_col;
      if (n >= K) continue;
      const float bval = bias ? bias[n] : 0.f;
#pragma unroll
      for (int rr = 0; rr < 4; ++rr) {
        const long long m = m_base + i * 16 + rgrp * 4 + rr;
        if (m >= M) continue;
        float v = acc[i][j][rr] + bval;
        if (fuse_silu) v = silu_f(v);
        y[m * K + n] = f32_to_bf16_bits(v);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Small-C conv (the UNet/VAE stem: C_in <= 8, e.g. 4 latents -> 320).
// MIOpen/CK fall back to very slow paths for NHWC C=4; this is a simple
// HBM-bound kernel: 32 pixels x 8 k-slots per 256-thread block, weights
// staged in LDS, per-pixel taps read once and reused across the k-slots
// via L1.
// ---------------------------------------------------------------------------

#define SC_KSLOTS 8

template <int RS, int C>
__global__ __launch_bounds__(256) void conv_smallc_kernel(
    const uint16_t* __restrict__ x, const uint16_t* __restrict__ wt,
    const float* __restrict__ bias, uint16_t* __restrict__ y, int B, int H,
    int W, int K, int fuse_silu) {
  constexpr int KD = RS * C;
  extern __shared__ __align__(16) uint16_t w_sm[];  // [K][KD]
  for (int i = threadIdx.x; i < K * KD; i += blockDim.x) w_sm[i] = wt[i];
  __syncthreads();

  const long long HW = (long long)H * W;
  const long long M = (long long)B * HW;
  const int kper = (K + SC_KSLOTS - 1) / SC_KSLOTS;
  const int pix_local = threadIdx.x / SC_KSLOTS;   // 0..31
  const int kslot = threadIdx.x % SC_KSLOTS;
  const long long p = (long long)blockIdx.x * 32 + pix_local;
  if (p >= M) return;
  const int b = (int)(p / HW);
  const int rem = (int)(p - (long long)b * HW);
  const int py = rem / W, px = rem % W;

  // taps fully unrolled -> registers (a runtime-indexed array would live
  // in scratch: 5x slowdown)
  float taps[KD];
#pragma unroll
  for (int rs = 0; rs < RS; ++rs) {
    const int r = RS == 1 ? 0 : rs / 3;
    const int s = RS == 1 ? 0 : rs % 3;
    const int sy = py + (RS == 1 ? 0 : r - 1);
    const int sx = px + (RS == 1 ? 0 : s - 1);
    const bool ok = sy >= 0 && sy < H && sx >= 0 && sx < W;
    const uint16_t* src =
        x + (((long long)b * H + sy) * W + sx) * C;
#pragma unroll
    for (int c = 0; c < C; ++c)
      taps[rs * C + c] = ok ? bf16_bits_to_f32(src[c]) : 0.f;
  }

  const int k0 = kslot * kper;
  const int k1 = min(k0 + kper, K);
  uint16_t* dst = y + p * K;
  for (int k = k0; k < k1; ++k) {
    const uint16_t* wrow = &w_sm[k * KD];
    float acc = bias ? bias[k] : 0.f;
#pragma unroll
    for (int i = 0; i < KD; ++i)
      acc += taps[i] * bf16_bits_to_f32(wrow[i]);
    if (fuse_silu) acc = silu_f(acc);
    dst[k] = f32_to_bf16_bits(acc);
  }
}

torch::Tensor conv_smallc(torch::Tensor x, torch::Tensor wt, torch::Tensor bias,
                          int64_t B, int64_t H, int64_t W, int64_t C,
                          int64_t K, int64_t rs, bool fuse_silu) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kBFloat16 && x.is_contiguous());
  TORCH_CHECK(C <= 8, "conv_smallc is for C <= 8");
  TORCH_CHECK(rs == 9 || rs == 1);
  auto y = torch::empty({B, H, W, K}, x.options());
  const long long M = B * H * W;
  dim3 grid((unsigned)((M + 31) / 32));
  dim3 block(256);
  const size_t smem = (size_t)K * rs * C * sizeof(uint16_t);
  TORCH_CHECK(smem <= 64 * 1024, "weights exceed LDS budget");
  auto stream = at::hip::getCurrentHIPStream();
  const float* bptr = nullptr;
  torch::Tensor bf;
  if (bias.defined() && bias.numel() > 0) {
    bf = bias.contiguous().to(at::kFloat);
    bptr = bf.data_ptr<float>();
  }
#define SC_LAUNCH(RS_, C_)                                                     \
  hipLaunchKernelGGL((conv_smallc_kernel<RS_, C_>), grid, block, smem, stream,\
                     (const uint16_t*)x.data_ptr(),                           \
                     (const uint16_t*)wt.data_ptr(), bptr,                    \
                     (uint16_t*)y.data_ptr(), (int)B, (int)H, (int)W, (int)K, \
                     fuse_silu ? 1 : 0)
  TORCH_CHECK(C == 3 || C == 4 || C == 8, "conv_smallc supports C in {3,4,8}");
  if (rs == 9) {
    if (C == 3) SC_LAUNCH(9, 3);
    else if (C == 4) SC_LAUNCH(9, 4);
    else SC_LAUNCH(9, 8);
  } else {
    if (C == 3) SC_LAUNCH(1, 3);
    else if (C == 4) SC_LAUNCH(1, 4);
    else SC_LAUNCH(1, 8);
  }
#undef SC_LAUNCH
  HIP_CHECK_LAUNCH();
  return y;
}

// ---------------------------------------------------------------------------
// NHWC fused GroupNorm(+SiLU): one block per (n, g); channels of a group are
// contiguous within each pixel (span Cg*2 bytes).
// ---------------------------------------------------------------------------

template <bool FUSE_SILU>
__global__ void groupnorm_nhwc_kernel(const uint16_t* __restrict__ x,
                                      uint16_t* __restrict__ y,
                                      const float* __restrict__ weight,
                                      const float* __restrict__ bias,
                                      int C, int G, long long HW, float eps) {
  const int n = blockIdx.x / G;
  const int g = blockIdx.x % G;
  const int Cg = C / G;
  const long long base = (long long)n * HW * C + (long long)g * Cg;

  __shared__ float scratch[16];
  __shared__ float s_mean, s_rstd;

  float sum = 0.f, sumsq = 0.f;
  // per-pixel traversal: each thread reads its pixel's Cg contiguous
  // channels (vector-8 where possible) — measured faster than a
  // flat-index scheme whose per-element divisions dominate
  for (long long pp = threadIdx.x; pp < HW; pp += blockDim.x) {
    const uint16_t* px = x + base + pp * C;
    int c = 0;
    for (; c + 7 < Cg; c += 8) {
      ushort8_t v = *reinterpret_cast<const ushort8_t*>(px + c);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float f = bf16_bits_to_f32(v[j]);
        sum += f;
        sumsq += f * f;
      }
    }
    for (; c < Cg; ++c) {
      float f = bf16_bits_to_f32(px[c]);
      sum += f;
      sumsq += f * f;
    }
  }
  sum = block_reduce(sum, scratch, SumOp{}, 0.f);
  sumsq = block_reduce(sumsq, scratch, SumOp{}, 0.f);
  if (threadIdx.x == 0) {
    const float count = (float)HW * Cg;
    float mean = sum / count;
    float var = sumsq / count - mean * mean;
    s_mean = mean;
    s_rstd = rsqrtf(fmaxf(var, 0.f) + eps);
  }
  __syncthreads();
  const float mean = s_mean, rstd = s_rstd;

  for (long long pp = threadIdx.x; pp < HW; pp += blockDim.x) {
    const uint16_t* px = x + base + pp * C;
    uint16_t* py = y + base + pp * C;
    int c = 0;
    for (; c + 7 < Cg; c += 8) {
      ushort8_t v = *reinterpret_cast<const ushort8_t*>(px + c);
      ushort8_t o;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int ch = g * Cg + c + j;
        float f = (bf16_bits_to_f32(v[j]) - mean) * rstd;
        f = f * weight[ch] + bias[ch];
        if (FUSE_SILU) f = silu_f(f);
        o[j] = f32_to_bf16_bits(f);
      }
      *reinterpret_cast<ushort8_t*>(py + c) = o;
    }
    for (; c < Cg; ++c) {
      const int ch = g * Cg + c;
      float f = (bf16_bits_to_f32(px[c]) - mean) * rstd;
      f = f * weight[ch] + bias[ch];
      if (FUSE_SILU) f = silu_f(f);
      py[c] = f32_to_bf16_bits(f);
    }
  }
}

// ---------------------------------------------------------------------------
// v2 GroupNorm: two fully-coalesced passes. The v1 kernel above (one block
// per (n, g)) reads each pixel's Cg*2 bytes at C*2-byte stride — ~6% of a
// 128-byte line useful at Cg=4..10, and rocprof r01 showed it at 13.4% of
// flagship kernel time (391 us avg). v2 streams the tensor linearly twice
// (stats, then fused normalize+affine+SiLU): 3 x bytes of traffic total,
// which is the memory-bound floor for an unfused GN.
// Pass 1: each block sweeps a contiguous stripe of ONE image, accumulating
// per-group partial sums in LDS (a 16-byte vector touches at most 2 groups
// when Cg >= 8), then flushes G*2 global fp32 atomics.
// ---------------------------------------------------------------------------

#define GN_MAXG 64

__global__ __launch_bounds__(256) void groupnorm_stats_kernel(
    const uint16_t* __restrict__ x, float* __restrict__ gsum,  // [B, G, 2]
    int C, int G, long long HW, int blocks_per_image) {
  const int b = blockIdx.y;
  const int Cg = C / G;
  const long long total = HW * C;                // elements in one image
  const long long nvec = total >> 3;             // 8-element vectors
  const long long per_block = (nvec + blocks_per_image - 1) / blocks_per_image;
  const long long v0 = (long long)blockIdx.x * per_block;
  const long long v1 = min(v0 + per_block, nvec);

  __shared__ float s_sum[GN_MAXG], s_sq[GN_MAXG];
  for (int g = threadIdx.x; g < G; g += blockDim.x) {
    s_sum[g] = 0.f;
    s_sq[g] = 0.f;
  }
  __syncthreads();

  const uint16_t* base = x + (long long)b * total;
  for (long long v = v0 + threadIdx.x; v < v1; v += blockDim.x) {
    const long long e = v << 3;
    const int c0 = (int)(e % C);
    ushort8_t vec = *reinterpret_cast<const ushort8_t*>(base + e);
    // the vector spans at most two groups (C % 8 == 0 and Cg >= 8); split
    // at the group boundary and do one LDS atomic per (run, stat)
    const int g0 = c0 / Cg;
    const int cut = min((g0 + 1) * Cg - c0, 8);  // elements in the first run
    float s0 = 0.f, q0 = 0.f, s1 = 0.f, q1 = 0.f;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = bf16_bits_to_f32(vec[j]);
      if (j < cut) {
        s0 += f;
        q0 += f * f;
      } else {
        s1 += f;
        q1 += f * f;
      }
    }
    atomicAdd(&s_sum[g0], s0);
    atomicAdd(&s_sq[g0], q0);
    if (cut < 8) {
      const int g1 = (c0 + 7) / Cg;
      atomicAdd(&s_sum[g1], s1);
      atomicAdd(&s_sq[g1], q1);
    }
  }
  __syncthreads();
  for (int g = threadIdx.x; g < G; g += blockDim.x) {
    if (s_sum[g] != 0.f || s_sq[g] != 0.f) {
      atomicAdd(&gsum[((long long)b * G + g) * 2 + 0], s_sum[g]);
      atomicAdd(&gsum[((long long)b * G + g) * 2 + 1], s_sq[g]);
    }
  }
}

template <bool FUSE_SILU>
__global__ __launch_bounds__(256) void groupnorm_apply_kernel(
    const uint16_t* __restrict__ x, uint16_t* __restrict__ y,
    const float* __restrict__ gsum, const float* __restrict__ weight,
    const float* __restrict__ bias, int C, int G, long long HW, float eps) {
  const int b = blockIdx.y;
  const int Cg = C / G;
  const float inv_count = 1.f / ((float)HW * Cg);
  const long long total = HW * C;
  const long long nvec = total >> 3;
  const long long stride = (long long)gridDim.x * blockDim.x;
  const uint16_t* src = x + (long long)b * total;
  uint16_t* dst = y + (long long)b * total;
  for (long long v = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       v < nvec; v += stride) {
    const long long e = v << 3;
    const int c0 = (int)(e % C);
    ushort8_t vec = *reinterpret_cast<const ushort8_t*>(src + e);
    ushort8_t out;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int c = c0 + j;
      const int g = c / Cg;
      const float sum = gsum[((long long)b * G + g) * 2 + 0];
      const float sq = gsum[((long long)b * G + g) * 2 + 1];
      const float mean = sum * inv_count;
      const float var = sq * inv_count - mean * mean;
      const float rstd = rsqrtf(fmaxf(var, 0.f) + eps);
      float f = (bf16_bits_to_f32(vec[j]) - mean) * rstd;
      f = f * weight[c] + bias[c];
      if (FUSE_SILU) f = silu_f(f);
      out[j] = f32_to_bf16_bits(f);
    }
    *reinterpret_cast<ushort8_t*>(dst + e) = out;
  }
}

torch::Tensor group_norm_nhwc(torch::Tensor x, int64_t groups,
                              torch::Tensor weight, torch::Tensor bias,
                              double eps, bool fuse_silu) {
  // x: [B, H, W, C] bf16 contiguous
  TORCH_CHECK(x.is_cuda() && x.dim() == 4 && x.scalar_type() == at::kBFloat16
              && x.is_contiguous());
  const int B = x.size(0), C = x.size(3);
  const long long HW = (long long)x.size(1) * x.size(2);
  TORCH_CHECK(C % groups == 0);
  const int G = (int)groups, Cg = C / G;
  auto w = weight.contiguous().to(at::kFloat);
  auto b = bias.contiguous().to(at::kFloat);
  auto y = torch::empty_like(x);
  auto stream = at::hip::getCurrentHIPStream();
  // Cg == 4 also satisfies the <=2-groups-per-16B-vector invariant (each
  // aligned 8-element vector covers exactly two complete groups); the VAE
  // C=128/G=32 stages live here and were the v1 kernel's worst case.
  if (C % 8 == 0 && (Cg >= 8 || Cg == 4) && G <= GN_MAXG) {
    auto gsum = torch::zeros({B, G, 2},
                             x.options().dtype(at::kFloat));
    const long long nvec = HW * C / 8;
    // ~2048 blocks across the chip, each confined to one image
    int bpi = (int)std::min<long long>(
        std::max<long long>(1, 2048 / std::max(B, 1)),
        std::max<long long>(1, nvec / 256));
    dim3 sgrid((unsigned)bpi, (unsigned)B);
    hipLaunchKernelGGL(groupnorm_stats_kernel, sgrid, dim3(256), 0, stream,
                       (const uint16_t*)x.data_ptr(), gsum.data_ptr<float>(),
                       C, G, HW, bpi);
    dim3 agrid((unsigned)bpi, (unsigned)B);
    if (fuse_silu)
      hipLaunchKernelGGL((groupnorm_apply_kernel<true>), agrid, dim3(256), 0,
                         stream, (const uint16_t*)x.data_ptr(),
                         (uint16_t*)y.data_ptr(), gsum.data_ptr<float>(),
                         w.data_ptr<float>(), b.data_ptr<float>(), C, G, HW,
                         (float)eps);
    else
      hipLaunchKernelGGL((groupnorm_apply_kernel<false>), agrid, dim3(256), 0,
                         stream, (const uint16_t*)x.data_ptr(),
                         (uint16_t*)y.data_ptr(), gsum.data_ptr<float>(),
                         w.data_ptr<float>(), b.data_ptr<float>(), C, G, HW,
                         (float)eps);
    HIP_CHECK_LAUNCH();
    return y;
  }
  // fallback (odd channel counts): v1 per-(n,g) kernel
  dim3 grid(B * (int)groups);
  const long long per_group = (long long)(C / groups) * HW;
  dim3 block(B * groups >= 128 ? 256 : (per_group > (1 << 20) ? 1024 : 256));
  if (fuse_silu)
    hipLaunchKernelGGL((groupnorm_nhwc_kernel<true>), grid, block, 0, stream,
                       (const uint16_t*)x.data_ptr(), (uint16_t*)y.data_ptr(),
                       w.data_ptr<float>(), b.data_ptr<float>(), C,
                       (int)groups, HW, (float)eps);
  else
    hipLaunchKernelGGL((groupnorm_nhwc_kernel<false>), grid, block, 0, stream,
                       (const uint16_t*)x.data_ptr(), (uint16_t*)y.data_ptr(),
                       w.data_ptr<float>(), b.data_ptr<float>(), C,
                       (int)groups, HW, (float)eps);
  HIP_CHECK_LAUNCH();
  return y;
}
