// Common device helpers for the CDNA4 (gfx950) kernels.
// All kernels in this extension are written directly for MI355X: wave64,
// LDS-staged tiles, MFMA for matmul-shaped work, bf16 activations with fp32
// accumulation. No CUDA-compat paths.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdint>

#define WAVE_SIZE 64

using bf16_t = __hip_bfloat16;

// Raw-bits bf16 <-> f32 (load path: shift; store path: RNE via builtin).
__device__ __forceinline__ float bf16_bits_to_f32(uint16_t u) {
  union { uint32_t u32; float f; } cvt;
  cvt.u32 = (uint32_t)u << 16;
  return cvt.f;
}

__device__ __forceinline__ uint16_t f32_to_bf16_bits(float f) {
  union { __hip_bfloat16 b; uint16_t u; } cvt;
  cvt.b = __float2bfloat16(f);  // round-to-nearest-even
  return cvt.u;
}

// Vector types for wide loads (G13: always vectorize bf16).
typedef uint16_t ushort8_t __attribute__((ext_vector_type(8)));
typedef float float4_t __attribute__((ext_vector_type(4)));

// Wave-wide reductions (64 lanes).
__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
  return v;
}

__device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, 64));
  return v;
}

// Block reduction over up to 1024 threads via LDS (caller provides scratch
// of >= blockDim.x / 64 floats).
template <typename Op>
__device__ __forceinline__ float block_reduce(float v, float* scratch, Op op,
                                              float identity) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int nwaves = (blockDim.x + 63) >> 6;
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = op(v, __shfl_xor(v, off, 64));
  if (lane == 0) scratch[wave] = v;
  __syncthreads();
  float r = identity;
  if (wave == 0) {
    r = (lane < nwaves) ? scratch[lane] : identity;
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) r = op(r, __shfl_xor(r, off, 64));
    if (lane == 0) scratch[0] = r;
  }
  __syncthreads();
  r = scratch[0];
  __syncthreads();
  return r;
}

struct SumOp {
  __device__ float operator()(float a, float b) const { return a + b; }
};
struct MaxOp {
  __device__ float operator()(float a, float b) const { return fmaxf(a, b); }
};

__device__ __forceinline__ float silu_f(float x) {
  return x / (1.0f + __expf(-x));
}

#define HIP_CHECK_LAUNCH()                                                   \
  do {                                                                       \
    hipError_t err = hipGetLastError();                                      \
    if (err != hipSuccess) {                                                 \
      TORCH_CHECK(false, "HIP kernel launch failed: ",                       \
                  hipGetErrorString(err));                                   \
    }                                                                        \
  } while (0)
