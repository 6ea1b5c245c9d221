#include "hip/hip_runtime.h"
// Single-wave MFMA layout self-test: C[16,16] = A[16,32] @ B[32,16] in
// bf16 with fp32 accumulation, using exactly the fragment load patterns the
// attention kernel relies on. A GPU test compares this against torch.matmul
// with an ASYMMETRIC B (a symmetric B would pass a transposed C-write — see
// the CDNA4 guide's transpose-detection rule).
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

typedef __attribute__((ext_vector_type(8))) short short8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

__global__ void mfma_selftest_kernel(const uint16_t* __restrict__ a,
                                     const uint16_t* __restrict__ b,
                                     float* __restrict__ c) {
  const int lane = threadIdx.x & 63;
  // A fragment: row = lane&15, k = (lane>>4)*8 .. +8   (A is [16][32])
  short8 af = *reinterpret_cast<const short8*>(a + (lane & 15) * 32 + (lane >> 4) * 8);
  // B fragment: col = lane&15, k = (lane>>4)*8 .. +8   (B is [32][16],
  // loaded from B^T storage [16][32] so k is contiguous: bt[col][k])
  short8 bf = *reinterpret_cast<const short8*>(b + (lane & 15) * 32 + (lane >> 4) * 8);
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bf, acc, 0, 0, 0);
  // C: col = lane&15, row = (lane>>4)*4 + r
#pragma unroll
  for (int r = 0; r < 4; ++r)
    c[((lane >> 4) * 4 + r) * 16 + (lane & 15)] = acc[r];
}

torch::Tensor mfma_selftest(torch::Tensor a, torch::Tensor bt) {
  TORCH_CHECK(a.is_cuda() && a.scalar_type() == at::kBFloat16);
  TORCH_CHECK(a.sizes() == torch::IntArrayRef({16, 32}), "a must be [16,32]");
  TORCH_CHECK(bt.sizes() == torch::IntArrayRef({16, 32}),
              "bt must be [16,32] (= B^T with B [32,16])");
  auto ac = a.contiguous();
  auto btc = bt.contiguous();
  auto c = torch::empty({16, 16}, a.options().dtype(at::kFloat));
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(mfma_selftest_kernel, dim3(1), dim3(64), 0, stream,
                     (const uint16_t*)ac.data_ptr(),
                     (const uint16_t*)btc.data_ptr(), c.data_ptr<float>());
  HIP_CHECK_LAUNCH();
  return c;
}
