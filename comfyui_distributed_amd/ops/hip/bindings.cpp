// Python bindings for the gfx950 kernel extension.
#include <torch/extension.h>

torch::Tensor group_norm_fused(torch::Tensor x, int64_t groups,
                               torch::Tensor weight, torch::Tensor bias,
                               double eps, bool fuse_silu);
torch::Tensor layer_norm_bf16(torch::Tensor x, torch::Tensor gamma,
                              torch::Tensor beta, double eps);
torch::Tensor act_mul_bf16(torch::Tensor a, torch::Tensor b, bool gelu);
torch::Tensor extract_resize(torch::Tensor src, int64_t x1, int64_t y1,
                             int64_t x2, int64_t y2, int64_t ow, int64_t oh);
void blend_tile(torch::Tensor canvas, torch::Tensor tile, int64_t x1,
                int64_t y1, int64_t x2, int64_t y2, int64_t mx1, int64_t my1,
                int64_t mx2, int64_t my2, double sigma);
torch::Tensor attn_fwd(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                       int64_t heads, int64_t kv_heads, int64_t nk_real,
                       double scale);
torch::Tensor attn_fwd_packed(torch::Tensor q, torch::Tensor k,
                              torch::Tensor v, int64_t heads,
                              int64_t kv_heads, double scale);
torch::Tensor attn_fwd_qkv(torch::Tensor qkv, int64_t heads, double scale);
torch::Tensor attn_fwd_q_kv(torch::Tensor q, torch::Tensor kv, int64_t heads,
                            double scale);
torch::Tensor mfma_selftest(torch::Tensor a, torch::Tensor b);
torch::Tensor conv_nhwc(torch::Tensor x, torch::Tensor wt, torch::Tensor bias,
                        int64_t B, int64_t H, int64_t W, int64_t C, int64_t K,
                        int64_t rs, bool fuse_silu);
torch::Tensor group_norm_nhwc(torch::Tensor x, int64_t groups,
                              torch::Tensor weight, torch::Tensor bias,
                              double eps, bool fuse_silu);
torch::Tensor conv_smallc(torch::Tensor x, torch::Tensor wt, torch::Tensor bias,
                          int64_t B, int64_t H, int64_t W, int64_t C,
                          int64_t K, int64_t rs, bool fuse_silu);
torch::Tensor gemm256_bf16(torch::Tensor x, torch::Tensor w,
                           torch::Tensor bias, bool fuse_silu);
torch::Tensor conv256_nhwc(torch::Tensor x, torch::Tensor wt,
                           torch::Tensor bias, torch::Tensor residual,
                           int64_t B, int64_t H,
                           int64_t W, int64_t C, int64_t K, int64_t rs,
                           int64_t stride, bool up2, bool fuse_silu);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("group_norm_fused", &group_norm_fused,
        "fused GroupNorm(+SiLU), NCHW bf16");
  m.def("layer_norm", &layer_norm_bf16, "LayerNorm over last dim, bf16");
  m.def("act_mul", &act_mul_bf16, "a * act(b) elementwise, bf16");
  m.def("extract_resize", &extract_resize,
        "crop region of [B,H,W,C] f32 -> Lanczos-3 resample");
  m.def("blend_tile", &blend_tile,
        "fused resample + blurred-rect mask + composite (in-place canvas)");
  m.def("attn_fwd", &attn_fwd, "flash attention forward, bf16 MFMA");
  m.def("attn_fwd_packed", &attn_fwd_packed,
        "flash attention on packed [B,N,H*D] QKV (no host reshapes)");
  m.def("attn_fwd_qkv", &attn_fwd_qkv,
        "self-attention on fused [B,N,3*H*D] projection output");
  m.def("attn_fwd_q_kv", &attn_fwd_q_kv,
        "cross-attention on q [B,Nq,H*D] + fused kv [B,Nk,2*H*D]");
  m.def("mfma_selftest", &mfma_selftest,
        "single-wave 16x16x32 bf16 MFMA with the kernel fragment layouts");
  m.def("conv_nhwc", &conv_nhwc,
        "implicit-GEMM 3x3/1x1 NHWC bf16 conv on MFMA (+fused SiLU)");
  m.def("group_norm_nhwc", &group_norm_nhwc,
        "fused GroupNorm(+SiLU), NHWC bf16");
  m.def("conv_smallc", &conv_smallc,
        "small-C NHWC conv (stem convs, C <= 8)");
  m.def("gemm256_bf16", &gemm256_bf16,
        "256x256x64 glds-staged MFMA GEMM, torch-Linear layout (+SiLU)");
  m.def("conv256_nhwc", &conv256_nhwc,
        "256-tile implicit-GEMM 3x3/1x1 NHWC conv on the same template");
}
