#!/usr/bin/env python3
"""Kernel microbenchmarks on MI355X: attention / groupnorm / tile ops.

Usage (on a GPU box):
    python tools/kernel_bench.py [--op attn|gn|all] [--iters 50]

Prints per-shape timings + achieved TFLOP/s (attention) / GB/s (norms),
and an eager-torch comparison where applicable.
"""

from __future__ import annotations

import argparse
import sys
import time
from pathlib import Path

import torch

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from comfyui_distributed_amd.ops import dispatch  # noqa: E402


def timeit(fn, iters=50, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def bench_attn(iters):
    print("== attention (bf16, fwd) ==")
    shapes = [
        # (name, BH, Nq, Nk, D, heads)  — SD1.5/SDXL/WAN hot shapes
        ("sd15 self 68x68 b16", 128, 4624, 4624, 40, 8),
        ("sd15 self 34x34 b16", 128, 1156, 1156, 80, 8),
        ("sd15 self 17x17 b16", 128, 289, 289, 160, 8),
        ("sd15 cross 68x68", 128, 4624, 77, 40, 8),
        ("sdxl self 64x64 b4", 80, 4096, 4096, 64, 20),
        ("wan self 32k d128", 16, 32760, 32760, 128, 16),
    ]
    for name, bh, nq, nk, d, heads in shapes:
        q = torch.randn(bh, nq, d, device="cuda", dtype=torch.bfloat16) / 2
        k = torch.randn(bh, nk, d, device="cuda", dtype=torch.bfloat16) / 2
        v = torch.randn(bh, nk, d, device="cuda", dtype=torch.bfloat16)

        t = timeit(lambda: dispatch.attention(q, k, v, heads=heads), iters)
        dp = dispatch._dpad_for(d)
        flops = 4 * bh * nq * nk * d  # 2 gemms, real D
        flops_pad = 4 * bh * nq * nk * dp
        print(f"{name:24s} D={d:3d}->{dp:3d}  {t*1e3:8.3f} ms  "
              f"{flops/t/1e12:7.1f} TF (real) {flops_pad/t/1e12:7.1f} TF (padded)")


def bench_gn(iters):
    print("== fused GroupNorm+SiLU NCHW (bf16) ==")
    from comfyui_distributed_amd.ops import ext

    mod = ext.get_ext(True)
    for shape in [(16, 320, 68, 68), (16, 640, 34, 34), (16, 1280, 17, 17),
                  (8, 512, 136, 136), (1, 128, 1088, 1088)]:
        x = torch.randn(*shape, device="cuda", dtype=torch.bfloat16)
        w = torch.randn(shape[1], device="cuda")
        b = torch.randn(shape[1], device="cuda")
        t = timeit(lambda: mod.group_norm_fused(x, 32, w, b, 1e-5, True), iters)
        gb = 2 * x.numel() * 2 / 1e9  # read + write bf16
        # eager comparison
        import torch.nn.functional as F

        xf = x
        te = timeit(lambda: F.silu(F.group_norm(xf.float(), 32, w, b, 1e-5)).to(torch.bfloat16), iters)
        print(f"{str(shape):24s} {t*1e3:7.3f} ms  {gb/t:7.0f} GB/s   "
              f"eager {te*1e3:7.3f} ms ({te/t:4.1f}x)")


def bench_gn_nhwc(iters):
    print("== fused GroupNorm+SiLU NHWC/channels-last (bf16) — the hot one ==")
    import torch.nn.functional as F

    from comfyui_distributed_amd.ops import ext

    mod = ext.get_ext(True)
    # (B, H, W, C): UNet trunk levels at tile_batch=16, VAE decode stages
    for shape in [(16, 68, 68, 320), (16, 34, 34, 640), (16, 17, 17, 1280),
                  (16, 136, 136, 512), (16, 272, 272, 512),
                  (16, 544, 544, 128), (1, 1088, 1088, 128)]:
        x = torch.randn(*shape, device="cuda", dtype=torch.bfloat16)
        C = shape[3]
        w = torch.randn(C, device="cuda")
        b = torch.randn(C, device="cuda")
        y = mod.group_norm_nhwc(x, 32, w, b, 1e-5, True)
        # reference: fp32 NCHW group_norm on the same data
        xf = x.permute(0, 3, 1, 2).float()
        ref = F.silu(F.group_norm(xf, 32, w, b, 1e-5)).permute(0, 2, 3, 1)
        err = (y.float() - ref).abs().max().item()
        t = timeit(lambda: mod.group_norm_nhwc(x, 32, w, b, 1e-5, True), iters)
        gb = 3 * x.numel() * 2 / 1e9  # 2 reads + 1 write bf16
        print(f"{str(shape):24s} {t*1e3:7.3f} ms  {gb/t:7.0f} GB/s  "
              f"maxerr {err:.4f}")


def bench_conv(iters):
    print("== conv3x3 bf16: ours (NHWC MFMA) vs MIOpen NCHW vs MIOpen CL ==")
    from comfyui_distributed_amd.ops import dispatch

    shapes = [
        # (B, C, H, W, K) — UNet/VAE hot shapes at tile batch 16
        (16, 320, 68, 68, 320),
        (16, 640, 34, 34, 640),
        (16, 1280, 17, 17, 1280),
        (1, 512, 136, 136, 512),
        (1, 256, 544, 544, 256),
        (1, 128, 1088, 1088, 128),
    ]
    for b, c, h, w, k in shapes:
        conv = torch.nn.Conv2d(c, k, 3, padding=1).cuda().to(torch.bfloat16)
        x = (torch.randn(b, c, h, w) / 4).cuda().to(torch.bfloat16)
        xcl = x.contiguous(memory_format=torch.channels_last)
        conv_cl = torch.nn.Conv2d(c, k, 3, padding=1).cuda().to(torch.bfloat16) \
            .to(memory_format=torch.channels_last)
        t_ours = timeit(lambda: dispatch.conv2d_mfma(xcl, conv), iters)
        t_nchw = timeit(lambda: conv(x), iters)
        t_cl = timeit(lambda: conv_cl(xcl), iters)
        flops = 2 * b * h * w * k * c * 9
        print(f"B{b} C{c} {h}x{w} K{k}: ours {t_ours*1e3:7.3f} ms "
              f"({flops/t_ours/1e12:6.1f} TF) | miopen-nchw {t_nchw*1e3:7.3f} "
              f"| miopen-cl {t_cl*1e3:7.3f}")


def bench_gemm(iters):
    print("== gemm256 bf16 (glds 256-tile) vs hipBLASLt ==")
    from comfyui_distributed_amd.ops import ext

    mod = ext.get_ext(True)
    shapes = [
        # (M, N, K) — UNet Linear hot shapes at tile_batch 16 (2x CFG batch)
        (32 * 4624, 320, 320),    # qkv proj level0 (N=960 fused)
        (32 * 4624, 960, 320),    # fused qkv
        (32 * 4624, 2560, 320),   # GEGLU in
        (32 * 4624, 320, 1280),   # GEGLU out
        (32 * 1156, 1920, 640),
        (32 * 289, 3840, 1280),
        (8192, 8192, 8192),       # reference square
    ]
    for (M, N, K) in shapes:
        x = (torch.randn(M, K, device="cuda") / 4).to(torch.bfloat16)
        w = (torch.randn(N, K, device="cuda") / 4).to(torch.bfloat16)
        b = torch.empty(0, device="cuda")
        t = timeit(lambda: mod.gemm256_bf16(x, w, b, False), iters)
        tl = timeit(lambda: torch.nn.functional.linear(x, w), iters)
        fl = 2.0 * M * N * K
        print(f"M{M} N{N} K{K}: ours {t*1e3:7.3f} ms ({fl/t/1e12:6.1f} TF) | "
              f"blaslt {tl*1e3:7.3f} ms ({fl/tl/1e12:6.1f} TF)")


def bench_conv256(iters):
    print("== conv256 (glds template) vs conv v2 vs MIOpen-CL ==")
    import os

    from comfyui_distributed_amd.ops import dispatch

    shapes = [
        (32, 320, 68, 68, 320),
        (32, 640, 34, 34, 640),
        (32, 1280, 17, 17, 1280),
        (16, 512, 136, 136, 512),
        (16, 256, 272, 272, 256),
        (16, 128, 544, 544, 128),
    ]
    for b, c, h, w, k in shapes:
        conv = torch.nn.Conv2d(c, k, 3, padding=1).cuda().to(torch.bfloat16)
        x = (torch.randn(b, c, h, w) / 4).cuda().to(torch.bfloat16)
        xcl = x.contiguous(memory_format=torch.channels_last)
        conv_cl = torch.nn.Conv2d(c, k, 3, padding=1).cuda().to(torch.bfloat16) \
            .to(memory_format=torch.channels_last)
        dispatch._CONV256 = True
        t256 = timeit(lambda: dispatch.conv2d_mfma(xcl, conv), iters)
        dispatch._CONV256 = False
        tv2 = timeit(lambda: dispatch.conv2d_mfma(xcl, conv), iters)
        dispatch._CONV256 = True
        tmi = timeit(lambda: conv_cl(xcl), iters)
        flops = 2.0 * b * h * w * k * c * 9
        print(f"B{b} C{c} {h}x{w} K{k}: 256 {t256*1e3:7.3f} ms "
              f"({flops/t256/1e12:6.1f} TF) | v2 {tv2*1e3:7.3f} "
              f"({flops/tv2/1e12:6.1f} TF) | miopen-cl {tmi*1e3:7.3f} "
              f"({flops/tmi/1e12:6.1f} TF)")


def bench_tiles(iters):
    print("== tile ops (f32) ==")
    from comfyui_distributed_amd.ops import ext

    mod = ext.get_ext(True)
    src = torch.rand(1, 4096, 4096, 3, device="cuda")
    t = timeit(lambda: mod.extract_resize(src, 512, 512, 1056, 1056, 544, 544), iters)
    print(f"extract 544 from 4K     {t*1e3:7.3f} ms")
    canvas = torch.rand(1, 4096, 4096, 3, device="cuda").contiguous()
    tile = torch.rand(1, 544, 544, 3, device="cuda")
    t = timeit(lambda: mod.blend_tile(canvas, tile, 512, 512, 1056, 1056,
                                      544, 544, 1024, 1024, 8.0), iters)
    print(f"blend 544 into 4K       {t*1e3:7.3f} ms")


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--op", default="all")
    ap.add_argument("--iters", type=int, default=50)
    args = ap.parse_args()
    assert torch.cuda.is_available()
    if args.op in ("attn", "all"):
        bench_attn(args.iters)
    if args.op in ("gn", "all"):
        bench_gn(args.iters)
    if args.op in ("gn_nhwc", "gn", "all"):
        bench_gn_nhwc(args.iters)
    if args.op in ("conv", "all"):
        bench_conv(args.iters)
    if args.op in ("gemm", "all"):
        bench_gemm(args.iters)
    if args.op in ("conv256", "all"):
        bench_conv256(args.iters)
    if args.op in ("tiles", "all"):
        bench_tiles(args.iters)


if __name__ == "__main__":
    main()
