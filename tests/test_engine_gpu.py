"""GPU end-to-end: models + engine on one MI355X."""

import pytest
import torch

pytestmark = pytest.mark.gpu


def test_tiny_stack_gpu_generation():
    from comfyui_distributed_amd.engine import GenParams, generate_images
    from comfyui_distributed_amd.models import create_diffusion_stack

    stack = create_diffusion_stack("tiny", device="cuda:0", dtype=torch.bfloat16)
    cond = stack.make_conditioning(0)
    p = GenParams(seed=1, steps=2, cfg=1.0, width=64, height=64, batch_size=2)
    imgs = generate_images(stack, cond, None, p)
    assert imgs.shape == (2, 64, 64, 3)
    assert torch.isfinite(imgs).all()


def test_sd15_tile_sample_gpu():
    from comfyui_distributed_amd.engine.usdu import (
        USDUParams,
        blend_results,
        plan_for_image,
        sample_tiles,
    )
    from comfyui_distributed_amd.models import create_diffusion_stack

    stack = create_diffusion_stack("sd15", device="cuda:0", dtype=torch.bfloat16)
    cond = stack.make_conditioning(0)
    uncond = stack.make_conditioning(1)
    p = USDUParams(seed=1, steps=2, cfg=8.0, denoise=0.5, tile_width=512,
                   tile_height=512, padding=32, mask_blur=8, tile_batch=2)
    canvas = torch.rand(1, 1024, 1024, 3, device="cuda:0")
    plans = plan_for_image(1024, 1024, p)
    assert len(plans) == 4
    res = sample_tiles(stack, cond, uncond, p, canvas, plans, [0, 3])
    assert set(res.keys()) == {(0, 0), (3, 0)}
    for t in res.values():
        assert t.shape == (1, 544, 544, 3)
        assert torch.isfinite(t).all()
    blend_results(canvas, res, plans, p)
    assert torch.isfinite(canvas).all()


def test_unet_bf16_vs_cpu_fp32_reference():
    """The whole GPU UNet (HIP attention + fused norms) against the CPU fp32
    path at a small spatial size: outputs must agree to bf16 tolerance."""
    from comfyui_distributed_amd.models import create_diffusion_stack

    stack_gpu = create_diffusion_stack("tiny", device="cuda:0", dtype=torch.bfloat16, seed=5)
    stack_cpu = create_diffusion_stack("tiny", device="cpu", dtype=torch.float32, seed=5)
    x = torch.randn(1, 4, 16, 16)
    t = torch.tensor([100.0])
    ctx = stack_cpu.make_conditioning(0)["context"]
    with torch.no_grad():
        out_cpu = stack_cpu.unet(x, t, ctx)
        out_gpu = stack_gpu.unet(
            x.cuda().to(torch.bfloat16), t.cuda(), ctx.cuda().to(torch.bfloat16)
        )
    err = (out_gpu.float().cpu() - out_cpu).abs().max().item()
    rel = err / out_cpu.abs().max().item()
    assert rel < 0.15, f"UNet GPU/CPU mismatch: abs {err}, rel {rel}"


def test_smoke_entry():
    import __graft_entry__

    __graft_entry__.smoke()


def test_wan_tiny_temporal_vae_gpu():
    from comfyui_distributed_amd.models import create_diffusion_stack
    from comfyui_distributed_amd.models.video import VideoGenParams, generate_video

    stack = create_diffusion_stack("wan_tiny", device="cuda:0",
                                   dtype=torch.bfloat16)
    cond = stack.make_conditioning(0)
    p = VideoGenParams(seed=3, steps=1, cfg=1.0, width=32, height=32, frames=9)
    assert stack.latent_frames(9) == 3  # causal 4x temporal compression
    out = generate_video(stack, cond, None, p)
    assert out.shape == (9, 32, 32, 3)
    assert torch.isfinite(out).all()


def test_flux_tiny_gpu_generation():
    from comfyui_distributed_amd.engine.generate import GenParams, generate_images
    from comfyui_distributed_amd.models import create_diffusion_stack

    stack = create_diffusion_stack("flux_tiny", device="cuda:0",
                                   dtype=torch.bfloat16)
    cond = stack.make_conditioning(0)
    p = GenParams(seed=4, steps=2, cfg=1.0, width=64, height=64, batch_size=2)
    imgs = generate_images(stack, cond, None, p)
    assert imgs.shape == (2, 64, 64, 3)
    assert torch.isfinite(imgs).all()


def test_flux_usdu_tile_gpu():
    from comfyui_distributed_amd.engine.usdu import USDUParams, process_single_gpu
    from comfyui_distributed_amd.models import create_diffusion_stack

    stack = create_diffusion_stack("flux_tiny", device="cuda:0",
                                   dtype=torch.bfloat16)
    cond = stack.make_conditioning(1)
    p = USDUParams(seed=5, steps=1, cfg=1.0, denoise=0.5, tile_width=64,
                   tile_height=64, padding=32, mask_blur=4, tile_batch=4)
    img = torch.rand(1, 128, 128, 3, generator=torch.Generator().manual_seed(2))
    out = process_single_gpu(stack, cond, None, p, img)
    assert out.shape == (1, 128, 128, 3)
    assert torch.isfinite(out).all()


def test_graphed_sampler_loop_matches_eager():
    """The whole-loop hipGraph fast path must match the eager sampler loop
    for every captured sampler (same seeds, same conditioning)."""
    import os

    from comfyui_distributed_amd.engine.usdu import (
        USDUParams,
        plan_for_image,
        sample_tiles,
    )
    from comfyui_distributed_amd.models import create_diffusion_stack

    stack = create_diffusion_stack("tiny", device="cuda:0",
                                   dtype=torch.bfloat16, seed=3)
    cond = stack.make_conditioning(0)
    uncond = stack.make_conditioning(1)
    canvas = torch.rand(1, 128, 128, 3, device="cuda:0",
                        generator=torch.Generator("cuda:0").manual_seed(7))
    for sampler in ("euler", "heun", "dpm_2", "dpmpp_2m"):
        p = USDUParams(seed=1, steps=3, cfg=7.0, denoise=0.6, tile_width=64,
                       tile_height=64, padding=16, mask_blur=4, tile_batch=4,
                       sampler_name=sampler)
        plans = plan_for_image(128, 128, p)
        os.environ["DISTGPU_GRAPH_LOOP"] = "0"
        try:
            eager = sample_tiles(stack, cond, uncond, p, canvas, plans, [0, 1])
        finally:
            os.environ["DISTGPU_GRAPH_LOOP"] = "1"
        graphed = sample_tiles(stack, cond, uncond, p, canvas, plans, [0, 1])
        for key in eager:
            a, b = eager[key], graphed[key]
            err = (a - b).abs().max().item()
            assert err < 0.05, f"{sampler} {key}: graphed vs eager err {err}"
