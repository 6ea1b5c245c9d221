"""Flux-family MMDiT: param counts, shapes, determinism, seed-parallel."""

import pytest
import torch

from comfyui_distributed_amd.models.registry import create_diffusion_stack


def test_flux_tiny_forward_shapes():
    stack = create_diffusion_stack("flux_tiny")
    cond = stack.make_conditioning(0)
    x = torch.randn(2, 16, 8, 8)
    t = torch.full((2,), 500.0)
    out = stack.model(x, t, cond["context"].expand(2, -1, -1),
                      cond["vec"].expand(2, -1))
    assert out.shape == x.shape
    assert torch.isfinite(out).all()


def test_flux12b_param_count():
    """Flux-dev scale: ~11-12B parameters at dim 3072, 19+38 blocks."""
    from comfyui_distributed_amd.models.flux import FLUX12B, FluxModel

    # count without materializing 12B floats: sum shapes via meta device
    with torch.device("meta"):
        m = FluxModel(FLUX12B)
    n = sum(p.numel() for p in m.parameters())
    assert 10.5e9 < n < 13.5e9, n / 1e9


def test_flux_generation_deterministic():
    from comfyui_distributed_amd.engine.generate import GenParams, generate_images

    stack = create_diffusion_stack("flux_tiny")
    cond = stack.make_conditioning(1)
    p = GenParams(seed=7, steps=2, cfg=1.0, width=32, height=32, batch_size=1)
    a = generate_images(stack, cond, None, p)
    b = generate_images(stack, cond, None, p)
    assert a.shape == (1, 32, 32, 3)
    assert torch.equal(a, b)
    c = generate_images(stack, cond, None, GenParams(
        seed=8, steps=2, cfg=1.0, width=32, height=32))
    assert not torch.allclose(a, c)


def test_flux_cfg_path():
    from comfyui_distributed_amd.engine.generate import GenParams, generate_images

    stack = create_diffusion_stack("flux_tiny")
    cond, uncond = stack.make_conditioning(1), stack.make_conditioning(2)
    p = GenParams(seed=7, steps=1, cfg=3.0, width=16, height=16)
    out = generate_images(stack, cond, uncond, p)
    assert torch.isfinite(out).all()


def test_flux_workflow_graph():
    from comfyui_distributed_amd.graph.executor import Executor

    prompt = {
        "1": {"class_type": "CheckpointLoader", "inputs": {"ckpt_name": "flux_tiny"}},
        "2": {"class_type": "CLIPTextEncode",
              "inputs": {"text": "a fox", "clip": ["1", 1]}},
        "3": {"class_type": "FluxGenerate", "inputs": {
            "model": ["1", 0], "positive": ["2", 0], "seed": 4, "steps": 1,
            "cfg": 1.0, "width": 16, "height": 16, "batch_size": 2}},
        "4": {"class_type": "PreviewImage", "inputs": {"images": ["3", 0]}},
    }
    previews = []
    ex = Executor(context={"preview_images": previews, "device": "cpu"})
    ex.execute(prompt)
    assert previews and previews[0].shape == (2, 16, 16, 3)


def test_flux_usdu_single_gpu_and_order_invariance():
    """Flux runs through the distributed tile upscaler: flow img2img per
    tile, same determinism contract as SD (tile order irrelevant)."""
    from comfyui_distributed_amd.engine.usdu import (
        USDUParams, blend_results, plan_for_image, process_single_gpu,
        sample_tiles)

    stack = create_diffusion_stack("flux_tiny")
    cond = stack.make_conditioning(1)
    p = USDUParams(seed=5, steps=1, cfg=1.0, denoise=0.5, tile_width=16,
                   tile_height=16, padding=16, mask_blur=2, tile_batch=4)
    img = torch.rand(1, 32, 32, 3, generator=torch.Generator().manual_seed(2))
    out = process_single_gpu(stack, cond, None, p, img)
    assert out.shape == (1, 32, 32, 3)
    assert torch.isfinite(out).all()

    # processing tiles in two disjoint calls (any order) blends identically
    plans = plan_for_image(32, 32, p)
    canvas = img.clone().float()
    r1 = sample_tiles(stack, cond, None, p, canvas, plans, [2, 3])
    r2 = sample_tiles(stack, cond, None, p, canvas, plans, [0, 1])
    blend_results(canvas, {**r1, **r2}, plans, p)
    assert torch.allclose(canvas, out, atol=1e-5)


def test_flux_img2img_node_path():
    from comfyui_distributed_amd.graph.builtin_nodes import FluxGenerate

    stack = create_diffusion_stack("flux_tiny")
    cond = stack.make_conditioning(1)
    src = torch.rand(1, 32, 32, 3, generator=torch.Generator().manual_seed(4))
    out = FluxGenerate().generate(stack, cond, seed=2, steps=1, cfg=1.0,
                                  width=64, height=64, image=src,
                                  denoise=0.4)[0]
    assert out.shape == (1, 32, 32, 3)  # follows the init image size
    assert torch.isfinite(out).all()
    # low denoise keeps the output near the init image reconstruction
    with torch.no_grad():
        recon = stack.vae.decode(stack.vae.encode(src))
    out_hi = FluxGenerate().generate(stack, cond, seed=2, steps=1, cfg=1.0,
                                     width=64, height=64, image=src,
                                     denoise=1.0)[0]
    assert (out - recon.float()).abs().mean() < (out_hi - recon.float()).abs().mean()
