import torch

from comfyui_distributed_amd.engine.conditioning import crop_tile_conditioning
from comfyui_distributed_amd.engine.usdu import USDUParams, process_single_gpu
from comfyui_distributed_amd.models import create_diffusion_stack


def test_non_spatial_passthrough_shares():
    cond = {"context": torch.randn(1, 4, 8)}
    out = crop_tile_conditioning(cond, (0, 0, 8, 8), (16, 16), (8, 8))
    assert out is cond  # shared, not cloned


def test_control_hint_cropped_and_resized():
    hint = torch.zeros(1, 3, 32, 32)
    hint[:, :, 8:16, 8:16] = 1.0
    cond = {"context": torch.randn(1, 4, 8), "control_hint": hint}
    out = crop_tile_conditioning(cond, (8, 8, 16, 16), (32, 32), (24, 24))
    assert out["control_hint"].shape == (1, 3, 24, 24)
    assert out["control_hint"].mean() > 0.95  # all inside the white rect
    # original untouched
    assert cond["control_hint"].shape == (1, 3, 32, 32)


def test_mask_crop_keeps_rank():
    cond = {"context": torch.randn(1, 4, 8), "mask": torch.ones(1, 32, 32)}
    out = crop_tile_conditioning(cond, (0, 0, 16, 16), (32, 32), (8, 8))
    assert out["mask"].shape == (1, 8, 8)


def test_area_intersection_and_rebase():
    cond = {"context": torch.randn(1, 4, 8), "area": (10, 10, 20, 20)}
    out = crop_tile_conditioning(cond, (0, 0, 16, 16), (32, 32), (16, 16))
    # intersect (10,10,16,16) -> rebased (10,10) size 6x6, scale 1
    assert out["area"] == (10, 10, 6, 6)
    out2 = crop_tile_conditioning(cond, (30, 30, 32, 32), (32, 32), (8, 8))
    assert out2["area"] is None  # disjoint


def test_reference_latents_latent_space_crop():
    cond = {"context": torch.randn(1, 4, 8),
            "reference_latents": torch.randn(1, 4, 8, 8)}  # 64px canvas /8
    out = crop_tile_conditioning(cond, (0, 0, 32, 32), (64, 64), (32, 32))
    assert out["reference_latents"].shape == (1, 4, 4, 4)


def test_usdu_runs_with_spatial_conditioning():
    """End-to-end: spatial conditioning forces per-tile denoisers and the
    pipeline still produces a finite canvas."""
    stack = create_diffusion_stack("tiny")
    cond = stack.make_conditioning(0)
    cond["control_hint"] = torch.rand(1, 3, 32, 32)
    p = USDUParams(seed=1, steps=1, cfg=1.0, denoise=0.5, tile_width=16,
                   tile_height=16, padding=16, mask_blur=2, tile_batch=4)
    img = torch.rand(1, 32, 32, 3)
    out = process_single_gpu(stack, cond, None, p, img)
    assert torch.isfinite(out).all()
