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


def test_gligen_boxes_clip_rebase_and_drop():
    emb_a, emb_b, emb_c = (torch.randn(4, 8) for _ in range(3))
    cond = {"context": torch.randn(1, 4, 8), "gligen": [
        (emb_a, (2, 2, 8, 8)),      # fully inside tile (0,0,16,16)
        (emb_b, (12, 12, 10, 10)),  # straddles the tile edge -> clipped
        (emb_c, (24, 24, 4, 4)),    # fully outside -> dropped
    ]}
    out = crop_tile_conditioning(cond, (0, 0, 16, 16), (32, 32), (16, 16))
    assert len(out["gligen"]) == 2
    (ea, box_a), (eb, box_b) = out["gligen"]
    assert ea is emb_a and box_a == (2, 2, 8, 8)        # scale 1, unchanged
    assert eb is emb_b and box_b == (12, 12, 4, 4)      # clipped at 16
    # all boxes disjoint -> key collapses to None
    out2 = crop_tile_conditioning(cond, (24, 0, 32, 8), (32, 32), (8, 8))
    assert out2["gligen"] is None
    # original list untouched
    assert len(cond["gligen"]) == 3


def test_gligen_boxes_scale_with_processing_size():
    cond = {"context": torch.randn(1, 4, 8),
            "gligen": [(torch.randn(2, 8), (4, 4, 8, 8))]}
    out = crop_tile_conditioning(cond, (0, 0, 16, 16), (32, 32), (32, 32))
    _, box = out["gligen"][0]
    assert box == (8, 8, 16, 16)  # 2x upscale of the tile


def test_model_patch_crop_and_restore():
    from comfyui_distributed_amd.engine.model_patch import (
        crop_model_patches, crop_patch_dict)

    patch = {"image": torch.rand(1, 3, 64, 64),
             "latent": torch.rand(1, 4, 8, 8),
             "strength": 0.7}
    out = crop_patch_dict(patch, (0, 0, 32, 32), (64, 64), (32, 32))
    assert out["image"].shape == (1, 3, 32, 32)
    assert torch.equal(out["image"], patch["image"][:, :, :32, :32])
    assert out["latent"].shape == (1, 4, 4, 4)
    assert out["strength"] == 0.7  # non-spatial passthrough

    class S:
        pass

    s = S()
    s.model_patches = {"ctrl": patch}
    with crop_model_patches(s, (0, 0, 32, 32), (64, 64), (16, 16)) as active:
        assert active
        assert s.model_patches["ctrl"]["image"].shape == (1, 3, 16, 16)
    assert s.model_patches["ctrl"] is patch  # restored

    s2 = S()  # no patches -> no-op
    with crop_model_patches(s2, (0, 0, 8, 8), (16, 16), (8, 8)) as active:
        assert not active


def test_usdu_runs_with_model_patches():
    stack = create_diffusion_stack("tiny")
    stack.model_patches = {"ctrl": {"image": torch.rand(1, 3, 32, 32)}}
    cond = stack.make_conditioning(0)
    p = USDUParams(seed=1, steps=1, cfg=1.0, denoise=0.5, tile_width=16,
                   tile_height=16, padding=16, mask_blur=2, tile_batch=4)
    img = torch.rand(1, 32, 32, 3)
    out = process_single_gpu(stack, cond, None, p, img)
    assert torch.isfinite(out).all()
    assert stack.model_patches["ctrl"]["image"].shape == (1, 3, 32, 32)  # restored
