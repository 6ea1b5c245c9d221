"""WAN video family tests (tiny config, CPU)."""

import pytest
import torch

from comfyui_distributed_amd.models import create_diffusion_stack
from comfyui_distributed_amd.models.video import (
    VideoGenParams,
    flow_sigmas,
    generate_video,
)


def test_flow_sigmas_monotone():
    s = flow_sigmas(10)
    assert len(s) == 11
    assert s[0] == 1.0 and s[-1] == 0.0
    assert (s.diff() < 0).all()


def test_wan_tiny_generate_video():
    stack = create_diffusion_stack("wan_tiny", seed=1)
    cond = stack.make_conditioning(0)
    p = VideoGenParams(seed=2, steps=2, cfg=1.0, width=16, height=16, frames=5)
    frames = generate_video(stack, cond, None, p)
    assert frames.shape == (5, 16, 16, 3)
    assert torch.isfinite(frames).all()


def test_wan_4n1_validation():
    stack = create_diffusion_stack("wan_tiny")
    with pytest.raises(ValueError):
        stack.validate_frames(8)
    stack.validate_frames(17)


def test_wan_frames_feed_batch_divider():
    from comfyui_distributed_amd.nodes.utilities import ImageBatchDivider

    stack = create_diffusion_stack("wan_tiny", seed=1)
    cond = stack.make_conditioning(0)
    p = VideoGenParams(seed=2, steps=1, cfg=1.0, width=16, height=16, frames=5)
    frames = generate_video(stack, cond, None, p)
    parts = ImageBatchDivider().divide_batch(frames, 2)
    assert parts[0].shape[0] == 3 and parts[1].shape[0] == 2


def test_per_frame_tile_upscale_pipeline():
    """BASELINE config 5 composition: WAN frames -> per-frame distributed
    tile upscale with an image diffusion stack (frames ride the batch dim
    through the USDU engine)."""
    from comfyui_distributed_amd.engine.usdu import USDUParams, process_single_gpu
    from comfyui_distributed_amd.ops import dispatch

    wan = create_diffusion_stack("wan_tiny", seed=1)
    cond_v = wan.make_conditioning(0)
    frames = generate_video(wan, cond_v, None,
                            VideoGenParams(seed=2, steps=1, cfg=1.0,
                                           width=16, height=16, frames=5))
    # 2x Lanczos pre-upscale (the USDU "No Upscale" contract)
    up = dispatch.extract_resize(frames, (0, 0, 16, 16), 32, 32)
    assert up.shape == (5, 32, 32, 3)

    sd = create_diffusion_stack("tiny", seed=3)
    cond_i = sd.make_conditioning(0)
    p = USDUParams(seed=4, steps=1, cfg=1.0, denoise=0.4, tile_width=16,
                   tile_height=16, padding=16, mask_blur=2, tile_batch=4)
    out = process_single_gpu(sd, cond_i, None, p, up)
    assert out.shape == (5, 32, 32, 3)
    assert torch.isfinite(out).all()
