"""Dynamic-mode worker protocol against an in-memory fake master (mirrors
the static-mode protocol test; reference upscale/modes/dynamic.py:213-313)."""

import pytest
import torch

from comfyui_distributed_amd.engine.usdu import USDUParams
from comfyui_distributed_amd.models import create_diffusion_stack
from comfyui_distributed_amd.nodes.runtime import NodeRuntime, set_runtime
from comfyui_distributed_amd.server import usdu_http


@pytest.fixture(autouse=True)
def fresh_runtime():
    set_runtime(None)
    yield
    set_runtime(None)


class FakeDynamicMaster(NodeRuntime):
    def __init__(self, n_images):
        super().__init__()
        self.pending = list(range(n_images))
        self.images = {}
        self.finished = False
        self.heartbeats = 0

    async def post_json(self, url, payload, timeout=60.0):
        if url.endswith("/distributed/job_status"):
            return {"ready": True}
        if url.endswith("/distributed/request_image"):
            if self.pending:
                idx = self.pending.pop(0)
                return {"image_idx": idx,
                        "estimated_remaining": len(self.pending)}
            return {"image_idx": None, "estimated_remaining": 0}
        if url.endswith("/distributed/heartbeat"):
            self.heartbeats += 1
            return {"status": "ok"}
        if url.endswith("/distributed/submit_image"):
            self.images[payload["image_idx"]] = payload
            if payload.get("is_last"):
                self.finished = True
            return {"status": "ok"}
        raise AssertionError(url)


def test_worker_dynamic_protocol():
    stack = create_diffusion_stack("tiny", seed=7)
    cond = stack.make_conditioning(0)
    p = USDUParams(seed=3, steps=1, cfg=1.0, denoise=0.5, tile_width=16,
                   tile_height=16, padding=16, mask_blur=2, tile_batch=4)
    g = torch.Generator().manual_seed(5)
    imgs = torch.rand(3, 32, 32, 3, generator=g)  # batch of 3 whole images
    rt = FakeDynamicMaster(n_images=3)
    set_runtime(rt)
    usdu_http.run_usdu_role(
        mode="dynamic", params=p, stack=stack, cond=cond, uncond=None,
        image=imgs, job_id="dj", is_worker=True,
        master_url="http://master:1", enabled_workers=["w1"], worker_id="w1",
    )
    assert rt.finished
    assert sorted(rt.images) == [0, 1, 2]
    assert rt.heartbeats == 3
    # each submitted image decodes to the right size
    from comfyui_distributed_amd.utils.image import decode_png_base64

    for idx, payload in rt.images.items():
        t = decode_png_base64(payload["image"])
        assert t.shape == (1, 32, 32, 3)
    # the last submission carried is_last (estimated_remaining == 0)
    assert rt.images[2]["is_last"] is True
