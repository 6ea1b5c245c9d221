"""End-to-end fault tolerance: a worker pulls tiles and dies without
submitting; the master times it out, requeues its tiles and finishes the
job itself (reference master-takeover behavior,
upscale/modes/static.py:354-363 + job_timeout.py)."""

import asyncio
import time

import pytest
import torch

from comfyui_distributed_amd.engine.usdu import USDUParams, process_single_gpu
from comfyui_distributed_amd.models import create_diffusion_stack
from comfyui_distributed_amd.nodes.runtime import NodeRuntime, set_runtime
from comfyui_distributed_amd.server import usdu_http
from comfyui_distributed_amd.utils import constants


@pytest.fixture(autouse=True)
def fresh_runtime():
    set_runtime(None)
    yield
    set_runtime(None)


@pytest.mark.timeout(120)
def test_master_takeover_after_worker_death(tmp_config, monkeypatch):
    monkeypatch.setattr(constants, "COLLECTOR_SLICE_TIMEOUT", 0.05)
    monkeypatch.setattr(constants, "HEARTBEAT_INTERVAL", 0.2)

    stack = create_diffusion_stack("tiny", seed=7)
    cond = stack.make_conditioning(0)
    # tile_batch=1: the master flow samples per tile, and conv numerics are
    # batch-size dependent — the reference must chunk identically
    p = USDUParams(seed=3, steps=1, cfg=1.0, denoise=0.5, tile_width=8,
                   tile_height=8, padding=8, mask_blur=2, tile_batch=1)
    g = torch.Generator().manual_seed(42)
    img = torch.rand(1, 32, 32, 3, generator=g)

    rt = NodeRuntime()

    async def dead_probe(wid):
        return None  # worker is gone

    rt.probe_worker = dead_probe
    set_runtime(rt)

    async def go():
        master = asyncio.create_task(
            usdu_http._master_static(
                "static", p, stack, cond, None, img, "ft_job", ["w1"]
            )
        )

        # "worker": grab two tiles, heartbeat once, then die silently
        async def doomed_worker():
            for _ in range(100):
                job = await rt.job_state.get_tile_job("ft_job")
                if job is not None:
                    break
                await asyncio.sleep(0.02)
            assert job is not None
            grabbed = []
            for _ in range(2):
                try:
                    idx = job.pending_tasks.get_nowait()
                    job.assigned_to_workers[idx] = "w1"
                    grabbed.append(idx)
                except asyncio.QueueEmpty:
                    break
            # back-date the heartbeat so the monitor times us out fast
            job.worker_status["w1"] = time.time() - 9999
            return grabbed

        grabbed = await doomed_worker()
        canvas = await asyncio.wait_for(master, timeout=90)
        return grabbed, canvas

    # use a short worker timeout via config
    from comfyui_distributed_amd.utils.config import load_config, save_config

    cfg = load_config()
    cfg["settings"]["worker_timeout_seconds"] = 1
    save_config(cfg)

    grabbed, canvas = asyncio.run(go())
    assert torch.isfinite(canvas).all()
    # the job completed despite the dead worker; result equals single-GPU
    ref = process_single_gpu(stack, cond, None, p, img)
    assert torch.allclose(canvas, ref, atol=3e-5), (
        (canvas - ref).abs().max().item()
    )
