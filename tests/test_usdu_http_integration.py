"""Tile-parallel USDU over real HTTP: master + worker servers in one
process, the full static-mode protocol (job_status poll -> request_image
pull -> submit_tiles -> heartbeat), and the determinism contract: the
distributed canvas equals the single-GPU canvas exactly."""

import asyncio

import pytest
import torch
from aiohttp.test_utils import TestClient, TestServer

from comfyui_distributed_amd.server.app import DistributedServer
from comfyui_distributed_amd.utils.config import load_config, save_config


@pytest.mark.timeout(180)
def test_distributed_usdu_static_over_http(tmp_config, monkeypatch):
    from comfyui_distributed_amd.utils import constants

    monkeypatch.setattr(constants, "COLLECTOR_SLICE_TIMEOUT", 0.1)
    monkeypatch.setattr(constants, "JOB_READY_POLL_INTERVAL", 0.2)

    async def go():
        worker_srv = DistributedServer(is_worker=True)
        worker_client = TestClient(TestServer(worker_srv.build_app()))
        await worker_client.start_server()

        master_srv = DistributedServer()
        previews: list = []
        master_srv.executor.context["preview_images"] = previews
        master_client = TestClient(TestServer(master_srv.build_app()))
        await master_client.start_server()

        cfg = load_config()
        cfg["workers"] = [{
            "id": "w1", "name": "worker1", "host": "127.0.0.1",
            "port": worker_client.server.port, "cuda_device": 0,
            "enabled": True, "type": "remote",
        }]
        cfg["master"]["host"] = "127.0.0.1"
        cfg["master"]["port"] = master_client.server.port
        save_config(cfg)

        prompt = {
            "1": {"class_type": "CheckpointLoader", "inputs": {"ckpt_name": "tiny"}},
            "2": {"class_type": "CLIPTextEncode",
                  "inputs": {"text": "detail", "clip": ["1", 1]}},
            "3": {"class_type": "CLIPTextEncode",
                  "inputs": {"text": "", "clip": ["1", 1]}},
            "4": {"class_type": "LoadImage", "inputs": {"image": "synthetic:32x32"}},
            "5": {"class_type": "UltimateSDUpscaleDistributed", "inputs": {
                "upscaled_image": ["4", 0], "model": ["1", 0],
                "positive": ["2", 0], "negative": ["3", 0], "vae": ["1", 2],
                "seed": 3, "steps": 1, "cfg": 1.0, "sampler_name": "euler",
                "scheduler": "karras", "denoise": 0.5, "tile_width": 16,
                "tile_height": 16, "padding": 16, "mask_blur": 2,
                "force_uniform_tiles": True, "tiled_decode": False}},
            "6": {"class_type": "DistributedCollector",
                  "inputs": {"images": ["5", 0], "load_balance": False}},
            "7": {"class_type": "PreviewImage", "inputs": {"images": ["6", 0]}},
        }

        resp = await master_client.post("/distributed/queue", json={
            "prompt": prompt, "client_id": "it", "enabled_worker_ids": ["w1"],
        })
        assert resp.status == 200

        for _ in range(600):
            if previews:
                break
            await asyncio.sleep(0.25)

        from comfyui_distributed_amd.server.network import close_client_session

        await close_client_session()
        await worker_client.close()
        await master_client.close()
        return previews

    previews = asyncio.run(go())
    assert previews, "distributed USDU never completed"
    canvas = previews[0]
    assert canvas.shape == (1, 32, 32, 3)
    assert torch.isfinite(canvas).all()

    # determinism contract: distributed == single-GPU, bit-for-bit-ish
    from comfyui_distributed_amd.engine.usdu import USDUParams, process_single_gpu
    from comfyui_distributed_amd.graph.builtin_nodes import _STACK_CACHE

    from comfyui_distributed_amd.graph.builtin_nodes import stable_text_seed

    stack = _STACK_CACHE[("tiny", "cpu")]
    cond = stack.make_conditioning(stable_text_seed("detail"))
    uncond = stack.make_conditioning(stable_text_seed(""))
    p = USDUParams(seed=3, steps=1, cfg=1.0, sampler_name="euler",
                   scheduler="karras", denoise=0.5, tile_width=16,
                   tile_height=16, padding=16, mask_blur=2)
    g = torch.Generator().manual_seed(0)
    img = torch.rand(1, 32, 32, 3, generator=g)  # synthetic:32x32 seed 0
    ref = process_single_gpu(stack, cond, uncond, p, img)
    # worker tiles cross the wire as PNG (uint8): tolerance covers that
    # quantization, and nothing else (a conditioning/weight/seed mismatch
    # shows up as O(0.5) diffs)
    assert torch.allclose(canvas, ref, atol=0.02), (
        (canvas - ref).abs().max().item()
    )


@pytest.mark.timeout(180)
def test_zombie_worker_tiles_requeued_over_http(tmp_config, monkeypatch):
    """Protocol-level fault injection: a 'worker' that exists only on the
    wire (answers probes with an idle queue) claims tiles via
    /distributed/request_image, heartbeats once, and never submits. The
    timeout monitor must probe it, requeue its tiles, and the master must
    finish the job with the exact single-GPU result."""
    from aiohttp import web

    from comfyui_distributed_amd.utils import constants

    monkeypatch.setattr(constants, "COLLECTOR_SLICE_TIMEOUT", 0.1)
    monkeypatch.setattr(constants, "HEARTBEAT_INTERVAL", 0.3)

    async def go():
        # fake worker: passes preflight + accepts dispatch + probes idle
        fake = web.Application()

        async def fake_get(_r):
            return web.json_response({"exec_info": {"queue_remaining": 0}})

        async def fake_post(_r):
            return web.json_response({"prompt_id": "fake"})

        fake.router.add_get("/prompt", fake_get)
        fake.router.add_post("/prompt", fake_post)
        fake_client = TestClient(TestServer(fake))
        await fake_client.start_server()

        master_srv = DistributedServer()
        previews: list = []
        master_srv.executor.context["preview_images"] = previews
        mc = TestClient(TestServer(master_srv.build_app()))
        await mc.start_server()

        cfg = load_config()
        cfg["workers"] = [{
            "id": "w2", "name": "zombie", "host": "127.0.0.1",
            "port": fake_client.server.port, "cuda_device": 0,
            "enabled": True, "type": "remote",
        }]
        cfg["master"]["host"] = "127.0.0.1"
        cfg["master"]["port"] = mc.server.port
        cfg["settings"]["worker_timeout_seconds"] = 1
        save_config(cfg)

        prompt = {
            "1": {"class_type": "CheckpointLoader", "inputs": {"ckpt_name": "tiny"}},
            "2": {"class_type": "CLIPTextEncode",
                  "inputs": {"text": "detail", "clip": ["1", 1]}},
            "3": {"class_type": "CLIPTextEncode",
                  "inputs": {"text": "", "clip": ["1", 1]}},
            "4": {"class_type": "LoadImage", "inputs": {"image": "synthetic:48x48"}},
            "5": {"class_type": "UltimateSDUpscaleDistributed", "inputs": {
                "upscaled_image": ["4", 0], "model": ["1", 0],
                "positive": ["2", 0], "negative": ["3", 0], "vae": ["1", 2],
                "seed": 3, "steps": 1, "cfg": 1.0, "sampler_name": "euler",
                "scheduler": "karras", "denoise": 0.5, "tile_width": 16,
                "tile_height": 16, "padding": 16, "mask_blur": 2,
                "force_uniform_tiles": True, "tiled_decode": False}},
            "6": {"class_type": "DistributedCollector",
                  "inputs": {"images": ["5", 0], "load_balance": False}},
            "7": {"class_type": "PreviewImage", "inputs": {"images": ["6", 0]}},
        }
        resp = await mc.post("/distributed/queue", json={
            "prompt": prompt, "client_id": "zt", "enabled_worker_ids": ["w2"]})
        assert resp.status == 200
        body = await resp.json()
        assert "w2" in body["participants"]
        job_id = body["job_ids"]["5"]

        # zombie protocol: wait for the tile job, claim up to 2 tiles, die
        claimed = []
        for _ in range(200):
            r = await mc.post("/distributed/job_status", json={"job_id": job_id})
            if r.status == 200 and (await r.json()).get("ready"):
                break
            await asyncio.sleep(0.05)
        for _ in range(2):
            r = await mc.post("/distributed/request_image", json={
                "job_id": job_id, "worker_id": "w2"})
            if r.status == 200:
                idx = (await r.json()).get("tile_idx")
                if idx is not None:
                    claimed.append(idx)
        await mc.post("/distributed/heartbeat", json={
            "job_id": job_id, "worker_id": "w2"})
        # ...and never submit anything again

        for _ in range(600):
            if previews:
                break
            await asyncio.sleep(0.25)

        from comfyui_distributed_amd.server.network import close_client_session

        await close_client_session()
        await fake_client.close()
        await mc.close()
        return previews, claimed

    previews, claimed = asyncio.run(go())
    assert previews, "job never completed after zombie worker death"
    canvas = previews[0]
    assert canvas.shape == (1, 48, 48, 3)

    # exactness: master-recovered canvas == single-GPU reference
    from comfyui_distributed_amd.engine.usdu import USDUParams, process_single_gpu
    from comfyui_distributed_amd.graph.builtin_nodes import (
        _STACK_CACHE, stable_text_seed)

    stack = _STACK_CACHE[("tiny", "cpu")]
    cond = stack.make_conditioning(stable_text_seed("detail"))
    uncond = stack.make_conditioning(stable_text_seed(""))
    p = USDUParams(seed=3, steps=1, cfg=1.0, sampler_name="euler",
                   scheduler="karras", denoise=0.5, tile_width=16,
                   tile_height=16, padding=16, mask_blur=2)
    g = torch.Generator().manual_seed(0)
    img = torch.rand(1, 48, 48, 3, generator=g)
    ref = process_single_gpu(stack, cond, uncond, p, img)
    assert torch.allclose(canvas, ref, atol=1e-5), (
        (canvas - ref).abs().max().item()
    )


@pytest.mark.timeout(180)
def test_distributed_usdu_dynamic_over_http(tmp_config, monkeypatch):
    """Image-parallel (dynamic) mode end-to-end over real HTTP: batch >=
    dynamic_threshold routes whole images through the pull queue; the
    worker returns full frames via /distributed/submit_image."""
    from comfyui_distributed_amd.utils import constants

    monkeypatch.setattr(constants, "COLLECTOR_SLICE_TIMEOUT", 0.1)
    monkeypatch.setattr(constants, "JOB_READY_POLL_INTERVAL", 0.2)

    async def go():
        worker_srv = DistributedServer(is_worker=True)
        wc = TestClient(TestServer(worker_srv.build_app()))
        await wc.start_server()
        master_srv = DistributedServer()
        previews: list = []
        master_srv.executor.context["preview_images"] = previews
        mc = TestClient(TestServer(master_srv.build_app()))
        await mc.start_server()

        cfg = load_config()
        cfg["workers"] = [{
            "id": "w1", "name": "worker1", "host": "127.0.0.1",
            "port": wc.server.port, "cuda_device": 0,
            "enabled": True, "type": "remote",
        }]
        cfg["master"]["host"] = "127.0.0.1"
        cfg["master"]["port"] = mc.server.port
        save_config(cfg)

        prompt = {
            "1": {"class_type": "CheckpointLoader", "inputs": {"ckpt_name": "tiny"}},
            "2": {"class_type": "CLIPTextEncode",
                  "inputs": {"text": "detail", "clip": ["1", 1]}},
            "3": {"class_type": "CLIPTextEncode",
                  "inputs": {"text": "", "clip": ["1", 1]}},
            "4": {"class_type": "EmptyLatentImage",
                  "inputs": {"width": 32, "height": 32, "batch_size": 2}},
            "5": {"class_type": "KSampler", "inputs": {
                "model": ["1", 0], "seed": 2, "steps": 1, "cfg": 1.0,
                "sampler_name": "euler", "scheduler": "karras",
                "positive": ["2", 0], "negative": ["3", 0],
                "latent_image": ["4", 0], "denoise": 1.0}},
            "6": {"class_type": "VAEDecode",
                  "inputs": {"samples": ["5", 0], "vae": ["1", 2]}},
            "7": {"class_type": "UltimateSDUpscaleDistributed", "inputs": {
                "upscaled_image": ["6", 0], "model": ["1", 0],
                "positive": ["2", 0], "negative": ["3", 0], "vae": ["1", 2],
                "seed": 3, "steps": 1, "cfg": 1.0, "sampler_name": "euler",
                "scheduler": "karras", "denoise": 0.5, "tile_width": 16,
                "tile_height": 16, "padding": 16, "mask_blur": 2,
                "force_uniform_tiles": True, "tiled_decode": False,
                "dynamic_threshold": 2}},
            "8": {"class_type": "DistributedCollector",
                  "inputs": {"images": ["7", 0], "load_balance": False}},
            "9": {"class_type": "PreviewImage", "inputs": {"images": ["8", 0]}},
        }
        resp = await mc.post("/distributed/queue", json={
            "prompt": prompt, "client_id": "dyn", "enabled_worker_ids": ["w1"]})
        assert resp.status == 200

        for _ in range(600):
            if previews:
                break
            await asyncio.sleep(0.25)

        from comfyui_distributed_amd.server.network import close_client_session

        await close_client_session()
        await wc.close()
        await mc.close()
        return previews

    previews = asyncio.run(go())
    assert previews, "dynamic USDU never completed"
    canvas = previews[0]
    assert canvas.shape == (2, 32, 32, 3)
    assert torch.isfinite(canvas).all()


@pytest.mark.timeout(180)
def test_zombie_worker_image_requeued_dynamic_mode(tmp_config, monkeypatch):
    """Dynamic-mode fault injection: the zombie claims a whole-image index
    via the pull queue and never submits; the monitor must requeue it and
    the master must finish every image."""
    from aiohttp import web

    from comfyui_distributed_amd.utils import constants

    monkeypatch.setattr(constants, "COLLECTOR_SLICE_TIMEOUT", 0.1)
    monkeypatch.setattr(constants, "HEARTBEAT_INTERVAL", 0.3)

    async def go():
        fake = web.Application()

        async def fake_get(_r):
            return web.json_response({"exec_info": {"queue_remaining": 0}})

        async def fake_post(_r):
            return web.json_response({"prompt_id": "fake"})

        fake.router.add_get("/prompt", fake_get)
        fake.router.add_post("/prompt", fake_post)
        fc = TestClient(TestServer(fake))
        await fc.start_server()

        master_srv = DistributedServer()
        previews: list = []
        master_srv.executor.context["preview_images"] = previews
        mc = TestClient(TestServer(master_srv.build_app()))
        await mc.start_server()

        cfg = load_config()
        cfg["workers"] = [{
            "id": "wz", "name": "zombie", "host": "127.0.0.1",
            "port": fc.server.port, "cuda_device": 0, "enabled": True,
            "type": "remote",
        }]
        cfg["master"]["host"] = "127.0.0.1"
        cfg["master"]["port"] = mc.server.port
        cfg["settings"]["worker_timeout_seconds"] = 1
        save_config(cfg)

        prompt = {
            "1": {"class_type": "CheckpointLoader", "inputs": {"ckpt_name": "tiny"}},
            "2": {"class_type": "CLIPTextEncode",
                  "inputs": {"text": "x", "clip": ["1", 1]}},
            "3": {"class_type": "EmptyLatentImage",
                  "inputs": {"width": 32, "height": 32, "batch_size": 2}},
            "4": {"class_type": "KSampler", "inputs": {
                "model": ["1", 0], "seed": 2, "steps": 1, "cfg": 1.0,
                "sampler_name": "euler", "scheduler": "karras",
                "positive": ["2", 0], "negative": ["2", 0],
                "latent_image": ["3", 0], "denoise": 1.0}},
            "5": {"class_type": "VAEDecode",
                  "inputs": {"samples": ["4", 0], "vae": ["1", 2]}},
            "6": {"class_type": "UltimateSDUpscaleDistributed", "inputs": {
                "upscaled_image": ["5", 0], "model": ["1", 0],
                "positive": ["2", 0], "negative": ["2", 0], "vae": ["1", 2],
                "seed": 3, "steps": 1, "cfg": 1.0, "sampler_name": "euler",
                "scheduler": "karras", "denoise": 0.5, "tile_width": 16,
                "tile_height": 16, "padding": 16, "mask_blur": 2,
                "force_uniform_tiles": True, "tiled_decode": False,
                "dynamic_threshold": 2}},
            "7": {"class_type": "DistributedCollector",
                  "inputs": {"images": ["6", 0], "load_balance": False}},
            "8": {"class_type": "PreviewImage", "inputs": {"images": ["7", 0]}},
        }
        resp = await mc.post("/distributed/queue", json={
            "prompt": prompt, "client_id": "dz", "enabled_worker_ids": ["wz"]})
        assert resp.status == 200
        job_id = (await resp.json())["job_ids"]["6"]

        # zombie: wait for readiness, claim one image index, vanish
        claimed = None
        for _ in range(200):
            r = await mc.post("/distributed/job_status", json={"job_id": job_id})
            if r.status == 200 and (await r.json()).get("ready"):
                break
            await asyncio.sleep(0.05)
        r = await mc.post("/distributed/request_image", json={
            "job_id": job_id, "worker_id": "wz"})
        if r.status == 200:
            claimed = (await r.json()).get("image_idx")
        await mc.post("/distributed/heartbeat", json={
            "job_id": job_id, "worker_id": "wz"})

        for _ in range(600):
            if previews:
                break
            await asyncio.sleep(0.25)
        from comfyui_distributed_amd.server.network import close_client_session

        await close_client_session()
        await fc.close()
        await mc.close()
        return previews, claimed

    previews, claimed = asyncio.run(go())
    assert previews, "dynamic job never completed after zombie claim"
    canvas = previews[0]
    assert canvas.shape == (2, 32, 32, 3)
    assert torch.isfinite(canvas).all()


@pytest.mark.timeout(180)
def test_video_upscale_pipeline_over_http(tmp_config, monkeypatch):
    """The distributed_upscale_video.json composition, downsized: WAN
    frames ride the batch dim into USDU, cross the dynamic threshold, and
    the frames come back image-parallel via a real worker server."""
    from comfyui_distributed_amd.utils import constants

    monkeypatch.setattr(constants, "COLLECTOR_SLICE_TIMEOUT", 0.1)
    monkeypatch.setattr(constants, "JOB_READY_POLL_INTERVAL", 0.2)

    async def go():
        worker_srv = DistributedServer(is_worker=True)
        wc = TestClient(TestServer(worker_srv.build_app()))
        await wc.start_server()
        master_srv = DistributedServer()
        previews: list = []
        master_srv.executor.context["preview_images"] = previews
        mc = TestClient(TestServer(master_srv.build_app()))
        await mc.start_server()

        cfg = load_config()
        cfg["workers"] = [{
            "id": "w1", "name": "w", "host": "127.0.0.1",
            "port": wc.server.port, "cuda_device": 0, "enabled": True,
            "type": "remote",
        }]
        cfg["master"]["host"] = "127.0.0.1"
        cfg["master"]["port"] = mc.server.port
        save_config(cfg)

        prompt = {
            "1": {"class_type": "CheckpointLoader",
                  "inputs": {"ckpt_name": "wan_tiny"}},
            "2": {"class_type": "CLIPTextEncode",
                  "inputs": {"text": "clip", "clip": ["1", 1]}},
            "3": {"class_type": "WanVideoGenerate", "inputs": {
                "model": ["1", 0], "positive": ["2", 0], "seed": 5,
                "steps": 1, "cfg": 1.0, "width": 32, "height": 32,
                "frames": 5}},
            "4": {"class_type": "CheckpointLoader",
                  "inputs": {"ckpt_name": "tiny"}},
            "5": {"class_type": "CLIPTextEncode",
                  "inputs": {"text": "sharp", "clip": ["4", 1]}},
            "6": {"class_type": "CLIPTextEncode",
                  "inputs": {"text": "", "clip": ["4", 1]}},
            "7": {"class_type": "UltimateSDUpscaleDistributed", "inputs": {
                "upscaled_image": ["3", 0], "model": ["4", 0],
                "positive": ["5", 0], "negative": ["6", 0], "vae": ["4", 2],
                "seed": 3, "steps": 1, "cfg": 1.0, "sampler_name": "euler",
                "scheduler": "karras", "denoise": 0.4, "tile_width": 16,
                "tile_height": 16, "padding": 16, "mask_blur": 2,
                "force_uniform_tiles": True, "tiled_decode": False,
                "dynamic_threshold": 2}},
            "8": {"class_type": "DistributedCollector",
                  "inputs": {"images": ["7", 0], "load_balance": False}},
            "9": {"class_type": "PreviewImage", "inputs": {"images": ["8", 0]}},
        }
        r = await mc.post("/distributed/queue", json={
            "prompt": prompt, "client_id": "vu", "enabled_worker_ids": ["w1"]})
        assert r.status == 200

        for _ in range(600):
            if previews:
                break
            await asyncio.sleep(0.25)
        from comfyui_distributed_amd.server.network import close_client_session

        await close_client_session()
        await wc.close()
        await mc.close()
        return previews

    previews = asyncio.run(go())
    assert previews, "video upscale pipeline never completed"
    frames = previews[0]
    assert frames.shape == (5, 32, 32, 3)
    assert torch.isfinite(frames).all()


@pytest.mark.timeout(120)
def test_interrupt_aborts_running_usdu(tmp_config, monkeypatch):
    """POST /interrupt mid-job: the master's collection loop aborts, the
    tile job is cleaned up (finally path), and no preview is produced."""
    from comfyui_distributed_amd.utils import constants

    monkeypatch.setattr(constants, "COLLECTOR_SLICE_TIMEOUT", 0.1)

    async def go():
        master_srv = DistributedServer()
        previews: list = []
        master_srv.executor.context["preview_images"] = previews
        mc = TestClient(TestServer(master_srv.build_app()))
        await mc.start_server()

        cfg = load_config()
        cfg["workers"] = []
        cfg["master"]["host"] = "127.0.0.1"
        cfg["master"]["port"] = mc.server.port
        save_config(cfg)

        # a deliberately slow local job: 64 tiles x 4 steps on the tiny
        # stack keeps the executor busy long enough to interrupt
        prompt = {
            "1": {"class_type": "CheckpointLoader", "inputs": {"ckpt_name": "tiny"}},
            "2": {"class_type": "CLIPTextEncode",
                  "inputs": {"text": "detail", "clip": ["1", 1]}},
            "3": {"class_type": "CLIPTextEncode",
                  "inputs": {"text": "", "clip": ["1", 1]}},
            "4": {"class_type": "LoadImage",
                  "inputs": {"image": "synthetic:128x128"}},
            "5": {"class_type": "UltimateSDUpscaleDistributed", "inputs": {
                "upscaled_image": ["4", 0], "model": ["1", 0],
                "positive": ["2", 0], "negative": ["3", 0], "vae": ["1", 2],
                "seed": 3, "steps": 4, "cfg": 2.0, "sampler_name": "euler",
                "scheduler": "karras", "denoise": 0.5, "tile_width": 16,
                "tile_height": 16, "padding": 16, "mask_blur": 2,
                "force_uniform_tiles": True, "tiled_decode": False}},
            "6": {"class_type": "PreviewImage", "inputs": {"images": ["5", 0]}},
        }
        r = await mc.post("/prompt", json={"prompt": prompt, "client_id": "i"})
        assert r.status == 200
        pid = (await r.json())["prompt_id"]
        await asyncio.sleep(0.8)  # let the job start chewing tiles
        r = await mc.post("/interrupt")
        assert r.status == 200
        # the prompt must finish (as an error) promptly, not run all tiles
        done = False
        for _ in range(200):
            body = await (await mc.get(f"/history/{pid}")).json()
            if body:
                done = True
                break
            await asyncio.sleep(0.1)
        assert done, "interrupted prompt never resolved"
        assert body[pid]["status"]["completed"] is False
        assert previews == []
        from comfyui_distributed_amd.nodes.runtime import get_runtime

        get_runtime().clear_interrupt()
        await mc.close()
        return True

    assert asyncio.run(go())
