"""Full master<->worker HTTP integration: one process, two real aiohttp
servers. A seed-parallel workflow is queued on the master; the master
prunes/overrides/dispatches the worker prompt over real HTTP, both sides
execute the graph (tiny model), the worker's collector POSTs base64-PNG
envelopes to the master's /distributed/job_complete, and the master's
collector combines master+worker batches.

The reference has no test like this (its suite stubs all
transport); this is the end-to-end wire check.
"""

import asyncio
import json

import pytest
import torch
from aiohttp.test_utils import TestClient, TestServer

from comfyui_distributed_amd.server.app import DistributedServer
from comfyui_distributed_amd.utils.config import load_config, save_config


@pytest.mark.timeout(120)
def test_seed_parallel_roundtrip_over_http(tmp_config, monkeypatch):
    from comfyui_distributed_amd.utils import constants

    monkeypatch.setattr(constants, "COLLECTOR_SLICE_TIMEOUT", 0.1)

    async def go():
        # worker first, master second: the master's NodeRuntime must be the
        # process-global one (its collector drains its own job_state)
        worker_srv = DistributedServer(is_worker=True)
        worker_client = TestClient(TestServer(worker_srv.build_app()))
        await worker_client.start_server()

        master_srv = DistributedServer()
        master_previews: list = []
        master_srv.executor.context["preview_images"] = master_previews
        master_srv.executor.context["saved_images"] = []
        master_client = TestClient(TestServer(master_srv.build_app()))
        await master_client.start_server()

        # register the worker in config with its real ephemeral port
        cfg = load_config()
        cfg["workers"] = [{
            "id": "w1", "name": "worker1", "host": "127.0.0.1",
            "port": worker_client.server.port, "cuda_device": 0,
            "enabled": True, "type": "remote",  # remote => real HTTP callbacks
        }]
        cfg["master"]["host"] = "127.0.0.1"
        cfg["master"]["port"] = master_client.server.port
        save_config(cfg)

        prompt = {
            "1": {"class_type": "CheckpointLoader", "inputs": {"ckpt_name": "tiny"}},
            "2": {"class_type": "CLIPTextEncode",
                  "inputs": {"text": "cat", "clip": ["1", 1]}},
            "3": {"class_type": "CLIPTextEncode",
                  "inputs": {"text": "", "clip": ["1", 1]}},
            "4": {"class_type": "DistributedSeed", "inputs": {"seed": 5}},
            "5": {"class_type": "EmptyLatentImage",
                  "inputs": {"width": 16, "height": 16, "batch_size": 1}},
            "6": {"class_type": "KSampler", "inputs": {
                "model": ["1", 0], "seed": ["4", 0], "steps": 1, "cfg": 1.0,
                "sampler_name": "euler", "scheduler": "karras",
                "positive": ["2", 0], "negative": ["3", 0],
                "latent_image": ["5", 0], "denoise": 1.0}},
            "7": {"class_type": "VAEDecode",
                  "inputs": {"samples": ["6", 0], "vae": ["1", 2]}},
            "8": {"class_type": "DistributedCollector",
                  "inputs": {"images": ["7", 0], "load_balance": False}},
            "9": {"class_type": "PreviewImage", "inputs": {"images": ["8", 0]}},
        }

        resp = await master_client.post("/distributed/queue", json={
            "prompt": prompt, "client_id": "it", "enabled_worker_ids": ["w1"],
        })
        assert resp.status == 200
        body = await resp.json()
        assert "w1" in body["participants"] and "master" in body["participants"]

        # wait for both executors to drain
        for _ in range(400):
            if (master_srv.prompt_queue.qsize() == 0 and not master_srv.executing
                    and worker_srv.prompt_queue.qsize() == 0
                    and not worker_srv.executing and master_previews):
                break
            await asyncio.sleep(0.25)

        from comfyui_distributed_amd.server.network import close_client_session

        await close_client_session()
        await worker_client.close()
        await master_client.close()
        return master_previews

    previews = asyncio.run(go())
    assert previews, "master collector never produced output"
    combined = previews[0]
    # master image + worker image, combined on the master
    assert combined.shape == (2, 16, 16, 3)
    assert torch.isfinite(combined).all()


@pytest.mark.timeout(180)
def test_three_concurrent_collector_jobs(tmp_config, monkeypatch):
    """Three distributed jobs queued back-to-back: per-job collector queues
    must stay isolated (no cross-job image leakage), and the sequential
    prompt loops on both servers must drain all of them."""
    from comfyui_distributed_amd.utils import constants

    monkeypatch.setattr(constants, "COLLECTOR_SLICE_TIMEOUT", 0.2)

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

        def prompt_for(size):
            return {
                "1": {"class_type": "DistributedSeed", "inputs": {"seed": size}},
                "2": {"class_type": "LoadImage",
                      "inputs": {"image": f"synthetic:{size}x{size}"}},
                "3": {"class_type": "DistributedCollector",
                      "inputs": {"images": ["2", 0], "load_balance": False}},
                "4": {"class_type": "PreviewImage", "inputs": {"images": ["3", 0]}},
            }

        sizes = [8, 12, 16]
        for s in sizes:
            r = await mc.post("/distributed/queue", json={
                "prompt": prompt_for(s), "client_id": f"c{s}",
                "enabled_worker_ids": ["w1"]})
            assert r.status == 200

        for _ in range(600):
            if len(previews) >= 3:
                break
            await asyncio.sleep(0.2)
        from comfyui_distributed_amd.server.network import close_client_session

        await close_client_session()
        await wc.close()
        await mc.close()
        return previews

    previews = asyncio.run(go())
    assert len(previews) == 3
    # jobs completed in submission order; each combined batch = 2 ranks of
    # the RIGHT size (no cross-job mixing)
    got = sorted(p.shape[1] for p in previews)
    assert got == [8, 12, 16]
    for p in previews:
        assert p.shape[0] == 2 and torch.isfinite(p).all()


@pytest.mark.timeout(120)
def test_simultaneous_queue_requests(tmp_config, monkeypatch):
    """Two /distributed/queue POSTs in flight at once: orchestration
    (probe, job-id minting, queue pre-create) must interleave safely."""
    from comfyui_distributed_amd.utils import constants

    monkeypatch.setattr(constants, "COLLECTOR_SLICE_TIMEOUT", 0.2)

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

        def prompt_for(seed):
            return {
                "1": {"class_type": "DistributedSeed", "inputs": {"seed": seed}},
                "2": {"class_type": "LoadImage",
                      "inputs": {"image": "synthetic:8x8"}},
                "3": {"class_type": "DistributedCollector",
                      "inputs": {"images": ["2", 0], "load_balance": False}},
                "4": {"class_type": "PreviewImage", "inputs": {"images": ["3", 0]}},
            }

        r1, r2 = await asyncio.gather(
            mc.post("/distributed/queue", json={
                "prompt": prompt_for(1), "client_id": "a",
                "enabled_worker_ids": ["w1"]}),
            mc.post("/distributed/queue", json={
                "prompt": prompt_for(2), "client_id": "b",
                "enabled_worker_ids": ["w1"]}),
        )
        assert r1.status == 200 and r2.status == 200
        j1, j2 = await r1.json(), await r2.json()
        # job ids unique across concurrent requests
        ids1, ids2 = set(j1["job_ids"].values()), set(j2["job_ids"].values())
        assert ids1 and ids2 and not (ids1 & ids2)

        for _ in range(400):
            if len(previews) >= 2:
                break
            await asyncio.sleep(0.2)
        from comfyui_distributed_amd.server.network import close_client_session

        await close_client_session()
        await wc.close()
        await mc.close()
        return previews

    previews = asyncio.run(go())
    assert len(previews) == 2
    for p in previews:
        assert p.shape == (2, 8, 8, 3)


@pytest.mark.timeout(120)
def test_delegate_master_end_to_end(tmp_config, monkeypatch):
    """delegate_master=true over real sockets: the worker generates, the
    master executes only the post-collector subgraph with a placeholder
    image and returns ONLY the worker's batch."""
    from comfyui_distributed_amd.utils import constants

    monkeypatch.setattr(constants, "COLLECTOR_SLICE_TIMEOUT", 0.2)

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
            "1": {"class_type": "DistributedSeed", "inputs": {"seed": 4}},
            "2": {"class_type": "LoadImage",
                  "inputs": {"image": "synthetic:8x8"}},
            "3": {"class_type": "DistributedCollector",
                  "inputs": {"images": ["2", 0], "load_balance": False}},
            "4": {"class_type": "PreviewImage", "inputs": {"images": ["3", 0]}},
        }
        r = await mc.post("/distributed/queue", json={
            "prompt": prompt, "client_id": "dm", "enabled_worker_ids": ["w1"],
            "delegate_master": True})
        assert r.status == 200
        body = await r.json()
        assert body["participants"] == ["w1"]  # master orchestrates only

        for _ in range(300):
            if previews:
                break
            await asyncio.sleep(0.2)
        from comfyui_distributed_amd.server.network import close_client_session

        await close_client_session()
        await wc.close()
        await mc.close()
        return previews

    previews = asyncio.run(go())
    assert previews, "delegate job never completed"
    # only the worker's single image — no master batch
    assert previews[0].shape == (1, 8, 8, 3)
