"""Audio over the collector wire, end-to-end on real sockets: each
participant produces a seed-offset waveform; the master's collector
concatenates them along the samples dim (reference collector.py audio
rides the last envelope; _combine_audio :121-174)."""

import asyncio

import pytest
import torch
from aiohttp.test_utils import TestClient, TestServer

from comfyui_distributed_amd.server.app import DistributedServer
from comfyui_distributed_amd.utils.config import load_config, save_config


@pytest.mark.timeout(120)
def test_seed_parallel_audio_over_http(tmp_config, monkeypatch):
    from comfyui_distributed_amd.utils import constants

    monkeypatch.setattr(constants, "COLLECTOR_SLICE_TIMEOUT", 0.2)

    async def go():
        worker_srv = DistributedServer(is_worker=True)
        wc = TestClient(TestServer(worker_srv.build_app()))
        await wc.start_server()
        master_srv = DistributedServer()
        previews: list = []
        audios: list = []
        master_srv.executor.context["preview_images"] = previews
        master_srv.executor.context["collected_audio"] = audios
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
            "1": {"class_type": "DistributedSeed", "inputs": {"seed": 5}},
            "2": {"class_type": "SyntheticAudio", "inputs": {
                "seed": ["1", 0], "seconds": 0.01, "sample_rate": 8000}},
            "3": {"class_type": "LoadImage",
                  "inputs": {"image": "synthetic:8x8"}},
            "4": {"class_type": "DistributedCollector", "inputs": {
                "images": ["3", 0], "audio": ["2", 0],
                "load_balance": False}},
            "5": {"class_type": "PreviewImage", "inputs": {"images": ["4", 0]}},
            "6": {"class_type": "AudioBatchDivider", "inputs": {
                "audio": ["4", 1], "divide_by": 2}},
        }
        resp = await mc.post("/distributed/queue", json={
            "prompt": prompt, "client_id": "au", "enabled_worker_ids": ["w1"]})
        assert resp.status == 200

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
    assert previews, "audio collector job never completed"
    # both ranks' images present; audio combination is validated at the
    # node level (test_collector_node) — here the job completing over the
    # wire with an audio payload attached is the claim under test
    assert previews[0].shape == (2, 8, 8, 3)


def test_synthetic_audio_node_deterministic():
    from comfyui_distributed_amd.graph.builtin_nodes import SyntheticAudio

    a = SyntheticAudio().generate(seed=3, seconds=0.01, sample_rate=8000)[0]
    b = SyntheticAudio().generate(seed=3, seconds=0.01, sample_rate=8000)[0]
    c = SyntheticAudio().generate(seed=4, seconds=0.01, sample_rate=8000)[0]
    assert a["waveform"].shape == (1, 2, 80)
    assert torch.equal(a["waveform"], b["waveform"])
    assert not torch.allclose(a["waveform"], c["waveform"])
