"""Collector node master collect-loop tests (async, in-memory — the
reference drives job_complete + queue state the same way,
tests/api/test_distributed_queue.py)."""

import asyncio
import json

import pytest
import torch

from comfyui_distributed_amd.nodes.collector import (
    DistributedCollectorNode,
    decode_job_complete_envelope,
)
from comfyui_distributed_amd.nodes.runtime import NodeRuntime, set_runtime
from comfyui_distributed_amd.utils.audio import encode_audio_payload
from comfyui_distributed_amd.utils.image import encode_png_base64


@pytest.fixture(autouse=True)
def fresh_runtime():
    set_runtime(None)
    yield
    set_runtime(None)


def make_item(worker_id, idx, is_last, value=0.5, audio=None):
    return {
        "worker_id": worker_id,
        "image_index": idx,
        "is_last": is_last,
        "tensor": torch.full((1, 4, 4, 3), value),
        "audio": audio,
    }


def test_collect_reorders_master_then_workers():
    rt = NodeRuntime()
    set_runtime(rt)
    node = DistributedCollectorNode()

    async def go():
        q = await rt.job_state.ensure_queue("j")
        # worker results arrive out of order
        await q.put(make_item("w2", 0, True, value=0.3))
        await q.put(make_item("w1", 1, False, value=0.2))
        await q.put(make_item("w1", 0, True, value=0.1))
        local = torch.full((2, 4, 4, 3), 0.9)
        images, audio = await node.collect_on_master(
            local, None, "j", json.dumps(["w1", "w2"]), False
        )
        return images

    images = asyncio.run(go())
    assert images.shape == (5, 4, 4, 3)
    # master batch first, then w1 (by image index), then w2
    vals = [round(images[i].mean().item(), 1) for i in range(5)]
    assert vals == [0.9, 0.9, 0.1, 0.2, 0.3]


def test_collect_timeout_drops_straggler(monkeypatch):
    from comfyui_distributed_amd.utils import constants

    monkeypatch.setattr(constants, "HEARTBEAT_TIMEOUT", 0.5)
    monkeypatch.setattr(constants, "COLLECTOR_SLICE_TIMEOUT", 0.05)
    rt = NodeRuntime()  # probe_worker returns None -> no grace
    set_runtime(rt)
    node = DistributedCollectorNode()

    async def go():
        q = await rt.job_state.ensure_queue("j2")
        await q.put(make_item("w1", 0, True, value=0.1))
        # w2 never reports
        local = torch.full((1, 4, 4, 3), 0.9)
        images, _ = await node.collect_on_master(
            local, None, "j2", json.dumps(["w1", "w2"]), False
        )
        return images

    images = asyncio.run(go())
    assert images.shape == (2, 4, 4, 3)  # master + w1 only


def test_collect_busy_probe_gets_grace_then_delivers(monkeypatch):
    from comfyui_distributed_amd.utils import constants

    monkeypatch.setattr(constants, "HEARTBEAT_TIMEOUT", 0.2)
    monkeypatch.setattr(constants, "COLLECTOR_SLICE_TIMEOUT", 0.05)
    rt = NodeRuntime()
    probes = {"n": 0}

    async def busy_probe(wid):
        probes["n"] += 1
        return {"exec_info": {"queue_remaining": 1}}

    rt.probe_worker = busy_probe
    set_runtime(rt)
    node = DistributedCollectorNode()

    async def go():
        q = await rt.job_state.ensure_queue("j3")

        async def late_delivery():
            await asyncio.sleep(0.6)
            await q.put(make_item("w1", 0, True, value=0.4))

        task = asyncio.create_task(late_delivery())
        local = torch.full((1, 4, 4, 3), 0.9)
        images, _ = await node.collect_on_master(
            local, None, "j3", json.dumps(["w1"]), False
        )
        await task
        return images

    images = asyncio.run(go())
    assert images.shape == (2, 4, 4, 3)
    assert probes["n"] >= 1  # straggler was probed and granted grace


def test_collect_audio_combined():
    rt = NodeRuntime()
    set_runtime(rt)
    node = DistributedCollectorNode()
    payload = encode_audio_payload(
        {"waveform": torch.ones(1, 2, 10), "sample_rate": 8000}
    )

    async def go():
        q = await rt.job_state.ensure_queue("j4")
        await q.put(make_item("w1", 0, True, audio=payload))
        local_audio = {"waveform": torch.zeros(1, 2, 5), "sample_rate": 8000}
        images, audio = await node.collect_on_master(
            torch.zeros(1, 4, 4, 3), local_audio, "j4", json.dumps(["w1"]), False
        )
        return audio

    audio = asyncio.run(go())
    assert audio["waveform"].shape == (1, 2, 15)


def test_delegate_only_excludes_local():
    rt = NodeRuntime()
    set_runtime(rt)
    node = DistributedCollectorNode()

    async def go():
        q = await rt.job_state.ensure_queue("j5")
        await q.put(make_item("w1", 0, True, value=0.7))
        placeholder = torch.zeros(0, 64, 64, 3)
        images, _ = await node.collect_on_master(
            placeholder, None, "j5", json.dumps(["w1"]), True
        )
        return images

    images = asyncio.run(go())
    assert images.shape == (1, 4, 4, 3)


def test_envelope_validation():
    with pytest.raises(ValueError):
        decode_job_complete_envelope({"job_id": "x"})
    img = torch.rand(1, 4, 4, 3)
    item = decode_job_complete_envelope({
        "job_id": "x", "worker_id": "w", "batch_idx": 2,
        "image": encode_png_base64(img), "is_last": True,
    })
    assert item["image_index"] == 2 and item["tensor"].shape == (1, 4, 4, 3)


def test_pass_through_returns_unchanged():
    node = DistributedCollectorNode()
    imgs = torch.rand(2, 4, 4, 3)
    out_imgs, out_audio = node.run(imgs, multi_job_id="j", pass_through=True)
    assert out_imgs is imgs


def test_interrupt_aborts_collect_and_clears():
    from comfyui_distributed_amd.utils import constants

    rt = NodeRuntime()
    set_runtime(rt)
    node = DistributedCollectorNode()

    async def go():
        await rt.job_state.ensure_queue("ji")
        rt.interrupt()
        with pytest.raises(InterruptedError):
            await node.collect_on_master(
                torch.zeros(1, 4, 4, 3), None, "ji", json.dumps(["w1"]), False
            )
        # queue cleaned up on abort
        assert "ji" not in rt.job_state.pending_jobs
        rt.clear_interrupt()
        rt.throw_if_interrupted()  # no raise after clear

    asyncio.run(go())
