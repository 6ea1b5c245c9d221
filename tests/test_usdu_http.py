"""HTTP-mode USDU flow tests (single process; the worker flow talks to an
in-memory fake master runtime — the reference tests its modes the same way,
driving endpoints/state machines without sockets)."""

import asyncio
import time

import pytest
import torch

from comfyui_distributed_amd.engine.usdu import USDUParams, process_single_gpu
from comfyui_distributed_amd.models import create_diffusion_stack
from comfyui_distributed_amd.nodes.runtime import NodeRuntime, set_runtime
from comfyui_distributed_amd.server import usdu_http
from comfyui_distributed_amd.utils import constants
from comfyui_distributed_amd.server.job_state import TileJobState


@pytest.fixture(autouse=True)
def fresh_runtime():
    set_runtime(None)
    yield
    set_runtime(None)


def tiny_setup():
    stack = create_diffusion_stack("tiny", seed=7)
    cond = stack.make_conditioning(0)
    p = USDUParams(seed=3, steps=2, cfg=1.0, denoise=0.5, tile_width=16,
                   tile_height=16, padding=16, mask_blur=2, tile_batch=2)
    g = torch.Generator().manual_seed(42)
    img = torch.rand(1, 32, 32, 3, generator=g)
    return stack, cond, p, img


def test_master_static_no_workers_matches_single_gpu():
    stack, cond, p, img = tiny_setup()
    out = usdu_http.run_usdu_role(
        mode="static", params=p, stack=stack, cond=cond, uncond=None,
        image=img, job_id="j1", is_worker=False, master_url="",
        enabled_workers=[], worker_id="",
    )
    ref = process_single_gpu(stack, cond, None, p, img)
    assert torch.allclose(out, ref, atol=3e-5)


def test_master_dynamic_no_workers_matches_single_gpu():
    stack, cond, p, img = tiny_setup()
    img2 = torch.cat([img, img.flip(1)], dim=0)  # batch of 2
    out = usdu_http.run_usdu_role(
        mode="dynamic", params=p, stack=stack, cond=cond, uncond=None,
        image=img2, job_id="j2", is_worker=False, master_url="",
        enabled_workers=[], worker_id="",
    )
    ref = process_single_gpu(stack, cond, None, p, img2)
    assert torch.allclose(out, ref, atol=3e-5)


class FakeMasterRuntime(NodeRuntime):
    """Routes the worker's HTTP calls to an in-memory master job."""

    def __init__(self, n_tiles):
        super().__init__()
        self.pending = list(range(n_tiles))
        self.submitted = []
        self.heartbeats = 0
        self.finished = False

    async def post_json(self, url, payload, timeout=60.0):
        if url.endswith("/distributed/job_status"):
            return {"ready": True}
        if url.endswith("/distributed/request_image"):
            if self.pending:
                idx = self.pending.pop(0)
                return {"tile_idx": idx, "estimated_remaining": len(self.pending)}
            return {"tile_idx": None, "estimated_remaining": 0}
        if url.endswith("/distributed/heartbeat"):
            self.heartbeats += 1
            return {"status": "ok"}
        if url.endswith("/distributed/submit_tiles"):
            self.submitted.extend(payload["tiles"])
            if payload.get("is_last"):
                self.finished = True
            return {"status": "ok"}
        raise AssertionError(f"unexpected url {url}")


def test_worker_static_protocol():
    stack, cond, p, img = tiny_setup()
    rt = FakeMasterRuntime(n_tiles=4)
    set_runtime(rt)
    usdu_http.run_usdu_role(
        mode="static", params=p, stack=stack, cond=cond, uncond=None,
        image=img, job_id="j3", is_worker=True,
        master_url="http://master:8188", enabled_workers=["w1"],
        worker_id="w1",
    )
    assert rt.finished
    assert rt.heartbeats == 4
    assert sorted(t["tile_idx"] for t in rt.submitted) == [0, 1, 2, 3]
    assert all("image" in t for t in rt.submitted)


def test_timeout_requeue_state_machine():
    async def go():
        job = TileJobState(job_id="x", total_tasks=3, batch_size=1)
        job.worker_status["w1"] = time.time() - 120
        job.assigned_to_workers[1] = "w1"
        job.assigned_to_workers[2] = "w1"
        job.completed_tasks[(2, 0)] = True  # tile 2 finished
        await usdu_http.check_and_requeue_timed_out_workers(job, timeout=60)
        assert "w1" not in job.worker_status
        requeued = []
        while not job.pending_tasks.empty():
            requeued.append(job.pending_tasks.get_nowait())
        return requeued

    requeued = asyncio.run(go())
    assert requeued == [1]  # only the incomplete assigned task


def test_timeout_grace_for_busy_worker():
    async def go():
        rt = NodeRuntime()

        async def busy_probe(wid):
            return {"exec_info": {"queue_remaining": 3}}

        rt.probe_worker = busy_probe
        set_runtime(rt)
        job = TileJobState(job_id="x", total_tasks=2, batch_size=1)
        job.worker_status["w1"] = time.time() - 120
        job.assigned_to_workers[0] = "w1"
        await usdu_http.check_and_requeue_timed_out_workers(job, timeout=60)
        assert "w1" in job.worker_status  # grace refreshed
        assert job.pending_tasks.empty()

    asyncio.run(go())


class GoneJobRuntime(NodeRuntime):
    """Master whose job is already cleaned up: request_image 404s."""

    def __init__(self):
        super().__init__()
        self.completions = []

    async def post_json(self, url, payload, timeout=60.0):
        if url.endswith("/distributed/job_status"):
            return {"ready": True}
        if url.endswith("/distributed/request_image"):
            err = RuntimeError("404, message='unknown job'")
            err.status = 404
            raise err
        if url.endswith("/distributed/submit_tiles"):
            self.completions.append(payload)
            return {"status": "ok"}
        if url.endswith("/distributed/heartbeat"):
            return {"status": "ok"}
        raise AssertionError(f"unexpected url {url}")


def test_worker_static_graceful_when_job_gone(monkeypatch):
    """A worker whose master finished and cleaned the job must exit
    gracefully AND still send the empty-batch completion signal (reference
    test_static_mode.py: flush_empty_final_still_sends_completion_signal +
    the 404-retry in worker_comms.py:124-188)."""
    monkeypatch.setattr(constants, "JOB_READY_POLL_ATTEMPTS", 1)
    stack, cond, p, img = tiny_setup()
    rt = GoneJobRuntime()
    set_runtime(rt)
    usdu_http.run_usdu_role(
        mode="static", params=p, stack=stack, cond=cond, uncond=None,
        image=img, job_id="gone", is_worker=True,
        master_url="http://master:8188", enabled_workers=["w1"],
        worker_id="w1",
    )
    assert len(rt.completions) == 1
    final = rt.completions[0]
    assert final["is_last"] is True and final["tiles"] == []


def test_master_static_cleans_job_on_failure(monkeypatch):
    """A master whose sampling raises must still remove the job entry
    (no stale tile jobs after interrupt/error)."""
    stack, cond, p, img = tiny_setup()
    rt = NodeRuntime()
    set_runtime(rt)

    def boom(*a, **k):
        raise RuntimeError("sampler exploded")

    monkeypatch.setattr(usdu_http, "sample_tiles", boom)
    with pytest.raises(RuntimeError, match="exploded"):
        usdu_http.run_usdu_role(
            mode="static", params=p, stack=stack, cond=cond, uncond=None,
            image=img, job_id="doomed", is_worker=False, master_url="",
            enabled_workers=[], worker_id="",
        )
    assert asyncio.run(rt.job_state.get_tile_job("doomed")) is None


def test_requeue_scales_to_1024_tiles():
    """SDXL-8K scale guard (BASELINE config 4 has 64 tiles; an 8K canvas
    with 256px tiles has 1024): the HTTP requeue path is in-memory dicts,
    so a dead worker holding hundreds of assignments requeues in
    milliseconds, not store-RTT-bound seconds (the round-1 advisor's
    O(total) concern on the RCCL path's assigned_incomplete)."""
    async def go():
        job = TileJobState(job_id="big", total_tasks=1024, batch_size=2)
        job.worker_status["w1"] = time.time() - 120
        for t in range(512):
            job.assigned_to_workers[t] = "w1"
            if t % 2 == 0:  # half its tiles fully done
                job.completed_tasks[(t, 0)] = True
                job.completed_tasks[(t, 1)] = True
        t0 = time.monotonic()
        await usdu_http.check_and_requeue_timed_out_workers(job, timeout=60)
        elapsed = time.monotonic() - t0
        requeued = []
        while not job.pending_tasks.empty():
            requeued.append(job.pending_tasks.get_nowait())
        return requeued, elapsed

    requeued, elapsed = asyncio.run(go())
    assert sorted(requeued) == list(range(1, 512, 2))  # odd = incomplete
    assert elapsed < 1.0, f"requeue took {elapsed:.2f}s at 1024-tile scale"
