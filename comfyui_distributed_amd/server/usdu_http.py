"""HTTP-mode USDU flows (remote/cloud workers over the wire protocol).

Reference counterparts: upscale/modes/static.py (tile pull-queue, master
participates, collection phase with heartbeat-driven requeue, leftover
takeover, deterministic blend), upscale/modes/dynamic.py (whole-image
queue), upscale/worker_comms.py (size-aware batched sends with retries),
upscale/job_timeout.py (timeout -> probe -> grace -> requeue).

Intra-node multi-GPU does NOT go through here (parallel/usdu_dist.py is
the RCCL path); this serves workers reached over HTTP.
"""

from __future__ import annotations

import asyncio
import time

import torch

from ..engine.usdu import USDUParams, blend_results, plan_for_image, sample_tiles
from ..nodes.runtime import get_runtime
from ..utils import constants
from ..utils.async_bridge import run_async_in_server_loop
from ..utils.config import get_worker_timeout_seconds
from ..utils.image import decode_png_base64, encode_png_base64
from ..utils.logging import debug_log, log


def run_usdu_role(*, mode, params, stack, cond, uncond, image, job_id,
                  is_worker, master_url, enabled_workers, worker_id):
    if is_worker:
        if mode == "dynamic":
            return _worker_dynamic(params, stack, cond, uncond, image, job_id,
                                   master_url, worker_id)
        return _worker_static(params, stack, cond, uncond, image, job_id,
                              master_url, worker_id)
    if mode == "dynamic":
        return run_async_in_server_loop(
            _master_dynamic(params, stack, cond, uncond, image, job_id,
                            enabled_workers),
            timeout=None,
        )
    return run_async_in_server_loop(
        _master_static(mode, params, stack, cond, uncond, image, job_id,
                       enabled_workers),
        timeout=None,
    )


# ---------------------------------------------------------------------------
# master
# ---------------------------------------------------------------------------


async def _master_static(mode, params: USDUParams, stack, cond, uncond, image,
                         job_id, enabled_workers):
    rt = get_runtime()
    state = rt.job_state
    canvas = image.to(stack.device, torch.float32).clone().contiguous()
    B, H, W, _ = canvas.shape
    plans = plan_for_image(W, H, params)
    job = await state.init_static_job(job_id, len(plans), batch_size=B)
    loop = asyncio.get_running_loop()
    try:
        return await _master_static_body(mode, params, stack, cond, uncond,
                                         canvas, B, plans, job, job_id, loop)
    finally:
        # interrupt/exception included: never leak the job entry
        await state.cleanup_job(job_id)


async def _master_static_body(mode, params, stack, cond, uncond, canvas, B,
                              plans, job, job_id, loop):
    rt = get_runtime()
    results: dict[tuple[int, int], torch.Tensor] = {}

    # -- master participates: pull tiles from its own queue ---------------
    while True:
        try:
            idx = job.pending_tasks.get_nowait()
        except asyncio.QueueEmpty:
            break
        job.assigned_to_workers[idx] = "master"
        res = await loop.run_in_executor(
            None, lambda i=idx: sample_tiles(stack, cond, uncond, params,
                                             canvas, plans, [i])
        )
        for key, tile in res.items():
            results[key] = tile
            job.completed_tasks[key] = True

    # -- collection phase: drain workers, requeue on timeout ---------------
    timeout = get_worker_timeout_seconds()
    last_check = time.monotonic()
    while len(results) < len(plans) * B:
        rt.throw_if_interrupted()
        # adopt any tasks requeued by the monitor
        try:
            idx = job.pending_tasks.get_nowait()
            job.assigned_to_workers[idx] = "master"
            res = await loop.run_in_executor(
                None, lambda i=idx: sample_tiles(stack, cond, uncond, params,
                                                 canvas, plans, [i])
            )
            for key, tile in res.items():
                results[key] = tile
                job.completed_tasks[key] = True
            continue
        except asyncio.QueueEmpty:
            pass
        try:
            item = await asyncio.wait_for(job.results.get(),
                                          constants.COLLECTOR_SLICE_TIMEOUT)
            t, b = int(item["tile_idx"]), int(item["batch_idx"])
            results[(t, b)] = item["tensor"].to(canvas.device, torch.float32)
            job.completed_tasks[(t, b)] = True
            continue
        except asyncio.TimeoutError:
            pass
        now = time.monotonic()
        if now - last_check >= constants.HEARTBEAT_INTERVAL:
            last_check = now
            await check_and_requeue_timed_out_workers(job, timeout)
            if not job.worker_status and job.pending_tasks.empty():
                # no live workers, nothing requeued and tiles still missing:
                # requeue every incomplete assigned task for master takeover
                missing = _incomplete_task_ids(job, plans, B)
                if missing:
                    for m in missing:
                        job.pending_tasks.put_nowait(m)
                    log(f"usdu {job_id}: master takeover of {len(missing)} tiles")
                else:
                    break

    blend_results(canvas, results, plans, params)
    return canvas


def _incomplete_task_ids(job, plans, batch: int) -> list[int]:
    """Tile ids with any missing (tile, batch) result — the batched-static
    per-tile-id completeness check (reference job_timeout.py:126-145)."""
    missing = []
    for t in range(len(plans)):
        if any((t, b) not in job.completed_tasks for b in range(batch)):
            missing.append(t)
    return missing


async def check_and_requeue_timed_out_workers(job, timeout: float):
    """Reference job_timeout.py:17-150: snapshot suspects under the loop,
    probe outside, grace when busy, requeue + drop otherwise."""
    rt = get_runtime()
    now = time.time()
    suspects = [wid for wid, ts in job.worker_status.items()
                if now - ts > timeout]
    for wid in suspects:
        info = await rt.probe_worker(wid)
        if info and info.get("exec_info", {}).get("queue_remaining", 0):
            job.worker_status[wid] = time.time()  # grace
            debug_log(f"usdu: worker {wid} busy — grace")
            continue
        requeued = []
        for task, owner in list(job.assigned_to_workers.items()):
            if owner != wid:
                continue
            batch = getattr(job, "batch_size", 1)
            if any((task, b) not in job.completed_tasks for b in range(batch)):
                job.pending_tasks.put_nowait(task)
                requeued.append(task)
            del job.assigned_to_workers[task]
        job.worker_status.pop(wid, None)
        log(f"usdu: worker {wid} timed out — requeued {requeued}")


async def _master_dynamic(params: USDUParams, stack, cond, uncond, image,
                          job_id, enabled_workers):
    """Whole-image parallelism for big batches (reference
    upscale/modes/dynamic.py:22-211): master participates by pulling image
    indices from its own queue; worker results arrive as full images."""
    rt = get_runtime()
    state = rt.job_state
    canvas = image.to(stack.device, torch.float32).clone().contiguous()
    B, H, W, _ = canvas.shape
    plans = plan_for_image(W, H, params)
    job = await state.init_dynamic_job(job_id, B)
    loop = asyncio.get_running_loop()
    try:
        return await _master_dynamic_body(params, stack, cond, uncond,
                                          canvas, B, plans, job, loop)
    finally:
        await state.cleanup_job(job_id)


async def _master_dynamic_body(params, stack, cond, uncond, canvas, B, plans,
                               job, loop):
    rt = get_runtime()
    done_images: dict[int, torch.Tensor] = {}
    timeout = get_worker_timeout_seconds()
    last_check = time.monotonic()

    def process_whole_image(b: int) -> torch.Tensor:
        one = canvas[b : b + 1].clone().contiguous()
        results = sample_tiles(stack, cond, uncond, params, one, plans,
                               list(range(len(plans))), batch_offset=b)
        blend_results(one, results, plans, params)
        return one

    while len(done_images) < B:
        rt.throw_if_interrupted()
        try:
            idx = job.pending_images.get_nowait()
            job.assigned_to_workers[idx] = "master"
            done_images[idx] = await loop.run_in_executor(
                None, process_whole_image, idx
            )
            job.completed_images[idx] = True
            continue
        except asyncio.QueueEmpty:
            pass
        try:
            item = await asyncio.wait_for(job.results.get(),
                                          constants.COLLECTOR_SLICE_TIMEOUT)
            i = int(item["image_idx"])
            done_images[i] = item["tensor"].to(canvas.device, torch.float32)
            job.completed_images[i] = True
            continue
        except asyncio.TimeoutError:
            pass
        now = time.monotonic()
        if now - last_check >= constants.HEARTBEAT_INTERVAL:
            last_check = now
            await _requeue_dynamic_timeouts(job, timeout)
            if not job.worker_status and job.pending_images.empty():
                missing = [i for i in range(B) if i not in done_images]
                for m in missing:
                    job.pending_images.put_nowait(m)
                if not missing:
                    break
    for i, img in done_images.items():
        canvas[i : i + 1] = img
    return canvas


async def _requeue_dynamic_timeouts(job, timeout: float):
    rt = get_runtime()
    now = time.time()
    for wid in [w for w, ts in job.worker_status.items() if now - ts > timeout]:
        info = await rt.probe_worker(wid)
        if info and info.get("exec_info", {}).get("queue_remaining", 0):
            job.worker_status[wid] = time.time()
            continue
        for task, owner in list(job.assigned_to_workers.items()):
            if owner == wid and task not in job.completed_images:
                job.pending_images.put_nowait(task)
                del job.assigned_to_workers[task]
        job.worker_status.pop(wid, None)
        log(f"usdu dynamic: worker {wid} timed out — dropped")


def _worker_dynamic(params: USDUParams, stack, cond, uncond, image, job_id,
                    master_url, worker_id):
    """Reference dynamic.py:213-313: pull image indices, process every tile
    locally, POST the finished full image."""
    rt = get_runtime()
    canvas = image.to(stack.device, torch.float32).clone().contiguous()
    B, H, W, _ = canvas.shape
    plans = plan_for_image(W, H, params)
    for _ in range(constants.JOB_READY_POLL_ATTEMPTS):
        status = run_async_in_server_loop(
            rt.post_json(f"{master_url}/distributed/job_status",
                         {"job_id": job_id}), timeout=30.0)
        if status.get("ready"):
            break
        time.sleep(constants.JOB_READY_POLL_INTERVAL)
    while True:
        resp = _request_work(rt, master_url, job_id, worker_id)
        idx = resp.get("image_idx") if resp is not None else None
        if idx is None:
            break
        idx = int(idx)
        one = canvas[idx : idx + 1].clone().contiguous()
        results = sample_tiles(stack, cond, uncond, params, one, plans,
                               list(range(len(plans))), batch_offset=idx)
        blend_results(one, results, plans, params)
        run_async_in_server_loop(
            rt.post_json(f"{master_url}/distributed/heartbeat",
                         {"job_id": job_id, "worker_id": worker_id}),
            timeout=30.0,
        )
        _post_with_retry(rt, f"{master_url}/distributed/submit_image", {
            "job_id": job_id, "worker_id": worker_id, "image_idx": idx,
            "image": encode_png_base64(one[0:1].cpu()),
            "is_last": int(resp.get("estimated_remaining", 0)) == 0,
        })
    return None


# ---------------------------------------------------------------------------
# worker
# ---------------------------------------------------------------------------


def _worker_static(params: USDUParams, stack, cond, uncond, image, job_id,
                   master_url, worker_id):
    rt = get_runtime()
    canvas = image.to(stack.device, torch.float32).clone().contiguous()
    B, H, W, _ = canvas.shape
    plans = plan_for_image(W, H, params)

    # job-ready poll (reference static.py:33-47)
    for _ in range(constants.JOB_READY_POLL_ATTEMPTS):
        status = run_async_in_server_loop(
            rt.post_json(f"{master_url}/distributed/job_status",
                         {"job_id": job_id}), timeout=30.0)
        if status.get("ready"):
            break
        time.sleep(constants.JOB_READY_POLL_INTERVAL)

    pending_payloads: list[dict] = []
    bytes_pending = 0

    def flush(is_last: bool):
        nonlocal pending_payloads, bytes_pending
        if not pending_payloads and not is_last:
            return
        body = {"job_id": job_id, "worker_id": worker_id,
                "tiles": pending_payloads, "is_last": is_last}
        _post_with_retry(rt, f"{master_url}/distributed/submit_tiles", body)
        pending_payloads = []
        bytes_pending = 0

    processed = 0
    while True:
        resp = _request_work(rt, master_url, job_id, worker_id)
        idx = resp.get("tile_idx") if resp is not None else None
        if idx is None:
            break
        res = sample_tiles(stack, cond, uncond, params, canvas, plans, [int(idx)])
        for (t, b), tile in sorted(res.items()):
            png = encode_png_base64(tile[0].cpu())
            pending_payloads.append({"tile_idx": t, "batch_idx": b, "image": png})
            bytes_pending += len(png)
        processed += 1
        run_async_in_server_loop(
            rt.post_json(f"{master_url}/distributed/heartbeat",
                         {"job_id": job_id, "worker_id": worker_id}),
            timeout=30.0,
        )
        if (len(pending_payloads) >= constants.MAX_BATCH
                or bytes_pending >= constants.MAX_PAYLOAD_SIZE - constants.PAYLOAD_HEADROOM):
            flush(False)
    flush(True)
    debug_log(f"usdu worker {worker_id}: processed {processed} tiles")
    return None


def _request_work(rt, master_url: str, job_id, worker_id) -> dict | None:
    """Pull the next work item. A 404 means the job is gone — either not
    yet initialized (transient, retried briefly) or already completed and
    cleaned up by the master; both end with a graceful None instead of
    crashing the worker's prompt (reference worker_comms.py:124-188 does a
    404-retry loop for the same race)."""
    for attempt in range(3):
        try:
            return run_async_in_server_loop(
                rt.post_json(f"{master_url}/distributed/request_image",
                             {"job_id": job_id, "worker_id": worker_id}),
                timeout=60.0,
            )
        except Exception as exc:  # noqa: BLE001
            if getattr(exc, "status", None) == 404:
                if attempt < 2:
                    time.sleep(0.2)
                    continue
                return None
            raise
    return None


def _post_with_retry(rt, url: str, body: dict):
    delay = 0.5
    last_exc = None
    for _ in range(constants.SEND_RETRY_ATTEMPTS):
        try:
            return run_async_in_server_loop(rt.post_json(url, body), timeout=120.0)
        except Exception as exc:  # noqa: BLE001
            last_exc = exc
            time.sleep(delay)
            delay *= 2
    raise RuntimeError(f"send to {url} failed after retries: {last_exc}")


def decode_tile_submission(body: dict) -> list[dict]:
    """submit_tiles body -> result-queue items (tensor decoded)."""
    out = []
    for tile in body.get("tiles", []):
        out.append({
            "tile_idx": int(tile["tile_idx"]),
            "batch_idx": int(tile.get("batch_idx", 0)),
            "tensor": decode_png_base64(tile["image"]),
            "worker_id": str(body.get("worker_id", "")),
        })
    return out
