"""Distributed USDU: tile pull-queue over all ranks, streamed RCCL result
transport, canonical blend on rank 0 — with live fault tolerance.

This is the intra-node re-architecture of the reference's static mode
(upscale/modes/static.py): the HTTP request_image/submit_tiles loop becomes
a TCPStore pull queue + per-chunk p2p sends over xGMI; the master
participates exactly like the reference master does (pulls tiles from its
own queue) while ALSO running the monitor loop — heartbeat age -> grace ->
requeue -> drop (reference upscale/job_timeout.py:17-150) — and finally
takes over any tile whose result never arrived (reference master-takeover,
upscale/modes/static.py:354-363,469-513). Determinism: extraction reads the
original canvas and blending runs in ascending (tile, batch) order on
rank 0, so the result is bit-identical to the single-GPU run regardless of
tile assignment, requeues, or duplicated work after a crash. (On GPU,
"bit-identical" is modulo one source of run-to-run noise that is
independent of assignment: the GroupNorm stats kernel's fp32 atomic
reduction order, ~1e-2 through a bf16 sampler chain — see
tests/test_distributed_gpu.py. The CPU path is exactly deterministic.)
"""

from __future__ import annotations

import threading
import time

import torch

from ..engine.usdu import USDUParams, blend_results, plan_for_image, sample_tiles
from ..utils.logging import debug_log, log
from .dist import DistContext, ResultMailbox, broadcast_tensor
from .tile_queue import TileQueue, TileScheduler


def _pop_chunk(queue: TileQueue, n: int) -> list[int]:
    ids: list[int] = []
    while len(ids) < n:
        idx = queue.pop()
        if idx is None:
            break
        ids.append(idx)
    return ids


def run_distributed_usdu(
    ctx: DistContext,
    store,
    stack,
    cond: dict,
    uncond: dict | None,
    params: USDUParams,
    image: torch.Tensor | None,
    job_id: str = "usdu",
    broadcast_input: bool = False,
    scheduler_timeout: float | None = None,
    recv_timeout: float = 30.0,
) -> torch.Tensor | None:
    """All ranks call this with the same input image (constructed
    identically per rank), or — with ``broadcast_input=True`` — only rank 0
    provides it and it is broadcast over RCCL/xGMI to the others. Returns
    the blended canvas on rank 0, None on workers.

    Fault tolerance (rank > 0 crash at ANY point after the job starts):
    the master's scheduler requeues a silent rank's unfinished tiles, the
    mailbox bounds every receive, and tiles whose results were lost in
    flight (marked done but never delivered) are reprocessed locally before
    the blend. A worker crash therefore costs time, never correctness.
    """
    if broadcast_input:
        image = broadcast_tensor(ctx, image if ctx.is_master else None)
    assert image is not None, "non-master ranks need broadcast_input=True"
    # .clone(): .to() is a no-copy alias when dtype/device already match,
    # and the blend pass mutates the canvas in place.
    canvas = image.to(stack.device, torch.float32).clone().contiguous()
    B, H, W, _ = canvas.shape
    plans = plan_for_image(W, H, params)

    queue = TileQueue(store, job_id, ctx.rank)
    if ctx.is_master:
        queue.init_job(len(plans))
    else:
        # job-ready poll instead of a barrier: one fewer collective that a
        # dead rank could hang (reference workers poll job_status the same
        # way, upscale/modes/static.py:33-47)
        deadline = time.monotonic() + recv_timeout
        while not queue.is_ready():
            if time.monotonic() > deadline:
                raise TimeoutError(f"job {job_id} never became ready")
            time.sleep(0.01)
    queue.heartbeat()  # initial stamp: a rank is monitorable from the start

    # pop enough tile ids per iteration to fill the sampler batch
    # (each tile id covers all B batch images)
    ids_per_iter = max(1, params.tile_batch // max(B, 1))
    mailbox = ResultMailbox(ctx, store, job_id, recv_timeout=recv_timeout)

    if not ctx.is_master:
        # liveness heartbeat thread: a chunk's first pass can take MINUTES
        # (MIOpen warmup, hipGraph capture) — per-chunk heartbeats alone
        # would get a healthy rank dropped by the master's scheduler. A
        # live process heartbeating == the reference's probe-alive grace
        # (upscale/job_timeout.py:85-100); a SIGKILLed rank's thread dies
        # with it, so crash detection is unaffected.
        stop_hb = threading.Event()

        def _hb_loop():
            while not stop_hb.wait(5.0):
                try:
                    queue.heartbeat()
                except Exception:
                    return

        hb_thread = threading.Thread(target=_hb_loop, daemon=True)
        hb_thread.start()
        done = 0
        try:
            while True:
                ids = _pop_chunk(queue, ids_per_iter)
                if not ids:
                    break
                res = sample_tiles(stack, cond, uncond, params, canvas,
                                   plans, ids)
                items = sorted(res.items())
                # post the chunk's sends BEFORE marking done: a crash
                # between the two leaves tiles un-done -> the scheduler
                # requeues them
                mailbox.send_chunk([img[0] for (_tb, img) in items],
                                   [tb for (tb, _img) in items])
                for idx in ids:
                    queue.mark_done(idx)
                queue.heartbeat()
                done += len(ids)
            mailbox.finish()
            queue.heartbeat()
        finally:
            stop_hb.set()
        debug_log(f"rank {ctx.rank}: processed {done} tiles")
        return None

    # ---- master: process + drain + monitor --------------------------------
    workers = list(range(1, ctx.world_size))
    scheduler = TileScheduler(queue, workers, timeout=scheduler_timeout)
    results: dict[tuple[int, int], torch.Tensor] = {}

    def _absorb(tensors, meta):
        for tensor, (t, b) in zip(tensors, meta):
            results[(int(t), int(b))] = tensor[None].float().to(canvas.device)

    done = 0
    while True:
        ids = _pop_chunk(queue, ids_per_iter)
        if ids:
            res = sample_tiles(stack, cond, uncond, params, canvas, plans, ids)
            _absorb([img[0] for (_tb, img) in sorted(res.items())],
                    [tb for (tb, _img) in sorted(res.items())])
            for idx in ids:
                queue.mark_done(idx)
            queue.heartbeat()
            done += len(ids)
        _absorb(*mailbox.drain())
        for r in list(mailbox.failed_ranks):
            scheduler.force_drop(r)
        scheduler.check_and_requeue()
        if not ids:
            if all(r in scheduler.dropped or mailbox.rank_finished(r)
                   for r in workers):
                _absorb(*mailbox.drain())
                break
            time.sleep(0.02)
    debug_log(f"rank 0: processed {done} tiles locally")

    # ---- takeover: reprocess anything whose result never arrived ----------
    missing_tiles = sorted({
        t for t in range(len(plans))
        if any((t, b) not in results for b in range(B))
    })
    if missing_tiles:
        log(f"master takeover: {len(missing_tiles)} tiles missing — "
            "reprocessing locally")
    for i in range(0, len(missing_tiles), ids_per_iter):
        ids = missing_tiles[i : i + ids_per_iter]
        res = sample_tiles(stack, cond, uncond, params, canvas, plans, ids)
        _absorb([img[0] for (_tb, img) in sorted(res.items())],
                [tb for (tb, _img) in sorted(res.items())])

    blend_results(canvas, results, plans, params)
    from ..engine.usdu import phase_timer
    phase_timer.flush(f" job={job_id}")
    return canvas
