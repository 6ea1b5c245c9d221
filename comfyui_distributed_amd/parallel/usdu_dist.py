"""Distributed USDU: tile pull-queue over all ranks, RCCL result gather,
canonical blend on rank 0.

This is the intra-node re-architecture of the reference's static mode
(upscale/modes/static.py): the HTTP request_image/submit_tiles loop becomes
a TCPStore pull queue + one batched p2p gather over xGMI; the master
participates exactly like the reference master does (pulls tiles from its
own queue). Determinism: extraction reads the original canvas and blending
runs in ascending (tile, batch) order on rank 0, so the result is
bit-identical to the single-GPU run regardless of tile assignment.
"""

from __future__ import annotations

import torch

from ..engine.usdu import USDUParams, blend_results, plan_for_image, sample_tiles
from ..utils.logging import debug_log
from .dist import DistContext, broadcast_tensor, gather_tensor_lists
from .tile_queue import TileQueue


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
) -> torch.Tensor | None:
    """All ranks call this with the same input image (constructed
    identically per rank), or — with ``broadcast_input=True`` — only rank 0
    provides it and it is broadcast over RCCL/xGMI to the others. Returns
    the blended canvas on rank 0, None on workers."""
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
    ctx.barrier()

    tensors: list[torch.Tensor] = []
    meta: list[tuple[int, int]] = []
    done = 0
    # pop enough tile ids per iteration to fill the sampler batch
    # (each tile id covers all B batch images)
    ids_per_iter = max(1, params.tile_batch // max(B, 1))
    while True:
        ids: list[int] = []
        while len(ids) < ids_per_iter:
            idx = queue.pop()
            if idx is None:
                break
            ids.append(idx)
        if not ids:
            break
        res = sample_tiles(stack, cond, uncond, params, canvas, plans, ids)
        for (t, b), img in sorted(res.items()):
            tensors.append(img[0])
            meta.append((t, b))
        for idx in ids:
            queue.mark_done(idx)
        queue.heartbeat()
        done += len(ids)
    debug_log(f"rank {ctx.rank}: processed {done} tiles")

    gathered = gather_tensor_lists(ctx, tensors, meta)
    if not ctx.is_master:
        return None
    all_tensors, all_meta = gathered
    results = {
        (int(t), int(b)): tensor[None].float().to(canvas.device)
        for tensor, (t, b) in zip(all_tensors, all_meta)
    }
    blend_results(canvas, results, plans, params)
    return canvas
