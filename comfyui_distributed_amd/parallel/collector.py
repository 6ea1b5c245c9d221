"""Seed-parallel generation + collector gather over RCCL.

Reference counterpart: DistributedSeed (nodes/utilities.py:52-75, worker
seed offset = worker_index + 1) + DistributedCollector (nodes/collector.py:
HTTP POST of base64 PNGs into the master's asyncio queue, then deterministic
reorder + torch.cat on the CPU). Here the whole gather is one RCCL pass
over xGMI, the reorder key is (rank, batch_index) — master first, workers
in rank order — matching the reference's "master batch first, then workers
in enabled order" (collector.py:193-236).
"""

from __future__ import annotations

from dataclasses import replace

import torch

from ..engine.generate import GenParams, generate_images
from .dist import DistContext, gather_tensor_lists


def seed_for_rank(base_seed: int, rank: int) -> int:
    """Master keeps the base seed; worker i (rank i, i>=1) adds its
    worker_index + 1 = rank (DistributedSeed semantics)."""
    return base_seed + rank


def seed_parallel_generate(
    ctx: DistContext,
    stack,
    cond: dict,
    uncond: dict | None,
    params: GenParams,
) -> torch.Tensor | None:
    """Every rank generates its batch with its offset seed; images are
    gathered to rank 0 and concatenated in (rank, batch_index) order.
    Returns [world*B, H, W, 3] float32 on rank 0, None elsewhere."""
    p = replace(params, seed=seed_for_rank(params.seed, ctx.rank))
    images = generate_images(stack, cond, uncond, p)
    tensors = [images[i] for i in range(images.shape[0])]
    meta = [(ctx.rank, i) for i in range(images.shape[0])]
    gathered = gather_tensor_lists(ctx, tensors, meta)
    if not ctx.is_master:
        return None
    all_tensors, all_meta = gathered
    order = sorted(range(len(all_meta)), key=lambda j: all_meta[j])
    return torch.stack([all_tensors[j] for j in order])
