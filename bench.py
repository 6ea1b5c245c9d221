#!/usr/bin/env python3
"""Flagship benchmark: SD1.5 Ultimate-SD-Upscale 4x -> 4K, 512px tiles
scattered over N GPUs via the RCCL tile pull-queue (BASELINE.json config 3).

Metric: tiles/sec aggregated over the whole node (one "tile" = one
(tile, image) sample: extract -> VAE encode -> 20-step CFG sampler at
544x544 -> VAE decode -> seam blend). Weak scaling: the image batch equals
the GPU count, so per-GPU work is fixed as N grows.

Single process:           python bench.py --steps 3 --warmup 1
N ranks (driver contract): python -m torch.distributed.run --nnodes=1 \
    --nproc-per-node N --master-addr 127.0.0.1 --master-port P bench.py \
    --gpus N --steps K --warmup W
"""

from __future__ import annotations

import argparse
import json
import os
import time

import torch

from comfyui_distributed_amd.engine.usdu import USDUParams, plan_for_image
from comfyui_distributed_amd.models import create_diffusion_stack
from comfyui_distributed_amd.parallel.dist import init_from_env
from comfyui_distributed_amd.parallel.tile_queue import LocalStore
from comfyui_distributed_amd.parallel.usdu_dist import run_distributed_usdu
from comfyui_distributed_amd.ops import dispatch as ops


CONFIG_PRESETS = {
    # BASELINE.json configs (3 is the flagship / default)
    "usdu-sd15-4k": {},
    "gen-sdxl": {"model": "sdxl"},
    "usdu-sdxl-8k": {"model": "sdxl", "src_size": 2048, "scale": 4,
                     "tile": 1024},
    "wan-t2v": {"model": "wan14b"},
    # beyond-BASELINE: Flux MMDiT seed-parallel generation
    "gen-flux": {"model": "flux12b", "cfg": 1.0},
}


def parse_args():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=3)
    ap.add_argument("--warmup", type=int, default=1)
    ap.add_argument("--config", default="usdu-sd15-4k",
                    choices=sorted(CONFIG_PRESETS))
    ap.add_argument("--model", default="sd15")
    ap.add_argument("--src-size", type=int, default=1024)
    ap.add_argument("--scale", type=int, default=4)
    ap.add_argument("--tile", type=int, default=512)
    ap.add_argument("--sampler-steps", type=int, default=20)
    ap.add_argument("--denoise", type=float, default=0.5)
    ap.add_argument("--cfg", type=float, default=8.0)
    ap.add_argument("--tile-batch", type=int, default=16)
    args = ap.parse_args()
    for key, value in CONFIG_PRESETS[args.config].items():
        if getattr(args, key) == ap.get_default(key):
            setattr(args, key, value)
    return args


def bench_generation(args, ctx, stack, cond, uncond):
    """Seed-parallel generation benches (BASELINE configs 2 and 5)."""
    import torch.distributed as dist

    from comfyui_distributed_amd.parallel.collector import seed_parallel_generate

    world = ctx.world_size
    if args.config == "wan-t2v":
        from comfyui_distributed_amd.models.video import (
            VideoGenParams,
            generate_video,
        )

        p = VideoGenParams(seed=3, steps=args.sampler_steps, cfg=args.cfg,
                           width=480, height=480, frames=17)

        def one_step(_s):
            # every rank generates its own seed-offset clip; frames gathered
            from dataclasses import replace

            from comfyui_distributed_amd.parallel.collector import seed_for_rank
            from comfyui_distributed_amd.parallel.dist import gather_tensor_lists

            local = generate_video(
                stack, cond, uncond, replace(p, seed=seed_for_rank(p.seed, ctx.rank))
            )
            tensors = [local[i] for i in range(local.shape[0])]
            meta = [(ctx.rank, i) for i in range(local.shape[0])]
            gather_tensor_lists(ctx, tensors, meta)

        unit, per_step = "frames/s", 17 * world
        model_name = "wan2.2-14B-dit-random-init"
        cfg_extra = {"frames": 17, "size": "480x480",
                     "sampler": f"flow-euler/{args.sampler_steps}steps/cfg{args.cfg}"}
    else:
        from comfyui_distributed_amd.engine.generate import GenParams

        p = GenParams(seed=3, steps=args.sampler_steps, cfg=args.cfg,
                      width=1024, height=1024, batch_size=1)

        def one_step(_s):
            seed_parallel_generate(ctx, stack, cond, uncond, p)

        unit, per_step = "images/s", world
        model_name = ("flux-mmdit-11.9B-random-init"
                      if args.config == "gen-flux"
                      else "sdxl-unet-2.6B-random-init")
        cfg_extra = {"size": "1024x1024",
                     "sampler": f"euler/{args.sampler_steps}steps/cfg{args.cfg}"}

    for w in range(args.warmup):
        one_step(-1 - w)
    ctx.barrier()
    ctx.sync_device()
    t0 = time.perf_counter()
    for s in range(args.steps):
        one_step(s)
    ctx.barrier()
    ctx.sync_device()
    elapsed = time.perf_counter() - t0
    if world > 1:
        dev = ctx.device if ctx.backend == "nccl" else torch.device("cpu")
        t = torch.tensor([elapsed], dtype=torch.float64, device=dev)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())
    if ctx.is_master:
        print(json.dumps({
            "metric": f"{args.config} whole-node throughput",
            "value": round(per_step * args.steps / elapsed, 3),
            "unit": unit, "n_gpus": world, "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1000.0, 1),
            "higher_is_better": True, "scaling": "weak", "vs_baseline": None,
            "dtype": "bf16" if ctx.device.type == "cuda" else "fp32",
            "data": "synthetic",
            "config": {"model": model_name, "global_batch": world,
                       "parallelism": f"seed-parallel dp{world} + RCCL collector",
                       **cfg_extra},
        }))


def main():
    args = parse_args()
    ctx = init_from_env()
    world = ctx.world_size
    n_gpus = max(args.gpus, world)
    device = ctx.device

    dtype = torch.bfloat16 if device.type == "cuda" else torch.float32
    stack = create_diffusion_stack(args.model, device=device, dtype=dtype, seed=0)
    cond = stack.make_conditioning(0)
    uncond = stack.make_conditioning(1)

    if args.config in ("gen-sdxl", "wan-t2v", "gen-flux"):
        bench_generation(args, ctx, stack, cond, uncond)
        return

    params = USDUParams(
        seed=7,
        steps=args.sampler_steps,
        cfg=args.cfg,
        sampler_name="euler",
        scheduler="normal",
        denoise=args.denoise,
        tile_width=args.tile,
        tile_height=args.tile,
        padding=32,
        mask_blur=8,
        tile_batch=args.tile_batch,
    )

    # synthetic source batch (= world size images), 4x Lanczos pre-upscale
    # on device (the USDU node's "No Upscale" contract takes the upscaled
    # canvas; we produce it with the resample kernel).
    canvas_size = args.src_size * args.scale
    g = torch.Generator().manual_seed(1234)
    src = torch.rand(world, args.src_size, args.src_size, 3, generator=g)
    src = src.to(device)
    canvas = ops.extract_resize(
        src, (0, 0, args.src_size, args.src_size), canvas_size, canvas_size
    )

    plans = plan_for_image(canvas_size, canvas_size, params)
    tiles_per_step = len(plans) * world  # (tile, image) samples per step

    if world > 1:
        import torch.distributed as dist

        port = int(os.environ.get("MASTER_PORT", "29500")) + 1
        store = dist.TCPStore(
            os.environ.get("MASTER_ADDR", "127.0.0.1"), port, world, ctx.is_master
        )
    else:
        store = LocalStore()

    def one_step(step_id: int):
        run_distributed_usdu(
            ctx, store, stack, cond, uncond, params, canvas,
            job_id=f"bench_{step_id}",
        )

    for w in range(args.warmup):
        one_step(-1 - w)

    ctx.barrier()
    ctx.sync_device()
    t0 = time.perf_counter()
    for s in range(args.steps):
        one_step(s)
    ctx.barrier()
    ctx.sync_device()
    elapsed = time.perf_counter() - t0

    # MAX over ranks
    if world > 1:
        import torch.distributed as dist

        dev = device if ctx.backend == "nccl" else torch.device("cpu")
        t = torch.tensor([elapsed], dtype=torch.float64, device=dev)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    model_names = {"sd15": "sd15-unet-860M-random-init",
                   "sdxl": "sdxl-unet-2.6B-random-init"}
    if ctx.is_master:
        value = tiles_per_step * args.steps / elapsed
        result = {
            "metric": f"tiles/sec (whole node), {args.model} USDU "
                      f"{args.scale}x->{canvas_size} {args.tile}px tiles",
            "value": round(value, 3),
            "unit": "tiles/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1000.0, 1),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16" if dtype == torch.bfloat16 else "fp32",
            "data": "synthetic",
            "config": {
                "model": model_names.get(args.model, args.model + "-random-init"),
                "global_batch": world,
                "canvas": f"{canvas_size}x{canvas_size}",
                "tile": f"{args.tile}+pad32->{plans[0].process_size[0]}",
                "tiles_per_image": len(plans),
                "sampler": f"euler/{args.sampler_steps}steps/denoise{args.denoise}/cfg{args.cfg}",
                "parallelism": f"tile-pull-queue dp{world} over RCCL/xGMI",
            },
        }
        print(json.dumps(result))


if __name__ == "__main__":
    main()
