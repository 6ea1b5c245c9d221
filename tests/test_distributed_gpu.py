"""Multi-process distributed USDU on a single GPU (gloo wire, cuda
compute): validates the rank>0 worker loop, ResultMailbox streaming and
the blend against the single-GPU canvas with REAL GPU tensors — the
closest 1-box rehearsal of the driver's multi-GPU SCALE run (RCCL itself
needs >1 device and is exercised there)."""

import os
import tempfile

import pytest
import torch
import torch.multiprocessing as mp

pytestmark = pytest.mark.gpu

PORT = 29930


def _rank_main(rank, world, out_dir, port):
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = "0"  # both ranks share cuda:0
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    import torch.distributed as tdist

    from comfyui_distributed_amd.engine.usdu import USDUParams, process_single_gpu
    from comfyui_distributed_amd.models import create_diffusion_stack
    from comfyui_distributed_amd.parallel.dist import init_from_env
    from comfyui_distributed_amd.parallel.usdu_dist import run_distributed_usdu

    ctx = init_from_env(backend="gloo")
    try:
        store = tdist.TCPStore("127.0.0.1", port + 1000, world, ctx.is_master)
        stack = create_diffusion_stack("tiny", device="cuda:0",
                                       dtype=torch.bfloat16, seed=7)
        cond = stack.make_conditioning(0)
        p = USDUParams(seed=3, steps=2, cfg=1.0, denoise=0.5, tile_width=32,
                       tile_height=32, padding=16, mask_blur=2, tile_batch=2)
        g = torch.Generator().manual_seed(99)
        img = torch.rand(1, 96, 96, 3, generator=g)  # 9 tiles
        out = run_distributed_usdu(ctx, store, stack, cond, None, p, img,
                                   job_id="gpu2rank")
        if ctx.is_master:
            ref = process_single_gpu(stack, cond, None, p, img)
            torch.save({"dist": out.cpu(), "ref": ref.cpu()},
                       os.path.join(out_dir, "result.pt"))
    finally:
        if tdist.is_initialized():
            tdist.destroy_process_group()


def test_two_rank_usdu_on_one_gpu_matches_single():
    out_dir = tempfile.mkdtemp()
    mp.spawn(_rank_main, args=(2, out_dir, PORT), nprocs=2, join=True)
    res = torch.load(os.path.join(out_dir, "result.pt"), weights_only=False)
    # GPU tolerance: the GroupNorm stats kernel accumulates with fp32
    # atomics, so two runs of the SAME tile differ by reduction order
    # (~1e-2 through a bf16 sampler chain). Tile->rank assignment adds no
    # additional variance (extraction reads the original canvas); the CPU
    # suite (test_distributed_cpu.py) checks exact equality on the
    # atomics-free path.
    assert torch.allclose(res["dist"], res["ref"], atol=0.12), \
        (res["dist"] - res["ref"]).abs().max().item()
