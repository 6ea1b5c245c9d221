"""Multi-process distributed tests on CPU (gloo backend, world_size=2):
collector gather ordering, distributed USDU vs single-GPU equivalence.
These exercise the same code paths RCCL runs on the GPU node."""

import os
import tempfile

import torch
import torch.multiprocessing as mp

PORT_BASE = 29710


def _run_rank(rank, world, fn_name, out_dir, port):
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    import torch.distributed as dist

    from comfyui_distributed_amd.parallel.dist import init_from_env

    ctx = init_from_env(backend="gloo")
    try:
        result = globals()[fn_name](ctx, port)
        if result is not None:
            torch.save(result, os.path.join(out_dir, f"rank{rank}.pt"))
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


def _spawn(fn_name, world=2, port=PORT_BASE):
    out_dir = tempfile.mkdtemp()
    mp.spawn(_run_rank, args=(world, fn_name, out_dir, port), nprocs=world,
             join=True)
    out = {}
    for r in range(world):
        path = os.path.join(out_dir, f"rank{r}.pt")
        if os.path.exists(path):
            out[r] = torch.load(path, weights_only=False)
    return out


# ---- bodies (run inside spawned processes) --------------------------------


def _body_gather(ctx, port):
    from comfyui_distributed_amd.parallel.dist import gather_tensor_lists

    n = 2 if ctx.rank == 0 else 3
    tensors = [torch.full((4, 4), float(ctx.rank * 10 + i)) for i in range(n)]
    meta = [(ctx.rank, i) for i in range(n)]
    res = gather_tensor_lists(ctx, tensors, meta)
    if ctx.is_master:
        all_t, all_m = res
        return {"meta": all_m, "sums": [float(t.sum()) for t in all_t]}
    return None


def _body_gather_empty_rank(ctx, port):
    from comfyui_distributed_amd.parallel.dist import gather_tensor_lists

    if ctx.rank == 0:
        tensors, meta = [], []
    else:
        tensors = [torch.ones(2, 2)]
        meta = [(ctx.rank, 0)]
    res = gather_tensor_lists(ctx, tensors, meta)
    if ctx.is_master:
        all_t, all_m = res
        return {"meta": all_m, "n": len(all_t)}
    return None


def _body_collector(ctx, port):
    from comfyui_distributed_amd.engine.generate import GenParams
    from comfyui_distributed_amd.models import create_diffusion_stack
    from comfyui_distributed_amd.parallel.collector import seed_parallel_generate

    stack = create_diffusion_stack("tiny", seed=123)
    cond, uncond = stack.make_conditioning(0), None
    p = GenParams(seed=5, steps=2, cfg=1.0, width=16, height=16, batch_size=2)
    out = seed_parallel_generate(ctx, stack, cond, uncond, p)
    if ctx.is_master:
        return {"images": out}
    return None


def _body_broadcast(ctx, port):
    from comfyui_distributed_amd.parallel.dist import broadcast_tensor

    src = torch.arange(24, dtype=torch.float32).reshape(2, 3, 4) if ctx.is_master else None
    out = broadcast_tensor(ctx, src)
    if not ctx.is_master:
        return {"ok": torch.equal(out, torch.arange(24, dtype=torch.float32).reshape(2, 3, 4))}
    return None


def _body_usdu_broadcast(ctx, port):
    import torch.distributed as tdist

    from comfyui_distributed_amd.engine.usdu import USDUParams, process_single_gpu
    from comfyui_distributed_amd.models import create_diffusion_stack
    from comfyui_distributed_amd.parallel.usdu_dist import run_distributed_usdu

    store = tdist.TCPStore("127.0.0.1", port + 1000, ctx.world_size,
                           ctx.is_master)
    stack = create_diffusion_stack("tiny", seed=7)
    cond = stack.make_conditioning(0)
    p = USDUParams(seed=3, steps=1, cfg=1.0, denoise=0.5, tile_width=16,
                   tile_height=16, padding=16, mask_blur=2, tile_batch=2)
    img = None
    if ctx.is_master:
        g = torch.Generator().manual_seed(77)
        img = torch.rand(1, 32, 32, 3, generator=g)
    out = run_distributed_usdu(ctx, store, stack, cond, None, p, img,
                               broadcast_input=True)
    if ctx.is_master:
        ref = process_single_gpu(stack, cond, None, p, img)
        return {"dist": out, "ref": ref}
    return None


def _body_usdu(ctx, port):
    import torch.distributed as tdist

    from comfyui_distributed_amd.engine.usdu import USDUParams, process_single_gpu
    from comfyui_distributed_amd.models import create_diffusion_stack
    from comfyui_distributed_amd.parallel.usdu_dist import run_distributed_usdu

    store = tdist.TCPStore("127.0.0.1", port + 1000, ctx.world_size,
                           ctx.is_master)
    stack = create_diffusion_stack("tiny", seed=7)
    cond = stack.make_conditioning(0)
    p = USDUParams(seed=3, steps=2, cfg=1.0, denoise=0.5, tile_width=16,
                   tile_height=16, padding=16, mask_blur=2, tile_batch=2)
    g = torch.Generator().manual_seed(99)
    img = torch.rand(1, 32, 32, 3, generator=g)
    out = run_distributed_usdu(ctx, store, stack, cond, None, p, img)
    if ctx.is_master:
        ref = process_single_gpu(stack, cond, None, p, img)
        return {"dist": out, "ref": ref}
    return None


# ---- tests ----------------------------------------------------------------


def test_gather_variable_counts():
    out = _spawn("_body_gather", port=PORT_BASE)
    res = out[0]
    assert sorted(res["meta"]) == [(0, 0), (0, 1), (1, 0), (1, 1), (1, 2)]
    assert len(res["sums"]) == 5


def test_gather_with_empty_master():
    out = _spawn("_body_gather_empty_rank", port=PORT_BASE + 1)
    res = out[0]
    assert res["n"] == 1 and res["meta"] == [(1, 0)]


def test_collector_ordering_and_shapes():
    out = _spawn("_body_collector", port=PORT_BASE + 2)
    images = out[0]["images"]
    # 2 ranks x batch 2 = 4 images, master's first
    assert images.shape == (4, 16, 16, 3)
    assert torch.isfinite(images).all()
    # different seeds per rank -> master and worker batches differ
    assert not torch.allclose(images[0], images[2])


def test_distributed_usdu_equals_single_gpu():
    out = _spawn("_body_usdu", port=PORT_BASE + 3)
    dist_c, ref_c = out[0]["dist"], out[0]["ref"]
    assert dist_c.shape == ref_c.shape
    assert torch.allclose(dist_c, ref_c, atol=1e-5)


def test_broadcast_tensor_gloo():
    out = _spawn("_body_broadcast", port=PORT_BASE + 4)
    assert out[1]["ok"] is True


def test_usdu_with_input_broadcast():
    out = _spawn("_body_usdu_broadcast", port=PORT_BASE + 5)
    dist_c, ref_c = out[0]["dist"], out[0]["ref"]
    assert torch.allclose(dist_c, ref_c, atol=1e-5)


def test_distributed_usdu_world4_equals_single_gpu():
    """4-rank tile-queue USDU (the shape the driver's 8-GPU scaling run
    exercises): result identical to the single-GPU canvas."""
    out = _spawn("_body_usdu", world=4, port=PORT_BASE + 6)
    dist_c, ref_c = out[0]["dist"], out[0]["ref"]
    assert torch.allclose(dist_c, ref_c, atol=1e-5)


def test_gather_variable_counts_world4():
    out = _spawn("_body_gather", world=4, port=PORT_BASE + 7)
    res = out[0]
    # rank0: 2 items, ranks 1-3: 3 items each
    assert len(res["meta"]) == 2 + 3 * 3
    assert sorted(res["meta"]) == sorted(
        [(0, i) for i in range(2)]
        + [(r, i) for r in range(1, 4) for i in range(3)])


def _body_flux_seed_parallel(ctx, port):
    from comfyui_distributed_amd.engine.generate import GenParams
    from comfyui_distributed_amd.models.registry import create_diffusion_stack
    from comfyui_distributed_amd.parallel.collector import seed_parallel_generate

    stack = create_diffusion_stack("flux_tiny", seed=7)
    cond = stack.make_conditioning(0)
    p = GenParams(seed=11, steps=1, cfg=1.0, width=16, height=16,
                  batch_size=1)
    out = seed_parallel_generate(ctx, stack, cond, None, p)
    if ctx.is_master:
        return {"images": out}
    return None


def test_flux_seed_parallel_world2():
    out = _spawn("_body_flux_seed_parallel", port=PORT_BASE + 8)
    images = out[0]["images"]
    assert images.shape == (2, 16, 16, 3)
    assert torch.isfinite(images).all()
    # per-rank seed offsets -> different images
    assert not torch.allclose(images[0], images[1])


def _body_wan_seed_parallel(ctx, port):
    from dataclasses import replace

    from comfyui_distributed_amd.models.registry import create_diffusion_stack
    from comfyui_distributed_amd.models.video import VideoGenParams, generate_video
    from comfyui_distributed_amd.parallel.collector import seed_for_rank
    from comfyui_distributed_amd.parallel.dist import gather_tensor_lists

    stack = create_diffusion_stack("wan_tiny", seed=7)
    cond = stack.make_conditioning(0)
    p = VideoGenParams(seed=3, steps=1, cfg=1.0, width=16, height=16,
                       frames=5)
    local = generate_video(stack, cond, None,
                           replace(p, seed=seed_for_rank(p.seed, ctx.rank)))
    tensors = [local[i] for i in range(local.shape[0])]
    meta = [(ctx.rank, i) for i in range(local.shape[0])]
    res = gather_tensor_lists(ctx, tensors, meta)
    if ctx.is_master:
        all_t, all_m = res
        return {"n": len(all_t), "meta": sorted(all_m)}
    return None


def test_wan_seed_parallel_frame_gather_world2():
    """The bench.py wan-t2v gather shape: every rank's 5 frames arrive on
    rank 0 with (rank, frame) metadata intact."""
    out = _spawn("_body_wan_seed_parallel", port=PORT_BASE + 9)
    res = out[0]
    assert res["n"] == 10
    assert res["meta"] == [(r, i) for r in range(2) for i in range(5)]


def _body_mailbox(ctx, port):
    """ResultMailbox contract: multi-chunk streaming, interleaved drains,
    done-marker termination, payload dtype preserved on the wire."""
    import torch.distributed as tdist

    from comfyui_distributed_amd.parallel.dist import ResultMailbox

    store = tdist.TCPStore("127.0.0.1", port + 1000, ctx.world_size,
                           ctx.is_master)
    mb = ResultMailbox(ctx, store, "mbjob", recv_timeout=30.0)
    if not ctx.is_master:
        for seq in range(3):
            tensors = [torch.full((4, 4), float(seq * 10 + i),
                                  dtype=torch.bfloat16) for i in range(2)]
            mb.send_chunk(tensors, [(seq, i) for i in range(2)])
        mb.finish()
        return None
    got_t, got_m = [], []
    import time as _t

    deadline = _t.monotonic() + 60
    while not mb.rank_finished(1):
        t, m = mb.drain()
        got_t += t
        got_m += m
        if _t.monotonic() > deadline:
            break
        _t.sleep(0.01)
    t, m = mb.drain()
    got_t += t
    got_m += m
    return {"meta": got_m,
            "vals": [float(x[0, 0]) for x in got_t],
            "dtypes": [str(x.dtype) for x in got_t]}


def test_result_mailbox_streams_chunks_in_order():
    out = _spawn("_body_mailbox", port=PORT_BASE + 10)
    res = out[0]
    assert res["meta"] == [(s, i) for s in range(3) for i in range(2)]
    assert res["vals"] == [0.0, 1.0, 10.0, 11.0, 20.0, 21.0]
    # source dtype preserved on the wire (bf16, not fp32-upcast)
    assert all(d == "torch.bfloat16" for d in res["dtypes"])
