"""Ultimate-SD-Upscale tile engine (the compute body of the USDU node).

Reference counterpart: upscale/tile_ops.py process_tile/process_tiles_batch
+ upscale/modes/single_gpu.py. Differences by design:

* The whole pipeline stays on-device: extract+resample, VAE encode, sampler
  loop, VAE decode and seam blend are HIP kernels / device tensors — the
  reference round-trips through PIL on the CPU for every tile.
* Tiles are batched along a combined (image, tile) axis — the reference
  batches only across the image batch, one tile at a time
  (tile_ops.py:239-287). With 288 GB HBM per GPU there is no reason not to
  fill the device.
* Per-tile noise is derived from (seed, tile_index, batch_index), so results
  are bit-identical no matter which GPU processes which tile — the
  reference has the same property because every worker re-seeds per tile.
"""

from __future__ import annotations

from dataclasses import dataclass

import torch

from ..models.sampling import CFGDenoiser, NoiseSchedule, sample
from ..ops import dispatch as ops
from ..utils import usdu_math
from ..utils.trace import trace_range

# --- wall-gap diagnosis (DISTGPU_PHASE_TIMING=1): host wall per phase with
# device syncs at the boundaries; prints a summary every flush() -------------
import os as _os
import time as _time


class _PhaseTimer:
    enabled = _os.environ.get("DISTGPU_PHASE_TIMING", "") == "1"

    def __init__(self):
        self.acc: dict[str, float] = {}
        self._t0 = None
        self._cur = None

    def mark(self, phase: str | None):
        if not self.enabled:
            return
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        now = _time.perf_counter()
        if self._cur is not None:
            self.acc[self._cur] = self.acc.get(self._cur, 0.0) + (now - self._t0)
        self._cur = phase
        self._t0 = now

    def flush(self, label: str = ""):
        if not self.enabled or not self.acc:
            return
        self.mark(None)
        total = sum(self.acc.values())
        parts = " ".join(f"{k}={v*1e3:.0f}ms" for k, v in
                         sorted(self.acc.items(), key=lambda kv: -kv[1]))
        print(f"[phase-timing]{label} total={total*1e3:.0f}ms {parts}",
              flush=True)
        self.acc.clear()


phase_timer = _PhaseTimer()


@dataclass
class USDUParams:
    seed: int = 0
    steps: int = 20
    cfg: float = 8.0
    sampler_name: str = "euler"
    scheduler: str = "normal"
    denoise: float = 0.35
    tile_width: int = 512
    tile_height: int = 512
    padding: int = 32
    mask_blur: int = 8
    force_uniform_tiles: bool = True
    tiled_decode: bool = False
    tile_batch: int = 16  # tiles sampled together per sampler call


def plan_for_image(width: int, height: int, p: USDUParams):
    return usdu_math.plan_tiles(
        width, height, p.tile_width, p.tile_height, p.padding,
        uniform=p.force_uniform_tiles,
    )


def _tile_noise(seed: int, tile_idx: int, batch_idx: int, shape) -> torch.Tensor:
    g = torch.Generator(device="cpu").manual_seed(
        (seed * 1_000_003 + tile_idx * 1009 + batch_idx) & 0x7FFFFFFF
    )
    return torch.randn(shape, generator=g)


def _flow_velocity(stack, cond, uncond, cfg_scale: float):
    """CFG velocity closure for flow-family stacks (Flux) inside USDU —
    the flow-matching counterpart of CFGDenoiser."""

    def velocity(x, t):
        b = x.shape[0]
        tt = (t * 1000.0).reshape(-1).to(x.device).expand(b).to(stack.dtype)
        ctx = cond["context"].expand(b, -1, -1).to(stack.dtype)
        vec = cond["vec"].expand(b, -1).to(stack.dtype)
        v = stack.model(x.to(stack.dtype), tt, ctx, vec)
        if cfg_scale != 1.0 and uncond is not None:
            vu = stack.model(
                x.to(stack.dtype), tt,
                uncond["context"].expand(b, -1, -1).to(stack.dtype),
                uncond["vec"].expand(b, -1).to(stack.dtype))
            v = vu + cfg_scale * (v - vu)
        return v.float()

    return velocity


def sample_tiles(
    stack,
    cond: dict,
    uncond: dict | None,
    params: USDUParams,
    canvas: torch.Tensor,
    plans: list,
    tile_indices: list[int],
    batch_offset: int = 0,
) -> dict[tuple[int, int], torch.Tensor]:
    """Sample the given tiles of ``canvas`` (all batch images) and return
    {(tile_idx, batch_idx): processed tile [1,Ph,Pw,C] float32}.

    Extraction always reads the ORIGINAL canvas, never partially-blended
    state, so results are bit-identical regardless of which rank processes
    which tiles (the reference's progressive local blend —
    upscale/modes/static.py:268-280 — makes worker output depend on its
    tile assignment; this framework deliberately drops that so the
    distributed result equals the single-GPU result exactly).
    """
    if not tile_indices:
        return {}
    B = canvas.shape[0]
    family = getattr(stack, "family", "sd")
    is_flow = family == "flux"
    if not is_flow and not hasattr(stack, "schedule"):
        raise ValueError(
            f"model family {family!r} cannot drive the tile upscaler: "
            "it has no image img2img path (video models upscale per-frame "
            "through an IMAGE model — see workflows/"
            "distributed_upscale_video.json)"
        )
    if is_flow:
        sigmas = base_denoiser = None  # flow models integrate velocity
    else:
        schedule: NoiseSchedule = stack.schedule
        sigmas = schedule.sigmas(
            params.steps, params.scheduler, params.denoise
        ).to(canvas.device)
        base_denoiser = CFGDenoiser(stack.unet, schedule, cond, uncond,
                                    params.cfg)

    from .conditioning import SPATIAL_KEYS, crop_tile_conditioning
    from .model_patch import crop_model_patches, stack_has_patches

    has_spatial = (
        any(k in (cond or {}) for k in SPATIAL_KEYS)
        or any(k in (uncond or {}) for k in SPATIAL_KEYS)
        or stack_has_patches(stack)
    )
    work = [(t, b) for t in sorted(tile_indices) for b in range(B)]
    # spatial conditioning / model patches differ per tile -> tiles can't
    # share a sampler batch (reference processes per tile for the same
    # reason: tile_ops.py crop context + crop_model_patch.py)
    step = 1 if has_spatial else max(params.tile_batch, 1)
    from ..nodes.runtime import get_runtime

    results: dict[tuple[int, int], torch.Tensor] = {}
    for i in range(0, len(work), step):
        # user interrupt aborts between chunks in EVERY mode (the
        # distributed loops also check; this covers single-GPU/local —
        # reference tile loops check comfy.model_management per tile)
        get_runtime().throw_if_interrupted()
        chunk = work[i : i + step]
        chunk_cond, chunk_uncond = cond, uncond
        if has_spatial:
            plan = plans[chunk[0][0]]
            chunk_cond = crop_tile_conditioning(
                cond, plan.crop_region, plan.canvas_size, plan.process_size
            )
            chunk_uncond = crop_tile_conditioning(
                uncond, plan.crop_region, plan.canvas_size, plan.process_size
            )
        denoiser = None
        if not is_flow:
            denoiser = (
                base_denoiser
                if not has_spatial
                else CFGDenoiser(stack.unet, schedule, chunk_cond,
                                 chunk_uncond, params.cfg)
            )
        # ---- extract + resample each (tile, batch) crop to process size ----
        phase_timer.mark("extract")
        crops = []
        for t, b in chunk:
            plan = plans[t]
            x1, y1, x2, y2 = plan.crop_region
            pw, ph = plan.process_size
            crops.append(ops.extract_resize(canvas[b : b + 1], (x1, y1, x2, y2), pw, ph))
        batch_img = torch.cat(crops, dim=0)
        # ---- encode -> img2img sample -> decode ----
        with torch.no_grad():
            phase_timer.mark("vae_encode")
            with trace_range("usdu.vae_encode"):
                latents = stack.vae.encode(batch_img)
            phase_timer.mark("noise")
            noise = torch.stack(
                [
                    _tile_noise(params.seed, t, b + batch_offset, latents.shape[1:])
                    for t, b in chunk
                ]
            ).to(latents.device)
            plan0 = plans[chunk[0][0]]
            phase_timer.mark("sample")
            with trace_range("usdu.sample"), crop_model_patches(
                stack, plan0.crop_region, plan0.canvas_size, plan0.process_size
            ):
                if is_flow:
                    from ..models.video import sample_flow

                    latent_out = sample_flow(
                        _flow_velocity(stack, chunk_cond, chunk_uncond,
                                       params.cfg),
                        noise,
                        params.steps,
                        shift=getattr(stack, "flow_shift", 3.0),
                        start_from_latent=latents.float(),
                        denoise=params.denoise,
                    )
                else:
                    latent_out = sample(
                        denoiser,
                        noise,
                        sigmas,
                        sampler=params.sampler_name,
                        seed=params.seed,
                        start_from_latent=latents.float(),
                    )
            phase_timer.mark("vae_decode")
            with trace_range("usdu.vae_decode"):
                if params.tiled_decode:
                    out_img = stack.vae.decode_tiled(latent_out.to(stack.dtype))
                else:
                    out_img = stack.vae.decode(latent_out.to(stack.dtype))
        phase_timer.mark("collect")
        for j, (t, b) in enumerate(chunk):
            results[(t, b)] = out_img[j : j + 1].float()
        phase_timer.mark(None)
    return results


def blend_results(
    canvas: torch.Tensor,
    results: dict[tuple[int, int], torch.Tensor],
    plans: list,
    params: USDUParams,
) -> None:
    """One canonical blend pass: ascending (tile_idx, batch_idx) — the
    deterministic order the reference enforces for worker tiles
    (upscale/modes/static.py:521-527)."""
    phase_timer.mark("blend")
    for (t, b) in sorted(results.keys()):
        blend_processed_tile(canvas, results[(t, b)], plans[t], params, batch_index=b)
    phase_timer.mark(None)


def process_tiles(
    stack, cond, uncond, params: USDUParams, canvas: torch.Tensor, plans: list,
    tile_indices: list[int],
) -> None:
    """sample_tiles + blend_results in place (single-rank convenience)."""
    results = sample_tiles(stack, cond, uncond, params, canvas, plans, tile_indices)
    blend_results(canvas, results, plans, params)


def blend_processed_tile(canvas, tile_img, plan, params: USDUParams, batch_index: int):
    """Blend one processed tile (at process size) into one canvas image
    (in place — canvas must be contiguous [B,H,W,C])."""
    assert canvas.is_contiguous()
    ops.blend_tile(
        canvas[batch_index : batch_index + 1],
        tile_img,
        plan.crop_region,
        plan.tile_rect,
        float(params.mask_blur),
    )


def process_single_gpu(stack, cond, uncond, params: USDUParams,
                       image: torch.Tensor) -> torch.Tensor:
    """Single-device USDU: all tiles locally (reference
    upscale/modes/single_gpu.py:8-72 semantics, batched-per-tile)."""
    canvas = image.to(stack.device, torch.float32).clone()
    B, H, W, _ = canvas.shape
    plans = plan_for_image(W, H, params)
    process_tiles(stack, cond, uncond, params, canvas, plans,
                  list(range(len(plans))))
    return canvas
