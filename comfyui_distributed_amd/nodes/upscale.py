"""UltimateSDUpscaleDistributed node front-end.

Signature parity with reference nodes/distributed_upscale.py:46-279
(required inputs incl. the hidden distributed fields and
``dynamic_threshold``). Role/mode dispatch parity:

* no enabled workers -> ``single_gpu`` (reference :259-267)
* batch >= dynamic_threshold -> ``dynamic`` (whole images per worker)
* else -> ``static`` (tile pull-queue)

The compute body is the on-device engine (engine/usdu.py); in-process
multi-GPU execution goes through parallel/usdu_dist.py (RCCL); the HTTP
master/worker flow for remote workers is served by server/app.py.
"""

from __future__ import annotations

import json

from ..engine.usdu import USDUParams, process_single_gpu
from ..models.sampling import SAMPLERS, SCHEDULERS
from ..utils.logging import debug_log


class UltimateSDUpscaleDistributed:
    @classmethod
    def INPUT_TYPES(cls):
        return {
            "required": {
                "upscaled_image": ("IMAGE",),
                "model": ("MODEL",),
                "positive": ("CONDITIONING",),
                "negative": ("CONDITIONING",),
                "vae": ("VAE",),
                "seed": ("INT", {"default": 0, "min": 0, "max": 0xFFFFFFFFFFFFFFFF}),
                "steps": ("INT", {"default": 20, "min": 1, "max": 10000}),
                "cfg": ("FLOAT", {"default": 8.0, "min": 0.0, "max": 100.0}),
                "sampler_name": (list(SAMPLERS),),
                "scheduler": (list(SCHEDULERS),),
                "denoise": ("FLOAT", {"default": 0.5, "min": 0.0, "max": 1.0, "step": 0.01}),
                "tile_width": ("INT", {"default": 512, "min": 64, "max": 2048, "step": 8}),
                "tile_height": ("INT", {"default": 512, "min": 64, "max": 2048, "step": 8}),
                "padding": ("INT", {"default": 32, "min": 0, "max": 256, "step": 8}),
                "mask_blur": ("INT", {"default": 8, "min": 0, "max": 256}),
                "force_uniform_tiles": ("BOOLEAN", {"default": True}),
                "tiled_decode": ("BOOLEAN", {"default": False}),
            },
            "hidden": {
                "multi_job_id": ("STRING", {"default": ""}),
                "is_worker": ("BOOLEAN", {"default": False}),
                "master_url": ("STRING", {"default": ""}),
                "enabled_worker_ids": ("STRING", {"default": "[]"}),
                "worker_id": ("STRING", {"default": ""}),
                "tile_indices": ("STRING", {"default": ""}),  # compat, unused
                "dynamic_threshold": ("INT", {"default": 8, "min": 1, "max": 64}),
            },
        }

    RETURN_TYPES = ("IMAGE",)
    FUNCTION = "run"
    CATEGORY = "image/upscaling"

    @classmethod
    def IS_CHANGED(cls, **kwargs):
        return float("nan")  # always re-execute

    @staticmethod
    def determine_processing_mode(batch_size: int, n_workers: int,
                                  dynamic_threshold: int = 8) -> str:
        """Reference :259-267: no workers -> single_gpu; large batches go
        image-parallel (dynamic), small go tile-parallel (static)."""
        if n_workers == 0:
            return "single_gpu"
        if batch_size >= dynamic_threshold:
            return "dynamic"
        return "static"

    @staticmethod
    def validate_4n1_batch(batch: int, model_family: str) -> None:
        """WAN/FLOW video models require 4n+1 frame batches
        (reference :125-162 validation)."""
        if model_family in ("wan", "flow") and batch % 4 != 1:
            raise ValueError(
                f"{model_family} models need a 4n+1 frame batch, got {batch}"
            )

    def run(self, upscaled_image, model, positive, negative, vae, seed, steps,
            cfg, sampler_name, scheduler, denoise, tile_width, tile_height,
            padding, mask_blur, force_uniform_tiles, tiled_decode,
            multi_job_id="", is_worker=False, master_url="",
            enabled_worker_ids="[]", worker_id="", tile_indices="",
            dynamic_threshold=8):
        params = USDUParams(
            seed=seed, steps=steps, cfg=cfg, sampler_name=sampler_name,
            scheduler=scheduler, denoise=denoise, tile_width=tile_width,
            tile_height=tile_height, padding=padding, mask_blur=mask_blur,
            force_uniform_tiles=force_uniform_tiles, tiled_decode=tiled_decode,
        )
        enabled = json.loads(enabled_worker_ids or "[]")
        if getattr(model, "family", "") in ("wan", "flow"):
            self.validate_4n1_batch(upscaled_image.shape[0],
                                    getattr(model, "family", ""))
        mode = self.determine_processing_mode(
            upscaled_image.shape[0], len(enabled), dynamic_threshold
        )
        debug_log(f"USDU node: mode={mode} job={multi_job_id} worker={is_worker}")

        # ``model`` is a DiffusionStack handle in this framework; positive /
        # negative are conditioning dicts ({"context": ..., "y": ...?}).
        stack = model
        cond = positive
        uncond = negative
        if mode == "single_gpu" or not multi_job_id:
            out = process_single_gpu(stack, cond, uncond, params, upscaled_image)
            return (out.cpu(),)
        # distributed HTTP modes are orchestrated by the server layer, which
        # calls into server/usdu_http.py with this node's params
        from ..server.usdu_http import run_usdu_role

        out = run_usdu_role(
            mode=mode, params=params, stack=stack, cond=cond, uncond=uncond,
            image=upscaled_image, job_id=multi_job_id, is_worker=is_worker,
            master_url=master_url, enabled_workers=[str(w) for w in enabled],
            worker_id=worker_id,
        )
        return (out.cpu() if out is not None else upscaled_image,)
