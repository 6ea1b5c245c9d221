"""Per-tile model-patch cropping.

Reference counterpart: utils/crop_model_patch.py (ModelPatchCropper /
crop_model_cond, :9-114) — some model families (DiffSynth/ZImage-style
control variants) attach spatial tensors to the MODEL rather than to the
conditioning: a pixel-space "image" patch and/or a pre-encoded "latent"
patch covering the whole canvas. When a tile is sampled, those patches
must be cropped to the tile's crop region (and resized to the processing
resolution) for the duration of that tile's sampler call, then restored.

Here patches live on the stack as ``stack.model_patches`` —
``dict[name, {"image": [B,C,H,W], "latent": [B,C,H/ds,W/ds]}]`` in
canvas coordinates. ``crop_model_patches`` is a context manager that
swaps in cropped views and restores the originals on exit (the reference
restores via ModelPatchCropper.__del__; an explicit ``finally`` is the
non-refcount-dependent equivalent).
"""

from __future__ import annotations

from contextlib import contextmanager

import torch.nn.functional as F

from ..utils.usdu_math import resize_region


def _crop_spatial(t, region, canvas_size, out_size, mode="bilinear"):
    x1, y1, x2, y2 = region
    cropped = t[:, :, y1:y2, x1:x2]
    if cropped.shape[-2:] == tuple(reversed(out_size)):
        return cropped
    return F.interpolate(
        cropped.float(), size=(out_size[1], out_size[0]), mode=mode,
        align_corners=False if mode == "bilinear" else None,
    ).to(t.dtype)


def crop_patch_dict(patch, region, canvas_size, process_size,
                    latent_downscale=8):
    """Crop one patch entry. ``image`` is cropped in pixel space; ``latent``
    in latent space (region scaled by the VAE downscale factor). Other keys
    pass through by reference."""
    out = dict(patch)
    img = patch.get("image")
    if img is not None:
        out["image"] = _crop_spatial(img, region, canvas_size, process_size)
    lat = patch.get("latent")
    if lat is not None:
        ds = latent_downscale
        lregion = resize_region(
            region, canvas_size,
            (canvas_size[0] // ds, canvas_size[1] // ds),
        )
        out["latent"] = _crop_spatial(
            lat, lregion, None, (process_size[0] // ds, process_size[1] // ds)
        )
    return out


@contextmanager
def crop_model_patches(stack, region, canvas_size, process_size,
                       latent_downscale=8):
    """Temporarily replace ``stack.model_patches`` with tile-cropped copies.

    No-op (yields False) when the stack carries no patches. Original patch
    dict object is restored on exit even if sampling raises.
    """
    patches = getattr(stack, "model_patches", None)
    if not patches:
        yield False
        return
    cropped = {
        name: crop_patch_dict(p, region, canvas_size, process_size,
                              latent_downscale)
        for name, p in patches.items()
    }
    stack.model_patches = cropped
    try:
        yield True
    finally:
        stack.model_patches = patches


def stack_has_patches(stack) -> bool:
    return bool(getattr(stack, "model_patches", None))
