"""Per-tile conditioning surgery.

Reference counterpart: utils/usdu_utils.py crop_cond (:297-502) +
upscale/conditioning.py clone_conditioning — when a workflow carries
spatially-anchored conditioning (ControlNet hints, attention masks, area
constraints, Flux-Kontext reference latents), each tile's sampler call must
see that conditioning cropped/remapped to the tile's crop region.

Conditioning here is a dict: {"context": [B,L,Dctx]} plus optional spatial
extras:
  control_hint       [B, C, H, W] pixel-space hint -> crop + resize
  mask               [B, H, W] or [B,1,H,W] pixel mask -> crop + resize
  area               (x, y, w, h) pixel rect -> intersect + rebase
  gligen             list of (emb [L,D], (x, y, w, h) pixel box) -> clip +
                     rebase each box to the tile; boxes fully outside drop
  reference_latents  [B, C, H/8, W/8] -> crop in latent space
Non-spatial keys pass through by reference (no copies of shared tensors —
the reference's clone_conditioning shares ControlNet models the same way).
"""

from __future__ import annotations

import torch.nn.functional as F

from ..utils.usdu_math import resize_region

SPATIAL_KEYS = ("control_hint", "mask", "area", "gligen", "reference_latents")


def crop_tile_conditioning(
    cond: dict | None,
    region: tuple[int, int, int, int],
    canvas_size: tuple[int, int],
    process_size: tuple[int, int],
    latent_downscale: int = 8,
) -> dict | None:
    """Crop/remap spatial conditioning to ``region`` of the canvas, sized
    for the tile's processing resolution."""
    if cond is None:
        return None
    if not any(k in cond for k in SPATIAL_KEYS):
        return cond  # nothing spatial: share as-is
    out = dict(cond)
    x1, y1, x2, y2 = region
    pw, ph = process_size

    hint = cond.get("control_hint")
    if hint is not None:
        cropped = hint[:, :, y1:y2, x1:x2]
        out["control_hint"] = F.interpolate(
            cropped.float(), size=(ph, pw), mode="bilinear", align_corners=False
        ).to(hint.dtype)

    mask = cond.get("mask")
    if mask is not None:
        squeeze = mask.dim() == 3
        m = mask[:, None] if squeeze else mask
        m = m[:, :, y1:y2, x1:x2]
        m = F.interpolate(m.float(), size=(ph, pw), mode="nearest").to(mask.dtype)
        out["mask"] = m[:, 0] if squeeze else m

    area = cond.get("area")
    if area is not None:
        ax, ay, aw, ah = area
        ix1, iy1 = max(ax, x1), max(ay, y1)
        ix2, iy2 = min(ax + aw, x2), min(ay + ah, y2)
        if ix2 <= ix1 or iy2 <= iy1:
            out["area"] = None  # tile outside the area: caller may skip cond
        else:
            # rebase into tile pixel space, then scale to processing size
            rx1, ry1, rx2, ry2 = resize_region(
                (ix1 - x1, iy1 - y1, ix2 - x1, iy2 - y1),
                (x2 - x1, y2 - y1), (pw, ph),
            )
            out["area"] = (rx1, ry1, rx2 - rx1, ry2 - ry1)

    gligen = cond.get("gligen")
    if gligen is not None:
        # reference crop_gligen (usdu_utils.py:335-378): clip each position
        # box to the tile, rebase to tile-local pixels, scale to processing
        # size; boxes with no overlap are dropped entirely
        kept = []
        for emb, box in gligen:
            bx, by, bw, bh = box
            ix1, iy1 = max(bx, x1), max(by, y1)
            ix2, iy2 = min(bx + bw, x2), min(by + bh, y2)
            if ix2 <= ix1 or iy2 <= iy1:
                continue
            rx1, ry1, rx2, ry2 = resize_region(
                (ix1 - x1, iy1 - y1, ix2 - x1, iy2 - y1),
                (x2 - x1, y2 - y1), (pw, ph),
            )
            kept.append((emb, (rx1, ry1, rx2 - rx1, ry2 - ry1)))
        out["gligen"] = kept or None

    ref = cond.get("reference_latents")
    if ref is not None:
        ds = latent_downscale
        lx1, ly1, lx2, ly2 = resize_region(
            region, canvas_size, (canvas_size[0] // ds, canvas_size[1] // ds)
        )
        cropped = ref[:, :, ly1:ly2, lx1:lx2]
        out["reference_latents"] = F.interpolate(
            cropped.float(), size=(ph // ds, pw // ds), mode="bilinear",
            align_corners=False,
        ).to(ref.dtype)

    return out
