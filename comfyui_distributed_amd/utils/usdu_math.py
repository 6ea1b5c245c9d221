"""Ultimate-SD-Upscale tile geometry (host-side math).

Capability parity with the geometry of reference utils/usdu_utils.py and
upscale/tile_ops.py:18-155 (A1111/USDU semantics: simple tile grid from
(0,0), mask-rect crop region with the one-pixel fix, uniform-tile aspect
expansion, in-bounds crop expansion to the processing size). All functions
are pure integer math; pixel work (resize/blend/mask) lives in the HIP
kernels under ops/.

Conventions: regions are (x1, y1, x2, y2) half-open pixel boxes on a
W x H canvas; tiles are indexed row-major.
"""

from __future__ import annotations

import math
from dataclasses import dataclass


def round_up(value: int, multiple: int = 8) -> int:
    return ((value + multiple - 1) // multiple) * multiple


def calculate_tiles(width: int, height: int, tile_w: int, tile_h: int) -> list[tuple[int, int]]:
    """Row-major grid of tile origins from (0,0): ceil(H/th) x ceil(W/tw)
    entries (reference tile_ops.py:18-32)."""
    cols = math.ceil(width / tile_w)
    rows = math.ceil(height / tile_h)
    return [(col * tile_w, row * tile_h) for row in range(rows) for col in range(cols)]


def tile_grid_shape(width: int, height: int, tile_w: int, tile_h: int) -> tuple[int, int]:
    return math.ceil(width / tile_w), math.ceil(height / tile_h)


def get_crop_region(
    rect: tuple[int, int, int, int], width: int, height: int, padding: int
) -> tuple[int, int, int, int]:
    """Pad a tile rect by ``padding`` and clamp to the canvas (the mask-rect
    -> crop-region step; reference usdu_utils.py:49-63).

    ``rect`` is the tile box (x1, y1, x2, y2) before padding.
    """
    x1, y1, x2, y2 = rect
    return (
        max(x1 - padding, 0),
        max(y1 - padding, 0),
        min(x2 + padding, width),
        min(y2 + padding, height),
    )


def fix_crop_region(
    region: tuple[int, int, int, int], width: int, height: int
) -> tuple[int, int, int, int]:
    """USDU's one-pixel correction: shave the far edge when the region does
    not reach the canvas border (reference usdu_utils.py:65-73)."""
    x1, y1, x2, y2 = region
    if x2 < width:
        x2 -= 1
    if y2 < height:
        y2 -= 1
    return x1, y1, x2, y2


def expand_crop(
    region: tuple[int, int, int, int],
    width: int,
    height: int,
    target_width: int,
    target_height: int,
) -> tuple[tuple[int, int, int, int], tuple[int, int]]:
    """Grow ``region`` to ``target_width x target_height`` while staying in
    bounds: try half the growth on the far edge, push the remainder to the
    near edge, then retry the far edge (reference usdu_utils.py:76-112).
    """
    x1, y1, x2, y2 = region

    diff = target_width - (x2 - x1)
    x2 = min(x2 + diff // 2, width)
    diff = target_width - (x2 - x1)
    x1 = max(x1 - diff, 0)
    diff = target_width - (x2 - x1)
    x2 = min(x2 + diff, width)

    diff = target_height - (y2 - y1)
    y2 = min(y2 + diff // 2, height)
    diff = target_height - (y2 - y1)
    y1 = max(y1 - diff, 0)
    diff = target_height - (y2 - y1)
    y2 = min(y2 + diff, height)

    return (x1, y1, x2, y2), (target_width, target_height)


def expand_to_aspect(
    region: tuple[int, int, int, int],
    width: int,
    height: int,
    aspect_w: int,
    aspect_h: int,
) -> tuple[int, int, int, int]:
    """Uniform-tiles aspect expansion: grow the short side of the region so
    its aspect matches the processing aspect (reference tile_ops.py:57-70)."""
    x1, y1, x2, y2 = region
    rw, rh = x2 - x1, y2 - y1
    if rw <= 0 or rh <= 0:
        return region
    if rw * aspect_h > rh * aspect_w:  # region wider than target aspect
        target_w, target_h = rw, round(rw * aspect_h / aspect_w)
    else:
        target_w, target_h = round(rh * aspect_w / aspect_h), rh
    (region, _size) = expand_crop(region, width, height, target_w, target_h)
    return region


def resize_region(
    region: tuple[int, int, int, int],
    init_size: tuple[int, int],
    resize_size: tuple[int, int],
) -> tuple[int, int, int, int]:
    """Map a region between canvas resolutions (floor near edge, ceil far
    edge; reference usdu_utils.py:115-124)."""
    x1, y1, x2, y2 = region
    iw, ih = init_size
    rw, rh = resize_size
    return (
        max(0, math.floor(x1 * rw / iw)),
        max(0, math.floor(y1 * rh / ih)),
        min(rw, math.ceil(x2 * rw / iw)),
        min(rh, math.ceil(y2 * rh / ih)),
    )


def processing_size(tile_w: int, tile_h: int, padding: int) -> tuple[int, int]:
    """Per-tile sampler resolution: tile + padding rounded up to a multiple
    of 8 (P = round8(tile + padding), e.g. 544 for 512+32)."""
    return round_up(tile_w + padding), round_up(tile_h + padding)


@dataclass(frozen=True)
class TilePlan:
    """Fully-resolved geometry of one tile of a USDU job."""

    index: int
    tile_rect: tuple[int, int, int, int]  # unpadded tile box on canvas
    crop_region: tuple[int, int, int, int]  # padded+fixed+aspect+expanded
    canvas_size: tuple[int, int]  # (W, H)
    process_size: tuple[int, int]  # sampler resolution (Pw, Ph)


def plan_tiles(
    width: int,
    height: int,
    tile_w: int,
    tile_h: int,
    padding: int,
    uniform: bool = True,
) -> list[TilePlan]:
    """Geometry for every tile of a canvas: grid -> pad -> one-pixel fix ->
    (uniform) aspect expansion -> expand to processing size."""
    pw, ph = processing_size(tile_w, tile_h, padding)
    plans = []
    for idx, (x, y) in enumerate(calculate_tiles(width, height, tile_w, tile_h)):
        rect = (x, y, min(x + tile_w, width), min(y + tile_h, height))
        region = get_crop_region(rect, width, height, padding)
        region = fix_crop_region(region, width, height)
        if uniform:
            region = expand_to_aspect(region, width, height, pw, ph)
        region, _ = expand_crop(region, width, height, pw, ph)
        plans.append(
            TilePlan(
                index=idx,
                tile_rect=rect,
                crop_region=region,
                canvas_size=(width, height),
                process_size=(pw, ph),
            )
        )
    return plans
