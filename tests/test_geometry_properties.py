"""Property-based tests (hypothesis) for the USDU tile geometry — the
invariants every mode and every rank depend on: full coverage, in-bounds
crops, round8 processing sizes, crop containing the tile."""

from hypothesis import given, settings
from hypothesis import strategies as st

from comfyui_distributed_amd.engine.usdu import USDUParams, plan_for_image


@st.composite
def geometry(draw):
    w = draw(st.integers(min_value=16, max_value=512))
    h = draw(st.integers(min_value=16, max_value=512))
    tile = draw(st.sampled_from([16, 32, 64, 128]))
    pad = draw(st.sampled_from([0, 8, 16, 32]))
    blur = draw(st.integers(min_value=0, max_value=16))
    uniform = draw(st.booleans())
    return w, h, tile, pad, blur, uniform


@given(geometry())
@settings(max_examples=120, deadline=None)
def test_plans_cover_canvas_in_bounds(geo):
    w, h, tile, pad, blur, uniform = geo
    p = USDUParams(seed=0, steps=1, cfg=1.0, tile_width=tile, tile_height=tile,
                   padding=pad, mask_blur=blur, force_uniform_tiles=uniform)
    plans = plan_for_image(w, h, p)
    assert plans, "no tiles planned"
    covered = [[False] * w for _ in range(h)]
    for plan in plans:
        x1, y1, x2, y2 = plan.crop_region
        tx1, ty1, tx2, ty2 = plan.tile_rect
        # crop within canvas
        assert 0 <= x1 < x2 <= w and 0 <= y1 < y2 <= h, plan.crop_region
        if not uniform:
            # crop contains the tile up to the A1111-compat 1px
            # fix_crop_region shrink (right/bottom edge -1) — the same
            # artifact USDU itself has at padding 0. (Uniform mode's
            # aspect expansion can legally shift the crop off a sliver
            # tile at the canvas edge; its geometry is pinned by the
            # golden-value tests in test_usdu_math.py instead.)
            assert (x1 <= tx1 and y1 <= ty1
                    and tx2 - 1 <= x2 and ty2 - 1 <= y2)
        # processing size: positive multiples of 8 (VAE/latent alignment)
        pw, ph = plan.process_size
        assert pw > 0 and ph > 0 and pw % 8 == 0 and ph % 8 == 0
        for yy in range(ty1, ty2):
            row = covered[yy]
            for xx in range(tx1, tx2):
                row[xx] = True
    assert all(all(row) for row in covered), "tiles do not cover the canvas"


@given(st.integers(2, 64), st.integers(2, 64),
       st.floats(0.3, 12.0), st.data())
@settings(max_examples=60, deadline=None)
def test_rect_mask_properties(h, w, sigma, data):
    """The analytic blurred-rect mask: in [0,1], ~1 deep inside the rect,
    ~0 far outside, monotone toward the rect."""
    import torch

    from comfyui_distributed_amd.ops.dispatch import rect_mask_cpu

    rx1 = data.draw(st.integers(0, w - 1))
    ry1 = data.draw(st.integers(0, h - 1))
    rx2 = data.draw(st.integers(rx1 + 1, w))
    ry2 = data.draw(st.integers(ry1 + 1, h))
    m = rect_mask_cpu(h, w, (rx1, ry1, rx2, ry2), sigma)
    assert m.shape == (h, w)
    assert torch.isfinite(m).all()
    assert (m >= -1e-5).all() and (m <= 1 + 1e-5).all()
    cx, cy = (rx1 + rx2) // 2, (ry1 + ry2) // 2
    # center of a rect much wider than sigma saturates to ~1
    if (rx2 - rx1) > 8 * sigma and (ry2 - ry1) > 8 * sigma:
        assert m[cy, cx] > 0.99
    # points further than 6 sigma outside are ~0
    far_x = rx2 + int(6 * sigma) + 1
    if far_x < w and ry1 <= cy < ry2:
        assert m[cy, far_x] < 0.01


@given(st.integers(4, 40), st.integers(4, 40), st.integers(4, 40),
       st.integers(4, 40))
@settings(max_examples=40, deadline=None)
def test_blend_stays_in_convex_hull(h, w, ow, oh):
    """Compositing under a [0,1] mask keeps every pixel inside the convex
    hull of (base, resampled tile) up to Lanczos ringing bounds."""
    import torch

    from comfyui_distributed_amd.ops.dispatch import blend_tile

    g = torch.Generator().manual_seed(h * 1000 + w)
    canvas = torch.rand(1, h, w, 3, generator=g)
    before = canvas.clone()
    tile = torch.full((1, oh, ow, 3), 0.5)
    region = (0, 0, w, h)
    blend_tile(canvas, tile, region, (1, 1, max(2, w - 1), max(2, h - 1)), 1.5)
    assert torch.isfinite(canvas).all()
    # uniform tile resamples to exactly 0.5 => blend is lerp(base, 0.5, m)
    lo = torch.minimum(before, torch.full_like(before, 0.5)) - 1e-4
    hi = torch.maximum(before, torch.full_like(before, 0.5)) + 1e-4
    assert (canvas >= lo).all() and (canvas <= hi).all()


@given(st.integers(1, 4), st.integers(1, 32), st.integers(1, 32))
@settings(max_examples=40, deadline=None)
def test_png_codec_roundtrip_exact_uint8(b, h, w):
    """PNG is lossless: any uint8-quantized image survives the wire
    bit-exactly (the collector determinism contract depends on it)."""
    import torch

    from comfyui_distributed_amd.utils.image import (
        decode_png_base64, encode_png_base64)

    g = torch.Generator().manual_seed(b * 7919 + h * 31 + w)
    raw = torch.randint(0, 256, (b, h, w, 3), generator=g, dtype=torch.uint8)
    img = raw.float() / 255.0
    for i in range(b):  # the wire sends one image per envelope
        out = decode_png_base64(encode_png_base64(img[i : i + 1]))
        assert out.shape == (1, h, w, 3)
        assert torch.equal((out * 255.0).round().to(torch.uint8),
                           raw[i : i + 1])


@given(st.integers(1, 2), st.integers(1, 2), st.integers(1, 5000),
       st.sampled_from([8000, 22050, 44100]))
@settings(max_examples=40, deadline=None)
def test_audio_envelope_roundtrip(b, c, n, sr):
    import torch

    from comfyui_distributed_amd.utils.audio import (
        decode_audio_payload, encode_audio_payload)

    g = torch.Generator().manual_seed(n)
    audio = {"waveform": torch.randn(b, c, n, generator=g),
             "sample_rate": sr}
    out = decode_audio_payload(encode_audio_payload(audio))
    assert out["sample_rate"] == sr
    assert torch.equal(out["waveform"], audio["waveform"])


@given(st.integers(8, 256), st.integers(8, 256), st.data())
@settings(max_examples=100, deadline=None)
def test_expand_crop_reaches_target_when_possible(w, h, data):
    from comfyui_distributed_amd.utils.usdu_math import expand_crop

    x1 = data.draw(st.integers(0, w - 2))
    y1 = data.draw(st.integers(0, h - 2))
    x2 = data.draw(st.integers(x1 + 1, w))
    y2 = data.draw(st.integers(y1 + 1, h))
    tw = data.draw(st.integers(x2 - x1, w))
    th = data.draw(st.integers(y2 - y1, h))
    (rx1, ry1, rx2, ry2), _ = expand_crop((x1, y1, x2, y2), w, h, tw, th)
    # stays in bounds and contains the original region
    assert 0 <= rx1 <= x1 and x2 <= rx2 <= w
    assert 0 <= ry1 <= y1 and y2 <= ry2 <= h
    # target fits the canvas -> exact size achieved
    assert rx2 - rx1 == tw and ry2 - ry1 == th


@given(st.integers(1, 512), st.integers(1, 512), st.integers(1, 512),
       st.integers(1, 512), st.data())
@settings(max_examples=100, deadline=None)
def test_resize_region_in_bounds_and_non_degenerate(iw, ih, rw, rh, data):
    from comfyui_distributed_amd.utils.usdu_math import resize_region

    x1 = data.draw(st.integers(0, iw - 1))
    y1 = data.draw(st.integers(0, ih - 1))
    x2 = data.draw(st.integers(x1 + 1, iw))
    y2 = data.draw(st.integers(y1 + 1, ih))
    nx1, ny1, nx2, ny2 = resize_region((x1, y1, x2, y2), (iw, ih), (rw, rh))
    assert 0 <= nx1 < nx2 <= rw
    assert 0 <= ny1 < ny2 <= rh
    # floor/ceil mapping never shrinks a region to nothing and covers the
    # scaled extent
    assert nx1 <= x1 * rw / iw + 1e-9 and nx2 >= x2 * rw / iw - 1e-9


@given(st.text(max_size=40), st.booleans(), st.integers(0, 12))
@settings(max_examples=100, deadline=None)
def test_distributed_value_never_crashes_on_fuzzed_json(raw, is_worker, idx):
    from comfyui_distributed_amd.nodes.utilities import DistributedValue

    out = DistributedValue().distribute(
        "def", raw, is_worker=is_worker, worker_id=f"worker_{idx}")
    assert isinstance(out, tuple) and len(out) == 1


@given(st.dictionaries(
    st.sampled_from(["prompt", "workflow", "client_id", "enabled_worker_ids",
                     "workers", "delegate_master", "auto_prepare",
                     "trace_execution_id", "junk"]),
    st.one_of(st.none(), st.booleans(), st.integers(), st.text(max_size=10),
              st.lists(st.text(max_size=5), max_size=3),
              st.dictionaries(st.text(max_size=4),
                              st.integers(), max_size=3)),
    max_size=6))
@settings(max_examples=150, deadline=None)
def test_queue_request_parser_total(payload):
    """parse_queue_request_payload is total: any JSON-shaped dict either
    parses or raises QueueRequestError — never an uncontrolled exception."""
    from comfyui_distributed_amd.server.queue_request import (
        QueueRequestError, parse_queue_request_payload)

    try:
        out = parse_queue_request_payload(payload)
        assert isinstance(out.prompt, dict)
        assert isinstance(out.enabled_worker_ids, list)
    except QueueRequestError:
        pass


@given(st.dictionaries(
    st.sampled_from(["job_id", "worker_id", "batch_idx", "image", "audio",
                     "is_last", "tiles"]),
    st.one_of(st.none(), st.booleans(), st.integers(-5, 5),
              st.text(max_size=12),
              st.lists(st.dictionaries(
                  st.sampled_from(["tile_idx", "batch_idx", "image"]),
                  st.one_of(st.integers(-2, 2), st.text(max_size=8)),
                  max_size=3), max_size=2)),
    max_size=6))
@settings(max_examples=150, deadline=None)
def test_wire_decoders_total(payload):
    """The two wire decoders raise only (TypeError|ValueError|KeyError) on
    junk — the route layer maps exactly those to 400s."""
    from comfyui_distributed_amd.nodes.collector import (
        decode_job_complete_envelope)
    from comfyui_distributed_amd.server.usdu_http import decode_tile_submission

    for fn in (decode_job_complete_envelope, decode_tile_submission):
        try:
            fn(dict(payload))
        except (TypeError, ValueError, KeyError):
            pass


@given(st.recursive(
    st.one_of(st.none(), st.booleans(), st.integers(-5, 5),
              st.text(max_size=6)),
    lambda children: st.dictionaries(st.text(max_size=6), children,
                                     max_size=4),
    max_leaves=12).filter(lambda v: isinstance(v, dict)))
@settings(max_examples=100, deadline=None)
def test_config_default_merge_properties(data):
    """_merge_defaults: every default key present; unknown user keys
    survive; user scalars win over defaults."""
    from comfyui_distributed_amd.utils.config import (
        DEFAULT_CONFIG, _merge_defaults)

    merged = _merge_defaults(DEFAULT_CONFIG, data)
    for key in DEFAULT_CONFIG:
        assert key in merged
    for key, value in data.items():
        assert key in merged
        if not isinstance(value, dict):
            assert merged[key] == value  # user scalar wins
    # defaults not clobbered when user section is a dict
    if isinstance(data.get("settings"), dict):
        for k in DEFAULT_CONFIG["settings"]:
            assert k in merged["settings"]
