from comfyui_distributed_amd.utils import usdu_math as m


def test_calculate_tiles_grid():
    tiles = m.calculate_tiles(1024, 768, 512, 512)
    assert m.tile_grid_shape(1024, 768, 512, 512) == (2, 2)
    assert tiles == [(0, 0), (512, 0), (0, 512), (512, 512)]


def test_calculate_tiles_uneven():
    tiles = m.calculate_tiles(1000, 500, 512, 512)
    assert tiles == [(0, 0), (512, 0)]


def test_crop_region_padding_and_clamp():
    region = m.get_crop_region((0, 0, 512, 512), 2048, 2048, 32)
    assert region == (0, 0, 544, 544)
    region = m.get_crop_region((1536, 1536, 2048, 2048), 2048, 2048, 32)
    assert region == (1504, 1504, 2048, 2048)


def test_fix_crop_region_one_pixel_rule():
    # interior region loses one pixel on the far edges
    assert m.fix_crop_region((0, 0, 544, 544), 2048, 2048) == (0, 0, 543, 543)
    # edges touching the canvas border stay
    assert m.fix_crop_region((1504, 1504, 2048, 2048), 2048, 2048) == (
        1504, 1504, 2048, 2048,
    )


def test_expand_crop_centered_growth():
    (region, size) = m.expand_crop((100, 100, 200, 200), 1000, 1000, 160, 160)
    assert size == (160, 160)
    x1, y1, x2, y2 = region
    assert (x2 - x1, y2 - y1) == (160, 160)
    # grows half right/bottom first, then left/top
    assert region == (70, 70, 230, 230)


def test_expand_crop_at_border():
    (region, _) = m.expand_crop((960, 960, 1000, 1000), 1000, 1000, 160, 160)
    assert region == (840, 840, 1000, 1000)
    (region, _) = m.expand_crop((0, 0, 40, 40), 1000, 1000, 160, 160)
    assert region == (0, 0, 160, 160)


def test_expand_to_aspect_widens_short_side():
    region = m.expand_to_aspect((0, 0, 200, 100), 1000, 1000, 100, 100)
    x1, y1, x2, y2 = region
    assert (x2 - x1) == (y2 - y1) == 200


def test_processing_size_round8():
    assert m.processing_size(512, 512, 32) == (544, 544)
    assert m.processing_size(500, 500, 30) == (536, 536)


def test_plan_tiles_4k_canvas():
    plans = m.plan_tiles(4096, 4096, 512, 512, 32)
    assert len(plans) == 64
    p0 = plans[0]
    assert p0.process_size == (544, 544)
    x1, y1, x2, y2 = p0.crop_region
    assert (x2 - x1, y2 - y1) == (544, 544)
    # every crop region must be exactly the processing size and in bounds
    for p in plans:
        x1, y1, x2, y2 = p.crop_region
        assert (x2 - x1, y2 - y1) == (544, 544)
        assert 0 <= x1 <= x2 <= 4096 and 0 <= y1 <= y2 <= 4096


def test_resize_region_scales_and_clamps():
    assert m.resize_region((10, 10, 20, 20), (100, 100), (200, 200)) == (20, 20, 40, 40)
    assert m.resize_region((0, 0, 33, 33), (100, 100), (50, 50)) == (0, 0, 17, 17)
