"""CPU-path op tests (the same math the GPU kernels implement; the GPU
numerics tests in test_ops_gpu.py compare the HIP kernels against these)."""


import torch
import torch.nn.functional as F

from comfyui_distributed_amd.ops import dispatch


def test_group_norm_silu_matches_torch():
    x = torch.randn(2, 8, 5, 5)
    w, b = torch.randn(8), torch.randn(8)
    y = dispatch.group_norm_silu(x, 4, w, b, eps=1e-5, silu=True)
    ref = F.silu(F.group_norm(x, 4, w, b, 1e-5))
    assert torch.allclose(y, ref, atol=1e-5)


def test_layer_norm_matches_torch():
    x = torch.randn(3, 7, 32)
    w, b = torch.randn(32), torch.randn(32)
    y = dispatch.layer_norm(x, w, b)
    ref = F.layer_norm(x, (32,), w, b)
    assert torch.allclose(y, ref, atol=1e-5)


def test_act_mul():
    a, b = torch.randn(100), torch.randn(100)
    assert torch.allclose(dispatch.act_mul(a, b), a * F.silu(b), atol=1e-5)
    assert torch.allclose(
        dispatch.act_mul(a, b, gelu=True), a * F.gelu(b, approximate="tanh"), atol=1e-5
    )


def test_attention_matches_sdpa():
    torch.manual_seed(0)
    q = torch.randn(4, 33, 40)
    k = torch.randn(4, 77, 40)
    v = torch.randn(4, 77, 40)
    out = dispatch.attention(q, k, v, heads=2)
    ref = F.scaled_dot_product_attention(q, k, v)
    assert torch.allclose(out, ref, atol=1e-4)


def test_attention_gqa():
    torch.manual_seed(0)
    b, h, hkv, n, d = 2, 4, 2, 16, 8
    q = torch.randn(b * h, n, d)
    k = torch.randn(b * hkv, n, d)
    v = torch.randn(b * hkv, n, d)
    out = dispatch.attention(q, k, v, heads=h, kv_heads=hkv)
    # manual expansion reference
    k_exp = k.reshape(b, hkv, 1, n, d).expand(b, hkv, 2, n, d).reshape(b * h, n, d)
    v_exp = v.reshape(b, hkv, 1, n, d).expand(b, hkv, 2, n, d).reshape(b * h, n, d)
    ref = F.scaled_dot_product_attention(q, k_exp, v_exp)
    assert torch.allclose(out, ref, atol=1e-4)


def test_extract_resize_identity():
    """Identity-size extraction of an interior region reproduces the crop."""
    torch.manual_seed(0)
    src = torch.rand(1, 64, 64, 3)
    out = dispatch.extract_resize(src, (16, 16, 48, 48), 32, 32)
    assert out.shape == (1, 32, 32, 3)
    # Lanczos at scale 1 with aligned grid is an identity (away from edges)
    assert torch.allclose(out[:, 3:-3, 3:-3], src[:, 19:45, 19:45], atol=1e-4)


def test_extract_resize_constant_preserved():
    src = torch.full((1, 32, 32, 3), 0.5)
    out = dispatch.extract_resize(src, (0, 0, 32, 32), 48, 48)
    assert torch.allclose(out, torch.full_like(out, 0.5), atol=1e-4)


def test_rect_mask_properties():
    m = dispatch.rect_mask_cpu(64, 64, (16, 16, 48, 48), sigma=4.0)
    assert m[32, 32] > 0.999  # deep inside
    assert m[0, 0] < 1e-4  # far outside
    assert abs(m[32, 16].item() - 0.5 * m[32, 32].item()) < 0.06  # edge ~ half
    m0 = dispatch.rect_mask_cpu(64, 64, (16, 16, 48, 48), sigma=0.0)
    assert m0[16, 16] == 1.0 and m0[15, 15] == 0.0


def test_blend_tile_full_mask_replaces_region():
    canvas = torch.zeros(1, 64, 64, 3)
    tile = torch.ones(1, 32, 32, 3)
    # sigma 0, mask covers the whole region -> hard replacement
    dispatch.blend_tile(canvas, tile, (16, 16, 48, 48), (16, 16, 48, 48), 0.0)
    assert torch.allclose(canvas[:, 20:44, 20:44], torch.ones(1, 24, 24, 3), atol=1e-4)
    assert torch.all(canvas[:, :16] == 0) and torch.all(canvas[:, 48:] == 0)


def test_blend_tile_soft_seam_monotone():
    canvas = torch.zeros(1, 64, 64, 3)
    tile = torch.ones(1, 32, 32, 3)
    dispatch.blend_tile(canvas, tile, (8, 8, 56, 56), (16, 16, 48, 48), 4.0)
    row = canvas[0, 32, :, 0]
    # ramp up into the rect, plateau ~1 inside
    assert row[32] > 0.99
    assert row[14] < row[16] < row[18] < row[20]
