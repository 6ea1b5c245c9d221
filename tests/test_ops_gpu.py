"""GPU numerics: HIP kernels vs plain fp32 PyTorch references.

All tests are gpu-marked; they fail loudly (KernelUnavailableError) if the
extension is missing on a GPU box — the HIP path must be the one that runs.
"""


import pytest
import torch
import torch.nn.functional as F

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def extmod():
    from comfyui_distributed_amd.ops import ext

    return ext.get_ext(required=True)


def test_mfma_fragment_layout(extmod):
    """C = A @ B with ASYMMETRIC operands (transpose-detecting)."""
    torch.manual_seed(0)
    a = torch.randn(16, 32).to(torch.bfloat16)
    b = torch.randn(32, 16).to(torch.bfloat16)
    bt = b.t().contiguous()  # kernel takes B^T storage
    c = extmod.mfma_selftest(a.cuda(), bt.cuda()).cpu()
    ref = a.float() @ b.float()
    assert torch.allclose(c, ref, atol=0.05, rtol=0.02), (
        (c - ref).abs().max().item()
    )


@pytest.mark.parametrize("d", [40, 64, 80, 128, 160])
@pytest.mark.parametrize("nq,nk", [(64, 64), (77, 77), (1156, 77), (1024, 1024)])
def test_attention_numerics(extmod, d, nq, nk):
    from comfyui_distributed_amd.ops import dispatch

    torch.manual_seed(d * 1000 + nq)
    bh = 4
    q = torch.randn(bh, nq, d) / 2
    k = torch.randn(bh, nk, d) / 2
    v = torch.randn(bh, nk, d)
    out_gpu = dispatch.attention(
        q.cuda().to(torch.bfloat16), k.cuda().to(torch.bfloat16),
        v.cuda().to(torch.bfloat16), heads=2,
    ).float().cpu()
    ref = dispatch.attention(q, k, v, heads=2)  # CPU fp32 reference
    err = (out_gpu - ref).abs().max().item()
    assert err < 0.03, f"max err {err} at d={d} nq={nq} nk={nk}"


def test_attention_gqa_gpu(extmod):
    from comfyui_distributed_amd.ops import dispatch

    torch.manual_seed(3)
    b, h, hkv, n, dh = 2, 8, 2, 128, 64
    q = torch.randn(b * h, n, dh) / 2
    k = torch.randn(b * hkv, n, dh) / 2
    v = torch.randn(b * hkv, n, dh)
    out = dispatch.attention(
        q.cuda().to(torch.bfloat16), k.cuda().to(torch.bfloat16),
        v.cuda().to(torch.bfloat16), heads=h, kv_heads=hkv,
    ).float().cpu()
    ref = dispatch.attention(q, k, v, heads=h, kv_heads=hkv)
    assert (out - ref).abs().max().item() < 0.03


@pytest.mark.parametrize("b,h,n,nk,d", [(2, 8, 640, 640, 40), (1, 4, 100, 77, 64),
                                        (2, 2, 256, 256, 80)])
def test_attention_packed_matches_reference(extmod, b, h, n, nk, d):
    """Packed [B,N,H*D] strided path vs CPU reference."""
    from comfyui_distributed_amd.ops import dispatch

    torch.manual_seed(b * 100 + d)
    q = torch.randn(b, n, h * d) / 2
    k = torch.randn(b, nk, h * d) / 2
    v = torch.randn(b, nk, h * d)
    out = dispatch.attention_packed(
        q.cuda().to(torch.bfloat16), k.cuda().to(torch.bfloat16),
        v.cuda().to(torch.bfloat16), heads=h,
    ).float().cpu()
    ref = dispatch.attention_packed(q, k, v, heads=h)
    err = (out - ref).abs().max().item()
    assert err < 0.03, f"packed err {err} b={b} h={h} n={n} d={d}"


def test_attention_fused_qkv_matches_reference(extmod):
    from comfyui_distributed_amd.ops import dispatch

    torch.manual_seed(11)
    b, h, n, d = 2, 8, 200, 40
    qkv = torch.randn(b, n, 3 * h * d) / 2
    out = dispatch.attention_qkv(qkv.cuda().to(torch.bfloat16), heads=h)
    ref = dispatch.attention_qkv(qkv, heads=h)
    assert (out.float().cpu() - ref).abs().max().item() < 0.03


def test_attention_fused_q_kv_matches_reference(extmod):
    from comfyui_distributed_amd.ops import dispatch

    torch.manual_seed(12)
    b, h, nq, nk, d = 2, 4, 150, 77, 64
    q = torch.randn(b, nq, h * d) / 2
    kv = torch.randn(b, nk, 2 * h * d) / 2
    out = dispatch.attention_q_kv(q.cuda().to(torch.bfloat16),
                                  kv.cuda().to(torch.bfloat16), heads=h)
    ref = dispatch.attention_q_kv(q, kv, heads=h)
    assert (out.float().cpu() - ref).abs().max().item() < 0.03


def test_conv_smallc_matches_torch(extmod):
    import torch.nn.functional as F

    from comfyui_distributed_amd.ops import dispatch

    torch.manual_seed(2)
    conv = torch.nn.Conv2d(4, 320, 3, padding=1).cuda().to(torch.bfloat16)
    x = (torch.randn(2, 4, 17, 19) / 2).cuda().to(torch.bfloat16)
    xcl = x.contiguous(memory_format=torch.channels_last)
    y = dispatch.conv2d_smallc(xcl, conv).float()
    ref = F.conv2d(x.float(), conv.weight.float(), conv.bias.float(), padding=1)
    err = (y - ref).abs().max().item()
    assert err / ref.abs().max().item() < 0.05


def test_attention_defer_max_rescale_branch(extmod):
    """Force the defer-max rescale branch (guide T13 hazard): a spiked K row
    in a LATER tile makes the running max jump past the threshold. Results
    must still match the exact CPU softmax."""
    from comfyui_distributed_amd.ops import dispatch

    torch.manual_seed(9)
    bh, n, d = 2, 256, 64
    q = torch.randn(bh, n, d) / 4
    k = torch.randn(bh, n, d) / 4
    v = torch.randn(bh, n, d)
    # tile size is 64: spike keys in tiles 2 and 3 so m jumps mid-stream
    k[:, 140] = q[:, 17] * 40.0  # huge dot for row 17 at tile 2
    k[:, 200] = q[:, 33] * 60.0  # even bigger at tile 3
    out = dispatch.attention(
        q.cuda().to(torch.bfloat16), k.cuda().to(torch.bfloat16),
        v.cuda().to(torch.bfloat16), heads=1,
    ).float().cpu()
    ref = dispatch.attention(q, k, v, heads=1)
    err = (out - ref).abs().max().item()
    assert err < 0.05, f"defer-max branch mismatch {err}"


def test_attention_softmax_rows_sum(extmod):
    """Uniform V exposes softmax normalization errors: out must equal V."""
    from comfyui_distributed_amd.ops import dispatch

    q = torch.randn(2, 100, 64).cuda().to(torch.bfloat16)
    k = torch.randn(2, 100, 64).cuda().to(torch.bfloat16)
    v = torch.ones(2, 100, 64).cuda().to(torch.bfloat16)
    out = dispatch.attention(q, k, v, heads=1).float()
    assert (out - 1.0).abs().max().item() < 0.01


@pytest.mark.parametrize("shape,groups", [((2, 320, 68, 68), 32), ((1, 512, 64, 64), 32), ((2, 33, 7, 7), 3)])
def test_groupnorm_silu_numerics(extmod, shape, groups):
    torch.manual_seed(1)
    x = torch.randn(*shape)
    w = torch.randn(shape[1])
    b = torch.randn(shape[1])
    y = extmod.group_norm_fused(
        x.cuda().to(torch.bfloat16), groups, w.cuda(), b.cuda(), 1e-5, True
    ).float().cpu()
    ref = F.silu(F.group_norm(x, groups, w, b, 1e-5))
    # bf16 output quantization: |err| <= atol + bf16-relative term
    bound = 0.02 + 0.01 * ref.abs()
    assert ((y - ref).abs() <= bound).all(), (y - ref).abs().max().item()


def test_groupnorm_no_silu(extmod):
    x = torch.randn(2, 64, 32, 32)
    w, b = torch.ones(64), torch.zeros(64)
    y = extmod.group_norm_fused(
        x.cuda().to(torch.bfloat16), 32, w.cuda(), b.cuda(), 1e-5, False
    ).float().cpu()
    ref = F.group_norm(x, 32, w, b, 1e-5)
    assert (y - ref).abs().max().item() < 0.05


@pytest.mark.parametrize("t,c", [(4624, 320), (77, 768), (100, 1280), (3, 2048)])
def test_layernorm_numerics(extmod, t, c):
    torch.manual_seed(2)
    x = torch.randn(t, c)
    w, b = torch.randn(c), torch.randn(c)
    y = extmod.layer_norm(x.cuda().to(torch.bfloat16), w.cuda(), b.cuda(), 1e-5)
    ref = F.layer_norm(x, (c,), w, b, 1e-5)
    assert (y.float().cpu() - ref).abs().max().item() < 0.08


def test_act_mul_numerics(extmod):
    a = torch.randn(1000)
    b = torch.randn(1000)
    y = extmod.act_mul(a.cuda().to(torch.bfloat16), b.cuda().to(torch.bfloat16), False)
    assert (y.float().cpu() - a * F.silu(b)).abs().max().item() < 0.05
    y = extmod.act_mul(a.cuda().to(torch.bfloat16), b.cuda().to(torch.bfloat16), True)
    assert (y.float().cpu() - a * F.gelu(b, approximate="tanh")).abs().max().item() < 0.05


def test_extract_resize_matches_cpu(extmod):
    from comfyui_distributed_amd.ops import dispatch

    torch.manual_seed(4)
    src = torch.rand(2, 128, 128, 3)
    for region, ow, oh in [((16, 16, 80, 80), 96, 96), ((0, 0, 128, 128), 64, 64),
                           ((100, 100, 128, 128), 40, 40)]:
        gpu = dispatch.extract_resize(src.cuda(), region, ow, oh).cpu()
        cpu = dispatch.extract_resize(src, region, ow, oh)
        err = (gpu - cpu).abs().max().item()
        assert err < 2e-3, f"{region} err {err}"


def test_blend_tile_matches_cpu(extmod):
    from comfyui_distributed_amd.ops import dispatch

    torch.manual_seed(5)
    canvas = torch.rand(1, 96, 96, 3)
    tile = torch.rand(1, 48, 48, 3)
    region = (16, 16, 60, 60)
    rect = (24, 24, 56, 56)
    cg = canvas.cuda().contiguous()
    dispatch.blend_tile(cg, tile.cuda(), region, rect, 4.0)
    cc = canvas.clone()
    dispatch.blend_tile(cc, tile, region, rect, 4.0)
    err = (cg.cpu() - cc).abs().max().item()
    assert err < 2e-3, err


def test_native_extension_is_loaded(extmod):
    """Paper trail: the in-tree .so is what's loaded."""
    from comfyui_distributed_amd.ops import ext

    assert ext.SO_PATH.exists()
    assert ext.get_ext(required=True) is not None
