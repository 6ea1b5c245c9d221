"""Op dispatch: HIP kernels on GPU, reference Python implementations on CPU.

On a GPU host the HIP extension is mandatory — a missing extension raises
``KernelUnavailableError`` instead of silently running eager PyTorch. The
CPU implementations exist for CPU-only test runs and for the reference
numerics the GPU tests compare against; they implement *the same math* as
the kernels (documented per-op).
"""

from __future__ import annotations

import math
import os

import torch
import torch.nn.functional as F

from . import ext

LANCZOS_A = 3
ATTN_DPADS = (48, 64, 96, 128, 160)


def hip_available() -> bool:
    return ext.get_ext(required=False) is not None


def _f32(t: torch.Tensor | None):
    """fp32 view of a (usually bf16) parameter, cached ON the tensor object
    — the GN/conv launchers used to re-cast weight+bias on every call:
    ~27k bf16->f32 copy kernels per flagship canvas (rocprof r02)."""
    if t is None or t.numel() == 0 or t.dtype == torch.float32:
        return t
    c = getattr(t, "_distgpu_f32", None)
    if c is None or c.device != t.device:
        c = t.detach().float().contiguous()
        try:
            t._distgpu_f32 = c
        except Exception:  # non-writable tensor subclass: fall back
            pass
    return c


def _on_gpu(*tensors: torch.Tensor) -> bool:
    return any(t.is_cuda for t in tensors if isinstance(t, torch.Tensor))


# ---------------------------------------------------------------------------
# norms / elementwise
# ---------------------------------------------------------------------------


def group_norm_silu(
    x: torch.Tensor,
    groups: int,
    weight: torch.Tensor,
    bias: torch.Tensor,
    eps: float = 1e-6,
    silu: bool = True,
) -> torch.Tensor:
    if _on_gpu(x):
        if (
            x.dim() == 4
            and x.is_contiguous(memory_format=torch.channels_last)
            and not x.is_contiguous()
        ):
            return group_norm_silu_cl(x, groups, weight, bias, eps, silu)
        return ext.get_ext(True).group_norm_fused(
            x.to(torch.bfloat16), groups, _f32(weight), _f32(bias), eps, silu
        )
    # manual GN (F.group_norm rejects 1-value-per-group shapes, e.g. a
    # batch-1 tensor at 1x1 spatial in deep tiny-config levels)
    b, c = x.shape[0], x.shape[1]
    xf = x.float().reshape(b, groups, -1)
    mean = xf.mean(-1, keepdim=True)
    var = xf.var(-1, unbiased=False, keepdim=True)
    y = ((xf - mean) / (var + eps).sqrt()).reshape(x.shape)
    shape = [1, c] + [1] * (x.dim() - 2)
    y = y * weight.float().view(shape) + bias.float().view(shape)
    if silu:
        y = F.silu(y)
    return y.to(x.dtype)


def layer_norm(
    x: torch.Tensor, weight: torch.Tensor, bias: torch.Tensor, eps: float = 1e-5
) -> torch.Tensor:
    if _on_gpu(x):
        return ext.get_ext(True).layer_norm(x.to(torch.bfloat16), weight, bias, eps)
    y = F.layer_norm(x.float(), (x.shape[-1],), weight.float(), bias.float(), eps)
    return y.to(x.dtype)


def act_mul(a: torch.Tensor, b: torch.Tensor, gelu: bool = False) -> torch.Tensor:
    """a * act(b): SiLU-mul / GEGLU gate."""
    if _on_gpu(a):
        return ext.get_ext(True).act_mul(a.to(torch.bfloat16), b.to(torch.bfloat16), gelu)
    bf = b.float()
    act = F.gelu(bf, approximate="tanh") if gelu else F.silu(bf)
    return (a.float() * act).to(a.dtype)


# ---------------------------------------------------------------------------
# channels-last conv path (hand-written implicit-GEMM MFMA kernels)
# ---------------------------------------------------------------------------


def conv_supported(conv) -> bool:
    """True when the MFMA conv kernel covers this nn.Conv2d."""
    k = conv.kernel_size
    stride_ok = conv.stride == (1, 1) or (
        # stride-2 3x3 (UNet/VAE Downsample) runs on the 256-tile kernel;
        # keeping it off MIOpen also kills its exhaustive-find warmup
        conv.stride == (2, 2) and k == (3, 3) and conv.in_channels % 64 == 0
    )
    # the 256-tile kernel bounds-masks any K_out (the UNet's K=4 out_conv
    # and the VAE's K=3/K=8 edge convs included — MIOpen's heuristic picks
    # for those are pathological); the v1/v2 kernels need K % 16
    kout_ok = (conv.out_channels % 16 == 0
               or (_CONV256 and conv.in_channels % 64 == 0))
    return (
        stride_ok
        and conv.in_channels % 32 == 0
        and kout_ok
        and ((k == (3, 3) and conv.padding == (1, 1))
             or (k == (1, 1) and conv.padding == (0, 0)))
        and conv.dilation == (1, 1)
        and conv.groups == 1
    )


def _repacked_weight(conv) -> torch.Tensor:
    """[K, R*S*C] bf16, (r,s,c) with c innermost — cached on the module."""
    wt = getattr(conv, "_distgpu_wt", None)
    if wt is None or wt.device != conv.weight.device:
        w = conv.weight.detach()  # [K, C, R, S]
        wt = w.permute(0, 2, 3, 1).reshape(w.shape[0], -1).contiguous()
        wt = wt.to(torch.bfloat16)
        conv._distgpu_wt = wt
    return wt


_CONV256 = os.environ.get("DISTGPU_CONV256", "1") == "1"
# hipBLASLt wins the plain-Linear shapes (measured gpurun_out/call3: 368-1614
# TF vs our 205-1009) — the 256-tile GEMM stays available for fused uses and
# future skinny-N tiles but is opt-in for nn.Linear routing
_GEMM256 = os.environ.get("DISTGPU_GEMM256", "0") == "1"
# epilogue residual fusion (ResBlock skip add): measured 1.5% SLOWER than
# the separate eager add on the flagship (same-box A/B, gpurun_out/call17:
# 7.915 vs 8.032 tiles/s — the dependent per-element load makes the
# store-issue-bound epilogue longer than the well-overlapped add pass).
# Kept available for shapes where it may win; off by default.
_FUSE_RES = os.environ.get("DISTGPU_FUSE_RESIDUAL", "0") == "1"


def conv2d_mfma(x: torch.Tensor, conv, fuse_silu: bool = False,
                up2: bool = False,
                residual: torch.Tensor | None = None) -> torch.Tensor:
    """x: NCHW tensor in channels_last memory format (bf16, on GPU) ->
    same layout. Dispatches to the implicit-GEMM NHWC kernel: the 256-tile
    glds template (gemm.hip) when shapes allow, else the v1/v2 kernels.
    ``up2`` fuses a nearest-2x upsample into the conv's tap addressing
    (the upsampled tensor never materializes). ``residual`` (a
    channels_last NCHW tensor of the OUTPUT shape) is added in the
    epilogue — the ResBlock skip connection without its own kernel."""
    assert x.is_cuda
    b, c, h, w = x.shape
    nhwc = x.permute(0, 2, 3, 1)  # view: contiguous when x is channels_last
    if not nhwc.is_contiguous():
        nhwc = nhwc.contiguous()
    wt = _repacked_weight(conv)
    rs = 9 if conv.kernel_size == (3, 3) else 1
    bias = _f32(conv.bias) if conv.bias is not None else torch.empty(0, device=x.device)
    stride = conv.stride[0]
    if residual is not None and not _FUSE_RES:
        return conv2d_mfma(x, conv, fuse_silu, up2) + residual
    if _CONV256 and c % 64 == 0 and (b * h * w >= 256 or up2):
        if residual is not None:
            res = residual.permute(0, 2, 3, 1)
            if not res.is_contiguous():
                res = res.contiguous()
            res = res.to(torch.bfloat16)
        else:
            res = torch.empty(0, device=x.device, dtype=torch.bfloat16)
        y = ext.get_ext(True).conv256_nhwc(
            nhwc.to(torch.bfloat16), wt, bias, res, b, h, w, c,
            conv.out_channels, rs, stride, up2, fuse_silu,
        )
        return y.permute(0, 3, 1, 2)
    assert stride == 1 and not up2, "strided/up2 conv needs the conv256 path"
    if residual is not None:
        return conv2d_mfma(x, conv, fuse_silu) + residual
    y = ext.get_ext(True).conv_nhwc(
        nhwc.to(torch.bfloat16), wt, bias, b, h, w, c, conv.out_channels, rs,
        fuse_silu,
    )
    return y.permute(0, 3, 1, 2)  # NCHW semantic, channels_last storage


def linear_mfma(x: torch.Tensor, weight: torch.Tensor,
                bias: torch.Tensor | None = None,
                fuse_silu: bool = False) -> torch.Tensor:
    """torch-Linear on the 256-tile MFMA GEMM when the shape fills it
    (tokens >= 1024, K % 64 == 0); hipBLASLt otherwise. x [..., K]."""
    K = x.shape[-1]
    M = x.numel() // K
    if (
        _GEMM256
        and _on_gpu(x)
        and M >= 1024
        and K % 64 == 0
        and weight.shape[0] >= 64
    ):
        x2 = x.reshape(M, K)
        if not x2.is_contiguous():
            x2 = x2.contiguous()
        b = bias if bias is not None else torch.empty(0, device=x.device)
        y = ext.get_ext(True).gemm256_bf16(
            x2.to(torch.bfloat16), weight.to(torch.bfloat16), b, fuse_silu)
        return y.reshape(*x.shape[:-1], weight.shape[0])
    y = F.linear(x, weight, bias)
    return F.silu(y) if fuse_silu else y


def conv2d_smallc(x: torch.Tensor, conv, fuse_silu: bool = False) -> torch.Tensor:
    """Stem conv (in_channels <= 8) on the dedicated HIP kernel — MIOpen/CK
    fall back to very slow paths for NHWC C=4. x: channels_last NCHW."""
    assert x.is_cuda and conv.in_channels <= 8
    b, c, h, w = x.shape
    nhwc = x.permute(0, 2, 3, 1)
    if not nhwc.is_contiguous():
        nhwc = nhwc.contiguous()
    wt = _repacked_weight(conv)
    rs = 9 if conv.kernel_size == (3, 3) else 1
    bias = _f32(conv.bias) if conv.bias is not None else torch.empty(0, device=x.device)
    y = ext.get_ext(True).conv_smallc(
        nhwc.to(torch.bfloat16), wt, bias, b, h, w, c, conv.out_channels, rs,
        fuse_silu,
    )
    return y.permute(0, 3, 1, 2)


def group_norm_silu_cl(x: torch.Tensor, groups: int, weight, bias,
                       eps: float = 1e-5, silu: bool = True) -> torch.Tensor:
    """GroupNorm(+SiLU) for channels_last NCHW tensors on GPU."""
    assert x.is_cuda
    nhwc = x.permute(0, 2, 3, 1)
    if not nhwc.is_contiguous():
        nhwc = nhwc.contiguous()
    y = ext.get_ext(True).group_norm_nhwc(
        nhwc.to(torch.bfloat16), groups, _f32(weight), _f32(bias), eps, silu
    )
    return y.permute(0, 3, 1, 2)


# ---------------------------------------------------------------------------
# attention
# ---------------------------------------------------------------------------


def _dpad_for(d: int) -> int:
    for p in ATTN_DPADS:
        if d <= p:
            return p
    raise ValueError(f"head dim {d} unsupported (max {ATTN_DPADS[-1]})")


def attention(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    heads: int,
    kv_heads: int | None = None,
    scale: float | None = None,
) -> torch.Tensor:
    """Multi-head attention over packed heads.

    q: [B*H, Nq, D], k/v: [B*Hkv, Nk, D]; returns [B*H, Nq, D].
    GPU path: the MFMA flash kernel (bf16, fp32 accumulation); dims are
    zero-padded to the kernel contract (D -> D_PAD, Nq -> 64k, Nk -> 32k)
    here, the kernel masks padded keys.
    """
    kv_heads = kv_heads or heads
    d = q.shape[-1]
    scale = scale if scale is not None else 1.0 / math.sqrt(d)
    if _on_gpu(q):
        _dpad_for(d)  # validates the supported range
        o = ext.get_ext(True).attn_fwd(
            q.to(torch.bfloat16).contiguous(),
            k.to(torch.bfloat16).contiguous(),
            v.to(torch.bfloat16).contiguous(),
            heads, kv_heads, k.shape[1], scale,
        )
        return o
    # CPU reference: fp32 math, GQA by repeating kv heads.
    qf, kf, vf = q.float(), k.float(), v.float()
    if kv_heads != heads:
        rep = heads // kv_heads
        b = q.shape[0] // heads
        kf = kf.reshape(b, kv_heads, 1, *kf.shape[1:]).expand(-1, -1, rep, -1, -1)
        kf = kf.reshape(b * heads, *k.shape[1:])
        vf = vf.reshape(b, kv_heads, 1, *vf.shape[1:]).expand(-1, -1, rep, -1, -1)
        vf = vf.reshape(b * heads, *v.shape[1:])
    s = torch.bmm(qf, kf.transpose(1, 2)) * scale
    p = torch.softmax(s, dim=-1)
    return torch.bmm(p, vf).to(q.dtype)


def attention_packed(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    heads: int,
    kv_heads: int | None = None,
    scale: float | None = None,
) -> torch.Tensor:
    """Attention on the packed projection layout: q [B, Nq, H*D],
    k/v [B, Nk, Hkv*D] -> [B, Nq, H*D]. On GPU this feeds the strided
    kernel directly (zero reshapes/copies); on CPU it unpacks and reuses
    the reference path."""
    kv_heads = kv_heads or heads
    d = q.shape[-1] // heads
    scale = scale if scale is not None else 1.0 / math.sqrt(d)
    if _on_gpu(q):
        return ext.get_ext(True).attn_fwd_packed(
            q.to(torch.bfloat16).contiguous(),
            k.to(torch.bfloat16).contiguous(),
            v.to(torch.bfloat16).contiguous(),
            heads, kv_heads, scale,
        )
    b, nq, _ = q.shape
    nk = k.shape[1]

    def split(u, h):
        return (u.reshape(b, -1, h, d).permute(0, 2, 1, 3)
                .reshape(b * h, -1, d))

    o = attention(split(q, heads), split(k, kv_heads), split(v, kv_heads),
                  heads=heads, kv_heads=kv_heads, scale=scale)
    return (o.reshape(b, heads, nq, d).permute(0, 2, 1, 3)
            .reshape(b, nq, heads * d))


def attention_qkv(qkv: torch.Tensor, heads: int,
                  scale: float | None = None) -> torch.Tensor:
    """Self-attention straight off a fused projection [B, N, 3*H*D]."""
    hd = qkv.shape[-1] // 3
    d = hd // heads
    scale = scale if scale is not None else 1.0 / math.sqrt(d)
    if _on_gpu(qkv):
        return ext.get_ext(True).attn_fwd_qkv(
            qkv.to(torch.bfloat16).contiguous(), heads, scale)
    q, k, v = qkv.split(hd, dim=-1)
    return attention_packed(q.contiguous(), k.contiguous(), v.contiguous(),
                            heads=heads, scale=scale)


def attention_q_kv(q: torch.Tensor, kv: torch.Tensor, heads: int,
                   scale: float | None = None) -> torch.Tensor:
    """Cross-attention: q [B,Nq,H*D] + fused kv [B,Nk,2*H*D]."""
    hd = q.shape[-1]
    d = hd // heads
    scale = scale if scale is not None else 1.0 / math.sqrt(d)
    if _on_gpu(q):
        return ext.get_ext(True).attn_fwd_q_kv(
            q.to(torch.bfloat16).contiguous(),
            kv.to(torch.bfloat16).contiguous(), heads, scale)
    k, v = kv.split(hd, dim=-1)
    return attention_packed(q, k.contiguous(), v.contiguous(), heads=heads,
                            scale=scale)


# ---------------------------------------------------------------------------
# tile pipeline (Lanczos-3 resample + erf-mask blend)
# ---------------------------------------------------------------------------


def _lanczos3(x: torch.Tensor) -> torch.Tensor:
    ax = x.abs()
    out = torch.where(
        ax < 1e-6,
        torch.ones_like(x),
        torch.sinc(x) * torch.sinc(x / LANCZOS_A),
    )
    return torch.where(ax >= LANCZOS_A, torch.zeros_like(x), out)


def _lanczos_weight_matrix(
    n_out: int, src_lo: float, scale: float, src_size: int
) -> torch.Tensor:
    """Dense [n_out, src_size] weight matrix matching the kernel's per-pixel
    tap enumeration (inclusive [floor(c-s+.5), floor(c+s+.5)], max 16 taps,
    edge clamp, renormalized)."""
    centers = src_lo + (torch.arange(n_out, dtype=torch.float64) + 0.5) * scale - 0.5
    # clamp like the HIP kernel: the full window must fit in 16 taps, or
    # truncation can leave a near-zero weight sum and the renormalization
    # explodes (downscales > 2.5x)
    fscale = min(max(scale, 1.0), (16 - 1) / (2.0 * LANCZOS_A))
    support = LANCZOS_A * fscale
    x0 = torch.floor(centers - support + 0.5)
    x1 = torch.floor(centers + support + 0.5)
    max_taps = min(int((x1 - x0).max().item()) + 1, 16)
    taps = x0[:, None] + torch.arange(max_taps, dtype=torch.float64)[None, :]
    w = _lanczos3(((taps - centers[:, None]) / fscale).float()).double()
    w = torch.where(taps <= x1[:, None], w, torch.zeros_like(w))
    w = w / w.sum(dim=1, keepdim=True).clamp_min(1e-12)
    idx = taps.long().clamp(0, src_size - 1)
    dense = torch.zeros(n_out, src_size, dtype=torch.float64)
    dense.scatter_add_(1, idx, w)
    return dense.float()


def _lanczos_resample_cpu(
    src: torch.Tensor, x1: int, y1: int, x2: int, y2: int, ow: int, oh: int
) -> torch.Tensor:
    """[B,H,W,C] f32 -> [B,oh,ow,C]: separable Lanczos-3 of region."""
    B, H, W, C = src.shape
    wx = _lanczos_weight_matrix(ow, float(x1), (x2 - x1) / ow, W)  # [ow, W]
    wy = _lanczos_weight_matrix(oh, float(y1), (y2 - y1) / oh, H)  # [oh, H]
    t = src.permute(0, 3, 1, 2).reshape(B * C, H, W).float()
    t = torch.einsum("oh,bhw->bow", wy, t)
    t = torch.einsum("pw,bow->bop", wx, t)
    return t.reshape(B, C, oh, ow).permute(0, 2, 3, 1).contiguous()


def extract_resize(
    src: torch.Tensor, region: tuple[int, int, int, int], ow: int, oh: int
) -> torch.Tensor:
    """Crop ``region`` of [B,H,W,C] f32 and Lanczos-3 resample to (ow, oh)."""
    x1, y1, x2, y2 = (int(r) for r in region)
    if _on_gpu(src):
        return ext.get_ext(True).extract_resize(src.float(), x1, y1, x2, y2, ow, oh)
    return _lanczos_resample_cpu(src.float(), x1, y1, x2, y2, ow, oh)


def rect_mask_cpu(
    h: int,
    w: int,
    rect: tuple[int, int, int, int],
    sigma: float,
    device=None,
) -> torch.Tensor:
    """Analytic Gaussian-blurred white-rect mask [h, w] (erf-product form)."""
    rx1, ry1, rx2, ry2 = (float(r) for r in rect)
    xs = torch.arange(w, dtype=torch.float32, device=device)
    ys = torch.arange(h, dtype=torch.float32, device=device)
    if sigma <= 0:
        gx = ((xs >= rx1) & (xs < rx2)).float()
        gy = ((ys >= ry1) & (ys < ry2)).float()
    else:
        inv_s = 1.0 / (sigma * math.sqrt(2.0))
        gx = 0.5 * (torch.erf((xs + 0.5 - rx1) * inv_s) - torch.erf((xs + 0.5 - rx2) * inv_s))
        gy = 0.5 * (torch.erf((ys + 0.5 - ry1) * inv_s) - torch.erf((ys + 0.5 - ry2) * inv_s))
    return gy[:, None] * gx[None, :]


def blend_tile(
    canvas: torch.Tensor,
    tile: torch.Tensor,
    region: tuple[int, int, int, int],
    mask_rect: tuple[int, int, int, int],
    sigma: float,
) -> None:
    """In-place: resample ``tile`` to ``region`` size and composite into
    ``canvas`` under the analytic blurred-rect mask."""
    x1, y1, x2, y2 = (int(r) for r in region)
    if _on_gpu(canvas):
        ext.get_ext(True).blend_tile(
            canvas, tile.float(), x1, y1, x2, y2,
            int(mask_rect[0]), int(mask_rect[1]), int(mask_rect[2]),
            int(mask_rect[3]), float(sigma),
        )
        return
    rw, rh = x2 - x1, y2 - y1
    t = _lanczos_resample_cpu(tile.float(), 0, 0, tile.shape[2], tile.shape[1], rw, rh)
    m_full = rect_mask_cpu(canvas.shape[1], canvas.shape[2], mask_rect, sigma)
    m = m_full[y1:y2, x1:x2][None, :, :, None]
    patch = canvas[:, y1:y2, x1:x2, :]
    canvas[:, y1:y2, x1:x2, :] = patch * (1 - m) + t * m
