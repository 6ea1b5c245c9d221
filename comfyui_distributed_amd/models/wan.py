"""WAN-2.2-family video diffusion transformer (t2v DiT).

The reference distributes WAN video workflows as opaque ComfyUI graphs
(README.md:98-112, workflows/distributed-wan.json: Collector +
ImageBatchDivider per-segment). This framework provides the model family
natively: a 3D-patchified DiT with per-block adaLN modulation, 3D RoPE
self-attention (the MFMA flash kernel, head_dim 128), cross-attention to
text context, RMSNorm, and GEGLU-free linear-SiLU FFN.

Config parity: ``wan14b`` matches WAN-2.2 14B dims (dim 5120, 40 blocks,
40 heads x 128, ffn 13824, ctx 4096, 16 latent channels, patch (1,2,2)).
``wan_tiny`` exists for CPU tests. Weights random-init (BASELINE.json).
The paired video VAE here is the image VAE applied per frame with x8
spatial compression (temporal compression deferred; documented design
simplification for round 1).
"""

from __future__ import annotations

import math
from dataclasses import dataclass

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops import dispatch as ops


@dataclass
class WanConfig:
    dim: int = 5120
    ffn_dim: int = 13824
    num_layers: int = 40
    num_heads: int = 40  # head_dim = dim // heads = 128
    text_dim: int = 4096
    in_channels: int = 16
    out_channels: int = 16
    patch: tuple = (1, 2, 2)  # (t, h, w)
    eps: float = 1e-6


WAN14B = WanConfig()
WAN_TINY = WanConfig(dim=64, ffn_dim=128, num_layers=2, num_heads=2,
                     text_dim=32, in_channels=4, out_channels=4)


class RMSNorm(nn.Module):
    def __init__(self, dim: int, eps: float = 1e-6):
        super().__init__()
        self.eps = eps
        self.weight = nn.Parameter(torch.ones(dim))

    def forward(self, x):
        xf = x.float()
        y = xf * torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + self.eps)
        return (y * self.weight.float()).to(x.dtype)


def rope_3d(t: int, h: int, w: int, head_dim: int, device, dtype=torch.float32):
    """3D rotary embeddings: head_dim split into (t, h, w) bands
    (2/6, 2/6, 2/6 of the dim pairs, t gets the remainder)."""
    n_pairs = head_dim // 2
    ph = pw = n_pairs // 3
    pt = n_pairs - ph - pw

    def band(n_pos, pairs):
        freqs = 1.0 / (10000.0 ** (torch.arange(pairs, device=device,
                                                dtype=torch.float64) / pairs))
        ang = torch.outer(torch.arange(n_pos, device=device,
                                       dtype=torch.float64), freqs)
        return ang

    at = band(t, pt)[:, None, None, :].expand(t, h, w, pt)
    ah = band(h, ph)[None, :, None, :].expand(t, h, w, ph)
    aw = band(w, pw)[None, None, :, :].expand(t, h, w, pw)
    ang = torch.cat([at, ah, aw], dim=-1).reshape(t * h * w, n_pairs)
    return ang.cos().to(dtype), ang.sin().to(dtype)


def apply_rope(x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor):
    """x: [B, N, H, D] (head-split view); rotate (even, odd) pairs of D."""
    x1, x2 = x[..., 0::2], x[..., 1::2]
    c = cos[None, : x.shape[1], None, :].to(x.dtype)
    s = sin[None, : x.shape[1], None, :].to(x.dtype)
    out = torch.empty_like(x)
    out[..., 0::2] = x1 * c - x2 * s
    out[..., 1::2] = x1 * s + x2 * c
    return out


class WanBlock(nn.Module):
    def __init__(self, cfg: WanConfig):
        super().__init__()
        d = cfg.dim
        self.heads = cfg.num_heads
        self.head_dim = d // cfg.num_heads
        self.norm1 = RMSNorm(d, cfg.eps)
        self.q = nn.Linear(d, d)
        self.k = nn.Linear(d, d)
        self.v = nn.Linear(d, d)
        self.o = nn.Linear(d, d)
        self.norm_q = RMSNorm(self.head_dim, cfg.eps)
        self.norm_k = RMSNorm(self.head_dim, cfg.eps)
        self.norm2 = RMSNorm(d, cfg.eps)
        self.cq = nn.Linear(d, d)
        self.ck = nn.Linear(cfg.text_dim, d)
        self.cv = nn.Linear(cfg.text_dim, d)
        self.co = nn.Linear(d, d)
        self.norm3 = RMSNorm(d, cfg.eps)
        self.ffn1 = nn.Linear(d, cfg.ffn_dim)
        self.ffn2 = nn.Linear(cfg.ffn_dim, d)
        # adaLN modulation: 6 gates/shifts/scales from the time embedding
        self.mod = nn.Parameter(torch.randn(6, d) / d**0.5)

    def forward(self, x, emb6, context, rope_cs):
        # emb6: [B, 6, dim] time modulation (per WAN: shared table + time MLP)
        b, n, _ = x.shape
        m = (emb6 + self.mod[None]).to(x.dtype)  # [B, 6, d]
        shift_a, scale_a, gate_a, shift_f, scale_f, gate_f = m.unbind(1)

        h = self.norm1(x) * (1 + scale_a[:, None]) + shift_a[:, None]
        # head-split VIEW for per-head RMSNorm + RoPE, then back to the
        # packed [B, N, H*D] layout the strided attention kernel reads
        hs = (b, n, self.heads, self.head_dim)
        cos, sin = rope_cs
        q = apply_rope(self.norm_q(self.q(h).reshape(hs)), cos, sin).reshape(b, n, -1)
        k = apply_rope(self.norm_k(self.k(h).reshape(hs)), cos, sin).reshape(b, n, -1)
        attn = ops.attention_packed(q.contiguous(), k.contiguous(), self.v(h),
                                    heads=self.heads)
        x = x + gate_a[:, None] * self.o(attn)

        # cross attention (no modulation per WAN): packed, zero reshapes
        h = self.norm2(x)
        attn = ops.attention_packed(self.cq(h), self.ck(context),
                                    self.cv(context), heads=self.heads)
        x = x + self.co(attn)

        h = self.norm3(x) * (1 + scale_f[:, None]) + shift_f[:, None]
        x = x + gate_f[:, None] * self.ffn2(F.silu(self.ffn1(h)))
        return x


class WanModel(nn.Module):
    def __init__(self, cfg: WanConfig = WAN14B):
        super().__init__()
        self.cfg = cfg
        pt, ph, pw = cfg.patch
        self.patch_embed = nn.Conv3d(cfg.in_channels, cfg.dim,
                                     kernel_size=cfg.patch, stride=cfg.patch)
        self.text_proj = nn.Sequential(
            nn.Linear(cfg.text_dim, cfg.text_dim), nn.GELU(),
            nn.Linear(cfg.text_dim, cfg.text_dim),
        )
        self.time_mlp = nn.Sequential(
            nn.Linear(256, cfg.dim), nn.SiLU(), nn.Linear(cfg.dim, cfg.dim)
        )
        self.time_mod = nn.Linear(cfg.dim, cfg.dim * 6)
        self.blocks = nn.ModuleList(WanBlock(cfg) for _ in range(cfg.num_layers))
        self.norm_out = RMSNorm(cfg.dim, cfg.eps)
        self.head = nn.Linear(cfg.dim, cfg.out_channels * pt * ph * pw)

    def forward(self, x, timesteps, context):
        """x: [B, C, T, H, W]; context: [B, L, text_dim]."""
        cfg = self.cfg
        dtype = self.head.weight.dtype
        x = x.to(dtype)
        context = self.text_proj(context.to(dtype))
        b, c, t, hh, ww = x.shape
        tokens = self.patch_embed(x)  # [B, dim, t', h', w']
        tp, hp, wp = tokens.shape[2:]
        tokens = tokens.flatten(2).transpose(1, 2)  # [B, N, dim]

        # sinusoidal(256) -> MLP -> 6-way modulation
        half = 128
        freqs = torch.exp(
            -math.log(10000.0)
            * torch.arange(half, device=x.device, dtype=torch.float32) / half
        )
        args = timesteps.float()[:, None] * freqs[None]
        temb = torch.cat([torch.cos(args), torch.sin(args)], dim=-1).to(dtype)
        temb = self.time_mlp(temb)
        emb6 = self.time_mod(F.silu(temb)).reshape(b, 6, cfg.dim)

        head_dim = cfg.dim // cfg.num_heads
        cos, sin = rope_3d(tp, hp, wp, head_dim, x.device)
        for blk in self.blocks:
            tokens = blk(tokens, emb6, context, (cos, sin))
        tokens = self.norm_out(tokens)
        out = self.head(tokens)  # [B, N, C*pt*ph*pw]
        pt_, ph_, pw_ = cfg.patch
        out = out.reshape(b, tp, hp, wp, cfg.out_channels, pt_, ph_, pw_)
        out = out.permute(0, 4, 1, 5, 2, 6, 3, 7).reshape(
            b, cfg.out_channels, tp * pt_, hp * ph_, wp * pw_
        )
        return out
