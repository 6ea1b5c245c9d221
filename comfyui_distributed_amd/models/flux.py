"""Flux-family MMDiT image model (rectified flow).

A third image family beyond SD1.5/SDXL (the reference host runs Flux
checkpoints the same way it runs SD ones, incl. Flux-Kontext reference
latents — its usdu_utils.py:445-502 crops them; this framework's
engine/conditioning.py does the same). Architecture follows the public
Flux design: patchified 16-channel latents, double-stream blocks (text
and image streams with joint attention), then single-stream blocks over
the concatenated sequence, multi-axis RoPE, adaLN modulation from
timestep (+ guidance + pooled vec) embeddings.

MI355X mapping: every attention is one `ops.attention_packed` call on the
[B, N, H*D] projection layout — the strided MFMA flash kernel consumes it
with zero repacking; QK RMSNorm and the gated residuals are elementwise
epilogues fused by the dispatch layer where available.
"""

from __future__ import annotations

import math
from dataclasses import dataclass

import torch
from torch import nn

from .. import ops


@dataclass
class FluxConfig:
    dim: int = 3072
    depth_double: int = 19
    depth_single: int = 38
    heads: int = 24
    context_dim: int = 4096
    vec_dim: int = 768
    in_channels: int = 16
    patch: int = 2
    mlp_ratio: float = 4.0
    axes_dim: tuple = (16, 56, 56)  # (txt-id, h, w) RoPE split of head_dim
    guidance_embed: bool = True

    @property
    def head_dim(self) -> int:
        return self.dim // self.heads


FLUX12B = FluxConfig()
# head_dim 64 (dim/heads), matching the attention kernel's validated
# D range (40..160) — a smaller head dim would leave the GPU path
# outside its tested contract
FLUX_TINY = FluxConfig(dim=128, depth_double=1, depth_single=1, heads=2,
                       context_dim=32, vec_dim=16, axes_dim=(16, 24, 24),
                       guidance_embed=False)


def timestep_embedding(t: torch.Tensor, dim: int, max_period=10000.0):
    half = dim // 2
    freqs = torch.exp(
        -math.log(max_period) * torch.arange(half, dtype=torch.float32,
                                             device=t.device) / half)
    args = t.float()[:, None] * freqs[None]
    return torch.cat([torch.cos(args), torch.sin(args)], dim=-1).to(t.dtype)


def rope_freqs(ids: torch.Tensor, axes_dim, theta=10000.0):
    """ids [N, n_axes] integer positions -> (cos, sin) [N, head_dim//2]."""
    outs_c, outs_s = [], []
    for a, d in enumerate(axes_dim):
        half = d // 2
        freqs = 1.0 / (theta ** (torch.arange(half, dtype=torch.float32,
                                              device=ids.device) / half))
        ang = ids[:, a].float()[:, None] * freqs[None]
        outs_c.append(torch.cos(ang))
        outs_s.append(torch.sin(ang))
    return torch.cat(outs_c, dim=-1), torch.cat(outs_s, dim=-1)


def apply_rope_packed(x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor,
                      heads: int) -> torch.Tensor:
    """Rotate pairs (even, odd) of each head's channels. x [B, N, H*D]."""
    b, n, hd = x.shape
    d = hd // heads
    xv = x.view(b, n, heads, d // 2, 2)
    x1, x2 = xv[..., 0], xv[..., 1]
    c = cos[None, :, None, :]
    s = sin[None, :, None, :]
    out = torch.stack([x1 * c - x2 * s, x1 * s + x2 * c], dim=-1)
    return out.reshape(b, n, hd)


class QKNorm(nn.Module):
    """Per-head RMSNorm of q and k (Flux's qk-norm)."""

    def __init__(self, head_dim: int):
        super().__init__()
        self.q_scale = nn.Parameter(torch.ones(head_dim))
        self.k_scale = nn.Parameter(torch.ones(head_dim))

    @staticmethod
    def _rms(x: torch.Tensor, scale: torch.Tensor, heads: int) -> torch.Tensor:
        b, n, hd = x.shape
        d = hd // heads
        xv = x.view(b, n, heads, d).float()
        xv = xv * torch.rsqrt(xv.pow(2).mean(-1, keepdim=True) + 1e-6)
        return (xv * scale.float()).reshape(b, n, hd).to(x.dtype)

    def forward(self, q, k, heads):
        return self._rms(q, self.q_scale, heads), self._rms(k, self.k_scale, heads)


class Modulation(nn.Module):
    def __init__(self, dim: int, n: int):
        super().__init__()
        self.n = n
        self.lin = nn.Linear(dim, n * dim)

    def forward(self, vec):
        return self.lin(nn.functional.silu(vec)).chunk(self.n, dim=-1)


def _mod(x, shift, scale):
    return x * (1 + scale[:, None]) + shift[:, None]


class DoubleStreamBlock(nn.Module):
    """Separate img/txt streams, one joint attention over both."""

    def __init__(self, cfg: FluxConfig):
        super().__init__()
        d, mlp = cfg.dim, int(cfg.dim * cfg.mlp_ratio)
        self.heads = cfg.heads
        for p in ("img", "txt"):
            setattr(self, f"{p}_mod", Modulation(d, 6))
            setattr(self, f"{p}_norm1", nn.LayerNorm(d, elementwise_affine=False))
            setattr(self, f"{p}_qkv", nn.Linear(d, 3 * d))
            setattr(self, f"{p}_qknorm", QKNorm(cfg.head_dim))
            setattr(self, f"{p}_proj", nn.Linear(d, d))
            setattr(self, f"{p}_norm2", nn.LayerNorm(d, elementwise_affine=False))
            setattr(self, f"{p}_mlp", nn.Sequential(
                nn.Linear(d, mlp), nn.GELU(approximate="tanh"), nn.Linear(mlp, d)))

    def _stream(self, prefix, x, vec):
        mod = getattr(self, f"{prefix}_mod")(vec)
        h = _mod(getattr(self, f"{prefix}_norm1")(x), mod[0], mod[1])
        qkv = getattr(self, f"{prefix}_qkv")(h)
        q, k, v = qkv.chunk(3, dim=-1)
        q, k = getattr(self, f"{prefix}_qknorm")(q, k, self.heads)
        return q, k, v, mod

    def forward(self, img, txt, vec, cos, sin):
        nt = txt.shape[1]
        iq, ik, iv, imod = self._stream("img", img, vec)
        tq, tk, tv, tmod = self._stream("txt", txt, vec)
        q = torch.cat([tq, iq], dim=1)
        k = torch.cat([tk, ik], dim=1)
        v = torch.cat([tv, iv], dim=1)
        q = apply_rope_packed(q, cos, sin, self.heads)
        k = apply_rope_packed(k, cos, sin, self.heads)
        attn = ops.attention_packed(q.contiguous(), k.contiguous(),
                                    v.contiguous(), heads=self.heads)
        ta, ia = attn[:, :nt], attn[:, nt:]
        img = img + imod[2][:, None] * self.img_proj(ia)
        img = img + imod[5][:, None] * self.img_mlp(
            _mod(self.img_norm2(img), imod[3], imod[4]))
        txt = txt + tmod[2][:, None] * self.txt_proj(ta)
        txt = txt + tmod[5][:, None] * self.txt_mlp(
            _mod(self.txt_norm2(txt), tmod[3], tmod[4]))
        return img, txt


class SingleStreamBlock(nn.Module):
    """Concatenated sequence; fused linear1 = qkv+mlp_in, linear2 =
    attn_out+mlp_out (Flux's parallel attention+MLP)."""

    def __init__(self, cfg: FluxConfig):
        super().__init__()
        d = cfg.dim
        self.heads = cfg.heads
        self.mlp_dim = int(d * cfg.mlp_ratio)
        self.mod = Modulation(d, 3)
        self.norm = nn.LayerNorm(d, elementwise_affine=False)
        self.linear1 = nn.Linear(d, 3 * d + self.mlp_dim)
        self.qknorm = QKNorm(cfg.head_dim)
        self.linear2 = nn.Linear(d + self.mlp_dim, d)
        self.act = nn.GELU(approximate="tanh")

    def forward(self, x, vec, cos, sin):
        shift, scale, gate = self.mod(vec)
        h = _mod(self.norm(x), shift, scale)
        proj = self.linear1(h)
        d = x.shape[-1]
        q, k, v, mlp = torch.split(proj, [d, d, d, self.mlp_dim], dim=-1)
        q, k = self.qknorm(q, k, self.heads)
        q = apply_rope_packed(q, cos, sin, self.heads)
        k = apply_rope_packed(k, cos, sin, self.heads)
        attn = ops.attention_packed(q.contiguous(), k.contiguous(),
                                    v.contiguous(), heads=self.heads)
        return x + gate[:, None] * self.linear2(
            torch.cat([attn, self.act(mlp)], dim=-1))


class FluxModel(nn.Module):
    def __init__(self, cfg: FluxConfig):
        super().__init__()
        self.cfg = cfg
        d = cfg.dim
        pc = cfg.in_channels * cfg.patch * cfg.patch
        self.img_in = nn.Linear(pc, d)
        self.txt_in = nn.Linear(cfg.context_dim, d)
        self.time_in = nn.Sequential(nn.Linear(256, d), nn.SiLU(), nn.Linear(d, d))
        self.vec_in = nn.Sequential(nn.Linear(cfg.vec_dim, d), nn.SiLU(),
                                    nn.Linear(d, d))
        self.guidance_in = (
            nn.Sequential(nn.Linear(256, d), nn.SiLU(), nn.Linear(d, d))
            if cfg.guidance_embed else None)
        self.double_blocks = nn.ModuleList(
            DoubleStreamBlock(cfg) for _ in range(cfg.depth_double))
        self.single_blocks = nn.ModuleList(
            SingleStreamBlock(cfg) for _ in range(cfg.depth_single))
        self.final_norm = nn.LayerNorm(d, elementwise_affine=False)
        self.final_mod = Modulation(d, 2)
        self.final_proj = nn.Linear(d, pc)

    def _ids(self, nt, h, w, device):
        txt_ids = torch.zeros(nt, 3, dtype=torch.long, device=device)
        ys, xs = torch.meshgrid(
            torch.arange(h, device=device), torch.arange(w, device=device),
            indexing="ij")
        img_ids = torch.stack(
            [torch.zeros_like(ys), ys, xs], dim=-1).reshape(h * w, 3)
        return torch.cat([txt_ids, img_ids])

    def forward(self, x, t, context, vec, guidance=None):
        """x [B, C, H/8, W/8] latents; t [B] in [0,1]*1000; context
        [B, L, Dctx]; vec [B, Dv] pooled conditioning."""
        cfg = self.cfg
        b, c, h, w = x.shape
        p = cfg.patch
        hp, wp = h // p, w // p
        img = x.reshape(b, c, hp, p, wp, p).permute(0, 2, 4, 1, 3, 5)
        img = img.reshape(b, hp * wp, c * p * p)
        img = self.img_in(img)
        txt = self.txt_in(context)
        mod_vec = self.time_in(timestep_embedding(t, 256)) + self.vec_in(vec)
        if self.guidance_in is not None:
            g = guidance if guidance is not None else torch.full(
                (b,), 4.0, device=x.device, dtype=x.dtype)
            mod_vec = mod_vec + self.guidance_in(timestep_embedding(g, 256))
        ids = self._ids(txt.shape[1], hp, wp, x.device)
        cos, sin = rope_freqs(ids, cfg.axes_dim)
        cos, sin = cos.to(x.dtype), sin.to(x.dtype)
        for blk in self.double_blocks:
            img, txt = blk(img, txt, mod_vec, cos, sin)
        seq = torch.cat([txt, img], dim=1)
        for blk in self.single_blocks:
            seq = blk(seq, mod_vec, cos, sin)
        img = seq[:, txt.shape[1]:]
        shift, scale = self.final_mod(mod_vec)
        img = self.final_proj(_mod(self.final_norm(img), shift, scale))
        img = img.reshape(b, hp, wp, c, p, p).permute(0, 3, 1, 4, 2, 5)
        return img.reshape(b, c, h, w)


FLUX_VAE_CHANNELS = 16


class FluxStack:
    """Flux family handle (same surface as DiffusionStack/WanStack where it
    matters: .device/.dtype/.vae/.make_conditioning; family-dispatched in
    engine.generate)."""

    family = "flux"
    context_tokens = 512
    flow_shift = 3.0

    def __init__(self, cfg: FluxConfig, device="cpu", dtype=torch.float32,
                 seed: int = 0, vae_variant=None):
        from .vae import VAE, VAEConfig

        if vae_variant is None:
            vae_variant = (
                VAEConfig(latent_channels=FLUX_VAE_CHANNELS)
                if cfg.dim >= 1024 else
                VAEConfig(latent_channels=FLUX_VAE_CHANNELS, base_channels=8,
                          channel_mult=(1, 1, 2, 2), num_res_blocks=1))
        torch.manual_seed(seed)
        self.cfg = cfg
        self.model = FluxModel(cfg).to(device=device, dtype=dtype).eval()
        self.vae = VAE(vae_variant).to(device=device, dtype=dtype).eval()
        self.device = torch.device(device)
        self.dtype = dtype

    def make_conditioning(self, prompt_seed: int = 0):
        g = torch.Generator().manual_seed(prompt_seed)
        ctx = torch.randn(1, self.context_tokens, self.cfg.context_dim,
                          generator=g).to(self.device, self.dtype)
        vec = torch.randn(1, self.cfg.vec_dim, generator=g).to(
            self.device, self.dtype)
        return {"context": ctx, "vec": vec}
