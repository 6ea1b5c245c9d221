"""Diffusion UNet (SD1.5 / SDXL architecture families), built on the gfx950
op set: fused GroupNorm+SiLU (HIP), MFMA flash attention (HIP), GEGLU
act-mul (HIP); convolutions go through torch/MIOpen.

The reference never opens the model — it calls ComfyUI's common_ksampler
(upscale/tile_ops.py:225-229). Here the UNet is part of the
framework. Weights are random-init (BASELINE.json: synthetic latents,
random-init UNet+VAE); the architectures match the named families so the
benchmark shapes are honest.
"""

from __future__ import annotations

import math
from dataclasses import dataclass

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops import dispatch as ops


@dataclass
class UNetConfig:
    in_channels: int = 4
    model_channels: int = 320
    out_channels: int = 4
    num_res_blocks: int = 2
    channel_mult: tuple = (1, 2, 4, 4)
    attn_levels: tuple = (0, 1, 2)
    transformer_depth: tuple = (1, 1, 1, 0)  # per level
    context_dim: int = 768
    num_heads: int = 8          # used when head_dim is None (SD1.5 style)
    head_dim: int | None = None  # SDXL style: fixed head dim
    adm_in_channels: int | None = None  # SDXL vector conditioning


SD15_UNET = UNetConfig()
SDXL_UNET = UNetConfig(
    model_channels=320,
    channel_mult=(1, 2, 4),
    attn_levels=(1, 2),
    transformer_depth=(0, 2, 10),
    context_dim=2048,
    head_dim=64,
    adm_in_channels=2816,
)


def timestep_embedding(t: torch.Tensor, dim: int, max_period: float = 10000.0):
    half = dim // 2
    freqs = torch.exp(
        -math.log(max_period) * torch.arange(half, dtype=torch.float32, device=t.device) / half
    )
    args = t.float()[:, None] * freqs[None]
    return torch.cat([torch.cos(args), torch.sin(args)], dim=-1).to(t.dtype)


class MfmaLinear(nn.Linear):
    """nn.Linear routed through the 256-tile MFMA GEMM on GPU (gemm.hip);
    parameter layout/init identical to nn.Linear so seeded CPU/GPU model
    construction stays in lockstep."""

    def forward(self, x):
        return ops.linear_mfma(x, self.weight, self.bias)


class MfmaConv2d(nn.Conv2d):
    """nn.Conv2d routed through the implicit-GEMM MFMA kernels for the
    supported shapes (stride 1, 3x3 pad1 / 1x1, C%32==0); MIOpen
    otherwise. Replaces the round-1 MIOpen igemm path (~33% of flagship
    kernel time, profiles/r01_prof_final_summary.csv)."""

    def forward(self, x):
        if (
            x.is_cuda
            and x.is_contiguous(memory_format=torch.channels_last)
            and ops.conv_supported(self)
        ):
            return ops.conv2d_mfma(x, self)
        return super().forward(x)


class FusedGroupNorm(nn.Module):
    """GroupNorm with optional fused SiLU via the HIP kernel."""

    def __init__(self, channels: int, groups: int = 32, silu: bool = False):
        super().__init__()
        if channels % groups != 0:  # tiny test configs
            groups = math.gcd(channels, groups)
        self.groups = groups
        self.silu = silu
        self.weight = nn.Parameter(torch.ones(channels))
        self.bias = nn.Parameter(torch.zeros(channels))

    def forward(self, x):
        return ops.group_norm_silu(x, self.groups, self.weight, self.bias, 1e-5, self.silu)


class FusedLayerNorm(nn.Module):
    def __init__(self, dim: int):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(dim))
        self.bias = nn.Parameter(torch.zeros(dim))

    def forward(self, x):
        return ops.layer_norm(x, self.weight, self.bias, 1e-5)


class ResBlock(nn.Module):
    def __init__(self, channels: int, emb_dim: int, out_channels: int | None = None):
        super().__init__()
        out_channels = out_channels or channels
        self.norm1 = FusedGroupNorm(channels, silu=True)
        self.conv1 = MfmaConv2d(channels, out_channels, 3, padding=1)
        self.emb_proj = nn.Linear(emb_dim, out_channels)
        self.norm2 = FusedGroupNorm(out_channels, silu=True)
        self.conv2 = MfmaConv2d(out_channels, out_channels, 3, padding=1)
        self.skip = (
            MfmaConv2d(channels, out_channels, 1) if out_channels != channels else nn.Identity()
        )

    def forward(self, x, emb):
        h = self.conv1(self.norm1(x))
        h = h + self.emb_proj(F.silu(emb))[:, :, None, None]
        h2 = self.norm2(h)
        skip = self.skip(x)
        if (h2.is_cuda and h2.is_contiguous(memory_format=torch.channels_last)
                and ops.conv_supported(self.conv2)):
            return ops.conv2d_mfma(h2, self.conv2, residual=skip)
        return self.conv2(h2) + skip


class CrossAttention(nn.Module):
    def __init__(self, dim: int, context_dim: int | None, heads: int, head_dim: int):
        super().__init__()
        inner = heads * head_dim
        context_dim = context_dim or dim
        self.heads = heads
        self.head_dim = head_dim
        self.to_q = MfmaLinear(dim, inner, bias=False)
        self.to_k = MfmaLinear(context_dim, inner, bias=False)
        self.to_v = MfmaLinear(context_dim, inner, bias=False)
        self.to_out = MfmaLinear(inner, dim)

    def _fused_weight(self, name: str, parts: list[torch.Tensor]) -> torch.Tensor:
        cached = getattr(self, name, None)
        if cached is None or cached.device != parts[0].device \
                or cached.dtype != parts[0].dtype:
            cached = torch.cat([p.detach() for p in parts], dim=0)
            object.__setattr__(self, name, cached)
        return cached

    def forward(self, x, context=None):
        # fused projections + strided attention: one GEMM feeds the kernel
        # directly — zero splits, pads or permutes on the host
        if context is None:  # self-attention: fused QKV
            w = self._fused_weight(
                "_wqkv", [self.to_q.weight, self.to_k.weight, self.to_v.weight]
            )
            o = ops.attention_qkv(ops.linear_mfma(x, w), heads=self.heads)
        else:  # cross-attention: fused KV
            w = self._fused_weight("_wkv", [self.to_k.weight, self.to_v.weight])
            o = ops.attention_q_kv(
                self.to_q(x), ops.linear_mfma(context, w), heads=self.heads
            )
        return self.to_out(o)


class GEGLUFeedForward(nn.Module):
    def __init__(self, dim: int, mult: int = 4):
        super().__init__()
        inner = dim * mult
        self.proj_in = MfmaLinear(dim, inner * 2)
        self.proj_out = MfmaLinear(inner, dim)

    def forward(self, x):
        a, gate = self.proj_in(x).chunk(2, dim=-1)
        return self.proj_out(ops.act_mul(a, gate, gelu=True))


class TransformerBlock(nn.Module):
    def __init__(self, dim: int, context_dim: int, heads: int, head_dim: int):
        super().__init__()
        self.norm1 = FusedLayerNorm(dim)
        self.attn1 = CrossAttention(dim, None, heads, head_dim)
        self.norm2 = FusedLayerNorm(dim)
        self.attn2 = CrossAttention(dim, context_dim, heads, head_dim)
        self.norm3 = FusedLayerNorm(dim)
        self.ff = GEGLUFeedForward(dim)

    def forward(self, x, context):
        x = x + self.attn1(self.norm1(x))
        x = x + self.attn2(self.norm2(x), context)
        x = x + self.ff(self.norm3(x))
        return x


class SpatialTransformer(nn.Module):
    def __init__(self, channels: int, context_dim: int, heads: int, head_dim: int,
                 depth: int = 1):
        super().__init__()
        self.norm = FusedGroupNorm(channels, silu=False)
        self.proj_in = MfmaLinear(channels, channels)
        self.blocks = nn.ModuleList(
            [TransformerBlock(channels, context_dim, heads, head_dim) for _ in range(depth)]
        )
        self.proj_out = MfmaLinear(channels, channels)

    def forward(self, x, context):
        b, c, h, w = x.shape
        residual = x
        t = self.norm(x).permute(0, 2, 3, 1).reshape(b, h * w, c)
        t = self.proj_in(t)
        for blk in self.blocks:
            t = blk(t, context)
        t = self.proj_out(t)
        return residual + t.reshape(b, h, w, c).permute(0, 3, 1, 2)


class StemConv(nn.Module):
    """conv_in (C<=8): dedicated HIP kernel on GPU channels_last inputs
    (MIOpen/CK have no fast NHWC solver for tiny C — see profiles/)."""

    def __init__(self, cin: int, cout: int):
        super().__init__()
        self.conv = nn.Conv2d(cin, cout, 3, padding=1)

    def forward(self, x):
        if (x.is_cuda and x.shape[1] <= 8
                and x.is_contiguous(memory_format=torch.channels_last)):
            return ops.conv2d_smallc(x, self.conv)
        return self.conv(x)


class Downsample(nn.Module):
    def __init__(self, channels):
        super().__init__()
        self.conv = MfmaConv2d(channels, channels, 3, stride=2, padding=1)

    def forward(self, x):
        return self.conv(x)


class Upsample(nn.Module):
    def __init__(self, channels):
        super().__init__()
        self.conv = MfmaConv2d(channels, channels, 3, padding=1)

    def forward(self, x, output_shape=None):
        # odd latent sizes (e.g. 68->34->17->9) need the exact skip shape on
        # the way back up, not blind 2x (ComfyUI passes output_shape the
        # same way)
        exact_2x = (output_shape is None
                    or tuple(output_shape) == (2 * x.shape[2], 2 * x.shape[3]))
        if (exact_2x and x.is_cuda
                and x.is_contiguous(memory_format=torch.channels_last)
                and x.shape[1] % 64 == 0  # up2 needs the conv256 path
                and ops.conv_supported(self.conv)):
            return ops.conv2d_mfma(x, self.conv, up2=True)
        if output_shape is not None:
            y = F.interpolate(x, size=tuple(output_shape), mode="nearest")
        else:
            y = F.interpolate(x, scale_factor=2, mode="nearest")
        return self.conv(y)


class _TimestepSequential(nn.ModuleList):
    def forward(self, x, emb, context, output_shape=None):
        for layer in self:
            if isinstance(layer, ResBlock):
                x = layer(x, emb)
            elif isinstance(layer, SpatialTransformer):
                x = layer(x, context)
            elif isinstance(layer, Upsample):
                x = layer(x, output_shape=output_shape)
            else:
                x = layer(x)
        return x


class UNetModel(nn.Module):
    """Eps-prediction UNet with the SD1.5/SDXL block structure."""

    def __init__(self, cfg: UNetConfig):
        super().__init__()
        self.cfg = cfg
        ch0 = cfg.model_channels
        emb_dim = ch0 * 4
        self.time_embed = nn.Sequential(
            nn.Linear(ch0, emb_dim), nn.SiLU(), nn.Linear(emb_dim, emb_dim)
        )
        if cfg.adm_in_channels:
            self.label_emb = nn.Sequential(
                nn.Linear(cfg.adm_in_channels, emb_dim), nn.SiLU(), nn.Linear(emb_dim, emb_dim)
            )

        def heads_for(ch):
            if cfg.head_dim is not None:
                return ch // cfg.head_dim, cfg.head_dim
            return cfg.num_heads, ch // cfg.num_heads

        self.input_blocks = nn.ModuleList(
            [_TimestepSequential([StemConv(cfg.in_channels, ch0)])]
        )
        skip_chans = [ch0]
        ch = ch0
        for level, mult in enumerate(cfg.channel_mult):
            out_ch = ch0 * mult
            for _ in range(cfg.num_res_blocks):
                layers = [ResBlock(ch, emb_dim, out_ch)]
                ch = out_ch
                if level in cfg.attn_levels and cfg.transformer_depth[level] > 0:
                    h, d = heads_for(ch)
                    layers.append(
                        SpatialTransformer(ch, cfg.context_dim, h, d, cfg.transformer_depth[level])
                    )
                self.input_blocks.append(_TimestepSequential(layers))
                skip_chans.append(ch)
            if level != len(cfg.channel_mult) - 1:
                self.input_blocks.append(_TimestepSequential([Downsample(ch)]))
                skip_chans.append(ch)

        h, d = heads_for(ch)
        mid_depth = cfg.transformer_depth[len(cfg.channel_mult) - 1] or 1
        self.middle_block = _TimestepSequential(
            [
                ResBlock(ch, emb_dim),
                SpatialTransformer(ch, cfg.context_dim, h, d, mid_depth),
                ResBlock(ch, emb_dim),
            ]
        )

        self.output_blocks = nn.ModuleList()
        for level, mult in reversed(list(enumerate(cfg.channel_mult))):
            out_ch = ch0 * mult
            for i in range(cfg.num_res_blocks + 1):
                layers = [ResBlock(ch + skip_chans.pop(), emb_dim, out_ch)]
                ch = out_ch
                if level in cfg.attn_levels and cfg.transformer_depth[level] > 0:
                    h, d = heads_for(ch)
                    layers.append(
                        SpatialTransformer(ch, cfg.context_dim, h, d, cfg.transformer_depth[level])
                    )
                if level != 0 and i == cfg.num_res_blocks:
                    layers.append(Upsample(ch))
                self.output_blocks.append(_TimestepSequential(layers))

        self.out_norm = FusedGroupNorm(ch0, silu=True)
        self.out_conv = MfmaConv2d(ch0, cfg.out_channels, 3, padding=1)

    def forward(self, x, timesteps, context, y=None):
        dtype = self.out_conv.weight.dtype
        x = x.to(dtype)
        if x.is_cuda:
            # channels_last end to end: MIOpen picks NHWC igemm kernels with
            # no batched_transpose pre/post passes, and the fused GroupNorm
            # dispatches to its NHWC variant (profiles/r01: transposes ~5%)
            x = x.contiguous(memory_format=torch.channels_last)
        context = context.to(dtype)
        if y is not None:
            y = y.to(dtype)
        emb = self.time_embed(
            timestep_embedding(timesteps, self.cfg.model_channels).to(dtype)
        )
        if self.cfg.adm_in_channels and y is not None:
            emb = emb + self.label_emb(y)
        hs = []
        h = x
        for block in self.input_blocks:
            h = block(h, emb, context)
            hs.append(h)
        h = self.middle_block(h, emb, context)
        for block in self.output_blocks:
            h = torch.cat([h, hs.pop()], dim=1)
            output_shape = hs[-1].shape[2:] if hs else None
            h = block(h, emb, context, output_shape=output_shape)
        return self.out_conv(self.out_norm(h))
