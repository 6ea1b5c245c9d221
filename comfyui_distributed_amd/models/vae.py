"""VAE encoder/decoder (SD autoencoder-KL architecture) on the gfx950 ops.

Reference counterpart: ComfyUI's VAEEncode/VAEDecode called from
upscale/tile_ops.py:212,232-235. GroupNorm+SiLU runs
on the fused HIP kernel, the mid-block attention on the MFMA flash kernel
(split into 64-dim heads); convs via torch/MIOpen.
"""

from __future__ import annotations

from dataclasses import dataclass

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops import dispatch as ops
from .unet import FusedGroupNorm, MfmaConv2d, StemConv


@dataclass
class VAEConfig:
    in_channels: int = 3
    latent_channels: int = 4
    base_channels: int = 128
    channel_mult: tuple = (1, 2, 4, 4)
    num_res_blocks: int = 2
    scale_factor: float = 0.18215


SD_VAE = VAEConfig()
SDXL_VAE = VAEConfig(scale_factor=0.13025)


class VAEResBlock(nn.Module):
    def __init__(self, cin: int, cout: int):
        super().__init__()
        self.norm1 = FusedGroupNorm(cin, silu=True)
        self.conv1 = nn.Conv2d(cin, cout, 3, padding=1)
        self.norm2 = FusedGroupNorm(cout, silu=True)
        self.conv2 = nn.Conv2d(cout, cout, 3, padding=1)
        self.skip = nn.Conv2d(cin, cout, 1) if cin != cout else nn.Identity()

    def _mfma_ok(self, x):
        return (
            x.is_cuda
            and ops.conv_supported(self.conv1)
            and ops.conv_supported(self.conv2)
            and (isinstance(self.skip, nn.Identity) or ops.conv_supported(self.skip))
        )

    def forward(self, x):
        if self._mfma_ok(x):
            # hand-written NHWC path: fused GN+SiLU and implicit-GEMM convs,
            # activations in channels_last storage end to end
            if not x.is_contiguous(memory_format=torch.channels_last):
                x = x.contiguous(memory_format=torch.channels_last)
            h = ops.group_norm_silu_cl(x, self.norm1.groups, self.norm1.weight,
                                       self.norm1.bias, 1e-5, True)
            h = ops.conv2d_mfma(h, self.conv1)
            h = ops.group_norm_silu_cl(h, self.norm2.groups, self.norm2.weight,
                                       self.norm2.bias, 1e-5, True)
            skip = x if isinstance(self.skip, nn.Identity) else ops.conv2d_mfma(x, self.skip)
            # skip add fused into conv2's epilogue (no separate add pass)
            return ops.conv2d_mfma(h, self.conv2, residual=skip)
        h = self.conv2(self.norm2(self.conv1(self.norm1(x))))
        return h + self.skip(x)


class VAEAttention(nn.Module):
    """Spatial self-attention; channels split into 64-dim heads so the MFMA
    flash kernel serves it."""

    def __init__(self, channels: int, head_dim: int = 64):
        super().__init__()
        if channels % head_dim != 0 or channels < head_dim:
            head_dim = channels  # tiny test configs
        assert channels % head_dim == 0
        self.heads = channels // head_dim
        self.head_dim = head_dim
        self.norm = FusedGroupNorm(channels, silu=False)
        self.qkv = nn.Linear(channels, channels * 3)
        self.proj = nn.Linear(channels, channels)

    def forward(self, x):
        b, c, h, w = x.shape
        t = self.norm(x).permute(0, 2, 3, 1).reshape(b, h * w, c)
        q, k, v = self.qkv(t).chunk(3, dim=-1)
        o = ops.attention_packed(q.contiguous(), k.contiguous(), v.contiguous(),
                                 heads=self.heads)
        return x + self.proj(o).reshape(b, h, w, c).permute(0, 3, 1, 2)


class VAEEncoder(nn.Module):
    def __init__(self, cfg: VAEConfig):
        super().__init__()
        ch = cfg.base_channels
        self.conv_in = StemConv(cfg.in_channels, ch)
        downs = []
        cin = ch
        for level, mult in enumerate(cfg.channel_mult):
            cout = ch * mult
            for _ in range(cfg.num_res_blocks):
                downs.append(VAEResBlock(cin, cout))
                cin = cout
            if level != len(cfg.channel_mult) - 1:
                downs.append(MfmaConv2d(cin, cin, 3, stride=2, padding=1))
        self.down = nn.ModuleList(downs)
        self.mid = nn.ModuleList(
            [VAEResBlock(cin, cin), VAEAttention(cin), VAEResBlock(cin, cin)]
        )
        self.norm_out = FusedGroupNorm(cin, silu=True)
        self.conv_out = MfmaConv2d(cin, cfg.latent_channels * 2, 3, padding=1)

    def forward(self, x):
        h = self.conv_in(x)
        for layer in self.down:
            h = layer(h)
        for layer in self.mid:
            h = layer(h)
        return self.conv_out(self.norm_out(h))


class VAEDecoder(nn.Module):
    def __init__(self, cfg: VAEConfig):
        super().__init__()
        ch = cfg.base_channels
        cin = ch * cfg.channel_mult[-1]
        self.conv_in = StemConv(cfg.latent_channels, cin)
        self.mid = nn.ModuleList(
            [VAEResBlock(cin, cin), VAEAttention(cin), VAEResBlock(cin, cin)]
        )
        ups = []
        for level, mult in reversed(list(enumerate(cfg.channel_mult))):
            cout = ch * mult
            for _ in range(cfg.num_res_blocks + 1):
                ups.append(VAEResBlock(cin, cout))
                cin = cout
            if level != 0:
                ups.append(_DecoderUpsample(cin))
        self.up = nn.ModuleList(ups)
        self.norm_out = FusedGroupNorm(cin, silu=True)
        self.conv_out = MfmaConv2d(cin, cfg.in_channels, 3, padding=1)

    def forward(self, z):
        h = self.conv_in(z)
        for layer in self.mid:
            h = layer(h)
        for layer in self.up:
            h = layer(h)
        return self.conv_out(self.norm_out(h))


class _DecoderUpsample(nn.Module):
    def __init__(self, channels):
        super().__init__()
        self.conv = MfmaConv2d(channels, channels, 3, padding=1)

    def forward(self, x):
        if (x.is_cuda and x.is_contiguous(memory_format=torch.channels_last)
                and x.shape[1] % 64 == 0  # up2 needs the conv256 path
                and ops.conv_supported(self.conv)):
            # nearest-2x fused into the conv's tap addressing: the 4x-sized
            # upsampled intermediate (the decoder's largest tensors) never
            # touches HBM
            return ops.conv2d_mfma(x, self.conv, up2=True)
        return self.conv(F.interpolate(x, scale_factor=2, mode="nearest"))


class VAE(nn.Module):
    """encode: [B,H,W,3] image in [0,1] -> scaled latent [B,4,H/8,W/8].
    decode: inverse. Deterministic encode (mean of the posterior), matching
    ComfyUI's VAEEncode behavior."""

    def __init__(self, cfg: VAEConfig = SD_VAE):
        super().__init__()
        self.cfg = cfg
        self.encoder = VAEEncoder(cfg)
        self.decoder = VAEDecoder(cfg)

    @property
    def downscale(self) -> int:
        return 2 ** (len(self.cfg.channel_mult) - 1)

    def encode(self, images: torch.Tensor) -> torch.Tensor:
        p = next(self.parameters())
        x = images.permute(0, 3, 1, 2).to(p.device, p.dtype) * 2.0 - 1.0
        if x.is_cuda:
            x = x.contiguous(memory_format=torch.channels_last)
        moments = self.encoder(x)
        mean = moments[:, : self.cfg.latent_channels]
        return mean * self.cfg.scale_factor

    def decode_tiled(self, latents: torch.Tensor, tile: int = 64,
                     overlap: int = 8) -> torch.Tensor:
        """Latent-tiled decode for very large canvases (the USDU node's
        ``tiled_decode`` option; reference delegates to ComfyUI's tiled
        VAE). Tiles overlap in latent space and are feathered together
        with linear ramps."""
        b, c, h, w = latents.shape
        if h <= tile and w <= tile:
            return self.decode(latents)
        ds = self.downscale
        step = tile - overlap
        out = None
        weight = None
        seen: set = set()
        for y0 in range(0, h, step):
            for x0 in range(0, w, step):
                y1, x1 = min(y0 + tile, h), min(x0 + tile, w)
                ys, xs = max(0, y1 - tile), max(0, x1 - tile)
                if (ys, xs) in seen:  # edge re-anchoring can repeat a tile
                    continue
                seen.add((ys, xs))
                piece = self.decode(latents[:, :, ys:y1, xs:x1])  # [B,ph,pw,3]
                ph, pw = piece.shape[1], piece.shape[2]
                if out is None:
                    out = torch.zeros(b, h * ds, w * ds, piece.shape[3],
                                      dtype=piece.dtype, device=piece.device)
                    weight = torch.zeros(1, h * ds, w * ds, 1,
                                         dtype=piece.dtype, device=piece.device)
                # linear feather over the overlap band (pixel space)
                ramp_y = torch.ones(ph, device=piece.device)
                ramp_x = torch.ones(pw, device=piece.device)
                band = overlap * ds
                if ys > 0:
                    ramp_y[:band] = torch.linspace(0, 1, band, device=piece.device)
                if y1 < h:
                    ramp_y[-band:] = torch.linspace(1, 0, band, device=piece.device)
                if xs > 0:
                    ramp_x[:band] = torch.linspace(0, 1, band, device=piece.device)
                if x1 < w:
                    ramp_x[-band:] = torch.linspace(1, 0, band, device=piece.device)
                m = (ramp_y[:, None] * ramp_x[None, :])[None, :, :, None]
                out[:, ys * ds:y1 * ds, xs * ds:x1 * ds] += piece * m
                weight[:, ys * ds:y1 * ds, xs * ds:x1 * ds] += m
                if y1 >= h and x1 >= w:
                    break
        return (out / weight.clamp_min(1e-8)).clamp(0, 1)

    def decode(self, latents: torch.Tensor) -> torch.Tensor:
        p = next(self.parameters())
        z = latents.to(p.device, p.dtype) / self.cfg.scale_factor
        if z.is_cuda:
            # channels_last so the decoder trunk runs the NHWC MFMA kernels
            z = z.contiguous(memory_format=torch.channels_last)
        x = self.decoder(z)
        img = (x.float() + 1.0) / 2.0
        return img.clamp(0, 1).permute(0, 2, 3, 1)
