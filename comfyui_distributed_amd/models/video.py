"""WAN video stack: flow-matching sampling + (2+1)D temporal VAE.

WAN-2.2 is a rectified-flow model: the network predicts velocity v at time
t in [0,1] with x_t = (1-t)*x0 + t*noise; sampling integrates dx/dt = v
from t=1 to 0 (Euler). Video latents are [B, C, 1+(T-1)/4, H/8, W/8]:
like the real WAN VAE, 8x spatial AND 4x temporal compression on 4n+1
frame clips.

The temporal stage is deliberately factorized out of the spatial convs
((2+1)D, not full 3D): every spatial conv stays a per-frame NHWC conv on
the MFMA implicit-GEMM path, and temporal mixing is a causal Conv1d over
the time axis batched across all (channel, pixel) positions - a
GEMM-shaped op that maps cleanly onto CDNA4 matrix cores, instead of the
awkward 3D-conv tilings a direct port of a causal-3D VAE would need.
"""

from __future__ import annotations

from dataclasses import dataclass

import torch
from torch import nn

from .vae import VAE, VAEConfig
from .wan import WAN14B, WAN_TINY, WanConfig, WanModel

#: WAN's VAE has 16 latent channels (x8 spatial, x4 temporal).
WAN_VAE = VAEConfig(latent_channels=16)
WAN_TINY_VAE = VAEConfig(latent_channels=4, base_channels=8,
                         channel_mult=(1, 1, 2, 2), num_res_blocks=1)

WAN_CONFIGS: dict[str, WanConfig] = {
    "wan14b": WAN14B,
    "wan_tiny": WAN_TINY,
}


class TemporalVAE(nn.Module):
    """(2+1)D video VAE: per-frame spatial VAE + causal temporal 4x
    resampling in latent space.

    encode: frames [T, H, W, 3] (T = 4n+1) -> z [Cz, 1+(T-1)/4, h, w]
    decode: z -> frames [T, H, W, 3]

    Causality: the temporal conv sees only past frames (front replicate
    pad of k-1), so streaming/chunked decode of long clips never needs
    future latents - the property the real WAN causal-3D VAE has.
    """

    t_down = 4

    def __init__(self, spatial: VAE):
        super().__init__()
        self.spatial = spatial
        cz = spatial.cfg.latent_channels
        k = self.t_down + 1
        self.tdown = nn.Conv1d(cz, cz, kernel_size=k, stride=self.t_down)
        # k=2*t_down, pad=t_down//2 -> exact 4x length; trimmed to T after
        self.tup = nn.ConvTranspose1d(cz, cz, kernel_size=2 * self.t_down,
                                      stride=self.t_down,
                                      padding=self.t_down // 2)
        self.latent_channels = cz

    def latent_frames(self, frames: int) -> int:
        if frames % self.t_down != 1:
            raise ValueError(f"need {self.t_down}n+1 frames, got {frames}")
        return 1 + (frames - 1) // self.t_down

    def _time_apply(self, conv, z):
        # z [C, T, h, w] -> conv over T batched across pixels.
        # Inputs are cast to the conv's own weight dtype: when the module is
        # .to(bf16) a caller-side fp32 tensor would otherwise be a dtype
        # mismatch inside Conv1d/ConvTranspose1d on GPU.
        c, t, h, w = z.shape
        seq = z.permute(2, 3, 0, 1).reshape(h * w, c, t).to(conv.weight.dtype)
        out = conv(seq)  # [h*w, C, T']
        return out.reshape(h, w, c, -1).permute(2, 3, 0, 1).to(z.dtype)

    def encode(self, frames: torch.Tensor) -> torch.Tensor:
        t = frames.shape[0]
        n_lat = self.latent_frames(t)
        zs = self.spatial.encode(frames)  # [T, Cz, h, w]
        z = zs.permute(1, 0, 2, 3)  # [Cz, T, h, w]
        # causal front pad (replicate frame 0), then stride-4 conv:
        # length T + (k-1) with k = t_down+1 gives exactly 1+(T-1)/4
        pad = self.tdown.kernel_size[0] - 1
        z = torch.cat([z[:, :1].expand(-1, pad, -1, -1), z], dim=1)
        z = self._time_apply(self.tdown, z)
        assert z.shape[1] == n_lat
        return z.to(frames.dtype)

    def decode(self, z: torch.Tensor, frames: int | None = None) -> torch.Tensor:
        """z [Cz, T_lat, h, w] -> frames [T, H, W, 3]; T defaults to the
        4n+1 clip length T_lat encodes."""
        t_lat = z.shape[1]
        t = frames if frames is not None else 1 + (t_lat - 1) * self.t_down
        zs = self._time_apply(self.tup, z)  # [Cz, 4*T_lat, h, w]
        zs = zs[:, :t]
        return self.spatial.decode(
            zs.permute(1, 0, 2, 3).to(z.dtype)
        )  # [T, H, W, 3]


@dataclass
class VideoGenParams:
    seed: int = 0
    steps: int = 20
    cfg: float = 5.0
    width: int = 480
    height: int = 480
    frames: int = 17  # 4n+1
    batch_size: int = 1


class WanStack:
    """Video model family handle (mirrors DiffusionStack's interface where
    it matters: .device/.dtype/.make_conditioning)."""

    family = "wan"
    context_tokens = 512

    def __init__(self, cfg: WanConfig, device="cpu", dtype=torch.float32,
                 seed: int = 0, vae_variant=None):
        if vae_variant is None:
            vae_variant = WAN_VAE if cfg.in_channels == 16 else WAN_TINY_VAE
        torch.manual_seed(seed)
        self.cfg = cfg
        self.model = WanModel(cfg).to(device=device, dtype=dtype).eval()
        self.vae = TemporalVAE(VAE(vae_variant)).to(
            device=device, dtype=dtype).eval()
        self.device = torch.device(device)
        self.dtype = dtype

    def make_conditioning(self, prompt_seed: int = 0):
        g = torch.Generator().manual_seed(prompt_seed)
        ctx = torch.randn(1, self.context_tokens, self.cfg.text_dim,
                          generator=g).to(self.device, self.dtype)
        return {"context": ctx}

    def validate_frames(self, frames: int) -> None:
        if frames % 4 != 1:
            raise ValueError(f"WAN needs 4n+1 frames, got {frames}")

    def latent_frames(self, frames: int) -> int:
        return self.vae.latent_frames(frames)


def flow_sigmas(steps: int, shift: float = 5.0) -> torch.Tensor:
    """Time schedule t: 1 -> 0 with the WAN timestep shift
    (t' = shift*t / (1 + (shift-1)*t))."""
    t = torch.linspace(1.0, 0.0, steps + 1)
    return shift * t / (1 + (shift - 1) * t)


class FlowCFGVelocity:
    def __init__(self, model, cond, uncond, cfg_scale: float):
        self.model = model
        self.cond = cond
        self.uncond = uncond
        self.cfg_scale = cfg_scale

    def __call__(self, x, t: torch.Tensor):
        b = x.shape[0]
        tt = (t * 1000.0).reshape(-1).to(x.device).expand(b)
        if self.cfg_scale != 1.0 and self.uncond is not None:
            ctx = torch.cat([
                self.cond["context"].expand(b, -1, -1),
                self.uncond["context"].expand(b, -1, -1),
            ])
            v = self.model(torch.cat([x] * 2), torch.cat([tt] * 2), ctx)
            vc, vu = v.chunk(2)
            return vu + self.cfg_scale * (vc - vu)
        return self.model(x, tt, self.cond["context"].expand(b, -1, -1))


def sample_flow(velocity_fn, noise: torch.Tensor, steps: int,
                shift: float = 5.0, start_from_latent: torch.Tensor | None = None,
                denoise: float = 1.0,
                pin_latent: torch.Tensor | None = None,
                pin_mask: torch.Tensor | None = None) -> torch.Tensor:
    """Euler integration of the rectified flow from noise (t=1) to data.

    img2img: with ``start_from_latent`` and ``denoise`` < 1 the schedule is
    truncated to [denoise, 0] and integration starts from the flow
    interpolant x_t0 = (1-t0)*x0 + t0*noise at the (shifted) strength t0 —
    the flow-matching analogue of the SD img2img start used by USDU.

    pinning (i2v / flow inpainting): ``pin_mask`` (broadcastable, 1 = keep
    pinned to ``pin_latent``, 0 = generate) resets the pinned region to
    its flow interpolant (1-t)*pin + t*noise before every velocity call;
    at t=0 it equals ``pin_latent`` exactly."""
    t_lin = torch.linspace(1.0, 0.0, steps + 1)
    if start_from_latent is not None and denoise < 1.0:
        t_lin = t_lin * denoise
    sig = shift * t_lin / (1 + (shift - 1) * t_lin)
    if start_from_latent is not None:
        t0 = sig[0].to(noise.device)
        x = (1 - t0) * start_from_latent.float() + t0 * noise.float()
    else:
        x = noise.float()

    def pin(xc, t):
        if pin_mask is None:
            return xc
        keep = (1 - t) * pin_latent.float() + t * noise.float()
        return xc * (1 - pin_mask) + keep * pin_mask

    from ..nodes.runtime import get_runtime

    rt = get_runtime()
    for i in range(steps):
        rt.throw_if_interrupted()  # user interrupt aborts per flow step
        t, t_next = sig[i], sig[i + 1]
        x = pin(x, t)
        v = velocity_fn(x, t)
        x = x + (t_next - t) * v.float()
    return pin(x, sig[-1])


def generate_video(stack: WanStack, cond, uncond, p: VideoGenParams,
                   start_image: torch.Tensor | None = None) -> torch.Tensor:
    """Returns frames [T, H, W, 3] float32 in [0,1] (ComfyUI IMAGE batch
    convention for video: frames along the batch dim, so the reference's
    ImageBatchDivider segments them directly).

    i2v: with ``start_image`` ([1, H, W, 3]) the first latent frame is
    pinned to the encoded image throughout the flow integration (temporal
    inpainting — the real WAN i2v conditions on the first frame the same
    way structurally, via a masked video latent)."""
    if getattr(stack, "family", "") != "wan":
        raise ValueError(
            f"model family {getattr(stack, 'family', '?')!r} is not a video "
            "stack: image models generate via engine.generate_images"
        )
    stack.validate_frames(p.frames)
    g = torch.Generator().manual_seed(p.seed)
    lat_t = stack.latent_frames(p.frames)  # 1 + (T-1)/4
    shape = (p.batch_size, stack.cfg.in_channels, lat_t, p.height // 8, p.width // 8)
    noise = torch.randn(shape, generator=g).to(stack.device)
    vel = FlowCFGVelocity(stack.model, cond, uncond, p.cfg)
    pin_latent = pin_mask = None
    if start_image is not None:
        with torch.no_grad():
            z0 = stack.vae.spatial.encode(
                start_image.to(stack.device, stack.dtype))  # [1, Cz, h, w]
        pin_latent = torch.zeros(shape, device=stack.device)
        pin_latent[:, :, 0] = z0[0].float()
        pin_mask = torch.zeros(1, 1, lat_t, 1, 1, device=stack.device)
        pin_mask[:, :, 0] = 1.0
    with torch.no_grad():
        lat = sample_flow(vel, noise, p.steps, pin_latent=pin_latent,
                          pin_mask=pin_mask)
        imgs = torch.cat([
            stack.vae.decode(lat[b].to(stack.dtype), frames=p.frames)
            for b in range(lat.shape[0])
        ])
    return imgs.float()
