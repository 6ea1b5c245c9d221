"""WAN video stack: flow-matching sampling + per-frame VAE decode.

WAN-2.2 is a rectified-flow model: the network predicts velocity v at time
t in [0,1] with x_t = (1-t)*x0 + t*noise; sampling integrates dx/dt = v
from t=1 to 0 (Euler). Video latents are [B, C, T, H/8, W/8].
"""

from __future__ import annotations

from dataclasses import dataclass

import torch

from .vae import VAE, VAEConfig
from .wan import WAN14B, WAN_TINY, WanConfig, WanModel

#: WAN's VAE has 16 latent channels (x8 spatial; temporal handled per-frame
#: this round).
WAN_VAE = VAEConfig(latent_channels=16)
WAN_TINY_VAE = VAEConfig(latent_channels=4, base_channels=8,
                         channel_mult=(1, 1, 2, 2), num_res_blocks=1)

WAN_CONFIGS: dict[str, WanConfig] = {
    "wan14b": WAN14B,
    "wan_tiny": WAN_TINY,
}


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
        self.vae = VAE(vae_variant).to(device=device, dtype=dtype).eval()
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
                shift: float = 5.0) -> torch.Tensor:
    """Euler integration of the rectified flow from noise (t=1) to data."""
    sig = flow_sigmas(steps, shift)
    x = noise.float()
    for i in range(steps):
        t, t_next = sig[i], sig[i + 1]
        v = velocity_fn(x, t)
        x = x + (t_next - t) * v.float()
    return x


def generate_video(stack: WanStack, cond, uncond, p: VideoGenParams) -> torch.Tensor:
    """Returns frames [T, H, W, 3] float32 in [0,1] (ComfyUI IMAGE batch
    convention for video: frames along the batch dim, so the reference's
    ImageBatchDivider segments them directly)."""
    stack.validate_frames(p.frames)
    g = torch.Generator().manual_seed(p.seed)
    lat_t = p.frames  # temporal compression 1 in this round's VAE
    shape = (p.batch_size, stack.cfg.in_channels, lat_t, p.height // 8, p.width // 8)
    noise = torch.randn(shape, generator=g).to(stack.device)
    vel = FlowCFGVelocity(stack.model, cond, uncond, p.cfg)
    with torch.no_grad():
        lat = sample_flow(vel, noise, p.steps)
        # per-frame VAE decode: [B*T, C, h, w] -> frames
        b, c, t, h, w = lat.shape
        frames = lat.permute(0, 2, 1, 3, 4).reshape(b * t, c, h, w)
        imgs = stack.vae.decode(frames.to(stack.dtype))
    return imgs.float()
