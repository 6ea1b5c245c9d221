"""txt2img generation (the KSampler+VAEDecode path a seed-parallel workflow
runs on every GPU; reference delegates this to ComfyUI — SURVEY.md §0)."""

from __future__ import annotations

from dataclasses import dataclass

import torch

from ..models.sampling import CFGDenoiser, sample


@dataclass
class GenParams:
    seed: int = 0
    steps: int = 20
    cfg: float = 7.5
    sampler_name: str = "euler"
    scheduler: str = "normal"
    width: int = 512
    height: int = 512
    batch_size: int = 1


def generate_latents(stack, cond, uncond, p: GenParams) -> torch.Tensor:
    if not hasattr(stack, "schedule"):
        raise ValueError(
            f"model family {getattr(stack, 'family', '?')!r} is not an "
            "image-generation stack: video models go through "
            "models.video.generate_video"
        )
    ds = stack.vae.downscale
    shape = (p.batch_size, stack.cfg.unet.in_channels, p.height // ds, p.width // ds)
    g = torch.Generator(device="cpu").manual_seed(p.seed)
    noise = torch.randn(shape, generator=g).to(stack.device)
    sigmas = stack.schedule.sigmas(p.steps, p.scheduler).to(stack.device)
    denoiser = CFGDenoiser(stack.unet, stack.schedule, cond, uncond, p.cfg)
    with torch.no_grad():
        return sample(denoiser, noise, sigmas, sampler=p.sampler_name, seed=p.seed)


def generate_latents_flux(stack, cond, uncond, p: GenParams,
                          init_latent: torch.Tensor | None = None,
                          denoise: float = 1.0) -> torch.Tensor:
    """Rectified-flow sampling for the Flux family (velocity prediction,
    Euler integration over the shifted time schedule). ``init_latent`` +
    ``denoise`` < 1 does flow img2img (truncated schedule from the
    interpolant, same math USDU uses per tile)."""
    from ..models.video import sample_flow

    if getattr(stack, "family", "") != "flux":
        raise ValueError(
            f"generate_latents_flux needs a flux stack, got "
            f"{getattr(stack, 'family', '?')!r}")
    b = p.batch_size
    shape = (b, stack.cfg.in_channels, p.height // 8, p.width // 8)
    g = torch.Generator(device="cpu").manual_seed(p.seed)
    noise = torch.randn(shape, generator=g).to(stack.device)
    ctx = cond["context"].expand(b, -1, -1).to(stack.dtype)
    vec = cond["vec"].expand(b, -1).to(stack.dtype)

    def velocity(x, t):
        tt = (t * 1000.0).reshape(-1).to(x.device).expand(b)
        v = stack.model(x.to(stack.dtype), tt.to(stack.dtype), ctx, vec)
        if p.cfg != 1.0 and uncond is not None:
            vu = stack.model(
                x.to(stack.dtype), tt.to(stack.dtype),
                uncond["context"].expand(b, -1, -1).to(stack.dtype),
                uncond["vec"].expand(b, -1).to(stack.dtype))
            v = vu + p.cfg * (v - vu)
        return v

    return sample_flow(velocity, noise, p.steps, shift=stack.flow_shift,
                       start_from_latent=init_latent, denoise=denoise)


def generate_images(stack, cond, uncond, p: GenParams) -> torch.Tensor:
    """Returns [B, H, W, 3] float32 in [0,1] on the stack device."""
    if getattr(stack, "family", "sd") == "flux":
        with torch.no_grad():
            latents = generate_latents_flux(stack, cond, uncond, p)
            return stack.vae.decode(latents.to(stack.dtype)).float()
    latents = generate_latents(stack, cond, uncond, p)
    with torch.no_grad():
        return stack.vae.decode(latents.to(stack.dtype)).float()
