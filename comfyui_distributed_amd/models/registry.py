"""Model family registry + random-init stack builder.

Families mirror BASELINE.json's configs: sd15 (512px), sdxl (1024px),
wan (video DiT). ``tiny`` exists for CPU tests. Weights are random-init
(no network access — BASELINE.json prescribes synthetic weights).
"""

from __future__ import annotations

from dataclasses import dataclass

import torch

from . import sampling
from .unet import SD15_UNET, SDXL_UNET, UNetConfig, UNetModel
from .vae import SD_VAE, SDXL_VAE, VAE, VAEConfig


@dataclass
class StackConfig:
    name: str
    unet: UNetConfig
    vae: VAEConfig
    context_tokens: int = 77
    native_size: int = 512


TINY_UNET = UNetConfig(
    model_channels=32,
    channel_mult=(1, 2),
    num_res_blocks=1,
    attn_levels=(0, 1),
    transformer_depth=(1, 1),
    context_dim=64,
    num_heads=2,
)
# 4 levels so the latent downscale matches the production VAEs (x8)
TINY_VAE = VAEConfig(base_channels=8, channel_mult=(1, 1, 2, 2), num_res_blocks=1)

MODEL_CONFIGS: dict[str, StackConfig] = {
    "sd15": StackConfig("sd15", SD15_UNET, SD_VAE, native_size=512),
    "sdxl": StackConfig("sdxl", SDXL_UNET, SDXL_VAE, native_size=1024),
    "tiny": StackConfig("tiny", TINY_UNET, TINY_VAE, native_size=64),
}


def model_names() -> list[str]:
    return list(MODEL_CONFIGS.keys())


class DiffusionStack:
    """UNet + VAE + schedule + synthetic conditioning for one model family."""

    def __init__(self, cfg: StackConfig, device="cpu", dtype=torch.float32,
                 seed: int = 0):
        self.cfg = cfg
        torch.manual_seed(seed)
        self.unet = UNetModel(cfg.unet).to(device=device, dtype=dtype).eval()
        self.vae = VAE(cfg.vae).to(device=device, dtype=dtype).eval()
        if torch.device(device).type == "cuda":
            # conv weights in channels_last so MIOpen runs pure-NHWC paths
            self.unet = self.unet.to(memory_format=torch.channels_last)
            self.vae = self.vae.to(memory_format=torch.channels_last)
        self.schedule = sampling.NoiseSchedule()
        self.device = torch.device(device)
        self.dtype = dtype

    def make_conditioning(self, prompt_seed: int = 0):
        """Synthetic text conditioning of the right shape (no encoder — no
        network for weights; BASELINE.json prescribes synthetic inputs)."""
        g = torch.Generator().manual_seed(prompt_seed)
        ctx = torch.randn(
            1, self.cfg.context_tokens, self.cfg.unet.context_dim, generator=g
        ).to(self.device, self.dtype)
        cond = {"context": ctx}
        if self.cfg.unet.adm_in_channels:
            cond["y"] = torch.randn(
                1, self.cfg.unet.adm_in_channels, generator=g
            ).to(self.device, self.dtype)
        return cond

    def parameters_bytes(self) -> int:
        return sum(
            p.numel() * p.element_size()
            for m in (self.unet, self.vae)
            for p in m.parameters()
        )


def create_diffusion_stack(name: str, device="cpu", dtype=None, seed: int = 0):
    if name in ("wan14b", "wan_tiny"):
        from .video import WAN_CONFIGS, WanStack

        if dtype is None:
            dtype = torch.bfloat16 if str(device).startswith("cuda") else torch.float32
        return WanStack(WAN_CONFIGS[name], device=device, dtype=dtype, seed=seed)
    if name in ("flux12b", "flux_tiny"):
        from .flux import FLUX12B, FLUX_TINY, FluxStack

        if dtype is None:
            dtype = torch.bfloat16 if str(device).startswith("cuda") else torch.float32
        cfg = FLUX12B if name == "flux12b" else FLUX_TINY
        return FluxStack(cfg, device=device, dtype=dtype, seed=seed)
    cfg = MODEL_CONFIGS[name]
    if dtype is None:
        dtype = torch.bfloat16 if (isinstance(device, str) and device.startswith("cuda")) or (
            isinstance(device, torch.device) and device.type == "cuda"
        ) else torch.float32
    return DiffusionStack(cfg, device=device, dtype=dtype, seed=seed)
