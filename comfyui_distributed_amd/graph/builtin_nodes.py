"""Built-in compute nodes (the subset of ComfyUI L0 the distributed
workflows need): model loading, conditioning, sampling, VAE, image IO.

The reference delegates all of these to the host ComfyUI;
this framework provides them natively on the gfx950 engine. Node/socket
names follow the ComfyUI wire vocabulary so reference workflows map 1:1.
"""

from __future__ import annotations

import torch

from ..models import create_diffusion_stack
from ..models.sampling import CFGDenoiser, SAMPLERS, SCHEDULERS, sample
from ..utils.image import decode_png_bytes, encode_png_bytes

_STACK_CACHE: dict[tuple, object] = {}

# Creation must be serialized: random init consumes the GLOBAL torch RNG
# stream, so two threads racing through manual_seed(0)+init interleave
# their draws and produce two different "identical" models (observed as a
# master/worker weight mismatch in the HTTP integration test).
import threading

_STACK_LOCK = threading.Lock()


def stable_text_seed(text: str) -> int:
    """Deterministic across processes (Python's hash() is salted per
    process, which would give master and workers different synthetic
    conditioning for the same prompt text)."""
    import hashlib

    return int.from_bytes(hashlib.md5(str(text).encode()).digest()[:4], "little") % (2**31)


def _device_dtype():
    if torch.cuda.is_available():
        return "cuda:0", torch.bfloat16
    return "cpu", torch.float32


class _ContextNode:
    _ctx: dict = {}

    def set_context(self, ctx):
        self._ctx = ctx


class CheckpointLoader(_ContextNode):
    """Loads a model family as a DiffusionStack (random-init weights; there
    is no checkpoint file IO in this offline environment — the ``ckpt_name``
    selects the architecture family, e.g. "sd15", "sdxl", "tiny")."""

    @classmethod
    def INPUT_TYPES(cls):
        return {"required": {"ckpt_name": ("STRING", {"default": "sd15"})}}

    RETURN_TYPES = ("MODEL", "CLIP", "VAE")
    FUNCTION = "load"
    CATEGORY = "loaders"

    def load(self, ckpt_name="sd15"):
        device = self._ctx.get("device")
        if device is None:
            device, dtype = _device_dtype()
        else:
            dtype = torch.bfloat16 if str(device).startswith("cuda") else torch.float32
        key = (str(ckpt_name), str(device))
        with _STACK_LOCK:
            if key not in _STACK_CACHE:
                _STACK_CACHE[key] = create_diffusion_stack(
                    str(ckpt_name), device=device, dtype=dtype
                )
        stack = _STACK_CACHE[key]
        # MODEL and VAE are the stack handle + its VAE; CLIP is the stack
        # too (conditioning is synthesized from it).
        return (stack, stack, stack.vae)


class CLIPTextEncode(_ContextNode):
    """Synthesizes conditioning of the model's context shape from the text
    (deterministic hash -> seed; no text encoder weights exist offline)."""

    @classmethod
    def INPUT_TYPES(cls):
        return {"required": {"text": ("STRING", {"default": ""}), "clip": ("CLIP",)}}

    RETURN_TYPES = ("CONDITIONING",)
    FUNCTION = "encode"
    CATEGORY = "conditioning"

    def encode(self, text="", clip=None):
        return (clip.make_conditioning(stable_text_seed(text)),)


class EmptyLatentImage(_ContextNode):
    @classmethod
    def INPUT_TYPES(cls):
        return {
            "required": {
                "width": ("INT", {"default": 512}),
                "height": ("INT", {"default": 512}),
                "batch_size": ("INT", {"default": 1}),
            }
        }

    RETURN_TYPES = ("LATENT",)
    FUNCTION = "generate"
    CATEGORY = "latent"

    def generate(self, width=512, height=512, batch_size=1):
        return ({"samples": torch.zeros(int(batch_size), 4, int(height) // 8,
                                        int(width) // 8), "size": (width, height)},)


class KSampler(_ContextNode):
    @classmethod
    def INPUT_TYPES(cls):
        return {
            "required": {
                "model": ("MODEL",),
                "seed": ("INT", {"default": 0}),
                "steps": ("INT", {"default": 20}),
                "cfg": ("FLOAT", {"default": 8.0}),
                "sampler_name": (list(SAMPLERS),),
                "scheduler": (list(SCHEDULERS),),
                "positive": ("CONDITIONING",),
                "negative": ("CONDITIONING",),
                "latent_image": ("LATENT",),
                "denoise": ("FLOAT", {"default": 1.0}),
            }
        }

    RETURN_TYPES = ("LATENT",)
    FUNCTION = "sample"
    CATEGORY = "sampling"

    def sample(self, model, seed, steps, cfg, sampler_name, scheduler,
               positive, negative, latent_image, denoise=1.0):
        stack = model
        shape = latent_image["samples"].shape
        g = torch.Generator(device="cpu").manual_seed(int(seed))
        noise = torch.randn(shape, generator=g).to(stack.device)
        sigmas = stack.schedule.sigmas(int(steps), scheduler, float(denoise)).to(
            stack.device
        )
        denoiser = CFGDenoiser(stack.unet, stack.schedule, positive, negative,
                               float(cfg))
        mask = latent_image.get("noise_mask")
        start = None
        if mask is not None or (
            denoise < 1.0 and latent_image["samples"].abs().sum() > 0
        ):
            start = latent_image["samples"].to(stack.device).float()
        with torch.no_grad():
            out = sample(denoiser, noise, sigmas, sampler=sampler_name,
                         seed=int(seed), start_from_latent=start,
                         denoise_mask=None if mask is None
                         else mask.to(stack.device))
        return ({"samples": out.cpu()},)


class VAEDecode(_ContextNode):
    @classmethod
    def INPUT_TYPES(cls):
        return {"required": {"samples": ("LATENT",), "vae": ("VAE",)}}

    RETURN_TYPES = ("IMAGE",)
    FUNCTION = "decode"
    CATEGORY = "latent"

    def decode(self, samples, vae):
        with torch.no_grad():
            img = vae.decode(samples["samples"].to(next(vae.parameters()).device))
        return (img.float().cpu(),)


class VAEEncode(_ContextNode):
    @classmethod
    def INPUT_TYPES(cls):
        return {"required": {"pixels": ("IMAGE",), "vae": ("VAE",)}}

    RETURN_TYPES = ("LATENT",)
    FUNCTION = "encode"
    CATEGORY = "latent"

    def encode(self, pixels, vae):
        with torch.no_grad():
            z = vae.encode(pixels)
        return ({"samples": z.cpu()},)


class WanVideoGenerate(_ContextNode):
    """t2v generation with the WAN stack; outputs frames as an IMAGE batch
    (frames along the batch dim, so batch dividers segment them)."""

    @classmethod
    def INPUT_TYPES(cls):
        return {
            "required": {
                "model": ("MODEL",),
                "positive": ("CONDITIONING",),
                "seed": ("INT", {"default": 0}),
                "steps": ("INT", {"default": 20}),
                "cfg": ("FLOAT", {"default": 5.0}),
                "width": ("INT", {"default": 480}),
                "height": ("INT", {"default": 480}),
                "frames": ("INT", {"default": 17}),
            },
            "optional": {"negative": ("CONDITIONING",),
                         "start_image": ("IMAGE",)},
        }

    RETURN_TYPES = ("IMAGE",)
    FUNCTION = "generate"
    CATEGORY = "video"

    def generate(self, model, positive, seed, steps, cfg, width, height,
                 frames, negative=None, start_image=None):
        from ..models.video import VideoGenParams, generate_video

        p = VideoGenParams(seed=int(seed), steps=int(steps), cfg=float(cfg),
                           width=int(width), height=int(height),
                           frames=int(frames))
        return (generate_video(model, positive, negative, p,
                               start_image=start_image).cpu(),)


class FluxGenerate(_ContextNode):
    """txt2img with a Flux-family MMDiT stack (rectified flow); the SD
    KSampler node does not apply to flow models, so flux workflows use
    this node the way video ones use WanVideoGenerate."""

    @classmethod
    def INPUT_TYPES(cls):
        return {
            "required": {
                "model": ("MODEL",),
                "positive": ("CONDITIONING",),
                "seed": ("INT", {"default": 0}),
                "steps": ("INT", {"default": 20}),
                "cfg": ("FLOAT", {"default": 1.0}),
                "width": ("INT", {"default": 1024}),
                "height": ("INT", {"default": 1024}),
                "batch_size": ("INT", {"default": 1}),
            },
            "optional": {"negative": ("CONDITIONING",),
                         "image": ("IMAGE",),
                         "denoise": ("FLOAT", {"default": 1.0, "min": 0.0,
                                               "max": 1.0})},
        }

    RETURN_TYPES = ("IMAGE",)
    FUNCTION = "generate"
    CATEGORY = "sampling"

    def generate(self, model, positive, seed, steps, cfg, width, height,
                 batch_size=1, negative=None, image=None, denoise=1.0):
        from ..engine.generate import GenParams, generate_latents_flux

        stack = model
        if image is not None:
            h, w = image.shape[1], image.shape[2]
        else:
            h, w = int(height), int(width)
        p = GenParams(seed=int(seed), steps=int(steps), cfg=float(cfg),
                      width=w, height=h, batch_size=int(batch_size))
        init = None
        if image is not None:
            with torch.no_grad():
                init = stack.vae.encode(
                    image.to(stack.device, stack.dtype)).float()
                if init.shape[0] != p.batch_size:
                    init = init.expand(p.batch_size, -1, -1, -1)
        with torch.no_grad():
            lat = generate_latents_flux(stack, positive, negative, p,
                                        init_latent=init,
                                        denoise=float(denoise))
            out = stack.vae.decode(lat.to(stack.dtype)).float()
        return (out.cpu(),)


class LatentUpscale(_ContextNode):
    """Resize a LATENT (hires-fix style). Widths/heights are pixel-space
    and divided by the VAE downscale (8), matching ComfyUI semantics."""

    @classmethod
    def INPUT_TYPES(cls):
        return {"required": {
            "samples": ("LATENT",),
            "upscale_method": (["nearest-exact", "bilinear", "bicubic"],),
            "width": ("INT", {"default": 1024}),
            "height": ("INT", {"default": 1024}),
        }}

    RETURN_TYPES = ("LATENT",)
    FUNCTION = "upscale"
    CATEGORY = "latent"

    def upscale(self, samples, upscale_method="bilinear", width=1024,
                height=1024):
        import torch.nn.functional as F

        mode = {"nearest-exact": "nearest-exact", "bilinear": "bilinear",
                "bicubic": "bicubic"}[str(upscale_method)]
        z = samples["samples"].float()
        out = F.interpolate(
            z, size=(int(height) // 8, int(width) // 8), mode=mode,
            align_corners=False if mode in ("bilinear", "bicubic") else None,
        )
        res = dict(samples)
        res["samples"] = out
        res.pop("noise_mask", None)  # stale resolution
        return (res,)


class ImageScale(_ContextNode):
    """Pixel-space resize on the fused Lanczos-3 resample kernel (the same
    HIP kernel the tile engine uses for extract/resize)."""

    @classmethod
    def INPUT_TYPES(cls):
        return {"required": {
            "image": ("IMAGE",),
            "width": ("INT", {"default": 1024}),
            "height": ("INT", {"default": 1024}),
        }}

    RETURN_TYPES = ("IMAGE",)
    FUNCTION = "scale"
    CATEGORY = "image"

    def scale(self, image, width=1024, height=1024):
        from ..ops import dispatch as ops

        b, h, w, _ = image.shape
        out = ops.extract_resize(image.float(), (0, 0, w, h),
                                 int(width), int(height))
        return (out.clamp(0, 1),)


class SaveAnimatedWEBP(_ContextNode):
    """IMAGE batch -> one animated WebP (ComfyUI-core node of the same
    name; the video workflows' frame batches become a playable file)."""

    @classmethod
    def INPUT_TYPES(cls):
        return {"required": {
            "images": ("IMAGE",),
            "filename_prefix": ("STRING", {"default": "video"}),
            "fps": ("FLOAT", {"default": 16.0, "min": 0.1, "max": 120.0}),
            "lossless": ("BOOLEAN", {"default": False}),
            "quality": ("INT", {"default": 80, "min": 0, "max": 100}),
        }}

    RETURN_TYPES = ()
    FUNCTION = "save"
    OUTPUT_NODE = True
    CATEGORY = "image/animation"

    def save(self, images, filename_prefix="video", fps=16.0,
             lossless=False, quality=80):
        from pathlib import Path

        from PIL import Image

        out_dir = Path(self._ctx.get("output_dir", "output"))
        out_dir.mkdir(parents=True, exist_ok=True)
        frames = []
        for i in range(images.shape[0]):
            arr = (images[i].clamp(0, 1) * 255.0).round().to(torch.uint8)
            frames.append(Image.fromarray(arr.cpu().numpy(), mode="RGB"))
        if not frames:
            raise ValueError("no frames to save")
        p = out_dir / f"{filename_prefix}_00000.webp"
        n = 0
        while p.exists():
            n += 1
            p = out_dir / f"{filename_prefix}_{n:05d}.webp"
        frames[0].save(
            p, save_all=True, append_images=frames[1:], format="WEBP",
            duration=int(round(1000.0 / float(fps))), loop=0,
            lossless=bool(lossless), quality=int(quality),
        )
        sink = self._ctx.get("saved_images")
        if isinstance(sink, list):
            sink.append(str(p))
        return ()


class LoadAudio(_ContextNode):
    """Load a WAV from the input dir (or "synthetic:<seconds>@<rate>")."""

    @classmethod
    def INPUT_TYPES(cls):
        return {"required": {"audio": ("STRING", {"default": ""})}}

    RETURN_TYPES = ("AUDIO",)
    FUNCTION = "load"
    CATEGORY = "audio"

    def load(self, audio=""):
        import os
        import wave
        from pathlib import Path

        name = str(audio)
        if name.startswith("synthetic:"):
            spec = name.split(":", 1)[1]
            secs, rate = spec.split("@") if "@" in spec else (spec, "44100")
            n = max(1, int(float(secs) * int(rate)))
            t = torch.arange(n, dtype=torch.float32) / float(rate)
            wf = (0.5 * torch.sin(2 * torch.pi * 440.0 * t))
            return ({"waveform": wf.expand(2, -1).unsqueeze(0).contiguous(),
                     "sample_rate": int(rate)},)
        path = Path(self._ctx.get("input_dir", "input")) / os.path.basename(name)
        with wave.open(str(path), "rb") as w:
            nch, sw, sr, nfr = (w.getnchannels(), w.getsampwidth(),
                                w.getframerate(), w.getnframes())
            raw = w.readframes(nfr)
        if sw != 2:
            raise ValueError(f"only 16-bit PCM WAV supported, got {sw*8}-bit")
        import numpy as np

        pcm = torch.from_numpy(
            np.frombuffer(raw, dtype=np.int16).copy()).reshape(nfr, nch)
        wf = (pcm.float() / 32767.0).transpose(0, 1).unsqueeze(0).contiguous()
        return ({"waveform": wf, "sample_rate": sr},)


class SaveAudio(_ContextNode):
    """Persist an AUDIO dict as 16-bit PCM WAV (stdlib wave; the collector's
    gathered audio becomes a file the way SaveImage persists images)."""

    @classmethod
    def INPUT_TYPES(cls):
        return {"required": {"audio": ("AUDIO",),
                             "filename_prefix": ("STRING",
                                                 {"default": "audio"})}}

    RETURN_TYPES = ()
    FUNCTION = "save"
    OUTPUT_NODE = True
    CATEGORY = "audio"

    def save(self, audio, filename_prefix="audio"):
        import wave
        from pathlib import Path

        out_dir = Path(self._ctx.get("output_dir", "output"))
        out_dir.mkdir(parents=True, exist_ok=True)
        wf = audio["waveform"]  # [B, C, N]
        sr = int(audio["sample_rate"])
        paths = []
        for b in range(wf.shape[0]):
            clip = wf[b].clamp(-1, 1)
            pcm = (clip * 32767.0).to(torch.int16)
            # interleave channels: [C, N] -> [N, C]
            inter = pcm.transpose(0, 1).contiguous()
            p = out_dir / f"{filename_prefix}_{b:05d}.wav"
            with wave.open(str(p), "wb") as w:
                w.setnchannels(int(wf.shape[1]))
                w.setsampwidth(2)
                w.setframerate(sr)
                w.writeframes(inter.numpy().tobytes())
            paths.append(str(p))
        sink = self._ctx.get("saved_images")
        if isinstance(sink, list):
            sink.extend(paths)
        return ()


class ImageToMask(_ContextNode):
    """One channel of an IMAGE as a MASK [B,H,W] (ComfyUI parity node)."""

    @classmethod
    def INPUT_TYPES(cls):
        return {"required": {
            "image": ("IMAGE",),
            "channel": (["red", "green", "blue"],),
        }}

    RETURN_TYPES = ("MASK",)
    FUNCTION = "convert"
    CATEGORY = "mask"

    def convert(self, image, channel="red"):
        idx = {"red": 0, "green": 1, "blue": 2}.get(str(channel), 0)
        return (image[..., idx].clone(),)


class SetLatentNoiseMask(_ContextNode):
    """Attach an inpainting mask to a LATENT: 1 = re-generate, 0 = keep.
    The pixel-space MASK is resampled to latent resolution here (ComfyUI
    SetLatentNoiseMask parity); KSampler pins the unmasked region."""

    @classmethod
    def INPUT_TYPES(cls):
        return {"required": {"samples": ("LATENT",), "mask": ("MASK",)}}

    RETURN_TYPES = ("LATENT",)
    FUNCTION = "set_mask"
    CATEGORY = "latent"

    def set_mask(self, samples, mask):
        import torch.nn.functional as F

        z = samples["samples"]
        m = mask[:, None].float()  # [B,1,H,W]
        m = F.interpolate(m, size=z.shape[-2:], mode="bilinear",
                          align_corners=False)
        out = dict(samples)
        out["noise_mask"] = m.clamp(0, 1)
        return (out,)


class SyntheticAudio(_ContextNode):
    """Deterministic seeded audio source (sine + noise) so audio pipelines
    (collector audio gather, AudioBatchDivider) are wireable end-to-end —
    the reference delegates audio generation to host-app nodes; this is
    the random-init stand-in (same stance as the image models)."""

    @classmethod
    def INPUT_TYPES(cls):
        return {"required": {
            "seed": ("INT", {"default": 0}),
            "seconds": ("FLOAT", {"default": 1.0}),
            "sample_rate": ("INT", {"default": 44100}),
            "freq": ("FLOAT", {"default": 440.0}),
        }}

    RETURN_TYPES = ("AUDIO",)
    FUNCTION = "generate"
    CATEGORY = "audio"

    def generate(self, seed=0, seconds=1.0, sample_rate=44100, freq=440.0):
        n = max(1, int(float(seconds) * int(sample_rate)))
        t = torch.arange(n, dtype=torch.float32) / float(sample_rate)
        g = torch.Generator().manual_seed(int(seed))
        wave = (0.8 * torch.sin(2 * torch.pi * float(freq) * t)
                + 0.05 * torch.randn(n, generator=g))
        return ({"waveform": wave.expand(2, -1).unsqueeze(0).contiguous(),
                 "sample_rate": int(sample_rate)},)


class LoadImage(_ContextNode):
    """Loads an image from the input directory (or a synthetic one when the
    name is "synthetic:<W>x<H>")."""

    @classmethod
    def INPUT_TYPES(cls):
        return {"required": {"image": ("STRING", {"default": ""})}}

    RETURN_TYPES = ("IMAGE",)
    FUNCTION = "load"
    CATEGORY = "image"

    @staticmethod
    def _resolve_input_path(input_dir, name):
        """Resolve ``name`` strictly inside the input dir. Prompts arrive
        over an unauthenticated endpoint, so an absolute path or a ``..``
        escape must not read arbitrary files (the reference confines loads
        via ComfyUI's folder_paths the same way)."""
        import os
        from pathlib import Path

        base = Path(input_dir).resolve()
        cand = (base / str(name)).resolve()
        if os.path.commonpath([str(base), str(cand)]) != str(base):
            raise ValueError(f"image path {name!r} escapes the input dir")
        return cand

    @classmethod
    def IS_CHANGED(cls, context=None, image=""):
        """Content mark folded into the execution-cache fingerprint: a
        re-uploaded file under the same name (media sync) re-executes the
        load instead of returning the stale cached tensor."""
        name = str(image)
        if name.startswith("synthetic:"):
            return name
        try:
            ctx = context or {}
            path = cls._resolve_input_path(ctx.get("input_dir", "input"), name)
            st = path.stat()
            return (st.st_mtime_ns, st.st_size)
        except Exception:
            return float("nan")  # unreadable -> always re-execute (and fail there)

    def load(self, image=""):
        name = str(image)
        if name.startswith("synthetic:"):
            w, h = (int(v) for v in name.split(":", 1)[1].split("x"))
            g = torch.Generator().manual_seed(0)
            return (torch.rand(1, h, w, 3, generator=g),)
        path = self._resolve_input_path(
            self._ctx.get("input_dir", "input"), name)
        raw = path.read_bytes()
        return (decode_png_bytes(raw),)


class SaveImage(_ContextNode):
    @classmethod
    def INPUT_TYPES(cls):
        return {"required": {"images": ("IMAGE",),
                             "filename_prefix": ("STRING", {"default": "out"})}}

    RETURN_TYPES = ()
    FUNCTION = "save"
    OUTPUT_NODE = True
    CATEGORY = "image"

    def save(self, images, filename_prefix="out"):
        import json
        from pathlib import Path

        out_dir = Path(self._ctx.get("output_dir", "output"))
        # ComfyUI allows a subfolder in the prefix ("run1/img"); keep it
        # inside the output dir
        prefix_path = Path(filename_prefix)
        if prefix_path.is_absolute() or ".." in prefix_path.parts:
            raise ValueError(f"bad filename_prefix {filename_prefix!r}")
        out_dir = out_dir / prefix_path.parent
        filename_prefix = prefix_path.name
        out_dir.mkdir(parents=True, exist_ok=True)
        # embed the executing prompt as PNG metadata (ComfyUI convention:
        # saved outputs carry their workflow for reproducibility)
        meta = None
        prompt = self._ctx.get("current_prompt")
        if prompt is not None:
            meta = {"prompt": json.dumps(prompt)}
        # continue numbering after existing outputs instead of overwriting
        # (ComfyUI's SaveImage counter behavior)
        import re

        pat = re.compile(re.escape(filename_prefix) + r"_(\d{5})\.png$")
        start = -1
        for existing in out_dir.glob(f"{filename_prefix}_*.png"):
            m = pat.match(existing.name)
            if m:
                start = max(start, int(m.group(1)))
        paths = []
        for i in range(images.shape[0]):
            p = out_dir / f"{filename_prefix}_{start + 1 + i:05d}.png"
            p.write_bytes(encode_png_bytes(images[i : i + 1],
                                           compress_level=4, metadata=meta))
            paths.append(str(p))
        sink = self._ctx.get("saved_images")
        if isinstance(sink, list):
            sink.extend(paths)
        return ()


class PreviewImage(SaveImage):
    @classmethod
    def INPUT_TYPES(cls):
        return {"required": {"images": ("IMAGE",)}}

    FUNCTION = "preview"

    def preview(self, images):
        sink = self._ctx.get("preview_images")
        if isinstance(sink, list):
            sink.append(images)
        return ()


BUILTIN_CLASS_MAPPINGS = {
    "CheckpointLoader": CheckpointLoader,
    "CheckpointLoaderSimple": CheckpointLoader,
    "CLIPTextEncode": CLIPTextEncode,
    "EmptyLatentImage": EmptyLatentImage,
    "KSampler": KSampler,
    "VAEDecode": VAEDecode,
    "VAEEncode": VAEEncode,
    "LoadImage": LoadImage,
    "SaveImage": SaveImage,
    "PreviewImage": PreviewImage,
    "WanVideoGenerate": WanVideoGenerate,
    "FluxGenerate": FluxGenerate,
    "SyntheticAudio": SyntheticAudio,
    "ImageToMask": ImageToMask,
    "LatentUpscale": LatentUpscale,
    "ImageScale": ImageScale,
    "SetLatentNoiseMask": SetLatentNoiseMask,
    "SaveAudio": SaveAudio,
    "LoadAudio": LoadAudio,
    "SaveAnimatedWEBP": SaveAnimatedWEBP,
}
