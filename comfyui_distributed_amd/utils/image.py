"""Image codecs for the edge (HTTP) API.

Intra-node transfers never touch this module — raw bf16/fp32 tensors move
over RCCL/xGMI. These codecs exist for wire parity with the reference's
base64-PNG envelopes (reference: utils/image.py:8-24, nodes/collector.py:95-98,
api/job_routes.py:291-307).

Canonical image tensor layout: float32 [B, H, W, C] in [0, 1] (the ComfyUI
IMAGE convention the reference uses throughout).
"""

from __future__ import annotations

import base64
import io

import numpy as np
import torch

try:
    from PIL import Image

    HAS_PIL = True
except ImportError:  # pragma: no cover - PIL present in this image
    HAS_PIL = False


def ensure_contiguous(t: torch.Tensor) -> torch.Tensor:
    return t if t.is_contiguous() else t.contiguous()


def tensor_to_uint8(t: torch.Tensor) -> np.ndarray:
    """[H,W,C] or [B,H,W,C] float in [0,1] -> uint8 numpy array."""
    arr = ensure_contiguous(t.detach()).to(torch.float32).cpu().numpy()
    return (np.clip(arr, 0.0, 1.0) * 255.0).round().astype(np.uint8)


def uint8_to_tensor(arr: np.ndarray) -> torch.Tensor:
    return torch.from_numpy(arr.astype(np.float32) / 255.0)


def tensor_to_pil(t: torch.Tensor):
    """One image [H,W,C] (or [1,H,W,C]) float -> PIL.Image."""
    if not HAS_PIL:
        raise RuntimeError("PIL not available")
    if t.dim() == 4:
        if t.shape[0] != 1:
            raise ValueError("tensor_to_pil expects a single image")
        t = t[0]
    arr = tensor_to_uint8(t)
    if arr.shape[-1] == 1:
        arr = arr[..., 0]
    return Image.fromarray(arr)


def pil_to_tensor(img) -> torch.Tensor:
    """PIL.Image -> [1,H,W,C] float32 tensor in [0,1]."""
    arr = np.array(img.convert("RGB"))
    return uint8_to_tensor(arr).unsqueeze(0)


def encode_png_base64(t: torch.Tensor, compress_level: int = 0) -> str:
    """One image tensor -> base64 PNG string (reference uses compress 0:
    bandwidth traded for CPU, nodes/collector.py:97)."""
    img = tensor_to_pil(t)
    buf = io.BytesIO()
    img.save(buf, format="PNG", compress_level=compress_level)
    return base64.b64encode(buf.getvalue()).decode("ascii")


def decode_png_base64(data: str) -> torch.Tensor:
    """base64 PNG string -> [1,H,W,3] float32 tensor (canonical decode,
    reference api/job_routes.py:291-307)."""
    if not HAS_PIL:
        raise RuntimeError("PIL not available")
    try:
        raw = base64.b64decode(data)
        img = Image.open(io.BytesIO(raw))
        img.load()
    except Exception as exc:  # PIL raises its own hierarchy -> one type
        raise ValueError(f"invalid image payload: {exc}") from exc
    return pil_to_tensor(img)


def encode_png_bytes(t: torch.Tensor, compress_level: int = 0,
                     metadata: dict | None = None) -> bytes:
    """``metadata`` (str -> str) is embedded as PNG tEXt chunks — the
    ComfyUI convention of storing the prompt in saved outputs."""
    img = tensor_to_pil(t)
    buf = io.BytesIO()
    pnginfo = None
    if metadata:
        from PIL.PngImagePlugin import PngInfo

        pnginfo = PngInfo()
        for key, value in metadata.items():
            pnginfo.add_text(str(key), str(value))
    img.save(buf, format="PNG", compress_level=compress_level,
             pnginfo=pnginfo)
    return buf.getvalue()


def decode_png_bytes(raw: bytes) -> torch.Tensor:
    if not HAS_PIL:
        raise RuntimeError("PIL not available")
    img = Image.open(io.BytesIO(raw))
    img.load()
    return pil_to_tensor(img)
