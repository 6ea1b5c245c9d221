"""Audio payload envelope.

Wire parity with reference utils/audio_payload.py:16-103: an AUDIO dict
``{"waveform": float32 [B, C, S] tensor, "sample_rate": int}`` round-trips
through the canonical JSON envelope
``{"sample_rate": int, "shape": [B, C, S], "dtype": "float32", "data": b64}``
with strict shape/byte validation and a configurable size cap.
"""

from __future__ import annotations

import base64

import numpy as np
import torch

from . import constants

CANONICAL_DTYPE = "float32"


class AudioPayloadError(ValueError):
    pass


def encode_audio_payload(audio: dict) -> dict:
    wf = audio.get("waveform")
    sr = audio.get("sample_rate")
    if wf is None or sr is None:
        raise AudioPayloadError("audio dict needs 'waveform' and 'sample_rate'")
    if not isinstance(wf, torch.Tensor) or wf.dim() != 3:
        raise AudioPayloadError("waveform must be a [B, C, S] tensor")
    arr = wf.detach().to(torch.float32).cpu().contiguous().numpy()
    if arr.nbytes > constants.MAX_AUDIO_PAYLOAD_BYTES:
        raise AudioPayloadError(
            f"audio payload {arr.nbytes} bytes exceeds cap "
            f"{constants.MAX_AUDIO_PAYLOAD_BYTES}"
        )
    return {
        "sample_rate": int(sr),
        "shape": list(arr.shape),
        "dtype": CANONICAL_DTYPE,
        "data": base64.b64encode(arr.tobytes()).decode("ascii"),
    }


def decode_audio_payload(payload: dict) -> dict:
    for key in ("sample_rate", "shape", "dtype", "data"):
        if key not in payload:
            raise AudioPayloadError(f"audio payload missing '{key}'")
    if payload["dtype"] != CANONICAL_DTYPE:
        raise AudioPayloadError(f"unsupported audio dtype {payload['dtype']!r}")
    shape = payload["shape"]
    if not (isinstance(shape, (list, tuple)) and len(shape) == 3):
        raise AudioPayloadError("audio shape must have 3 dims [B, C, S]")
    shape = tuple(int(x) for x in shape)
    if any(x < 0 for x in shape):
        raise AudioPayloadError("negative audio shape")
    raw = base64.b64decode(payload["data"])
    expected = int(np.prod(shape)) * 4
    if len(raw) != expected:
        raise AudioPayloadError(
            f"audio byte count {len(raw)} != shape-implied {expected}"
        )
    if len(raw) > constants.MAX_AUDIO_PAYLOAD_BYTES:
        raise AudioPayloadError("audio payload exceeds cap")
    arr = np.frombuffer(raw, dtype=np.float32).reshape(shape).copy()
    return {
        "waveform": torch.from_numpy(arr),
        "sample_rate": int(payload["sample_rate"]),
    }


def concat_audio(audios: list[dict]) -> dict:
    """Concatenate AUDIO dicts along the samples dim (collector semantics,
    reference nodes/collector.py:121-174)."""
    if not audios:
        raise AudioPayloadError("no audio to combine")
    sr = audios[0]["sample_rate"]
    waves = []
    for a in audios:
        if a["sample_rate"] != sr:
            raise AudioPayloadError("sample-rate mismatch in audio combine")
        waves.append(a["waveform"])
    ref_shape = waves[0].shape[:2]
    for w in waves:
        if w.shape[:2] != ref_shape:
            raise AudioPayloadError("channel/batch mismatch in audio combine")
    return {"waveform": torch.cat(waves, dim=-1), "sample_rate": sr}
