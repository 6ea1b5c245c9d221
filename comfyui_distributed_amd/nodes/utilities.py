"""Utility nodes: seed/value distribution, batch dividers, placeholders.

Signature + behavior parity with reference nodes/utilities.py:
DistributedSeed (worker offset = index+1, :52-75), DistributedValue (typed
1-indexed JSON overrides, :86-162), DistributedModelName (:164-224), the
batch dividers (balanced divmod chunking, :235-329), DistributedEmptyImage
(zero-batch placeholder, :332-354).
"""

from __future__ import annotations

import json

import torch

from ..utils.logging import debug_log


class AnyType(str):
    """Wildcard type token: compares equal to every type name."""

    def __ne__(self, other) -> bool:  # noqa: D105
        return False


any_type = AnyType("*")


class ByPassTypeTuple(tuple):
    """Indexing past the end returns the wildcard (variable socket counts)."""

    def __getitem__(self, index):
        if isinstance(index, int) and index >= len(self):
            return any_type
        return super().__getitem__(index)


def parse_worker_index(worker_id: str) -> int:
    """'worker_N' or bare int string -> N."""
    if isinstance(worker_id, str) and worker_id.startswith("worker_"):
        return int(worker_id.split("_")[1])
    return int(worker_id)


def chunk_bounds(total: int, parts: int) -> list[tuple[int, int]]:
    """Balanced contiguous chunks via divmod (reference :7-20)."""
    base, extra = divmod(total, parts)
    bounds = []
    start = 0
    for i in range(parts):
        size = base + (1 if i < extra else 0)
        bounds.append((start, start + size))
        start += size
    return bounds


class DistributedSeed:
    @classmethod
    def INPUT_TYPES(cls):
        return {
            "required": {
                "seed": ("INT", {"default": 1125899906842, "min": 0,
                                 "max": 1125899906842624, "forceInput": False}),
            },
            "hidden": {
                "is_worker": ("BOOLEAN", {"default": False}),
                "worker_id": ("STRING", {"default": ""}),
            },
        }

    RETURN_TYPES = ("INT",)
    RETURN_NAMES = ("seed",)
    FUNCTION = "distribute"
    CATEGORY = "utils"

    def distribute(self, seed, is_worker=False, worker_id=""):
        if not is_worker:
            return (seed,)
        try:
            offset = parse_worker_index(worker_id) + 1
            return (seed + offset,)
        except (ValueError, IndexError) as exc:
            debug_log(f"DistributedSeed: bad worker_id {worker_id!r}: {exc}")
            return (seed,)


class DistributedValue:
    @classmethod
    def INPUT_TYPES(cls):
        return {
            "required": {
                "default_value": ("STRING", {"default": ""}),
                "worker_values": ("STRING", {"default": "{}"}),
            },
            "hidden": {
                "is_worker": ("BOOLEAN", {"default": False}),
                "worker_id": ("STRING", {"default": ""}),
            },
        }

    RETURN_TYPES = (any_type,)
    RETURN_NAMES = ("value",)
    FUNCTION = "distribute"
    CATEGORY = "utils"

    @staticmethod
    def _coerce(value, value_type):
        if value_type == "INT":
            return int(float(value))
        if value_type == "FLOAT":
            return float(value)
        return value  # STRING / COMBO stay strings

    @classmethod
    def _coerce_safe(cls, value, value_type):
        try:
            return cls._coerce(value, value_type)
        except (TypeError, ValueError):
            return value

    def distribute(self, default_value, worker_values="{}", is_worker=False,
                   worker_id=""):
        try:
            values = (
                json.loads(worker_values)
                if isinstance(worker_values, str)
                else worker_values
            )
            if not isinstance(values, dict):
                values = {}
        except json.JSONDecodeError:
            values = {}
        value_type = values.get("_type", "STRING")
        default = self._coerce_safe(default_value, value_type)
        if not is_worker:
            return (default,)
        try:
            idx = parse_worker_index(worker_id)
            raw = values.get(str(idx + 1))  # 1-indexed worker keys
            # falsy values (0, 0.0) are legitimate overrides — only a
            # missing key or empty string falls back to the default
            if raw is not None and raw != "":
                return (self._coerce(raw, value_type),)
        except (ValueError, IndexError):
            pass
        return (default,)


class DistributedModelName:
    """Output node: stringifies a model path so workers can substitute
    their own local model files, AND writes the resolved value back into
    the executing workflow's node entry (reference :164-224 updates
    widgets_values the same way — the resolved name then travels in saved
    outputs' embedded workflow metadata)."""

    @classmethod
    def INPUT_TYPES(cls):
        return {
            "required": {"model_name": (any_type,)},
            "hidden": {"unique_id": "UNIQUE_ID",
                       "extra_pnginfo": "EXTRA_PNGINFO"},
        }

    RETURN_TYPES = ("STRING",)
    RETURN_NAMES = ("name",)
    FUNCTION = "log_input"
    OUTPUT_NODE = True
    CATEGORY = "utils"

    @staticmethod
    def _stringify(value):
        if isinstance(value, str):
            return value
        if isinstance(value, (int, float, bool)):
            return str(value)
        try:
            return json.dumps(value, indent=4)
        except (TypeError, ValueError):
            return str(value)

    def log_input(self, model_name, unique_id=None, extra_pnginfo=None):
        values = ([self._stringify(v) for v in model_name]
                  if isinstance(model_name, list)
                  else [self._stringify(model_name)])
        # write-back: keep the resolved display value in the workflow
        workflow = (extra_pnginfo or {}).get("workflow") \
            if isinstance(extra_pnginfo, dict) else None
        if workflow is not None and unique_id is not None:
            node = workflow.get(str(unique_id))
            if isinstance(node, dict):
                node["widgets_values"] = list(values)
        return (values[0] if len(values) == 1 else values,)


class ImageBatchDivider:
    MAX_SPLITS = 10

    @classmethod
    def INPUT_TYPES(cls):
        return {
            "required": {
                "images": ("IMAGE",),
                "divide_by": ("INT", {"default": 2, "min": 1, "max": 10, "step": 1}),
            }
        }

    RETURN_TYPES = ByPassTypeTuple(("IMAGE",))
    RETURN_NAMES = ByPassTypeTuple(tuple(f"batch_{i+1}" for i in range(10)))
    FUNCTION = "divide_batch"
    OUTPUT_NODE = True
    CATEGORY = "image"

    def divide_batch(self, images, divide_by):
        parts = max(1, min(int(divide_by), self.MAX_SPLITS))
        empty = images[:0]
        outputs = [
            images[s:e] if e > s else empty
            for s, e in chunk_bounds(images.shape[0], parts)
        ]
        outputs += [empty] * (self.MAX_SPLITS - len(outputs))
        return tuple(outputs[: self.MAX_SPLITS])


class AudioBatchDivider:
    MAX_SPLITS = 10

    @classmethod
    def INPUT_TYPES(cls):
        return {
            "required": {
                "audio": ("AUDIO",),
                "divide_by": ("INT", {"default": 2, "min": 1, "max": 10, "step": 1}),
            }
        }

    RETURN_TYPES = ByPassTypeTuple(("AUDIO",))
    RETURN_NAMES = ByPassTypeTuple(tuple(f"audio_{i+1}" for i in range(10)))
    FUNCTION = "divide_audio"
    OUTPUT_NODE = True
    CATEGORY = "audio"

    def divide_audio(self, audio, divide_by):
        parts = max(1, min(int(divide_by), self.MAX_SPLITS))
        wf = audio["waveform"]
        sr = audio["sample_rate"]
        outputs = []
        for s, e in chunk_bounds(wf.shape[-1], parts):
            outputs.append({"waveform": wf[..., s:e], "sample_rate": sr})
        empty = {"waveform": wf[..., :0], "sample_rate": sr}
        outputs += [empty] * (self.MAX_SPLITS - len(outputs))
        return tuple(outputs[: self.MAX_SPLITS])


class DistributedEmptyImage:
    """Zero-batch IMAGE placeholder for delegate-only master graphs."""

    @classmethod
    def INPUT_TYPES(cls):
        return {
            "required": {
                "width": ("INT", {"default": 64, "min": 8, "max": 16384}),
                "height": ("INT", {"default": 64, "min": 8, "max": 16384}),
            }
        }

    RETURN_TYPES = ("IMAGE",)
    FUNCTION = "create"
    CATEGORY = "image"

    def create(self, width=64, height=64):
        return (torch.zeros(0, int(height), int(width), 3),)
