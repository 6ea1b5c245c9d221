from .dispatch import (
    attention,
    attention_packed,
    act_mul,
    blend_tile,
    extract_resize,
    group_norm_silu,
    layer_norm,
    hip_available,
)

__all__ = [
    "attention",
    "act_mul",
    "blend_tile",
    "extract_resize",
    "group_norm_silu",
    "layer_norm",
    "hip_available",
]
