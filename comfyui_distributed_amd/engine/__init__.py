from .usdu import USDUParams, process_tiles, process_single_gpu
from .generate import GenParams, generate_images

__all__ = [
    "USDUParams",
    "process_tiles",
    "process_single_gpu",
    "GenParams",
    "generate_images",
]
