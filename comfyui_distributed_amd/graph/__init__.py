from .prompt import PromptGraph, NODE_CLASS_COLLECTOR, NODE_CLASS_UPSCALE
from . import transform

__all__ = ["PromptGraph", "transform", "NODE_CLASS_COLLECTOR", "NODE_CLASS_UPSCALE"]
