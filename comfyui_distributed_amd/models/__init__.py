from .registry import MODEL_CONFIGS, create_diffusion_stack, model_names

__all__ = ["MODEL_CONFIGS", "create_diffusion_stack", "model_names"]
