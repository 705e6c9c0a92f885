from .clip import CLIPTextEncoder
from .registry import ModelBundle, available_models, clear_cache, load_model
from .unet import UNetConfig, UNetModel
from .vae import AutoencoderKL, VAEConfig

__all__ = [
    "CLIPTextEncoder",
    "ModelBundle",
    "available_models",
    "clear_cache",
    "load_model",
    "UNetConfig",
    "UNetModel",
    "AutoencoderKL",
    "VAEConfig",
]
