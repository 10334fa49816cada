from .unet import UNet2DConditionModel, UNetConfig
from .vae import AutoencoderKL, VAEConfig, DiagonalGaussianDistribution
from .clip_text import CLIPTextModel, CLIPTextConfig

__all__ = [
    "UNet2DConditionModel", "UNetConfig",
    "AutoencoderKL", "VAEConfig", "DiagonalGaussianDistribution",
    "CLIPTextModel", "CLIPTextConfig",
]
