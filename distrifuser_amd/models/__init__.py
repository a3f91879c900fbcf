from .unet import UNet2DConditionNative, UNetConfig, SDXL_UNET, SD15_UNET, TINY_UNET
from .distri_unet import DistriUNet
from .vae import VAEDecoder, VAEDecoderConfig, SDXL_VAE, SD_VAE, TINY_VAE
from .clip import CLIPTextEncoder, CLIPTextConfig, CLIP_VIT_L, OPEN_CLIP_BIG_G, TINY_CLIP

__all__ = [
    "UNet2DConditionNative",
    "UNetConfig",
    "SDXL_UNET",
    "SD15_UNET",
    "TINY_UNET",
    "DistriUNet",
    "VAEDecoder",
    "VAEDecoderConfig",
    "SDXL_VAE",
    "SD_VAE",
    "TINY_VAE",
    "CLIPTextEncoder",
    "CLIPTextConfig",
    "CLIP_VIT_L",
    "OPEN_CLIP_BIG_G",
    "TINY_CLIP",
]
