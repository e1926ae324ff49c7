from .unet import UNetConfig, UNet2DCondition
from .taesd import TAESDEncoder, TAESDDecoder, TinyVAE
from .text_encoder import TextEncoder

__all__ = [
    "UNetConfig",
    "UNet2DCondition",
    "TAESDEncoder",
    "TAESDDecoder",
    "TinyVAE",
    "TextEncoder",
]
