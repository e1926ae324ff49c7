from .unet import UNetConfig, UNet2DCondition
from .taesd import TAESDEncoder, TAESDDecoder, TinyVAE
from .text_encoder import DualTextEncoder, TextEncoder

__all__ = [
    "UNetConfig",
    "UNet2DCondition",
    "TAESDEncoder",
    "TAESDDecoder",
    "TinyVAE",
    "DualTextEncoder",
    "TextEncoder",
]
