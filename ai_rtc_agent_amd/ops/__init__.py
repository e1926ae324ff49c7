from .interface import (
    attention,
    conv2d_nhwc,
    geglu,
    group_norm_silu_nhwc,
    hip_ext,
    hip_available,
    layer_norm,
    linear,
    postprocess_to_u8,
    preprocess_from_u8,
    silu,
    upsample_nearest2x_nhwc,
)

__all__ = [
    "attention",
    "conv2d_nhwc",
    "geglu",
    "group_norm_silu_nhwc",
    "hip_ext",
    "hip_available",
    "layer_norm",
    "linear",
    "postprocess_to_u8",
    "preprocess_from_u8",
    "silu",
    "upsample_nearest2x_nhwc",
]
