"""TinyVAE (TAESD-class) encoder/decoder — from scratch, NHWC.

Replaces the reference's TensorRT-compiled TAESD engines (SURVEY.md §2.2
N6/N7; reference build path lib/wrapper.py:815-860, TinyVAE swap at
lib/wrapper.py:699-707, model id "madebyollin/taesd" in download.py:17-21).

Architecture (TAESD-style): small conv stacks of 64-channel residual blocks
with ReLU, stride-2 convs down / nearest-2x up; tanh-clamp on the latent
input of the decoder. Hot ops route through ai_rtc_agent_amd.ops so the GPU
path shares the implicit-GEMM conv kernel with the UNet.

Latent contract: encode() returns SD-scale latents (x * 0.18215 applied),
decode() accepts the same — matching how the engine feeds UNet latents.
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from .. import ops
from .unet import Conv2d

SD_LATENT_SCALE = 0.18215


class _Block(nn.Module):
    """conv-relu conv-relu conv + skip, then fused relu."""

    def __init__(self, c: int):
        super().__init__()
        self.c1 = Conv2d(c, c, 3)
        self.c2 = Conv2d(c, c, 3)
        self.c3 = Conv2d(c, c, 3)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        # relu fused into each conv epilogue; the skip add is fused into c3
        # (epilogue order: relu(conv + bias + residual))
        h = self.c1(x, act=ops.ACT_RELU)
        h = self.c2(h, act=ops.ACT_RELU)
        return self.c3(h, residual=x, act=ops.ACT_RELU)


def fp8_flag_convs(vae) -> dict:
    """fp8 chain roles inside each _Block (all convs are 64ch, IC%64==0):
    c1/c2 emit e4m3 codes (out scales), every conv gets an input scale so
    split-K shapes that refuse q8 OUTPUT still consume f16 via the inline
    encode. conv_in/out, downs and upsamples stay f16 (tiny or IC%64!=0).
    The engine calibrates the amaxes on its first frames and freezes."""
    convs, outs = [], []
    for m in vae.modules():
        if isinstance(m, _Block):
            convs += [m.c1, m.c2, m.c3]
            outs += [m.c1, m.c2]
    return {"convs": convs, "outs": outs}


class TAESDEncoder(nn.Module):
    """(B,H,W,3) in [-1,1] -> (B,H/8,W/8,4) scaled latent."""

    def __init__(self, width: int = 64, latent_channels: int = 4):
        super().__init__()
        w = width
        self.conv_in = Conv2d(3, w, 3)
        self.stage1 = _Block(w)
        self.down1 = Conv2d(w, w, 3, stride=2, bias=False)
        self.stage2 = nn.ModuleList([_Block(w) for _ in range(3)])
        self.down2 = Conv2d(w, w, 3, stride=2, bias=False)
        self.stage3 = nn.ModuleList([_Block(w) for _ in range(3)])
        self.down3 = Conv2d(w, w, 3, stride=2, bias=False)
        self.stage4 = nn.ModuleList([_Block(w) for _ in range(3)])
        self.conv_out = Conv2d(w, latent_channels, 3)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        h = self.stage1(self.conv_in(x))
        h = self.down1(h)
        for b in self.stage2:
            h = b(h)
        h = self.down2(h)
        for b in self.stage3:
            h = b(h)
        h = self.down3(h)
        for b in self.stage4:
            h = b(h)
        return self.conv_out(h)


class TAESDDecoder(nn.Module):
    """(B,h,w,4) scaled latent -> (B,8h,8w,3) in [-1,1]."""

    def __init__(self, width: int = 64, latent_channels: int = 4):
        super().__init__()
        w = width
        self.conv_in = Conv2d(latent_channels, w, 3)
        self.stage1 = nn.ModuleList([_Block(w) for _ in range(3)])
        self.up1 = Conv2d(w, w, 3, bias=False)
        self.stage2 = nn.ModuleList([_Block(w) for _ in range(3)])
        self.up2 = Conv2d(w, w, 3, bias=False)
        self.stage3 = nn.ModuleList([_Block(w) for _ in range(3)])
        self.up3 = Conv2d(w, w, 3, bias=False)
        self.stage4 = _Block(w)
        self.conv_out = Conv2d(w, 3, 3)

    def forward(self, z: torch.Tensor) -> torch.Tensor:
        # tanh clamp keeps extreme latents in the trained range (TAESD-style)
        z = torch.tanh(z / 3.0) * 3.0
        h = self.conv_in(z, act=ops.ACT_RELU)
        for b in self.stage1:
            h = b(h)
        h = self.up1(ops.upsample_nearest2x_nhwc(h))
        for b in self.stage2:
            h = b(h)
        h = self.up2(ops.upsample_nearest2x_nhwc(h))
        for b in self.stage3:
            h = b(h)
        h = self.up3(ops.upsample_nearest2x_nhwc(h))
        h = self.stage4(h)
        return self.conv_out(h)


class TinyVAE(nn.Module):
    """Paired encoder/decoder with the SD latent-scale contract."""

    def __init__(self, width: int = 64, latent_channels: int = 4):
        super().__init__()
        self.encoder = TAESDEncoder(width, latent_channels)
        self.decoder = TAESDDecoder(width, latent_channels)

    def encode(self, img: torch.Tensor) -> torch.Tensor:
        return self.encoder(img) * SD_LATENT_SCALE

    def decode(self, z: torch.Tensor) -> torch.Tensor:
        return self.decoder(z / SD_LATENT_SCALE)
