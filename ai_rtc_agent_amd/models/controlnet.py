"""ControlNet — spatial conditioning side-network.

Parity target: the reference wrapper accepts use_controlnet +
controlnet_model and compiles a ControlNet TRT engine
(reference lib/wrapper.py:617-643, 787-795); the agent never enables it
(SURVEY.md §7 phase 4 notes it as optional surface). Implemented natively:
the standard ControlNet shape — a copy of the UNet's down+mid path fed by a
hint encoder, emitting zero-conv residuals that add onto the UNet's skip
stack and mid activation.

Shares every building block (ResnetBlock, SpatialTransformer, Conv2d) with
the UNet so the HIP kernels cover it with no new ops.
"""
from __future__ import annotations

from typing import List, Tuple

import torch
import torch.nn as nn

from .. import ops
from .unet import (
    Conv2d,
    Downsample,
    Linear,
    ResnetBlock,
    SpatialTransformer,
    UNetConfig,
    timestep_embedding,
)


class HintEncoder(nn.Module):
    """RGB hint (B,H,W,3) at full resolution -> (B,H/8,W/8,C0)."""

    def __init__(self, c0: int):
        super().__init__()
        self.conv1 = Conv2d(3, 16, 3)
        self.conv2 = Conv2d(16, 32, 3, stride=2)
        self.conv3 = Conv2d(32, 64, 3, stride=2)
        self.conv4 = Conv2d(64, c0, 3, stride=2)

    def forward(self, hint: torch.Tensor) -> torch.Tensor:
        h = self.conv1(hint, act=ops.ACT_SILU)
        h = self.conv2(h, act=ops.ACT_SILU)
        h = self.conv3(h, act=ops.ACT_SILU)
        return self.conv4(h)


class ZeroConv(Conv2d):
    """1x1 conv initialised to zero (ControlNet's 'do no harm at init')."""

    def __init__(self, c: int):
        super().__init__(c, c, 1)
        nn.init.zeros_(self.weight)
        nn.init.zeros_(self.bias)


class ControlNet(nn.Module):
    def __init__(self, cfg: UNetConfig):
        super().__init__()
        self.cfg = cfg
        chans = cfg.block_out_channels
        temb_dim = chans[0] * cfg.time_embed_dim_mult
        self.time_proj_dim = chans[0]
        self.time_embed = nn.ModuleList([Linear(chans[0], temb_dim), Linear(temb_dim, temb_dim)])
        self.conv_in = Conv2d(cfg.in_channels, chans[0], 3)
        self.hint_encoder = HintEncoder(chans[0])

        self.down_resnets = nn.ModuleList()
        self.down_attns = nn.ModuleList()
        self.downsamplers = nn.ModuleList()
        self.zero_convs = nn.ModuleList([ZeroConv(chans[0])])
        cin = chans[0]
        for bi, cout in enumerate(chans):
            depth = cfg.transformer_depth[bi]
            for _ in range(cfg.layers_per_block):
                self.down_resnets.append(ResnetBlock(cin, cout, temb_dim))
                self.down_attns.append(
                    SpatialTransformer(cout, cfg.cross_attention_dim, cfg.heads_for(cout), depth, cfg.use_linear_projection)
                    if depth > 0 else None
                )
                self.zero_convs.append(ZeroConv(cout))
                cin = cout
            if bi < len(chans) - 1:
                self.downsamplers.append(Downsample(cout))
                self.zero_convs.append(ZeroConv(cout))
            else:
                self.downsamplers.append(None)

        cmid = chans[-1]
        mid_depth = max(1, cfg.transformer_depth[-1])
        self.mid_res1 = ResnetBlock(cmid, cmid, temb_dim)
        self.mid_attn = SpatialTransformer(cmid, cfg.cross_attention_dim, cfg.heads_for(cmid), mid_depth, cfg.use_linear_projection)
        self.mid_res2 = ResnetBlock(cmid, cmid, temb_dim)
        self.mid_zero = ZeroConv(cmid)

    def forward(
        self,
        sample: torch.Tensor,
        timesteps: torch.Tensor,
        encoder_hidden_states: torch.Tensor,
        hint: torch.Tensor,
        scale: float = 1.0,
    ) -> Tuple[List[torch.Tensor], torch.Tensor]:
        """Returns (skip_residuals — one per UNet skip entry, mid_residual)."""
        cfg = self.cfg
        temb = timestep_embedding(timesteps, self.time_proj_dim).to(sample.dtype)
        temb = self.time_embed[1](ops.silu(self.time_embed[0](temb)))
        h = self.conv_in(sample) + self.hint_encoder(hint)
        outs = [self.zero_convs[0](h) * scale]
        zi = 1
        ri = 0
        for bi in range(len(cfg.block_out_channels)):
            for _ in range(cfg.layers_per_block):
                h = self.down_resnets[ri](h, temb)
                if self.down_attns[ri] is not None:
                    h = self.down_attns[ri](h, encoder_hidden_states)
                outs.append(self.zero_convs[zi](h) * scale)
                zi += 1
                ri += 1
            if self.downsamplers[bi] is not None:
                h = self.downsamplers[bi](h)
                outs.append(self.zero_convs[zi](h) * scale)
                zi += 1
        h = self.mid_res1(h, temb)
        h = self.mid_attn(h, encoder_hidden_states)
        h = self.mid_res2(h, temb)
        return outs, self.mid_zero(h) * scale
