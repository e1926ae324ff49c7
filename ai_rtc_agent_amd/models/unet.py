"""UNet2DCondition — from-scratch, NHWC-functional, MFMA-friendly.

Implements the SD-family conditional UNet the reference runs through its
TensorRT engine (usage contract at reference
lib/wrapper.py:463-465: latent (B,4,64,64) + timestep (B)
+ text embeds (B,77,ctx) -> noise pred (B,4,64,64)).

Design notes (MI355X-first):
- Activations are NHWC end-to-end: implicit-GEMM conv gathers contiguous
  channel runs; the transformer blocks view (B,H,W,C) as (B,H*W,C) with
  zero copies.
- Every hot op routes through ai_rtc_agent_amd.ops (HIP kernels on GPU,
  torch reference on CPU).
- Families: sd15 (ctx 768, heads=8), sd21/SD-Turbo (ctx 1024, head_dim 64,
  linear projections), sdxl (ctx 2048, deep transformer stacks, additional
  pooled-text/time-id embedding).
"""
from __future__ import annotations

import math
from dataclasses import dataclass, field
from typing import List, Optional

import torch
import torch.nn as nn

from .. import ops


def _fuse_gn() -> bool:
    """AIRTC_FUSE_GN=1 routes GroupNorm+SiLU through the convs'
    fused input-affine path — A/B'd slower at SD shapes (see ResnetBlock),
    kept for experiments."""
    import os

    return os.environ.get("AIRTC_FUSE_GN", "0") == "1"



@dataclass
class UNetConfig:
    in_channels: int = 4
    out_channels: int = 4
    block_out_channels: List[int] = field(default_factory=lambda: [320, 640, 1280, 1280])
    layers_per_block: int = 2
    cross_attention_dim: int = 768
    attention_head_dim: Optional[int] = None  # None -> fixed 8 heads (sd15)
    num_heads: int = 8
    transformer_depth: List[int] = field(default_factory=lambda: [1, 1, 1, 0])
    use_linear_projection: bool = False
    time_embed_dim_mult: int = 4
    # sdxl extras
    addition_embed_dim: int = 0  # 2816 for sdxl (pooled 1280 + 6*256 time ids)

    @staticmethod
    def sd15() -> "UNetConfig":
        return UNetConfig()

    @staticmethod
    def sd21() -> "UNetConfig":
        """SD 2.1 base geometry == SD-Turbo (distilled from SD 2.1)."""
        return UNetConfig(
            cross_attention_dim=1024,
            attention_head_dim=64,
            use_linear_projection=True,
        )

    @staticmethod
    def sdxl() -> "UNetConfig":
        return UNetConfig(
            block_out_channels=[320, 640, 1280],
            cross_attention_dim=2048,
            attention_head_dim=64,
            use_linear_projection=True,
            transformer_depth=[0, 2, 10],
            addition_embed_dim=2816,
        )

    @staticmethod
    def tiny_xl(ctx: int = 64) -> "UNetConfig":
        """Small sdxl-shaped config (addition-embedding path) for CPU tests."""
        cfg = UNetConfig.tiny(ctx)
        cfg.addition_embed_dim = 2816  # 1280 pooled + 6 x 256 time ids
        return cfg

    @staticmethod
    def tiny(ctx: int = 64) -> "UNetConfig":
        """Small config for CPU tests."""
        return UNetConfig(
            block_out_channels=[32, 64],
            layers_per_block=1,
            cross_attention_dim=ctx,
            attention_head_dim=16,
            transformer_depth=[1, 1],
            use_linear_projection=True,
        )

    def heads_for(self, channels: int) -> int:
        if self.attention_head_dim is None:
            return self.num_heads
        return max(1, channels // self.attention_head_dim)


def timestep_embedding(t: torch.Tensor, dim: int, max_period: int = 10000) -> torch.Tensor:
    """Sinusoidal embedding, (B,) -> (B, dim). fp32 for accuracy."""
    half = dim // 2
    freqs = torch.exp(
        -math.log(max_period) * torch.arange(half, dtype=torch.float32, device=t.device) / half
    )
    args = t.float()[:, None] * freqs[None]
    return torch.cat([torch.cos(args), torch.sin(args)], dim=-1)


class Linear(nn.Module):
    def __init__(self, din: int, dout: int, bias: bool = True):
        super().__init__()
        self.weight = nn.Parameter(torch.randn(dout, din) * (1.0 / math.sqrt(din)))
        self.bias = nn.Parameter(torch.zeros(dout)) if bias else None

    def forward(self, x: torch.Tensor, residual: torch.Tensor | None = None) -> torch.Tensor:
        return ops.linear(x, self.weight, self.bias, residual=residual)


class Conv2d(nn.Module):
    """3x3/1x1 conv over NHWC via ops.conv2d_nhwc. Weight kept OIHW.

    fp8 tier state (set by the engine's calibrate/freeze pass on chained
    fp8 layers, e.g. the TAESD conv stacks):
      _fp8_calibrate  record input/output absmax on the f16 path
      _fp8_in_scale   quantize an f16 INPUT inline at this scale (chain
                      heads whose producer is not fp8)
      _fp8_out_scale  ask the epilogue for e4m3 codes at this scale (the
                      GPU may still return f16 on split-K shapes — callers
                      dispatch on the returned dtype)
    """

    _fp8_calibrate = False
    _fp8_in_scale: float | None = None
    _fp8_out_scale: float | None = None
    _fp8_in_amax: float = 0.0
    _fp8_out_amax: float = 0.0

    def __init__(self, cin: int, cout: int, k: int = 3, stride: int = 1, bias: bool = True):
        super().__init__()
        self.stride = stride
        self.padding = k // 2
        self.weight = nn.Parameter(torch.randn(cout, cin, k, k) * (1.0 / math.sqrt(cin * k * k)))
        self.bias = nn.Parameter(torch.zeros(cout)) if bias else None

    def forward(
        self,
        x: torch.Tensor,
        fuse_silu: bool = False,
        act: int | None = None,
        residual: torch.Tensor | None = None,
        channel_bias: torch.Tensor | None = None,
        in_affine: torch.Tensor | None = None,
        in_act: int = 0,
    ) -> torch.Tensor:
        act_r = act if act is not None else (ops.ACT_SILU if fuse_silu else ops.ACT_NONE)
        a_scale = None
        if x.dtype == torch.uint8:
            # producer-quantized e4m3 codes: the producer attached its scale
            a_scale = getattr(x, "_airtc_fp8_scale", None)
            if a_scale is None:
                raise ValueError(
                    "u8 conv input must carry _airtc_fp8_scale (e4m3 codes "
                    "from a producer-quantized fp8 layer)")
        elif self._fp8_in_scale is not None and not self._fp8_calibrate:
            a_scale = self._fp8_in_scale  # chain head: inline encode
        if a_scale is not None:
            y = ops.conv2d_fp8_nhwc(
                x, self.weight, a_scale, self.bias, self.stride, self.padding,
                act=act_r, residual=residual, channel_bias=channel_bias,
                out_fp8_scale=self._fp8_out_scale,
            )
            if y.dtype == torch.uint8:
                y._airtc_fp8_scale = self._fp8_out_scale
            return y
        y = ops.conv2d_nhwc(
            x, self.weight, self.bias, self.stride, self.padding, fuse_silu,
            act=act, residual=residual, channel_bias=channel_bias,
            in_affine=in_affine, in_act=in_act,
        )
        if self._fp8_calibrate:
            self._fp8_in_amax = max(self._fp8_in_amax,
                                    x.float().abs().max().item())
            self._fp8_out_amax = max(self._fp8_out_amax,
                                     y.float().abs().max().item())
        return y


class GroupNormSiLU(nn.Module):
    # fp8 serving tier state (set by the engine's calibrate/freeze pass on
    # resnet norms whose consumer conv qualifies — see engine._fp8_freeze):
    #   _fp8_calibrate: record the running absmax of the f16 output
    #   _fp8_scale: emit e4m3 codes (u8) at this per-layer scale instead
    _fp8_calibrate = False
    _fp8_scale: float | None = None
    _fp8_amax: float = 0.0

    def __init__(self, channels: int, groups: int = 32, eps: float = 1e-5, silu: bool = True):
        super().__init__()
        # SD uses 32 groups (channels >= 320 so Cg >= 10). For toy test dims
        # keep Cg >= 8 so the half2-vectorized GN kernel stays applicable.
        g = groups
        while g > 1 and (channels % g != 0 or channels // g < 8):
            g //= 2
        self.groups = max(1, g)
        self.eps = eps
        self.silu = silu
        self.weight = nn.Parameter(torch.ones(channels))
        self.bias = nn.Parameter(torch.zeros(channels))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if self._fp8_scale is not None and not self._fp8_calibrate:
            q = ops.group_norm_silu_nhwc(x, self.groups, self.weight,
                                         self.bias, self.eps, self.silu,
                                         fp8_scale=self._fp8_scale)
            q._airtc_fp8_scale = self._fp8_scale
            return q
        y = ops.group_norm_silu_nhwc(x, self.groups, self.weight, self.bias, self.eps, self.silu)
        if self._fp8_calibrate:
            self._fp8_amax = max(self._fp8_amax, y.float().abs().max().item())
        return y

    def coeffs(self, x: torch.Tensor) -> torch.Tensor:
        """(B, C, 2) input-affine pairs for the fused GN->conv path
        (ops.conv2d_nhwc in_affine); the SiLU moves to the conv's in_act."""
        return ops.group_norm_coeffs(x, self.groups, self.weight, self.bias, self.eps)


class LayerNorm(nn.Module):
    def __init__(self, dim: int, eps: float = 1e-5):
        super().__init__()
        self.eps = eps
        self.weight = nn.Parameter(torch.ones(dim))
        self.bias = nn.Parameter(torch.zeros(dim))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return ops.layer_norm(x, self.weight, self.bias, self.eps)


class CrossAttention(nn.Module):
    """Attention with fused projections and WEIGHT-level head-dim padding.

    Head dims that the MFMA kernel doesn't serve natively (sd15's 40/80)
    are padded to the next supported size by inserting ZERO ROWS into the
    q/k/v projection weights (and zero input-columns into the out
    projection) ONCE at cache build — the per-frame path then runs the
    native kernel with no rearrange copies. Zero rows leave every dot
    product and the softmax unchanged, so numerics are identical.
    """

    _DIMS = (32, 64, 96, 128, 160)

    def __init__(self, dim: int, ctx_dim: int, heads: int):
        super().__init__()
        self.heads = heads
        self.head_dim = dim // heads
        self.dpad = (
            self.head_dim
            if self.head_dim in self._DIMS
            else min(x for x in self._DIMS if x >= self.head_dim)
        )
        self.to_q = Linear(dim, dim, bias=False)
        self.to_k = Linear(ctx_dim, dim, bias=False)
        self.to_v = Linear(ctx_dim, dim, bias=False)
        self.to_out = Linear(dim, dim)

    def _pad_heads(self, w: torch.Tensor) -> torch.Tensor:
        """(heads*d, src) -> (heads*dpad, src) with zero rows per head."""
        if self.dpad == self.head_dim:
            return w
        o, src = w.shape
        wp = w.view(self.heads, self.head_dim, src)
        wp = torch.nn.functional.pad(wp, (0, 0, 0, self.dpad - self.head_dim))
        return wp.reshape(self.heads * self.dpad, src)

    def _cached_w(self, name: str, ref: torch.Tensor, build) -> torch.Tensor:
        w = getattr(self, name, None)
        if w is None or w.device != ref.device or w.dtype != ref.dtype:
            w = build().detach().to(ref.device, ref.dtype)
            setattr(self, name, w)
        return w

    def compute_kv(self, ctx: torch.Tensor) -> torch.Tensor:
        """Fused K|V projection of a context. The engine calls this at
        prepare()/update_prompt() to PRECOMPUTE text K/V into a static
        buffer (`static_kv`): cross-attention K/V depend only on the prompt,
        so the per-frame graph skips these GEMMs entirely (the reference
        keeps prompt encoding out of its TRT hot path the same way,
        lib/pipeline.py:44-45)."""
        wkv = self._cached_w(
            "_wkv", ctx,
            lambda: torch.cat([self._pad_heads(p.weight) for p in (self.to_k, self.to_v)]),
        )
        return ops.linear(ctx, wkv)

    def forward(
        self,
        x: torch.Tensor,
        ctx: torch.Tensor | None = None,
        residual: torch.Tensor | None = None,
    ) -> torch.Tensor:
        if ctx is None:
            wqkv = self._cached_w(
                "_wqkv", x,
                lambda: torch.cat([self._pad_heads(p.weight) for p in (self.to_q, self.to_k, self.to_v)]),
            )
            qkv = ops.linear(x, wqkv)
            q, k, v = qkv.chunk(3, dim=-1)
        else:
            wq = self._cached_w("_wq", x, lambda: self._pad_heads(self.to_q.weight))
            q = ops.linear(x, wq)
            kv = getattr(self, "static_kv", None)
            if kv is None:
                kv = self.compute_kv(ctx)
            k, v = kv.chunk(2, dim=-1)
        o = ops.attention(q, k, v, self.heads, scale=1.0 / math.sqrt(self.head_dim))
        # out projection consumes the padded layout (zero input columns)
        wout = self._cached_w(
            "_wout", x,
            lambda: self._pad_heads(
                self.to_out.weight.t().contiguous()
            ).reshape(self.heads * self.dpad, -1).t().contiguous(),
        )
        return ops.linear(o, wout, self.to_out.bias, residual=residual)


class FeedForwardGEGLU(nn.Module):
    def __init__(self, dim: int, mult: int = 4):
        super().__init__()
        inner = dim * mult
        self.proj = Linear(dim, inner * 2)
        self.out = Linear(inner, dim)

    def forward(self, x: torch.Tensor, residual: torch.Tensor | None = None) -> torch.Tensor:
        return self.out(ops.geglu(self.proj(x)), residual=residual)


class BasicTransformerBlock(nn.Module):
    def __init__(self, dim: int, ctx_dim: int, heads: int):
        super().__init__()
        self.norm1 = LayerNorm(dim)
        self.attn1 = CrossAttention(dim, dim, heads)  # self
        self.norm2 = LayerNorm(dim)
        self.attn2 = CrossAttention(dim, ctx_dim, heads)  # cross
        self.norm3 = LayerNorm(dim)
        self.ff = FeedForwardGEGLU(dim)

    def forward(self, x: torch.Tensor, ctx: torch.Tensor) -> torch.Tensor:
        # every residual add is fused into the closing projection's epilogue
        x = self.attn1(self.norm1(x), residual=x)
        x = self.attn2(self.norm2(x), ctx, residual=x)
        x = self.ff(self.norm3(x), residual=x)
        return x


class SpatialTransformer(nn.Module):
    def __init__(self, channels: int, ctx_dim: int, heads: int, depth: int, linear_proj: bool):
        super().__init__()
        self.norm = GroupNormSiLU(channels, 32, eps=1e-6, silu=False)
        self.linear_proj = linear_proj
        if linear_proj:
            self.proj_in = Linear(channels, channels)
            self.proj_out = Linear(channels, channels)
        else:
            self.proj_in = Conv2d(channels, channels, k=1)
            self.proj_out = Conv2d(channels, channels, k=1)
        self.blocks = nn.ModuleList(
            [BasicTransformerBlock(channels, ctx_dim, heads) for _ in range(depth)]
        )

    def forward(self, x: torch.Tensor, ctx: torch.Tensor) -> torch.Tensor:
        b, h, w, c = x.shape
        res = x
        x = self.norm(x)
        if self.linear_proj:
            x = self.proj_in(x.view(b, h * w, c))
        else:
            x = self.proj_in(x).view(b, h * w, c)
        for blk in self.blocks:
            x = blk(x, ctx)
        if self.linear_proj:
            # spatial residual fused into proj_out's epilogue
            return self.proj_out(x, residual=res.view(b, h * w, c)).view(b, h, w, c)
        return self.proj_out(x.view(b, h, w, c), residual=res)


class ResnetBlock(nn.Module):
    # per-resnet precomputed time-emb projection (static-timestep serving
    # path; UNet2DCondition.precompute_time_embeddings fills it)
    _temb_b_static: torch.Tensor | None = None

    def __init__(self, cin: int, cout: int, temb_dim: int):
        super().__init__()
        self.norm1 = GroupNormSiLU(cin)
        self.conv1 = Conv2d(cin, cout, 3)
        self.time_emb_proj = Linear(temb_dim, cout)
        self.norm2 = GroupNormSiLU(cout)
        self.conv2 = Conv2d(cout, cout, 3)
        self.shortcut = Conv2d(cin, cout, 1) if cin != cout else None

    def forward(self, x: torch.Tensor, temb: torch.Tensor,
                temb_b: torch.Tensor | None = None) -> torch.Tensor:
        # time-emb add fused into conv1's epilogue; skip add fused into
        # conv2's. The DEEPER fusion (GN apply + SiLU inside the conv's
        # A-load, in_affine) measured SLOWER end-to-end on MI355X
        # (123.1 -> 88.3 fps headline; sd15 64.4 -> 35.1): the per-element
        # coefficient gather + transform sits in the register-staging
        # critical path of kernels that are already issue/latency-bound —
        # same regime as the rejected halo-tiled loads (ladder). Kept
        # env-gated for experiments: AIRTC_FUSE_GN=1.
        # temb_b: the precomputed per-resnet projection (static timesteps
        # — see UNet2DCondition.precompute_time_embeddings).
        if temb_b is None:
            temb_b = self.time_emb_proj(ops.silu(temb))
        if _fuse_gn():
            h = self.conv1(x, channel_bias=temb_b,
                           in_affine=self.norm1.coeffs(x), in_act=ops.ACT_SILU)
            skip = self.shortcut(x) if self.shortcut is not None else x
            return self.conv2(h, residual=skip,
                              in_affine=self.norm2.coeffs(h), in_act=ops.ACT_SILU)
        h = self.conv1(self.norm1(x), channel_bias=temb_b)
        skip = self.shortcut(x) if self.shortcut is not None else x
        return self.conv2(self.norm2(h), residual=skip)


def fp8_eligible_norms(unet: nn.Module) -> list:
    """The GNs whose consumer conv can run the fp8 MX-MFMA path: resnet
    norm1/norm2 with conv input channels % 64 == 0 (all SD/SDXL resnets;
    the engine calibrates + freezes per-layer scales on these)."""
    out = []
    for m in unet.modules():
        if isinstance(m, ResnetBlock):
            if m.conv1.weight.shape[1] % 64 == 0:
                out.append(m.norm1)
            if m.conv2.weight.shape[1] % 64 == 0:
                out.append(m.norm2)
    return out


class Downsample(nn.Module):
    def __init__(self, channels: int):
        super().__init__()
        self.conv = Conv2d(channels, channels, 3, stride=2)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.conv(x)


class Upsample(nn.Module):
    def __init__(self, channels: int):
        super().__init__()
        self.conv = Conv2d(channels, channels, 3)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.conv(ops.upsample_nearest2x_nhwc(x))


class UNet2DCondition(nn.Module):
    # static-timestep serving caches (see precompute_time_embeddings)
    _temb_static: torch.Tensor | None = None
    _temb_src: torch.Tensor | None = None

    def __init__(self, cfg: UNetConfig):
        super().__init__()
        self.cfg = cfg
        chans = cfg.block_out_channels
        temb_dim = chans[0] * cfg.time_embed_dim_mult

        self.time_proj_dim = chans[0]
        self.time_embed = nn.ModuleList([Linear(chans[0], temb_dim), Linear(temb_dim, temb_dim)])
        if cfg.addition_embed_dim:
            self.add_embed = nn.ModuleList(
                [Linear(cfg.addition_embed_dim, temb_dim), Linear(temb_dim, temb_dim)]
            )
        else:
            self.add_embed = None

        self.conv_in = Conv2d(cfg.in_channels, chans[0], 3)

        # -- down --
        self.down_resnets = nn.ModuleList()
        self.down_attns = nn.ModuleList()
        self.downsamplers = nn.ModuleList()
        skip_chans = [chans[0]]
        cin = chans[0]
        for bi, cout in enumerate(chans):
            depth = cfg.transformer_depth[bi]
            for _ in range(cfg.layers_per_block):
                self.down_resnets.append(ResnetBlock(cin, cout, temb_dim))
                self.down_attns.append(
                    SpatialTransformer(cout, cfg.cross_attention_dim, cfg.heads_for(cout), depth, cfg.use_linear_projection)
                    if depth > 0
                    else None
                )
                skip_chans.append(cout)
                cin = cout
            if bi < len(chans) - 1:
                self.downsamplers.append(Downsample(cout))
                skip_chans.append(cout)
            else:
                self.downsamplers.append(None)

        # -- mid --
        cmid = chans[-1]
        mid_depth = max(1, cfg.transformer_depth[-1]) if len(chans) else 1
        self.mid_res1 = ResnetBlock(cmid, cmid, temb_dim)
        self.mid_attn = SpatialTransformer(
            cmid, cfg.cross_attention_dim, cfg.heads_for(cmid), mid_depth, cfg.use_linear_projection
        )
        self.mid_res2 = ResnetBlock(cmid, cmid, temb_dim)

        # -- up --
        self.up_resnets = nn.ModuleList()
        self.up_attns = nn.ModuleList()
        self.upsamplers = nn.ModuleList()
        cin = cmid
        for bi, cout in enumerate(reversed(chans)):
            orig_bi = len(chans) - 1 - bi
            depth = cfg.transformer_depth[orig_bi]
            for _ in range(cfg.layers_per_block + 1):
                skip = skip_chans.pop()
                self.up_resnets.append(ResnetBlock(cin + skip, cout, temb_dim))
                self.up_attns.append(
                    SpatialTransformer(cout, cfg.cross_attention_dim, cfg.heads_for(cout), depth, cfg.use_linear_projection)
                    if depth > 0
                    else None
                )
                cin = cout
            if bi < len(chans) - 1:
                self.upsamplers.append(Upsample(cout))
            else:
                self.upsamplers.append(None)

        self.norm_out = GroupNormSiLU(chans[0])
        self.conv_out = Conv2d(chans[0], cfg.out_channels, 3)

    # ------------------------------------------------------------------
    @torch.no_grad()
    def precompute_time_embeddings(self, timesteps: torch.Tensor,
                                   added_cond: torch.Tensor | None = None,
                                   dtype: torch.dtype = torch.float16) -> None:
        """Serving fast path: the engine's timesteps are STATIC between
        prepare()/updates, so the whole time-embedding pipeline (~40 small
        kernels per frame: sinusoid + MLP + one silu+linear per resnet)
        runs ONCE here. Buffers refresh IN PLACE when shapes match so a
        captured hipGraph keeps reading the same storage; callers must
        re-invoke after weight refresh / t-index / added-cond updates.
        forward() uses the cache only when handed the SAME timesteps
        tensor object (identity check)."""
        temb = timestep_embedding(timesteps, self.time_proj_dim).to(dtype)
        temb = self.time_embed[1](ops.silu(self.time_embed[0](temb)))
        if self.add_embed is not None and added_cond is not None:
            temb = temb + self.add_embed[1](
                ops.silu(self.add_embed[0](added_cond.to(dtype))))

        def keep(buf, val):
            if buf is not None and buf.shape == val.shape and buf.dtype == val.dtype:
                buf.copy_(val)
                return buf
            return val

        self._temb_static = keep(self._temb_static, temb)
        self._temb_src = timesteps
        st = ops.silu(self._temb_static)
        for m in self.modules():
            if isinstance(m, ResnetBlock):
                m._temb_b_static = keep(m._temb_b_static,
                                        m.time_emb_proj(st).contiguous())

    def forward(
        self,
        sample: torch.Tensor,
        timesteps: torch.Tensor,
        encoder_hidden_states: torch.Tensor,
        added_cond: torch.Tensor | None = None,
        control: tuple | None = None,
    ) -> torch.Tensor:
        """sample: (B,H,W,C_in) NHWC latent; timesteps: (B,);
        encoder_hidden_states: (B,77,ctx)."""
        cfg = self.cfg
        # static-timestep fast path: the serving engine precomputes the
        # whole time-embedding pipeline (and each resnet's projection) once
        # per prepare/update — identity-checked so direct callers with
        # other timesteps still compute normally
        use_static = (getattr(self, "_temb_src", None) is timesteps
                      and self._temb_static is not None)
        if use_static:
            temb = self._temb_static
        else:
            temb = timestep_embedding(timesteps, self.time_proj_dim).to(sample.dtype)
            temb = self.time_embed[1](ops.silu(self.time_embed[0](temb)))
            if self.add_embed is not None and added_cond is not None:
                temb = temb + self.add_embed[1](ops.silu(self.add_embed[0](added_cond.to(sample.dtype))))

        x = self.conv_in(sample)
        skips = [x]
        ri = 0
        for bi in range(len(cfg.block_out_channels)):
            for _ in range(cfg.layers_per_block):
                x = self.down_resnets[ri](
                    x, temb,
                    self.down_resnets[ri]._temb_b_static if use_static else None)
                if self.down_attns[ri] is not None:
                    x = self.down_attns[ri](x, encoder_hidden_states)
                skips.append(x)
                ri += 1
            if self.downsamplers[bi] is not None:
                x = self.downsamplers[bi](x)
                skips.append(x)

        if control is not None:
            # ControlNet residuals: one per skip entry + one for mid
            skip_res, mid_res = control
            skips = [s + c for s, c in zip(skips, skip_res)]
        x = self.mid_res1(x, temb,
                          self.mid_res1._temb_b_static if use_static else None)
        x = self.mid_attn(x, encoder_hidden_states)
        x = self.mid_res2(x, temb,
                          self.mid_res2._temb_b_static if use_static else None)
        if control is not None:
            x = x + mid_res

        ri = 0
        for bi in range(len(cfg.block_out_channels)):
            for _ in range(cfg.layers_per_block + 1):
                skip = skips.pop()
                x = torch.cat([x, skip], dim=-1)
                x = self.up_resnets[ri](
                    x, temb,
                    self.up_resnets[ri]._temb_b_static if use_static else None)
                if self.up_attns[ri] is not None:
                    x = self.up_attns[ri](x, encoder_hidden_states)
                ri += 1
            if self.upsamplers[bi] is not None:
                x = self.upsamplers[bi](x)

        if _fuse_gn():
            return self.conv_out(x, in_affine=self.norm_out.coeffs(x),
                                 in_act=ops.ACT_SILU)
        return self.conv_out(self.norm_out(x))
