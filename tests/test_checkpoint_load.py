"""Diffusers-format checkpoint mapping tests (synthetic state dicts).

Real checkpoints are unfetchable offline; these synthesize diffusers-keyed
state dicts for the tiny config and verify full key coverage + value
placement (asymmetric values so a transposed/mis-mapped weight fails).
"""
import torch

from ai_rtc_agent_amd.models import UNet2DCondition, UNetConfig
from ai_rtc_agent_amd.models.load import (
    diffusers_unet_key_map,
    load_diffusers_unet,
)


def synth_diffusers_sd(cfg: UNetConfig, model: UNet2DCondition):
    """Build a diffusers-shaped dict whose values are distinct per key."""
    own = dict(model.state_dict())
    sd = {}
    for src, dst in diffusers_unet_key_map(cfg):
        if dst not in own:
            continue
        t = own[dst]
        g = torch.Generator().manual_seed(abs(hash(src)) % (2**31))
        sd[src] = torch.randn(t.shape, generator=g)
    return sd


def test_key_map_covers_every_parameter():
    cfg = UNetConfig.tiny()
    model = UNet2DCondition(cfg)
    mapped_dst = {dst for _, dst in diffusers_unet_key_map(cfg)}
    own = set(dict(model.state_dict()).keys())
    unmapped = {k for k in own if k not in mapped_dst}
    assert not unmapped, f"parameters with no diffusers mapping: {sorted(unmapped)[:10]}"


def test_load_places_values():
    cfg = UNetConfig.tiny()
    model = UNet2DCondition(cfg)
    sd = synth_diffusers_sd(cfg, model)
    n = load_diffusers_unet(model, sd, strict=True)
    assert n == len(sd)
    own = dict(model.state_dict())
    for src, dst in diffusers_unet_key_map(cfg):
        if src in sd and dst in own:
            v = sd[src]
            if v.shape != own[dst].shape:
                v = v[:, :, 0, 0] if v.dim() == 4 else v[:, :, None, None]
            assert torch.allclose(own[dst].float(), v.float(), atol=1e-6), dst


def test_load_changes_forward():
    cfg = UNetConfig.tiny()
    model = UNet2DCondition(cfg).eval()
    x = torch.randn(1, 16, 16, 4)
    t = torch.tensor([10])
    c = torch.randn(1, 77, cfg.cross_attention_dim)
    with torch.no_grad():
        before = model(x, t, c)
        load_diffusers_unet(model, synth_diffusers_sd(cfg, model), strict=True)
        after = model(x, t, c)
    assert not torch.allclose(before, after)


def test_sdxl_map_builds():
    cfg = UNetConfig.sdxl()
    pairs = diffusers_unet_key_map(cfg)
    srcs = [s for s, _ in pairs]
    assert "add_embedding.linear_1.weight" in srcs
    assert any("transformer_blocks.9" in s for s in srcs), "depth-10 mid stack"
