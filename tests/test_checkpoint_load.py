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


def _synth_taesd_sd(vae, scheme: str):
    """Build a TAESD-shaped state dict (diffusers AutoencoderTiny or raw
    madebyollin naming) with distinct values, shaped from our own params."""
    from ai_rtc_agent_amd.models.load import (
        TAESD_SEQ_DECODER_DIFFUSERS,
        TAESD_SEQ_DECODER_RAW,
        TAESD_SEQ_ENCODER,
    )

    dec_table = TAESD_SEQ_DECODER_DIFFUSERS if scheme == "diffusers" else TAESD_SEQ_DECODER_RAW
    mid = "layers." if scheme == "diffusers" else ""
    sd = {}
    for half, table, model in (
        ("encoder", TAESD_SEQ_ENCODER, vae.encoder),
        ("decoder", dec_table, vae.decoder),
    ):
        own = dict(model.state_dict())
        for src_i, dst, is_block in table:
            subs = [(f"conv.{ci}", our) for ci, our in (("0", "c1"), ("2", "c2"), ("4", "c3"))] \
                if is_block else [("", "")]
            for sub_src, sub_dst in subs:
                for p in ("weight", "bias"):
                    ours = f"{dst}.{sub_dst}.{p}".replace("..", ".") if sub_dst else f"{dst}.{p}"
                    if ours not in own:
                        continue
                    key = f"{half}.{mid}{src_i}.{sub_src}.{p}".replace("..", ".") \
                        if sub_src else f"{half}.{mid}{src_i}.{p}"
                    g = torch.Generator().manual_seed(abs(hash(key)) % (2**31))
                    sd[key] = torch.randn(own[ours].shape, generator=g)
    return sd


def test_taesd_load_covers_all_params_both_schemes():
    from ai_rtc_agent_amd.models.load import load_taesd
    from ai_rtc_agent_amd.models.taesd import TinyVAE

    for scheme in ("diffusers", "raw"):
        vae = TinyVAE(width=16)
        n_params = len(dict(vae.state_dict()))
        sd = _synth_taesd_sd(vae, scheme)
        n_enc, n_dec = load_taesd(vae, sd)
        assert n_enc + n_dec == len(sd) == n_params, (
            scheme, n_enc, n_dec, len(sd), n_params)


def test_taesd_load_changes_decoder_forward():
    from ai_rtc_agent_amd.models.load import load_taesd
    from ai_rtc_agent_amd.models.taesd import TinyVAE

    vae = TinyVAE(width=16).eval()
    z = torch.randn(1, 8, 8, 4)
    with torch.no_grad():
        before = vae.decode(z)
        load_taesd(vae, _synth_taesd_sd(vae, "diffusers"))
        after = vae.decode(z)
    assert not torch.allclose(before, after)


def test_load_model_dir_loads_unet_and_taesd(tmp_path):
    from safetensors.torch import save_file

    from ai_rtc_agent_amd.models.load import load_model_dir
    from ai_rtc_agent_amd.models.taesd import TinyVAE

    class _Eng:
        pass

    eng = _Eng()
    eng.unet = UNet2DCondition(UNetConfig.tiny())
    eng.vae = TinyVAE(width=16)
    unet_sd = synth_diffusers_sd(eng.unet.cfg, eng.unet)
    (tmp_path / "unet").mkdir()
    save_file(unet_sd, str(tmp_path / "unet" / "diffusion_pytorch_model.safetensors"))
    # AutoencoderTiny file at snapshot top level, as madebyollin/taesd ships it
    save_file(_synth_taesd_sd(eng.vae, "diffusers"),
              str(tmp_path / "diffusion_pytorch_model.safetensors"))
    dec_before = eng.vae.decoder.conv_out.weight.clone()
    assert load_model_dir(eng, str(tmp_path))
    assert not torch.allclose(dec_before, eng.vae.decoder.conv_out.weight)


def test_load_model_dir_skips_autoencoder_kl(tmp_path):
    from safetensors.torch import save_file

    from ai_rtc_agent_amd.models.load import load_model_dir
    from ai_rtc_agent_amd.models.taesd import TinyVAE

    class _Eng:
        pass

    eng = _Eng()
    eng.unet = UNet2DCondition(UNetConfig.tiny())
    eng.vae = TinyVAE(width=16)
    (tmp_path / "vae").mkdir()
    save_file({"encoder.down_blocks.0.resnets.0.conv1.weight": torch.zeros(4, 4, 3, 3)},
              str(tmp_path / "vae" / "diffusion_pytorch_model.safetensors"))
    dec_before = eng.vae.decoder.conv_out.weight.clone()
    load_model_dir(eng, str(tmp_path))
    # KL vae detected and skipped: decoder untouched (stays random-init)
    assert torch.allclose(dec_before, eng.vae.decoder.conv_out.weight)


def test_sdxl_map_builds():
    cfg = UNetConfig.sdxl()
    pairs = diffusers_unet_key_map(cfg)
    srcs = [s for s, _ in pairs]
    assert "add_embedding.linear_1.weight" in srcs
    assert any("transformer_blocks.9" in s for s in srcs), "depth-10 mid stack"
