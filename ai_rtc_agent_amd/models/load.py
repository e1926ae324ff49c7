"""Checkpoint loading: diffusers-format weights -> our NHWC-functional models.

A user of the reference points `--model-id` at a diffusers repo
(lykon/dreamshaper-8) and weights load through the diffusers stack
(reference lib/wrapper.py:645-707). We load the same artifacts directly:
safetensors state dicts in diffusers' UNet2DConditionModel / TAESD naming,
remapped to this package's module tree. No diffusers dependency.

Offline note: real checkpoints are unfetchable in the build environment;
the mapping is exercised by tests that synthesize a diffusers-shaped state
dict for the tiny config and assert numerically-identical forwards are
impossible to get wrong silently (strict key coverage).
"""
from __future__ import annotations

import os
from typing import Dict, List, Tuple

import torch

from .unet import UNet2DCondition, UNetConfig


def _resnet_map(prefix_src: str, prefix_dst: str) -> List[Tuple[str, str]]:
    return [
        (f"{prefix_src}.norm1.weight", f"{prefix_dst}.norm1.weight"),
        (f"{prefix_src}.norm1.bias", f"{prefix_dst}.norm1.bias"),
        (f"{prefix_src}.conv1.weight", f"{prefix_dst}.conv1.weight"),
        (f"{prefix_src}.conv1.bias", f"{prefix_dst}.conv1.bias"),
        (f"{prefix_src}.time_emb_proj.weight", f"{prefix_dst}.time_emb_proj.weight"),
        (f"{prefix_src}.time_emb_proj.bias", f"{prefix_dst}.time_emb_proj.bias"),
        (f"{prefix_src}.norm2.weight", f"{prefix_dst}.norm2.weight"),
        (f"{prefix_src}.norm2.bias", f"{prefix_dst}.norm2.bias"),
        (f"{prefix_src}.conv2.weight", f"{prefix_dst}.conv2.weight"),
        (f"{prefix_src}.conv2.bias", f"{prefix_dst}.conv2.bias"),
        (f"{prefix_src}.conv_shortcut.weight", f"{prefix_dst}.shortcut.weight"),
        (f"{prefix_src}.conv_shortcut.bias", f"{prefix_dst}.shortcut.bias"),
    ]


def _attnblock_map(src: str, dst: str) -> List[Tuple[str, str]]:
    """diffusers Transformer2DModel -> our SpatialTransformer."""
    pairs = [
        (f"{src}.norm.weight", f"{dst}.norm.weight"),
        (f"{src}.norm.bias", f"{dst}.norm.bias"),
        (f"{src}.proj_in.weight", f"{dst}.proj_in.weight"),
        (f"{src}.proj_in.bias", f"{dst}.proj_in.bias"),
        (f"{src}.proj_out.weight", f"{dst}.proj_out.weight"),
        (f"{src}.proj_out.bias", f"{dst}.proj_out.bias"),
    ]
    return pairs


def _basic_block_map(src: str, dst: str) -> List[Tuple[str, str]]:
    pairs = []
    for a, b in (("attn1", "attn1"), ("attn2", "attn2")):
        pairs += [
            (f"{src}.{a}.to_q.weight", f"{dst}.{b}.to_q.weight"),
            (f"{src}.{a}.to_k.weight", f"{dst}.{b}.to_k.weight"),
            (f"{src}.{a}.to_v.weight", f"{dst}.{b}.to_v.weight"),
            (f"{src}.{a}.to_out.0.weight", f"{dst}.{b}.to_out.weight"),
            (f"{src}.{a}.to_out.0.bias", f"{dst}.{b}.to_out.bias"),
        ]
    pairs += [
        (f"{src}.norm1.weight", f"{dst}.norm1.weight"),
        (f"{src}.norm1.bias", f"{dst}.norm1.bias"),
        (f"{src}.norm2.weight", f"{dst}.norm2.weight"),
        (f"{src}.norm2.bias", f"{dst}.norm2.bias"),
        (f"{src}.norm3.weight", f"{dst}.norm3.weight"),
        (f"{src}.norm3.bias", f"{dst}.norm3.bias"),
        (f"{src}.ff.net.0.proj.weight", f"{dst}.ff.proj.weight"),
        (f"{src}.ff.net.0.proj.bias", f"{dst}.ff.proj.bias"),
        (f"{src}.ff.net.2.weight", f"{dst}.ff.out.weight"),
        (f"{src}.ff.net.2.bias", f"{dst}.ff.out.bias"),
    ]
    return pairs


def diffusers_unet_key_map(cfg: UNetConfig) -> List[Tuple[str, str]]:
    """Full (diffusers key -> our key) table for a UNet2DConditionModel."""
    pairs: List[Tuple[str, str]] = [
        ("conv_in.weight", "conv_in.weight"),
        ("conv_in.bias", "conv_in.bias"),
        ("time_embedding.linear_1.weight", "time_embed.0.weight"),
        ("time_embedding.linear_1.bias", "time_embed.0.bias"),
        ("time_embedding.linear_2.weight", "time_embed.1.weight"),
        ("time_embedding.linear_2.bias", "time_embed.1.bias"),
        ("conv_norm_out.weight", "norm_out.weight"),
        ("conv_norm_out.bias", "norm_out.bias"),
        ("conv_out.weight", "conv_out.weight"),
        ("conv_out.bias", "conv_out.bias"),
    ]
    if cfg.addition_embed_dim:
        pairs += [
            ("add_embedding.linear_1.weight", "add_embed.0.weight"),
            ("add_embedding.linear_1.bias", "add_embed.0.bias"),
            ("add_embedding.linear_2.weight", "add_embed.1.weight"),
            ("add_embedding.linear_2.bias", "add_embed.1.bias"),
        ]

    nblocks = len(cfg.block_out_channels)
    ri = 0  # flat resnet index in our down list
    for bi in range(nblocks):
        depth = cfg.transformer_depth[bi]
        for li in range(cfg.layers_per_block):
            src_r = f"down_blocks.{bi}.resnets.{li}"
            pairs += _resnet_map(src_r, f"down_resnets.{ri}")
            if depth > 0:
                src_a = f"down_blocks.{bi}.attentions.{li}"
                dst_a = f"down_attns.{ri}"
                pairs += _attnblock_map(src_a, dst_a)
                for d in range(depth):
                    pairs += _basic_block_map(
                        f"{src_a}.transformer_blocks.{d}", f"{dst_a}.blocks.{d}"
                    )
            ri += 1
        if bi < nblocks - 1:
            pairs += [
                (f"down_blocks.{bi}.downsamplers.0.conv.weight", f"downsamplers.{bi}.conv.weight"),
                (f"down_blocks.{bi}.downsamplers.0.conv.bias", f"downsamplers.{bi}.conv.bias"),
            ]

    pairs += _resnet_map("mid_block.resnets.0", "mid_res1")
    pairs += _resnet_map("mid_block.resnets.1", "mid_res2")
    pairs += _attnblock_map("mid_block.attentions.0", "mid_attn")
    mid_depth = max(1, cfg.transformer_depth[-1])
    for d in range(mid_depth):
        pairs += _basic_block_map(
            f"mid_block.attentions.0.transformer_blocks.{d}", f"mid_attn.blocks.{d}"
        )

    ri = 0
    for ui, bi in enumerate(reversed(range(nblocks))):
        depth = cfg.transformer_depth[bi]
        for li in range(cfg.layers_per_block + 1):
            pairs += _resnet_map(f"up_blocks.{ui}.resnets.{li}", f"up_resnets.{ri}")
            if depth > 0:
                src_a = f"up_blocks.{ui}.attentions.{li}"
                dst_a = f"up_attns.{ri}"
                pairs += _attnblock_map(src_a, dst_a)
                for d in range(depth):
                    pairs += _basic_block_map(
                        f"{src_a}.transformer_blocks.{d}", f"{dst_a}.blocks.{d}"
                    )
            ri += 1
        if ui < nblocks - 1:
            pairs += [
                (f"up_blocks.{ui}.upsamplers.0.conv.weight", f"upsamplers.{ui}.conv.weight"),
                (f"up_blocks.{ui}.upsamplers.0.conv.bias", f"upsamplers.{ui}.conv.bias"),
            ]
    return pairs


def load_diffusers_unet(
    model: UNet2DCondition, sd: Dict[str, torch.Tensor], strict: bool = True
) -> int:
    """Load a diffusers UNet2DConditionModel state dict into our model.

    Handles the proj_in/proj_out conv(1x1)-vs-linear difference: diffusers
    stores (C, C, 1, 1) convs for non-linear-projection models; ours stores
    whichever the config says.
    """
    own = dict(model.state_dict())
    loaded = 0
    missing: List[str] = []
    for src, dst in diffusers_unet_key_map(model.cfg):
        if src not in sd:
            if "shortcut" in dst:  # optional (only when channels change)
                continue
            missing.append(src)
            continue
        v = sd[src]
        tgt = own.get(dst)
        if tgt is None:
            if "shortcut" in dst:
                continue
            missing.append(src)
            continue
        if v.shape != tgt.shape:
            if v.dim() == 4 and v.shape[2] == 1 and tgt.dim() == 2:
                v = v[:, :, 0, 0]  # 1x1 conv -> linear
            elif v.dim() == 2 and tgt.dim() == 4:
                v = v[:, :, None, None]
            else:
                raise ValueError(f"shape mismatch {src}: {v.shape} vs {tgt.shape}")
        own[dst].copy_(v.to(own[dst].dtype))
        loaded += 1
    if strict and missing:
        raise KeyError(f"missing {len(missing)} keys, e.g. {missing[:5]}")
    return loaded


# TAESD sequential layouts. madebyollin/taesd ships two namings for the SAME
# flat-Sequential architecture:
#   raw   ("taesd_encoder/decoder.safetensors"):  encoder.N... / decoder.N...
#   diffusers AutoencoderTiny ("diffusion_pytorch_model.safetensors" at the
#   snapshot TOP LEVEL): encoder.layers.N... / decoder.layers.N...
# Encoder indices coincide (no parameter-free modules before conv_out);
# decoder indices differ by 1 (raw has a Clamp at 0 and a ReLU at 2, diffusers
# folds both into forward()). Each entry: (seq index, our module, is_block).
TAESD_SEQ_ENCODER = [
    (0, "conv_in", False),
    (1, "stage1", True),
    (2, "down1", False), (3, "stage2.0", True), (4, "stage2.1", True), (5, "stage2.2", True),
    (6, "down2", False), (7, "stage3.0", True), (8, "stage3.1", True), (9, "stage3.2", True),
    (10, "down3", False), (11, "stage4.0", True), (12, "stage4.1", True), (13, "stage4.2", True),
    (14, "conv_out", False),
]
TAESD_SEQ_DECODER_RAW = [
    (1, "conv_in", False),
    (3, "stage1.0", True), (4, "stage1.1", True), (5, "stage1.2", True),
    (7, "up1", False),
    (8, "stage2.0", True), (9, "stage2.1", True), (10, "stage2.2", True),
    (12, "up2", False),
    (13, "stage3.0", True), (14, "stage3.1", True), (15, "stage3.2", True),
    (17, "up3", False),
    (18, "stage4", True),
    (19, "conv_out", False),
]
TAESD_SEQ_DECODER_DIFFUSERS = [(i - 1, dst, b) for i, dst, b in TAESD_SEQ_DECODER_RAW]


def _load_taesd_seq(model, sd: Dict[str, torch.Tensor], prefix: str, table) -> int:
    """Copy a flat-Sequential TAESD state dict into one of our TAESD halves.

    Blocks are madebyollin _Block / diffusers AutoencoderTinyBlock: a `conv`
    Sequential whose parameterized entries are conv.0/conv.2/conv.4 → our
    c1/c2/c3."""
    own = dict(model.state_dict())
    n = 0
    for src_i, dst, is_block in table:
        if is_block:
            for ci, our in (("0", "c1"), ("2", "c2"), ("4", "c3")):
                for p in ("weight", "bias"):
                    k = f"{prefix}{src_i}.conv.{ci}.{p}"
                    ours = f"{dst}.{our}.{p}"
                    if k in sd and ours in own:
                        own[ours].copy_(sd[k].to(own[ours].dtype))
                        n += 1
        else:
            for p in ("weight", "bias"):
                k = f"{prefix}{src_i}.{p}"
                ours = f"{dst}.{p}"
                if k in sd and ours in own:
                    own[ours].copy_(sd[k].to(own[ours].dtype))
                    n += 1
    return n


def load_taesd_encoder(model, sd: Dict[str, torch.Tensor], prefix: str = "encoder.") -> int:
    if any(k.startswith(prefix + "layers.") for k in sd):
        prefix = prefix + "layers."
    return _load_taesd_seq(model, sd, prefix, TAESD_SEQ_ENCODER)


def load_taesd_decoder(model, sd: Dict[str, torch.Tensor], prefix: str = "decoder.") -> int:
    if any(k.startswith(prefix + "layers.") for k in sd):
        return _load_taesd_seq(model, sd, prefix + "layers.", TAESD_SEQ_DECODER_DIFFUSERS)
    return _load_taesd_seq(model, sd, prefix, TAESD_SEQ_DECODER_RAW)


def load_taesd(vae, sd: Dict[str, torch.Tensor]) -> Tuple[int, int]:
    """Load both halves of a TinyVAE from a combined TAESD state dict
    (either naming scheme). Returns (n_encoder, n_decoder) tensors loaded."""
    return (
        load_taesd_encoder(vae.encoder, sd, prefix="encoder."),
        load_taesd_decoder(vae.decoder, sd, prefix="decoder."),
    )


def _is_autoencoder_kl(sd: Dict[str, torch.Tensor]) -> bool:
    return any(k.startswith("encoder.down_blocks.") for k in sd)


# ---------------------------------------------------------------------------
# CLIP text encoder (transformers CLIPTextModel naming, as shipped in a
# diffusers snapshot's text_encoder/ directory — reference loads the same
# artifact via CLIPTextModel.from_pretrained, lib/wrapper.py:468-473)
# ---------------------------------------------------------------------------
def clip_text_key_map(n_layers: int) -> List[Tuple[str, str]]:
    pairs: List[Tuple[str, str]] = [
        ("text_model.embeddings.token_embedding.weight", "token_emb.weight"),
        ("text_model.embeddings.position_embedding.weight", "pos_emb.weight"),
        ("text_model.final_layer_norm.weight", "final_ln.weight"),
        ("text_model.final_layer_norm.bias", "final_ln.bias"),
    ]
    for i in range(n_layers):
        src = f"text_model.encoder.layers.{i}"
        dst = f"blocks.{i}"
        for a, b in (
            ("layer_norm1", "ln1"), ("layer_norm2", "ln2"),
            ("self_attn.q_proj", "q_proj"), ("self_attn.k_proj", "k_proj"),
            ("self_attn.v_proj", "v_proj"), ("self_attn.out_proj", "out_proj"),
            ("mlp.fc1", "fc1"), ("mlp.fc2", "fc2"),
        ):
            for p in ("weight", "bias"):
                pairs.append((f"{src}.{a}.{p}", f"{dst}.{b}.{p}"))
    pairs.append(("text_projection.weight", "pooled_proj.weight"))
    return pairs


def load_clip_text_encoder(model, sd: Dict[str, torch.Tensor],
                           strict: bool = False) -> int:
    """Load a CLIPTextModel state dict into our TextEncoder. With strict,
    every expected key (except the optional text_projection) must match."""
    own = dict(model.state_dict())
    n = 0
    missing: List[str] = []
    for src, dst in clip_text_key_map(len(model.blocks)):
        optional = dst.startswith("pooled_proj")
        # transformers <= 4.x serializes CLIPTextModel with a "text_model."
        # prefix (what shipped diffusers snapshots contain); 5.x dropped it
        v = sd.get(src)
        if v is None:
            v = sd.get(src.removeprefix("text_model."))
        if v is None or dst not in own:
            if not optional:
                missing.append(src)
            continue
        if v.shape != own[dst].shape:
            raise ValueError(f"shape mismatch {src}: {v.shape} vs {own[dst].shape}")
        own[dst].copy_(v.to(own[dst].dtype))
        n += 1
    if strict and missing:
        raise KeyError(f"missing {len(missing)} CLIP keys, e.g. {missing[:5]}")
    return n


def load_model_dir(engine, model_dir: str) -> bool:
    """Load UNet (+ TAESD enc+dec) safetensors from a local diffusers-style
    directory; returns False when nothing was found (random init stays).

    TAESD probe order matches what download.py actually fetches for
    madebyollin/taesd: the diffusers AutoencoderTiny file sits at the
    snapshot TOP LEVEL; the raw split files are taesd_encoder/decoder;
    dreamshaper-8's vae/ subdir is a full AutoencoderKL that cannot populate
    a TinyVAE — it is detected and skipped with a warning."""
    import logging

    from safetensors.torch import load_file

    log = logging.getLogger(__name__)
    found = False
    for sub in ("unet/diffusion_pytorch_model.safetensors", "unet.safetensors"):
        p = os.path.join(model_dir, sub)
        if os.path.exists(p):
            n = load_diffusers_unet(engine.unet, load_file(p), strict=False)
            if n == 0:
                log.warning("UNet checkpoint %s matched 0 tensors", p)
            else:
                found = True
            break
    n_enc = n_dec = 0
    for sub in (
        "taesd.safetensors",
        "diffusion_pytorch_model.safetensors",  # AutoencoderTiny, top level
        "vae/diffusion_pytorch_model.safetensors",
    ):
        p = os.path.join(model_dir, sub)
        if not os.path.exists(p):
            continue
        sd = load_file(p)
        if _is_autoencoder_kl(sd):
            log.warning("%s is a full AutoencoderKL, not TAESD; skipping "
                        "(fetch madebyollin/taesd for the TinyVAE weights)", p)
            continue
        n_enc, n_dec = load_taesd(engine.vae, sd)
        if n_enc or n_dec:
            found = True
            break
    # raw split files (madebyollin/taesd also ships these)
    if n_enc == 0:
        p = os.path.join(model_dir, "taesd_encoder.safetensors")
        if os.path.exists(p):
            n_enc = load_taesd_encoder(engine.vae.encoder, load_file(p), prefix="")
            found = found or n_enc > 0
    if n_dec == 0:
        p = os.path.join(model_dir, "taesd_decoder.safetensors")
        if os.path.exists(p):
            n_dec = load_taesd_decoder(engine.vae.decoder, load_file(p), prefix="")
            found = found or n_dec > 0
    if found and n_enc == 0 and n_dec == 0:
        log.warning("model dir %s: UNet loaded but no TAESD weights matched — "
                    "VAE stays random-init", model_dir)
    # CLIP text encoder(s) + BPE tokenizer(s)
    # (reference lib/wrapper.py:468-473; sdxl adds text_encoder_2)
    te = getattr(engine, "text_encoder", None)
    if te is not None:
        targets = [(te, "text_encoder")]
        if hasattr(te, "enc1") and hasattr(te, "enc2"):  # sdxl dual
            targets = [(te.enc1, "text_encoder"), (te.enc2, "text_encoder_2")]
        for mod, sub in targets:
            for fname in ("model.safetensors", "pytorch_model.safetensors"):
                p = os.path.join(model_dir, sub, fname)
                if os.path.exists(p):
                    try:
                        n = load_clip_text_encoder(mod, load_file(p))
                        if n:
                            found = True
                        else:
                            log.warning("CLIP checkpoint %s matched 0 tensors", p)
                    except ValueError as e:
                        log.warning("CLIP checkpoint %s skipped: %s", p, e)
                    break
        if hasattr(te, "load_tokenizer_dir") and te.load_tokenizer_dir(model_dir):
            log.info("BPE tokenizer loaded from %s", model_dir)
    return found
