"""LoRA loading & weight fusion.

Parity target: the reference fuses LCM-LoRA plus an arbitrary dict of LoRA
files with per-file scales at engine build time (reference
lib/wrapper.py:645-697; build.py:18-32 uses a Civitai "ghibli" LoRA at scale
1.0; download.py:23-41 fetches it). Fusion happens BEFORE kernel-plan
capture, so the hot path never sees LoRA math — same design as the
reference's TRT engines compiled from the fused model.

Formats: a state dict of (down, up, alpha) triples keyed like
"<module_path>.lora_down.weight" / ".lora_up.weight" / ".alpha"
(kohya-style), or "<module_path>.lora_A.weight"/"lora_B.weight" (PEFT-style).
Module paths are matched against named_modules() of our UNet.

W_fused = W + scale * (alpha/rank) * (up @ down)
"""
from __future__ import annotations

import math
import re
from typing import Dict

import torch
import torch.nn as nn


def _collect_pairs(sd: Dict[str, torch.Tensor]) -> Dict[str, dict]:
    out: Dict[str, dict] = {}
    for k, v in sd.items():
        for marker, slot in (
            (".lora_down.weight", "down"),
            (".lora_up.weight", "up"),
            (".lora_A.weight", "down"),
            (".lora_B.weight", "up"),
            (".alpha", "alpha"),
        ):
            if k.endswith(marker):
                base = k[: -len(marker)]
                out.setdefault(base, {})[slot] = v
                break
    return out


def fuse_lora_state_dict(
    model: nn.Module, sd: Dict[str, torch.Tensor], scale: float = 1.0
) -> int:
    """Fuse a LoRA state dict into matching Linear/Conv weights in-place.

    Returns the number of modules fused. Unmatched LoRA keys are skipped
    (the reference behaves the same: fusing is best-effort by name).
    """
    pairs = _collect_pairs(sd)
    by_name = dict(model.named_modules())
    n = 0
    for base, slots in pairs.items():
        if "down" not in slots or "up" not in slots:
            continue
        # normalise separators: kohya uses '_' where modules use '.'
        cand = [base, base.replace("lora_unet_", "").replace("_", ".")]
        target = None
        for c in cand:
            if c in by_name and hasattr(by_name[c], "weight"):
                target = by_name[c]
                break
        if target is None:
            continue
        target_dev = target.weight.device
        down = slots["down"].float().to(target_dev)
        up = slots["up"].float().to(target_dev)
        rank = down.shape[0]
        alpha = float(slots.get("alpha", torch.tensor(float(rank))))
        w = target.weight.data
        if w.dim() == 4:  # conv OIHW: lora stored as (r, I*k*k) / (O, r)
            delta = (up.flatten(1) @ down.flatten(1)).view_as(w)
        else:
            delta = up @ down
        target.weight.data = (w.float() + scale * (alpha / rank) * delta).to(w.dtype)
        n += 1
    return n


def load_lora_file(path: str) -> Dict[str, torch.Tensor]:
    """Load a .safetensors or torch-serialised LoRA file."""
    if path.endswith(".safetensors"):
        from safetensors.torch import load_file

        return load_file(path)
    return torch.load(path, map_location="cpu")


def make_random_lora(
    model: nn.Module, rank: int = 4, seed: int = 0, limit: int = 8
) -> Dict[str, torch.Tensor]:
    """Synthesize a LoRA dict targeting the first `limit` Linear modules —
    offline stand-in for LCM-LoRA (no network: SURVEY.md §7 env note)."""
    g = torch.Generator().manual_seed(seed)
    sd: Dict[str, torch.Tensor] = {}
    count = 0
    for name, m in model.named_modules():
        if count >= limit:
            break
        w = getattr(m, "weight", None)
        if w is None or w.dim() != 2:
            continue
        o, i = w.shape
        sd[f"{name}.lora_down.weight"] = torch.randn(rank, i, generator=g) * 0.01
        sd[f"{name}.lora_up.weight"] = torch.randn(o, rank, generator=g) * 0.01
        sd[f"{name}.alpha"] = torch.tensor(float(rank))
        count += 1
    return sd
