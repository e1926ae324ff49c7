#!/usr/bin/env python3
"""AOT engine build — parity with reference build.py:11-32.

The reference instantiates its wrapper with the production config
(dreamshaper-8 + LoRA at scale 1.0, LCM-LoRA, TinyVAE, img2img, fp16,
cfg "self") to force TensorRT engine compilation into the cache. Ours
builds the MI355X kernel plan: instantiate the engine (fusing LoRAs),
warm every kernel-side weight transform, capture the hipGraph once when a
GPU is present, and serialize the plan under ENGINES_CACHE/engines--<model>
(same directory contract as lib/wrapper.py:593-597).

    python build.py [--model-id ...] [--family sd15|sd21|sdxl] [--lora path:scale]
"""
from __future__ import annotations

import argparse
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from ai_rtc_agent_amd.config import EngineConfig
from ai_rtc_agent_amd.engine import StreamDiffusionEngine
from ai_rtc_agent_amd.engine.plan import save_plan


def build(
    model_id: str = "lykon/dreamshaper-8",
    family: str = "sd15",
    lora: dict | None = None,
    width: int = 512,
) -> str:
    cfg = EngineConfig(
        model_id=model_id,
        model_family=family,
        width=width,
        height=width,
        use_lcm_lora=True,
        use_tiny_vae=True,
        cfg_type="self",
        mode="img2img",
        lora_dict=lora,
        device="cuda" if torch.cuda.is_available() else "cpu",
        use_hip_graph=torch.cuda.is_available(),
    )
    eng = StreamDiffusionEngine(cfg)
    eng.prepare()
    if torch.cuda.is_available():
        # warm the kernel-side weight transforms + capture the graph once
        frame = torch.zeros((width, width, 3), dtype=torch.uint8, device=cfg.device)
        eng(frame)
        torch.cuda.synchronize()
    out = save_plan(eng)
    print(f"engine plan written to {out}")
    return out


if __name__ == "__main__":
    p = argparse.ArgumentParser()
    p.add_argument("--model-id", default="lykon/dreamshaper-8")
    p.add_argument("--family", default="sd15", choices=["sd15", "sd21", "sdxl", "tiny"])
    p.add_argument("--width", type=int, default=512)
    p.add_argument("--lora", default=None, help="path:scale[,path:scale...]")
    a = p.parse_args()
    lora = None
    if a.lora:
        lora = {}
        for item in a.lora.split(","):
            path, _, scale = item.partition(":")
            lora[path] = float(scale or 1.0)
    build(a.model_id, a.family, lora, a.width)
