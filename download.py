#!/usr/bin/env python3
"""Model asset fetcher — parity with reference download.py.

The reference fetches HF snapshots (lykon/dreamshaper-8,
latent-consistency/lcm-lora-sdv1-5, madebyollin/taesd — download.py:17-21)
plus a Civitai LoRA (id 6526 / version 7657, download.py:23-41).

This environment has no network, so --offline (auto-detected on failure)
synthesizes random-init weights IN THE SAME CACHE LAYOUT so every
downstream path (build.py, the agent, LoRA fusion) exercises the real
loading code. On a connected deployment box the HF/Civitai paths run.
"""
from __future__ import annotations

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from ai_rtc_agent_amd.utils.paths import civitai_model_path

HF_MODELS = [
    "lykon/dreamshaper-8",
    "latent-consistency/lcm-lora-sdv1-5",
    "madebyollin/taesd",
]
CIVITAI_MODEL_ID = 6526
CIVITAI_VERSION_ID = 7657


def download_hf(model_id: str) -> bool:
    try:
        from huggingface_hub import snapshot_download

        snapshot_download(model_id)
        print(f"fetched {model_id}")
        return True
    except Exception as e:
        print(f"HF fetch failed for {model_id}: {e}")
        return False


def download_civitai(model_id: int, version_id: int) -> bool:
    import requests

    path = civitai_model_path(model_id, version_id)
    if os.path.exists(path):
        return True
    os.makedirs(os.path.dirname(path), exist_ok=True)
    try:
        url = f"https://civitai.com/api/download/models/{version_id}"
        r = requests.get(url, timeout=60)
        r.raise_for_status()
        with open(path, "wb") as f:
            f.write(r.content)
        print(f"fetched civitai {model_id}/{version_id}")
        return True
    except Exception as e:
        print(f"civitai fetch failed: {e}")
        return False


def synthesize_offline_assets(seed: int = 0) -> None:
    """Random-init stand-ins in the real cache layout (offline builds)."""
    import torch
    from safetensors.torch import save_file

    from ai_rtc_agent_amd.models import UNet2DCondition, UNetConfig
    from ai_rtc_agent_amd.models.lora import make_random_lora

    torch.manual_seed(seed)
    path = civitai_model_path(CIVITAI_MODEL_ID, CIVITAI_VERSION_ID)
    os.makedirs(os.path.dirname(path), exist_ok=True)
    if not os.path.exists(path):
        lora = make_random_lora(UNet2DCondition(UNetConfig.tiny()), rank=4, seed=seed)
        save_file(lora, path)
        print(f"synthesized offline LoRA at {path}")


def download(offline: bool = False) -> None:
    ok = True
    if not offline:
        for m in HF_MODELS:
            ok = download_hf(m) and ok
        ok = download_civitai(CIVITAI_MODEL_ID, CIVITAI_VERSION_ID) and ok
    if offline or not ok:
        print("falling back to offline synthetic assets")
        synthesize_offline_assets()


if __name__ == "__main__":
    p = argparse.ArgumentParser()
    p.add_argument("--offline", action="store_true")
    download(p.parse_args().offline)
