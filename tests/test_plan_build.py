"""AOT plan cache + build/download CLIs (tiny configs, CPU).

Contract parity: engines--<model> directory naming (reference
lib/wrapper.py:593-597), load-else-build ladder (lib/wrapper.py:611-615).
"""
import os

import torch

from ai_rtc_agent_amd.config import EngineConfig
from ai_rtc_agent_amd.engine import StreamDiffusionEngine
from ai_rtc_agent_amd.engine.plan import load_plan, plan_dir, save_plan


def tiny_cfg(tmp):
    return EngineConfig(
        model_id="test/tiny-model",
        model_family="tiny",
        width=64,
        height=64,
        device="cpu",
        use_hip_graph=False,
        use_lcm_lora=False,
    )


def test_plan_dir_naming(tmp_path):
    d = plan_dir("lykon/dreamshaper-8", str(tmp_path))
    assert d.endswith("engines--lykon--dreamshaper-8")


def test_plan_save_load_roundtrip(tmp_path):
    cfg = tiny_cfg(tmp_path)
    eng = StreamDiffusionEngine(cfg)
    eng.prepare()
    frame = torch.randint(0, 256, (64, 64, 3), dtype=torch.uint8)
    out1 = eng(frame)

    d = save_plan(eng, str(tmp_path))
    assert os.path.exists(os.path.join(d, "plan.json"))
    assert os.path.exists(os.path.join(d, "unet.safetensors"))

    cfg2 = tiny_cfg(tmp_path)
    eng2 = load_plan(cfg2, str(tmp_path))
    assert eng2 is not None
    eng2.prepare()
    out2 = eng2(frame)
    assert torch.equal(out1, out2), "plan-restored engine must reproduce outputs"


def test_plan_load_missing_returns_none(tmp_path):
    cfg = tiny_cfg(tmp_path)
    assert load_plan(cfg, str(tmp_path)) is None


def test_build_cli(tmp_path, monkeypatch):
    monkeypatch.setenv("ENGINES_CACHE", str(tmp_path))
    import build as build_mod

    out = build_mod.build(model_id="test/build-tiny", family="tiny", width=64)
    assert os.path.exists(os.path.join(out, "plan.json"))


def test_download_offline(tmp_path, monkeypatch):
    monkeypatch.setenv("CIVITAI_CACHE", str(tmp_path))
    import importlib

    import download as dl

    importlib.reload(dl)
    dl.download(offline=True)
    from ai_rtc_agent_amd.utils.paths import civitai_model_path

    assert os.path.exists(civitai_model_path(dl.CIVITAI_MODEL_ID, dl.CIVITAI_VERSION_ID))


def test_pipeline_uses_plan_cache(tmp_path, monkeypatch):
    """Serving path picks up a pre-built plan (load-else-build ladder)."""
    monkeypatch.setenv("ENGINES_CACHE", str(tmp_path))
    import build as build_mod

    build_mod.build(model_id="test/pipe-tiny", family="tiny", width=64)
    from ai_rtc_agent_amd.config import EngineConfig
    from ai_rtc_agent_amd.pipeline import StreamDiffusionPipeline

    cfg = EngineConfig(
        model_id="test/pipe-tiny", model_family="tiny", width=64, height=64,
        device="cpu", use_hip_graph=False, use_lcm_lora=False,
    )
    p = StreamDiffusionPipeline(cfg=cfg)
    out = p(torch.randint(0, 256, (64, 64, 3), dtype=torch.uint8))
    assert out.shape == (64, 64, 3)
