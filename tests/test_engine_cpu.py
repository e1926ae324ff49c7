"""End-to-end CPU engine tests (BASELINE config[0]: plumbing, no GPU).

Covers the reference behavioural surface: stream-batch law, per-frame
__call__, runtime prompt / t_index updates (reference lib/wrapper.py:389-407,
lib/pipeline.py:44-48), txt2img, pipelined-latency semantics (SURVEY.md §3.4
note).
"""
import pytest
import torch

from ai_rtc_agent_amd.config import EngineConfig, sd_turbo_config
from ai_rtc_agent_amd.engine import StreamDiffusionEngine


def make_engine(tiny_cfg, **kw):
    for k, v in kw.items():
        setattr(tiny_cfg, k, v)
    e = StreamDiffusionEngine(tiny_cfg)
    e.prepare()
    return e


def frame(h=64, w=64, seed=0):
    g = torch.Generator().manual_seed(seed)
    return torch.randint(0, 256, (h, w, 3), generator=g, dtype=torch.uint8)


def test_unet_batch_law(tiny_cfg):
    # B = len(t_index) * frame_buffer (reference lib/wrapper.py:159-163)
    assert tiny_cfg.unet_batch == 4
    tiny_cfg.frame_buffer_size = 2
    assert tiny_cfg.unet_batch == 8
    tiny_cfg.cfg_type = "initialize"
    assert tiny_cfg.unet_batch == 10
    tiny_cfg.cfg_type = "full"
    assert tiny_cfg.unet_batch == 16


def test_img2img_shapes_and_determinism(tiny_cfg):
    e = make_engine(tiny_cfg)
    out1 = e(frame(seed=1))
    assert out1.shape == (64, 64, 3) and out1.dtype == torch.uint8

    e2 = make_engine(
        EngineConfig(
            model_family="tiny", width=64, height=64, device="cpu",
            use_hip_graph=False, use_lcm_lora=False,
        )
    )
    out2 = e2(frame(seed=1))
    assert torch.equal(out1, out2), "same seed+input must be deterministic"


def test_stream_batch_pipelining(tiny_cfg):
    """A frame's content takes len(t_index) frame-times to drain."""
    e = make_engine(tiny_cfg)
    buf_before = e._x_t_buffer.clone()
    e(frame(seed=3))
    assert not torch.equal(buf_before, e._x_t_buffer), "FIFO must advance"
    # throughput: one output per call regardless of denoise depth
    for i in range(3):
        out = e(frame(seed=4 + i))
        assert out.shape == (64, 64, 3)


def test_update_prompt_changes_output(tiny_cfg):
    e = make_engine(tiny_cfg)
    f = frame(seed=7)
    base = [e(f) for _ in range(5)][-1]
    e.update_prompt("a completely different prompt with other words")
    after = [e(f) for _ in range(5)][-1]
    assert not torch.equal(base, after)


def test_update_t_index_list_contract(tiny_cfg):
    e = make_engine(tiny_cfg)
    old = e._coeff["alpha_prod_t_sqrt"].clone()
    e.update_t_index_list([18, 26, 35, 45])  # unchanged -> no-op
    assert torch.equal(old, e._coeff["alpha_prod_t_sqrt"])
    e.update_t_index_list([10, 20, 30, 40])  # same length -> in-place coeff update
    assert not torch.equal(old, e._coeff["alpha_prod_t_sqrt"])
    assert e._coeff["sub_timesteps"] == e.scheduler.sub_timesteps([10, 20, 30, 40])
    e.update_t_index_list([5, 25])  # length change -> full re-prepare
    assert e._x_t_buffer.shape[0] == 1 * (2 - 1)


def test_txt2img(tiny_cfg):
    tiny_cfg.mode = "txt2img"
    tiny_cfg.t_index_list = [0, 16, 32, 48]
    e = make_engine(tiny_cfg)
    out = e.txt2img()
    assert out.shape == (1, 64, 64, 3)


def test_sd_turbo_single_step_config():
    cfg = sd_turbo_config(
        model_family="tiny", width=64, height=64, device="cpu",
        use_hip_graph=False,
    )
    assert cfg.t_index_list == [0] and cfg.unet_batch == 1
    e = StreamDiffusionEngine(cfg)
    e.prepare()
    out = e(frame())
    assert out.shape == (64, 64, 3)
    # 1-step: no in-flight buffer at all (batch depth 1 -> zero added latency)
    assert e._x_t_buffer.shape[0] == 0


def test_stats_surface(tiny_cfg):
    e = make_engine(tiny_cfg)
    e(frame())
    s = e.stats()
    assert s["frames"] == 1
    assert "diffusion" in s["stages_ms"]


def test_non_square_resolution():
    from ai_rtc_agent_amd.config import EngineConfig
    cfg = EngineConfig(
        model_family="tiny", width=96, height=64, device="cpu",
        use_hip_graph=False, use_lcm_lora=False, t_index_list=[0, 25],
    )
    e = StreamDiffusionEngine(cfg)
    e.prepare()
    out = e(torch.randint(0, 256, (64, 96, 3), dtype=torch.uint8))
    assert out.shape == (64, 96, 3)


def test_lora_hot_swap():
    """Identical engines diverge exactly when one hot-swaps a LoRA (the
    stream-batch FIFO advances every call, so the control engine isolates
    the LoRA effect)."""
    from ai_rtc_agent_amd.models.lora import make_random_lora

    def fresh():
        cfg = EngineConfig(
            model_family="tiny", width=64, height=64, device="cpu",
            use_hip_graph=False, use_lcm_lora=False,
        )
        e = StreamDiffusionEngine(cfg)
        e.prepare()
        return e

    e1, e2 = fresh(), fresh()
    for i in range(3):
        a, b = e1(frame(seed=i)), e2(frame(seed=i))
        assert torch.equal(a, b), "engines must match before the swap"
    sd = make_random_lora(e2.unet, rank=2, seed=99, limit=20)
    assert e2.load_lora(sd, scale=2.0) > 0
    a, b = e1(frame(seed=7)), e2(frame(seed=7))
    assert not torch.equal(a, b), "hot-swapped LoRA must change output"


def test_frame_buffer_batching(tiny_cfg):
    """frame_buffer_size > 1: batched frames in, batched frames out
    (reference frame_bff_size, lib/wrapper.py:159-163)."""
    tiny_cfg.frame_buffer_size = 2
    e = make_engine(tiny_cfg)
    assert e.cfg.unet_batch == 8
    frames = torch.randint(0, 256, (2, 64, 64, 3), dtype=torch.uint8)
    out = e(frames)
    assert out.shape == (2, 64, 64, 3)


def test_t_index_same_length_update_refreshes_fused_coeffs_and_temb():
    """Same-length t-index updates write IN PLACE (no re-capture): the
    fused-scheduler f32 coefficient arrays and the static time-embedding
    caches must follow."""
    from ai_rtc_agent_amd.config import EngineConfig
    from ai_rtc_agent_amd.engine.engine import StreamDiffusionEngine

    cfg = EngineConfig(model_id="none", model_family="tiny", width=64,
                       height=64, t_index_list=[30], cfg_type="none",
                       use_lcm_lora=False, device="cpu",
                       acceleration="eager", use_hip_graph=False)
    cfg.similarity_filter.enabled = False
    eng = StreamDiffusionEngine(cfg)
    eng.prepare()
    temb_before = eng.unet._temb_static.clone()
    eng.update_t_index_list([10])
    fresh = eng.scheduler.coefficients([10], cfg.frame_buffer_size,
                                       eng.device, eng.dtype)
    for k in ("alpha_f32", "beta_f32", "c_skip_f32", "c_out_f32"):
        assert torch.allclose(eng._coeff[k], fresh[k]), k
    assert not torch.equal(eng.unet._temb_static, temb_before), \
        "static temb must refresh with the new timesteps"
    assert eng.unet._temb_src is eng._unet_batch_timesteps()
    # serving still works after the in-place update
    frame = torch.randint(0, 255, (64, 64, 3), dtype=torch.uint8)
    out = eng(frame)
    assert out.shape == (64, 64, 3)


def test_t_index_update_with_cfg_full_rebuilds_batched_ts():
    """cfg full doubles the ts batch via a separate cat buffer — a
    same-length t-index update must rebuild it (stale-cache regression)."""
    from ai_rtc_agent_amd.config import EngineConfig
    from ai_rtc_agent_amd.engine.engine import StreamDiffusionEngine

    cfg = EngineConfig(model_id="none", model_family="tiny", width=64,
                       height=64, t_index_list=[30], cfg_type="full",
                       guidance_scale=1.5, use_lcm_lora=False, device="cpu",
                       acceleration="eager", use_hip_graph=False)
    cfg.similarity_filter.enabled = False
    eng = StreamDiffusionEngine(cfg)
    eng.prepare()
    assert eng.rcfg.active
    eng.update_t_index_list([10])
    ts = eng._unet_batch_timesteps()
    want = int(eng.scheduler.timesteps[10])
    assert (ts == want).all(), f"batched ts stale: {ts.tolist()} != {want}"
    frame = torch.randint(0, 255, (64, 64, 3), dtype=torch.uint8)
    out = eng(frame)
    assert out.shape == (64, 64, 3)
