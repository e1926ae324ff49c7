"""FP8 (OCP e4m3) serving-tier tests.

CPU part: the quantization math (per-OC weight scaling, activation
round-trip error law, conv emulation accuracy vs f32). GPU part
(test_fp8_gpu.py): the MX-MFMA kernel against this exact emulation.

The fp8 tier is an MI355X-native addition (no reference counterpart):
v_mfma_scale_f32_16x16x128_f8f6f4 runs at 2x the f16 MFMA rate and the
fp8 weights halve the weight-bound layers' HBM traffic (ROADMAP item 1).
"""
import math

import pytest
import torch
import torch.nn.functional as F

from ai_rtc_agent_amd import ops


def test_weight_quant_shapes_and_scale_law():
    g = torch.Generator().manual_seed(0)
    w = torch.randn(64, 32, 3, 3, generator=g).half()
    q, s = ops.quantize_weight_fp8(w)
    assert q.dtype == torch.uint8 and q.shape == (64, 32 * 9)
    assert s.dtype == torch.float32 and s.shape == (64,)
    # per-OC absmax maps to +-448: decode magnitude max equals absmax (RNE
    # of the extreme is exact by construction: absmax/scale = 448 exactly)
    dec = q.view(torch.float8_e4m3fn).to(torch.float32) * s[:, None]
    wf = w.float().permute(0, 2, 3, 1).reshape(64, -1)
    assert torch.allclose(dec.abs().amax(dim=1), wf.abs().amax(dim=1), rtol=1e-6)


def test_weight_quant_relative_error_bound():
    g = torch.Generator().manual_seed(1)
    w = torch.randn(16, 64, 3, 3, generator=g).half()
    q, s = ops.quantize_weight_fp8(w)
    dec = (q.view(torch.float8_e4m3fn).to(torch.float32) * s[:, None])
    wf = w.float().permute(0, 2, 3, 1).reshape(16, -1)
    err = (dec - wf).abs()
    # e4m3: 3 mantissa bits -> rel err <= 2^-4 for normals; subnormal floor
    # is scale * 2^-10 absolute
    bound = wf.abs() * 2 ** -4 + s[:, None] * 2 ** -9
    assert (err <= bound + 1e-9).all()


def test_activation_roundtrip_error():
    g = torch.Generator().manual_seed(2)
    x = torch.randn(4096, generator=g).float() * 3.0
    scale = x.abs().max().item() / ops.FP8_MAX
    y = ops.fp8_roundtrip(x, scale)
    err = (y - x).abs()
    assert (err <= x.abs() * 2 ** -4 + scale * 2 ** -9 + 1e-9).all()
    # SNR sanity: quantization noise well below signal
    snr = 10 * math.log10((x ** 2).mean().item() / ((y - x) ** 2).mean().item())
    assert snr > 25, f"e4m3 round-trip SNR {snr:.1f} dB too low"


@pytest.mark.parametrize("ic,oc,h,r,act", [
    (64, 64, 16, 3, ops.ACT_SILU),
    (128, 64, 8, 1, ops.ACT_NONE),
])
def test_conv_fp8_emulation_close_to_f32(ic, oc, h, r, act):
    g = torch.Generator().manual_seed(3)
    x = (torch.randn(2, h, h, ic, generator=g)).half()
    w = (torch.randn(oc, ic, r, r, generator=g) / math.sqrt(ic * r * r)).half()
    b = torch.randn(oc, generator=g).half()
    a_scale = x.float().abs().max().item() / ops.FP8_MAX
    y8 = ops.conv2d_fp8_nhwc(x, w, a_scale, b, padding=r // 2, act=act)
    yf = ops.conv2d_nhwc(x, w, b, padding=r // 2, act=act)
    # fp8 noise at K = ic*r*r accumulation: compare in SNR terms
    num = (yf.float() ** 2).mean().item()
    den = ((y8.float() - yf.float()) ** 2).mean().item()
    snr = 10 * math.log10(num / max(den, 1e-20))
    assert snr > 20, f"fp8 conv emulation SNR {snr:.1f} dB vs f16 path"


def test_gn_fp8_output_and_prequantized_conv():
    """Producer-side quantization: GN writes e4m3 codes, conv consumes u8."""
    g = torch.Generator().manual_seed(5)
    ic, oc, h = 64, 64, 8
    x = torch.randn(2, h, h, ic, generator=g).half()
    gamma = torch.randn(ic, generator=g).float() * 0.3 + 1.0
    beta = torch.randn(ic, generator=g).float() * 0.1
    w = (torch.randn(oc, ic, 3, 3, generator=g) / math.sqrt(ic * 9)).half()
    ref = ops.group_norm_silu_nhwc(x, 8, gamma, beta).float()
    sa = ref.abs().max().item() / ops.FP8_MAX  # calibrated scale
    q = ops.group_norm_silu_nhwc(x, 8, gamma, beta, fp8_scale=sa)
    assert q.dtype == torch.uint8 and q.shape == x.shape
    # decode matches the f16 GN output within e4m3 noise
    dec = q.view(torch.float8_e4m3fn).to(torch.float32) * sa
    snr = 10 * math.log10((ref ** 2).mean().item() /
                          ((dec - ref) ** 2).mean().item() + 1e-20)
    assert snr > 25, f"GN fp8 codes SNR {snr:.1f} dB"
    # q8-input conv emulation == quantize-then-conv
    y_q8 = ops.conv2d_fp8_nhwc(q, w, sa)
    assert y_q8.dtype == torch.float16
    y_f16 = ops.conv2d_fp8_nhwc(ref.half(), w, sa)
    d = (y_q8.float() - y_f16.float()).abs().max().item()
    # same codes modulo the f16 round-trip of ref -> tiny
    assert d < 0.05, f"pre-quantized vs inline-quantized conv differ by {d}"


def test_sched_fused_matches_scheduler_math():
    """CPU parity of the fused scheduler ops vs StreamScheduler methods."""
    from ai_rtc_agent_amd.engine.scheduler import StreamScheduler

    sch = StreamScheduler(num_inference_steps=50)
    co = sch.coefficients([18, 26, 35, 45], 1, torch.device("cpu"), torch.float32)
    g = torch.Generator().manual_seed(0)
    x0 = torch.randn(4, 8, 8, 4, generator=g)
    nz = torch.randn(4, 8, 8, 4, generator=g)
    eps = torch.randn(4, 8, 8, 4, generator=g)
    got = ops.sched_add_noise(x0, nz, co["alpha_f32"], co["beta_f32"])
    ref = sch.add_noise(x0, nz, co["alpha_prod_t_sqrt"], co["beta_prod_t_sqrt"])
    assert torch.allclose(got, ref, atol=1e-5)
    got2 = ops.sched_blend(x0, eps, co["alpha_f32"], co["beta_f32"],
                           co["c_out_f32"], co["c_skip_f32"])
    ref2 = sch.step_batch(eps, x0, co)
    assert torch.allclose(got2, ref2, atol=1e-5)


def test_engine_fp8_calibrate_freeze_gate():
    """Engine lifecycle: calibrate on first frames -> freeze scales ->
    quality gate -> fp8 active (CPU emulation of the same math)."""
    from ai_rtc_agent_amd.config import EngineConfig
    from ai_rtc_agent_amd.engine.engine import StreamDiffusionEngine

    cfg = EngineConfig(
        model_id="none", model_family="tiny", width=64, height=64,
        t_index_list=[30], cfg_type="none", use_lcm_lora=False,
        device="cpu", acceleration="eager", use_hip_graph=False,
        use_fp8=True, fp8_calib_frames=2, fp8_min_snr_db=10.0,
    )
    cfg.similarity_filter.enabled = False
    eng = StreamDiffusionEngine(cfg)
    eng.prepare()
    assert eng._fp8_calib_left == 2 and len(eng._fp8_norms) > 0
    frame = torch.randint(0, 255, (64, 64, 3), dtype=torch.uint8)
    eng(frame)
    assert eng._fp8_calib_left == 1 and not eng.fp8_active
    out = eng(frame)
    assert eng._fp8_calib_left == 0
    assert eng.fp8_snr_db is not None
    assert eng.fp8_active, f"gate failed at {eng.fp8_snr_db:.1f} dB"
    # scales frozen on eligible norms, and serving still works
    assert all(n._fp8_scale is not None and n._fp8_scale > 0
               for n in eng._fp8_norms)
    out2 = eng(frame)
    assert out2.shape == out.shape and out2.dtype == torch.uint8
    st = eng.stats()
    assert st["fp8"]["active"] and st["fp8"]["layers"] == len(eng._fp8_norms)


def test_taesd_fp8_chain_emulation():
    """TAESD block chain in fp8: c1 f16->q8, c2 q8->q8, c3 q8->f16 with the
    f16 skip residual; decode output tracks the f16 path."""
    from ai_rtc_agent_amd.models.taesd import TinyVAE, fp8_flag_convs

    torch.manual_seed(3)
    vae = TinyVAE().eval()
    flags = fp8_flag_convs(vae)
    assert len(flags["convs"]) % 3 == 0 and len(flags["convs"]) > 0
    lat = torch.randn(1, 8, 8, 4)
    with torch.no_grad():
        ref = vae.decode(lat).float()
        # calibrate amaxes through one f16 pass
        for c in flags["convs"]:
            c._fp8_calibrate = True
        vae.decode(lat)
        for c in flags["convs"]:
            c._fp8_calibrate = False
            c._fp8_in_scale = c._fp8_in_amax * 1.5 / ops.FP8_MAX
        for c in flags["outs"]:
            c._fp8_out_scale = c._fp8_out_amax * 1.5 / ops.FP8_MAX
        got = vae.decode(lat).float()
    snr = 10 * math.log10((ref ** 2).mean().item() /
                          max(((got - ref) ** 2).mean().item(), 1e-20))
    assert snr > 16, f"TAESD fp8 chain SNR {snr:.1f} dB"
    # cleanup class-level-shadowing instance attrs not needed (fresh vae)


def test_engine_fp8_gate_fallback():
    """An absurd quality threshold must fall back to f16 serving."""
    from ai_rtc_agent_amd.config import EngineConfig
    from ai_rtc_agent_amd.engine.engine import StreamDiffusionEngine

    cfg = EngineConfig(
        model_id="none", model_family="tiny", width=64, height=64,
        t_index_list=[30], cfg_type="none", use_lcm_lora=False,
        device="cpu", acceleration="eager", use_hip_graph=False,
        use_fp8=True, fp8_calib_frames=1, fp8_min_snr_db=200.0,
    )
    cfg.similarity_filter.enabled = False
    eng = StreamDiffusionEngine(cfg)
    eng.prepare()
    frame = torch.randint(0, 255, (64, 64, 3), dtype=torch.uint8)
    eng(frame)
    assert not eng.fp8_active
    assert all(n._fp8_scale is None for n in eng._fp8_norms)
    out = eng(frame)  # still serves, in f16
    assert out.dtype == torch.uint8


def test_conv_fp8_emulation_with_affine_residual_cbias():
    g = torch.Generator().manual_seed(4)
    ic, oc, h = 64, 64, 8
    x = torch.randn(2, h, h, ic, generator=g).half()
    w = (torch.randn(oc, ic, 3, 3, generator=g) / math.sqrt(ic * 9)).half()
    aff = torch.randn(2, ic, 2, generator=g).float() * 0.2 + 0.6
    res = torch.randn(2, h, h, oc, generator=g).half()
    cb = torch.randn(2, oc, generator=g).half()
    a_scale = 4.0 / ops.FP8_MAX
    y8 = ops.conv2d_fp8_nhwc(x, w, a_scale, None, act=ops.ACT_SILU,
                             residual=res, channel_bias=cb, in_affine=aff,
                             in_act=ops.ACT_SILU)
    yf = ops.conv2d_nhwc(x, w, None, act=ops.ACT_SILU, residual=res,
                         channel_bias=cb, in_affine=aff, in_act=ops.ACT_SILU)
    num = (yf.float() ** 2).mean().item()
    den = ((y8.float() - yf.float()) ** 2).mean().item()
    snr = 10 * math.log10(num / max(den, 1e-20))
    assert snr > 18, f"fused-epilogue fp8 emulation SNR {snr:.1f} dB"


def test_fp8_recalibrates_after_weight_refresh():
    """LoRA hot-swap shifts activation stats: refresh_weights must drop the
    fp8 tier back to calibration and re-gate on fresh frames."""
    from ai_rtc_agent_amd.config import EngineConfig
    from ai_rtc_agent_amd.engine.engine import StreamDiffusionEngine
    from ai_rtc_agent_amd.models.lora import fuse_lora_state_dict, make_random_lora

    cfg = EngineConfig(
        model_id="none", model_family="tiny", width=64, height=64,
        t_index_list=[30], cfg_type="none", use_lcm_lora=False,
        device="cpu", acceleration="eager", use_hip_graph=False,
        use_fp8=True, fp8_calib_frames=1, fp8_min_snr_db=10.0,
    )
    cfg.similarity_filter.enabled = False
    eng = StreamDiffusionEngine(cfg)
    eng.prepare()
    frame = torch.randint(0, 255, (64, 64, 3), dtype=torch.uint8)
    eng(frame)
    assert eng.fp8_active
    fuse_lora_state_dict(eng.unet, make_random_lora(eng.unet, rank=2, seed=9), scale=0.5)
    eng.refresh_weights()
    assert not eng.fp8_active and eng._fp8_calib_left == 1
    eng(frame)  # one calibration frame -> freeze + gate again
    assert eng.fp8_active, f"re-gate failed at {eng.fp8_snr_db} dB"


def test_fp8_composes_with_sdxl_added_cond_path():
    """tiny_xl engine (addition-embedding path) with fp8 + the static temb
    cache: calibrate, gate, serve — the added-cond contribution must live
    inside the precomputed temb and survive a prompt update."""
    from ai_rtc_agent_amd.config import EngineConfig
    from ai_rtc_agent_amd.engine.engine import StreamDiffusionEngine

    cfg = EngineConfig(
        model_id="none", model_family="tiny_xl", width=64, height=64,
        t_index_list=[30], cfg_type="none", use_lcm_lora=False,
        device="cpu", acceleration="eager", use_hip_graph=False,
        use_fp8=True, fp8_calib_frames=1, fp8_min_snr_db=8.0,
    )
    cfg.similarity_filter.enabled = False
    eng = StreamDiffusionEngine(cfg)
    eng.prepare()
    assert eng._added_cond is not None
    frame = torch.randint(0, 255, (64, 64, 3), dtype=torch.uint8)
    eng(frame)
    assert eng.fp8_active
    temb0 = eng.unet._temb_static.clone()
    eng.update_prompt("a different prompt")
    assert not torch.equal(eng.unet._temb_static, temb0), \
        "prompt update changes pooled added-cond -> static temb must refresh"
    out = eng(frame)
    assert out.shape == (64, 64, 3) and out.dtype == torch.uint8


def test_fp8_with_frame_buffer_batching():
    """fp8 tier under the multi-stream batched serving shape (fbs=2)."""
    from ai_rtc_agent_amd.config import EngineConfig
    from ai_rtc_agent_amd.engine.engine import StreamDiffusionEngine

    cfg = EngineConfig(
        model_id="none", model_family="tiny", width=64, height=64,
        t_index_list=[30], cfg_type="none", use_lcm_lora=False,
        device="cpu", acceleration="eager", use_hip_graph=False,
        frame_buffer_size=2, use_fp8=True, fp8_calib_frames=1,
        fp8_min_snr_db=8.0,
    )
    cfg.similarity_filter.enabled = False
    eng = StreamDiffusionEngine(cfg)
    eng.prepare()
    batch = torch.randint(0, 255, (2, 64, 64, 3), dtype=torch.uint8)
    eng(batch)
    assert eng.fp8_active
    out = eng(batch)
    assert out.shape == (2, 64, 64, 3) and out.dtype == torch.uint8


def test_fp8_covers_controlnet_resnets():
    """ControlNet shares ResnetBlock — its GN->conv pairs join the tier."""
    from ai_rtc_agent_amd.config import EngineConfig
    from ai_rtc_agent_amd.engine.engine import StreamDiffusionEngine
    from ai_rtc_agent_amd.models.unet import fp8_eligible_norms

    cfg = EngineConfig(
        model_id="none", model_family="tiny", width=64, height=64,
        t_index_list=[30], cfg_type="none", use_lcm_lora=False,
        device="cpu", acceleration="eager", use_hip_graph=False,
        use_controlnet=True, use_fp8=True, fp8_calib_frames=1,
        fp8_min_snr_db=8.0,
    )
    cfg.similarity_filter.enabled = False
    eng = StreamDiffusionEngine(cfg)
    eng.prepare()
    n_unet = len(fp8_eligible_norms(eng.unet))
    assert len(eng._fp8_norms) > n_unet, "controlnet norms must be included"
    frame = torch.randint(0, 255, (64, 64, 3), dtype=torch.uint8)
    eng(frame)
    assert eng.fp8_active
    out = eng(frame)
    assert out.shape == (64, 64, 3)
    # controlnet norms that saw activations have frozen scales
    cn = [n for n in eng._fp8_norms[n_unet:] if n._fp8_scale is not None]
    assert len(cn) > 0
