"""ControlNet conditioning + safety checker (reference option surface:
lib/wrapper.py:617-643 controlnet, :930-942 safety checker)."""
import torch

from ai_rtc_agent_amd.config import EngineConfig
from ai_rtc_agent_amd.engine import StreamDiffusionEngine
from ai_rtc_agent_amd.models import UNet2DCondition, UNetConfig
from ai_rtc_agent_amd.models.controlnet import ControlNet
from ai_rtc_agent_amd.models.safety import SafetyChecker


def test_controlnet_shapes_and_zero_init():
    cfg = UNetConfig.tiny()
    cn = ControlNet(cfg).eval()
    unet = UNet2DCondition(cfg).eval()
    x = torch.randn(1, 16, 16, 4)
    t = torch.tensor([100])
    ctx = torch.randn(1, 77, cfg.cross_attention_dim)
    hint = torch.rand(1, 128, 128, 3) * 2 - 1
    with torch.no_grad():
        skips, mid = cn(x, t, ctx, hint)
        # zero-init convs: residuals are exactly zero at init -> UNet output
        # unchanged ("do no harm")
        assert all(torch.count_nonzero(s) == 0 for s in skips)
        assert torch.count_nonzero(mid) == 0
        base = unet(x, t, ctx)
        with_cn = unet(x, t, ctx, control=(skips, mid))
        assert torch.equal(base, with_cn)


def test_controlnet_conditions_after_training_signal():
    cfg = UNetConfig.tiny()
    cn = ControlNet(cfg).eval()
    # un-zero the zero-convs (simulating trained weights)
    for zc in list(cn.zero_convs) + [cn.mid_zero]:
        torch.nn.init.normal_(zc.weight, std=0.1)
    unet = UNet2DCondition(cfg).eval()
    x = torch.randn(1, 16, 16, 4)
    t = torch.tensor([100])
    ctx = torch.randn(1, 77, cfg.cross_attention_dim)
    with torch.no_grad():
        s1 = cn(x, t, ctx, torch.rand(1, 128, 128, 3))
        s2 = cn(x, t, ctx, torch.rand(1, 128, 128, 3))
        y1 = unet(x, t, ctx, control=s1)
        y2 = unet(x, t, ctx, control=s2)
    assert not torch.allclose(y1, y2), "hint must condition the output"


def test_engine_with_controlnet_end_to_end():
    cfg = EngineConfig(
        model_family="tiny", width=64, height=64, device="cpu",
        use_hip_graph=False, use_lcm_lora=False, use_controlnet=True,
        t_index_list=[0, 20], num_inference_steps=50,
    )
    e = StreamDiffusionEngine(cfg)
    e.prepare()
    out = e(torch.randint(0, 256, (64, 64, 3), dtype=torch.uint8))
    assert out.shape == (64, 64, 3)


def test_safety_checker_blanks_flagged_frames():
    sc = SafetyChecker(threshold=0.5).eval()
    img = torch.rand(2, 64, 64, 3) * 2 - 1
    with torch.no_grad():
        scores = sc.score(img)
        assert scores.shape == (2,) and ((scores >= 0) & (scores <= 1)).all()
        sc.threshold = -1.0  # force-flag everything
        out = sc.filter(img)
        assert torch.allclose(out, torch.full_like(out, -1.0)), "flagged -> black"
        sc.threshold = 2.0  # pass everything
        out2 = sc.filter(img)
        assert torch.allclose(out2, img, atol=1e-3)


def test_engine_with_safety_checker():
    cfg = EngineConfig(
        model_family="tiny", width=64, height=64, device="cpu",
        use_hip_graph=False, use_lcm_lora=False, use_safety_checker=True,
    )
    e = StreamDiffusionEngine(cfg)
    e.prepare()
    out = e(torch.randint(0, 256, (64, 64, 3), dtype=torch.uint8))
    assert out.shape == (64, 64, 3)
