"""Golden-tensor tests: HIP kernels vs plain-PyTorch fp32 references
(SURVEY.md §4 item b). Every op in the hot path is compared on the GPU
against the fp32 torch implementation of the same op.

Tolerance notes: f16 storage with f32 accumulation; conv/attention compare
at ~2e-2 abs (K up to ~11k accumulation in f32, inputs N(0,1)).
"""
import math

import pytest
import torch
import torch.nn.functional as F

from ai_rtc_agent_amd import ops

pytestmark = pytest.mark.gpu

DEV = "cuda"


def rnd(*shape, seed=0, scale=1.0):
    g = torch.Generator().manual_seed(seed)
    return (torch.randn(*shape, generator=g) * scale).half().to(DEV)


def assert_close(got, ref, atol, name):
    diff = (got.float() - ref.float()).abs()
    rel = diff.max().item()
    assert rel <= atol, f"{name}: max abs err {rel} > {atol}"


def test_ext_loads():
    assert ops.hip_ext() is not None, "HIP extension must load on the GPU box"


@pytest.mark.parametrize("ic,oc,h,stride,pad,r", [
    (32, 64, 16, 1, 1, 3),      # basic 3x3
    (64, 64, 32, 2, 1, 3),      # stride-2 downsample
    (320, 320, 16, 1, 1, 3),    # SD resnet shape
    (320, 640, 8, 1, 0, 1),     # 1x1 projection
    (4, 32, 16, 1, 1, 3),       # small-IC direct path (VAE conv_in)
    (96, 40, 12, 1, 1, 3),      # ragged OC, non-pow2 spatial
    (1280, 1280, 8, 1, 1, 3),   # 8x8 wide UNet layer -> split-K path
    (640, 1280, 16, 2, 1, 3),   # stride-2 into split-K territory
    (64, 64, 64, 1, 1, 3),      # TAESD-shaped: BM128 large-spatial path
    (320, 64, 64, 1, 0, 1),     # 1x1 conv on the BM128 path
    (3, 64, 160, 1, 1, 3),      # tiny-IC at large spatial: pad-to-32 MFMA route
])
def test_conv2d_vs_torch(ic, oc, h, stride, pad, r):
    x = rnd(2, h, h, ic, seed=ic + oc)
    w = rnd(oc, ic, r, r, seed=1, scale=1.0 / math.sqrt(ic * r * r))
    b = torch.randn(oc, generator=torch.Generator().manual_seed(2)).half().to(DEV)
    y = ops.conv2d_nhwc(x, w, b, stride=stride, padding=pad)
    ref = F.conv2d(x.permute(0, 3, 1, 2).float(), w.float(), b.float(),
                   stride=stride, padding=pad).permute(0, 2, 3, 1)
    assert y.shape == ref.shape
    assert_close(y, ref, 2e-2 * max(1.0, math.sqrt(ic * r * r) / 8), "conv2d")


def test_conv2d_fused_residual_and_channel_bias():
    # epilogue law: y = act(conv + bias + channel_bias + residual)
    x = rnd(2, 8, 8, 64, seed=20)
    w = rnd(32, 64, 3, 3, seed=21, scale=0.04)
    cb = rnd(2, 32, seed=22)
    res = rnd(2, 8, 8, 32, seed=23)
    y = ops.conv2d_nhwc(x, w, None, residual=res, channel_bias=cb, act=ops.ACT_RELU)
    base = F.conv2d(x.permute(0, 3, 1, 2).float(), w.float(), padding=1).permute(0, 2, 3, 1)
    ref = F.relu(base + cb.float()[:, None, None, :] + res.float())
    assert_close(y, ref, 2e-2, "conv fused epilogue")


def test_conv2d_splitk_fused_epilogue():
    # split-K path (small spatial, wide channels) must apply the same epilogue
    x = rnd(1, 8, 8, 1280, seed=30, scale=0.2)
    w = rnd(1280, 1280, 3, 3, seed=31, scale=0.01)
    res = rnd(1, 8, 8, 1280, seed=32)
    y = ops.conv2d_nhwc(x, w, None, residual=res)
    base = F.conv2d(x.permute(0, 3, 1, 2).float(), w.float(), padding=1).permute(0, 2, 3, 1)
    assert_close(y, base + res.float(), 5e-2, "conv splitk epilogue")


def test_conv2d_fused_silu():
    x = rnd(1, 8, 8, 32)
    w = rnd(32, 32, 3, 3, scale=0.05)
    y = ops.conv2d_nhwc(x, w, None, fuse_silu=True)
    ref = F.silu(F.conv2d(x.permute(0, 3, 1, 2).float(), w.float(), padding=1)).permute(0, 2, 3, 1)
    assert_close(y, ref, 2e-2, "conv+silu")


def test_group_norm_silu_vs_torch():
    x = rnd(2, 16, 16, 320, seed=5)
    g = torch.randn(320).float().to(DEV)
    b = torch.randn(320).float().to(DEV)
    y = ops.group_norm_silu_nhwc(x, 32, g, b, silu=True)
    ref = F.silu(F.group_norm(x.permute(0, 3, 1, 2).float(), 32, g, b)).permute(0, 2, 3, 1)
    assert_close(y, ref, 1e-2, "group_norm_silu")


def test_layer_norm_vs_torch():
    x = rnd(4, 77, 320, seed=6)
    g = torch.randn(320).float().to(DEV)
    b = torch.randn(320).float().to(DEV)
    y = ops.layer_norm(x, g, b)
    ref = F.layer_norm(x.float(), (320,), g, b)
    assert_close(y, ref, 1e-2, "layer_norm")


@pytest.mark.parametrize("lq,lk,c,heads", [
    (64, 64, 64, 1),      # single head/tile
    (4096, 4096, 320, 5), # sd21 self-attn @64x64, d=64
    (1024, 77, 640, 10),  # cross-attn vs text tokens, d=64
    (256, 256, 320, 8),   # sd15 d=40 (padded path)
    (100, 77, 128, 2),    # ragged Lq tail
    (64, 64, 32, 2),      # d=16 (tiny config padded path)
])
def test_attention_vs_sdpa(lq, lk, c, heads):
    q = rnd(2, lq, c, seed=lq)
    k = rnd(2, lk, c, seed=lk + 1)
    v = rnd(2, lk, c, seed=lk + 2)
    y = ops.attention(q, k, v, heads)
    d = c // heads
    qh = q.view(2, lq, heads, d).permute(0, 2, 1, 3).float()
    kh = k.view(2, lk, heads, d).permute(0, 2, 1, 3).float()
    vh = v.view(2, lk, heads, d).permute(0, 2, 1, 3).float()
    ref = F.scaled_dot_product_attention(qh, kh, vh).permute(0, 2, 1, 3).reshape(2, lq, c)
    assert_close(y, ref, 1e-2, "attention")


def test_geglu_silu_add():
    x = rnd(3, 16, 256, seed=9)
    assert_close(ops.geglu(x), (lambda a, b: a.float() * F.gelu(b.float()))(*x.chunk(2, -1)), 5e-3, "geglu")
    assert_close(ops.silu(x), F.silu(x.float()), 5e-3, "silu")
    y = rnd(3, 16, 256, seed=10)
    assert_close(ops.add_act(x, y, ops.ACT_RELU), F.relu(x.float() + y.float()), 5e-3, "add_relu")


def test_upsample2x():
    x = rnd(2, 8, 8, 64, seed=11)
    y = ops.upsample_nearest2x_nhwc(x)
    ref = F.interpolate(x.permute(0, 3, 1, 2).float(), scale_factor=2).permute(0, 2, 3, 1)
    assert_close(y, ref, 0, "upsample2x")


def test_pre_post_process():
    u8 = torch.randint(0, 256, (1, 64, 64, 3), dtype=torch.uint8, device=DEV)
    f = ops.preprocess_from_u8(u8, torch.float16)
    ref = (u8.float() / 127.5 - 1.0)
    assert_close(f, ref, 1e-2, "preprocess")
    back = ops.postprocess_to_u8(f)
    assert (back.int() - u8.int()).abs().max().item() <= 1


@pytest.mark.parametrize("ic,oc,h,groups", [
    (64, 64, 16, 32),      # BM64 register-staged path
    (320, 320, 16, 32),    # SD resnet shape (split-K)
    (64, 64, 64, 32),      # BM128 large-spatial path
    (4, 32, 16, 2),        # small-IC direct path
])
def test_fused_gn_conv_vs_unfused(ic, oc, h, groups):
    """in_affine fusion (GN apply + SiLU inside the conv's A-load) must
    match gn_silu -> conv bit-for-bit-ish on every conv path, including
    the zero-padding rule (padding applies AFTER the transform)."""
    x = rnd(2, h, h, ic, seed=ic + h)
    w = rnd(oc, ic, 3, 3, seed=2, scale=1.0 / math.sqrt(ic * 9))
    gamma = (torch.randn(ic, generator=torch.Generator().manual_seed(3)) * 0.3 + 1).to(DEV)
    beta = (torch.randn(ic, generator=torch.Generator().manual_seed(4)) * 0.2).to(DEV)

    ref = ops.conv2d_nhwc(
        ops.group_norm_silu_nhwc(x, groups, gamma, beta, 1e-5, True), w)
    aff = ops.group_norm_coeffs(x, groups, gamma, beta, 1e-5)
    assert aff.shape == (2, ic, 2)
    got = ops.conv2d_nhwc(x, w, in_affine=aff, in_act=ops.ACT_SILU)
    assert_close(got, ref, 2e-2, f"fused gn-conv ic{ic} h{h}")


def test_group_norm_coeffs_vs_cpu():
    x = rnd(2, 8, 8, 64, seed=9)
    gamma = (torch.randn(64, generator=torch.Generator().manual_seed(5)) + 1).to(DEV)
    beta = torch.randn(64, generator=torch.Generator().manual_seed(6)).to(DEV)
    got = ops.group_norm_coeffs(x, 32, gamma, beta, 1e-5)
    ref = ops.group_norm_coeffs(x.cpu(), 32, gamma.cpu(), beta.cpu(), 1e-5)
    assert_close(got, ref.to(DEV), 2e-2, "gn coeffs")


def test_sched_fused_kernels_vs_fp32():
    """Fused scheduler kernels (one launch each) vs the fp32 scheduler math."""
    from ai_rtc_agent_amd.engine.scheduler import StreamScheduler

    sch = StreamScheduler(num_inference_steps=50)
    co = sch.coefficients([18, 26, 35, 45], 2, torch.device(DEV), torch.float32)
    B = 8
    x0 = rnd(B, 16, 16, 4, seed=70)
    nz = rnd(B, 16, 16, 4, seed=71)
    eps = rnd(B, 16, 16, 4, seed=72)
    got = ops.sched_add_noise(x0, nz, co["alpha_f32"], co["beta_f32"])
    ref = (co["alpha_f32"].view(-1, 1, 1, 1) * x0.float()
           + co["beta_f32"].view(-1, 1, 1, 1) * nz.float())
    assert_close(got, ref, 2e-3, "sched_add_noise")
    got2 = ops.sched_blend(x0, eps, co["alpha_f32"], co["beta_f32"],
                           co["c_out_f32"], co["c_skip_f32"])
    a = co["alpha_f32"].view(-1, 1, 1, 1)
    b = co["beta_f32"].view(-1, 1, 1, 1)
    x0p = (x0.float() - b * eps.float()) / a
    ref2 = co["c_out_f32"].view(-1, 1, 1, 1) * x0p \
        + co["c_skip_f32"].view(-1, 1, 1, 1) * x0.float()
    assert_close(got2, ref2, 3e-2, "sched_blend")
