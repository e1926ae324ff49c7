"""GPU numerics for the fp8 (MX-scaled MFMA) conv path.

The kernel quantizes activations with the same RNE encode as torch's
float8_e4m3fn cast (verified by tools/fp8_probe.py), so kernel-vs-emulation
should differ only by f32 accumulation order — compared at tight SNR.
A second comparison against the plain f32 conv bounds the total
quantization noise.
"""
import math

import numpy as np
import pytest
import torch
import torch.nn.functional as F

from ai_rtc_agent_amd import ops

pytestmark = pytest.mark.gpu

DEV = "cuda"


def rnd(*shape, seed=0, scale=1.0):
    g = torch.Generator().manual_seed(seed)
    return (torch.randn(*shape, generator=g) * scale).half().to(DEV)


def snr_db(got, ref):
    num = (ref.float() ** 2).mean().item()
    den = ((got.float() - ref.float()) ** 2).mean().item()
    return 10 * math.log10(num / max(den, 1e-20))


def test_mx_mfma_layout_regression():
    """The raw-fragment probe facts the conv kernel is built on."""
    C = ops.hip_ext()
    rng = np.random.default_rng(7)
    vals = np.array([0.0, 0.5, -0.5, 1.0, -1.0, 2.0, -2.0, 4.0], np.float32)
    Amat = vals[rng.integers(0, 8, size=(16, 128))]
    Bmat = vals[rng.integers(0, 8, size=(16, 128))]

    def f8(x):
        return torch.tensor(x, dtype=torch.float32).to(torch.float8_e4m3fn).view(torch.uint8).numpy()

    def place(mat):
        buf = np.zeros(2048, np.uint8)
        for l in range(64):
            buf[l * 32:(l + 1) * 32] = f8(mat[l & 15, (l >> 4) * 32:(l >> 4) * 32 + 32])
        return torch.tensor(buf, device=DEV)

    d = C.fp8_mx_probe(place(Amat), place(Bmat), 127, 127)
    torch.cuda.synchronize()
    draw = d.cpu().numpy()
    got = np.zeros((16, 16), np.float32)
    for l in range(64):
        for j in range(4):
            got[(l >> 4) * 4 + j, l & 15] = draw[l * 4 + j]
    assert np.abs(got - Amat @ Bmat.T).max() == 0.0


def test_cvt_overflow_nan_and_kernel_clamps():
    """Measured fact: the HW encode does NOT saturate (overflow -> NaN).
    The conv kernel therefore clamps to +-448*sa before encoding; feed it
    activations far beyond the calibrated scale and require finite,
    saturated-but-sane output."""
    C = ops.hip_ext()
    fin = torch.tensor([1000.0, -1000.0], dtype=torch.float16, device=DEV)
    enc_in = torch.zeros(2, dtype=torch.uint8, device=DEV)
    enc, _ = C.fp8_cvt_probe(fin, 1.0, enc_in)
    torch.cuda.synchronize()
    dec = enc.cpu().view(torch.float8_e4m3fn).to(torch.float32)
    assert torch.isnan(dec).all(), \
        f"expected the documented NaN-on-overflow encode (got {dec.tolist()})"

    ic, oc, h = 64, 64, 16
    x = rnd(1, h, h, ic, seed=21)
    x[0, 0, 0, :] = 60000.0  # far beyond any calibrated absmax
    w = rnd(oc, ic, 3, 3, seed=22, scale=1.0 / math.sqrt(ic * 9))
    a_scale = 4.0 / ops.FP8_MAX  # deliberately small: forces clamping
    y = ops.conv2d_fp8_nhwc(x, w, a_scale)
    assert torch.isfinite(y.float()).all(), "kernel must clamp, not NaN"
    # the clamped row contributes at most 448*a_scale per element
    ref = ops.conv2d_fp8_nhwc(x.cpu(), w.cpu(), a_scale)
    assert snr_db(y.cpu(), ref) > 40  # f16-intermediate inline encode


@pytest.mark.parametrize("ic,oc,h,stride,pad,r", [
    (320, 320, 16, 1, 1, 3),    # SD resnet shape (BM64 + split-K)
    (1280, 1280, 8, 1, 1, 3),   # 8x8 wide layer, deep split-K
    (320, 640, 8, 1, 0, 1),     # 1x1 projection
    (640, 640, 16, 2, 1, 3),    # stride-2 downsample
    (64, 64, 64, 1, 1, 3),      # TAESD: BM128 path, K=576 tail tile
    (64, 128, 32, 1, 1, 3),     # BM128, ragged-K tail with OC tile > 1
])
def test_conv2d_fp8_vs_emulation(ic, oc, h, stride, pad, r):
    x = rnd(2, h, h, ic, seed=ic + oc, scale=2.0)
    w = rnd(oc, ic, r, r, seed=1, scale=1.0 / math.sqrt(ic * r * r))
    b = torch.randn(oc, generator=torch.Generator().manual_seed(2)).half().to(DEV)
    a_scale = x.float().abs().max().item() / ops.FP8_MAX
    y = ops.conv2d_fp8_nhwc(x, w, a_scale, b, stride=stride, padding=pad)
    # emulation golden (CPU path of the same function)
    ref = ops.conv2d_fp8_nhwc(x.cpu(), w.cpu(), a_scale, b.cpu(),
                              stride=stride, padding=pad)
    assert y.shape == ref.shape
    s = snr_db(y.cpu(), ref)
    # kernel encodes through an f16 intermediate (packed clamp+mul) — up to
    # 2^-10 relative double-rounding vs the f32 emulation, plus f32
    # accumulation-order differences
    assert s > 40, f"kernel vs emulation SNR {s:.1f} dB"
    # and against the plain f32 conv: total quantization noise bound
    f32 = F.conv2d(x.permute(0, 3, 1, 2).float(), w.float(), b.float(),
                   stride=stride, padding=pad).permute(0, 2, 3, 1)
    s2 = snr_db(y.cpu(), f32.cpu())
    assert s2 > 20, f"fp8 conv vs f32 SNR {s2:.1f} dB"


def test_conv2d_fp8_fused_epilogue_and_affine():
    ic, oc, h = 320, 320, 16
    x = rnd(2, h, h, ic, seed=5)
    w = rnd(oc, ic, 3, 3, seed=6, scale=1.0 / math.sqrt(ic * 9))
    g = torch.Generator().manual_seed(7)
    aff = (torch.randn(2, ic, 2, generator=g).float() * 0.2 + 0.5).to(DEV)
    res = rnd(2, h, h, oc, seed=8)
    cb = rnd(2, oc, seed=9)
    a_scale = 6.0 / ops.FP8_MAX
    y = ops.conv2d_fp8_nhwc(x, w, a_scale, None, act=ops.ACT_SILU,
                            residual=res, channel_bias=cb, in_affine=aff,
                            in_act=ops.ACT_SILU)
    ref = ops.conv2d_fp8_nhwc(x.cpu(), w.cpu(), a_scale, None,
                              act=ops.ACT_SILU, residual=res.cpu(),
                              channel_bias=cb.cpu(), in_affine=aff.cpu(),
                              in_act=ops.ACT_SILU)
    s = snr_db(y.cpu(), ref)
    assert s > 55, f"fused fp8 epilogue SNR {s:.1f} dB vs emulation"


@pytest.mark.parametrize("ic,oc,h", [(320, 320, 16), (64, 64, 64)])
def test_gn_fp8_to_conv_q8_path(ic, oc, h):
    """Producer-quantized fast path: GN writes codes, conv stages bytes."""
    x = rnd(2, h, h, ic, seed=31)
    g = torch.Generator().manual_seed(32)
    gamma = (torch.randn(ic, generator=g).float() * 0.3 + 1.0).to(DEV)
    beta = (torch.randn(ic, generator=g).float() * 0.1).to(DEV)
    w = rnd(oc, ic, 3, 3, seed=33, scale=1.0 / math.sqrt(ic * 9))
    gn = ops.group_norm_silu_nhwc(x, 32, gamma, beta)
    sa = gn.float().abs().max().item() / ops.FP8_MAX
    q = ops.group_norm_silu_nhwc(x, 32, gamma, beta, fp8_scale=sa)
    assert q.dtype == torch.uint8
    y = ops.conv2d_fp8_nhwc(q, w, sa)
    # reference: CPU emulation fed the same decoded codes
    ref = ops.conv2d_fp8_nhwc(q.cpu(), w.cpu(), sa)
    s = snr_db(y.cpu(), ref)
    assert s > 55, f"q8-path conv vs emulation SNR {s:.1f} dB"
    # end-to-end noise vs the all-f16 path stays fp8-bounded
    yf = ops.conv2d_nhwc(gn, w)
    s2 = snr_db(y.cpu(), yf.cpu())
    assert s2 > 20, f"GN-fp8->conv vs f16 path SNR {s2:.1f} dB"


def test_conv2d_fp8_batch_gt1_and_graph_capture():
    """fp8 conv must be hipGraph-capturable (the serving engine replays it)."""
    ic, oc, h = 640, 640, 16
    x = rnd(4, h, h, ic, seed=11)
    w = rnd(oc, ic, 3, 3, seed=12, scale=1.0 / math.sqrt(ic * 9))
    a_scale = x.float().abs().max().item() / ops.FP8_MAX
    y0 = ops.conv2d_fp8_nhwc(x, w, a_scale)  # warm caches outside capture
    torch.cuda.synchronize()
    gph = torch.cuda.CUDAGraph()
    with torch.cuda.graph(gph):
        y = ops.conv2d_fp8_nhwc(x, w, a_scale)
    gph.replay()
    torch.cuda.synchronize()
    assert snr_db(y, y0) > 80  # identical inputs -> identical quantized math
