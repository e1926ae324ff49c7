"""CPU-path op tests: shape/semantics of the dispatch layer reference ops.

These torch reference implementations are the goldens the HIP kernels are
compared against in tests/test_ops_gpu.py (SURVEY.md §4 item b).
"""
import torch
import torch.nn.functional as F

from ai_rtc_agent_amd import ops


def test_conv2d_nhwc_matches_torch():
    x = torch.randn(2, 8, 8, 16)
    w = torch.randn(32, 16, 3, 3)
    b = torch.randn(32)
    y = ops.conv2d_nhwc(x, w, b, stride=1, padding=1)
    ref = F.conv2d(x.permute(0, 3, 1, 2), w, b, padding=1).permute(0, 2, 3, 1)
    assert y.shape == (2, 8, 8, 32)
    assert torch.allclose(y, ref, atol=1e-5)


def test_conv2d_stride2():
    x = torch.randn(1, 8, 8, 4)
    w = torch.randn(8, 4, 3, 3)
    y = ops.conv2d_nhwc(x, w, None, stride=2, padding=1)
    assert y.shape == (1, 4, 4, 8)


def test_group_norm_silu():
    x = torch.randn(2, 4, 4, 32)
    gamma, beta = torch.randn(32), torch.randn(32)
    y = ops.group_norm_silu_nhwc(x, 8, gamma, beta, silu=True)
    ref = F.silu(F.group_norm(x.permute(0, 3, 1, 2), 8, gamma, beta)).permute(0, 2, 3, 1)
    assert torch.allclose(y, ref, atol=1e-5)


def test_attention_matches_sdpa():
    q = torch.randn(2, 16, 64)
    k = torch.randn(2, 9, 64)
    v = torch.randn(2, 9, 64)
    y = ops.attention(q, k, v, num_heads=4)
    assert y.shape == (2, 16, 64)
    # manual reference
    d = 16
    qh = q.view(2, 16, 4, d).transpose(1, 2)
    kh = k.view(2, 9, 4, d).transpose(1, 2)
    vh = v.view(2, 9, 4, d).transpose(1, 2)
    s = torch.softmax(qh @ kh.transpose(-1, -2) / d**0.5, dim=-1)
    ref = (s @ vh).transpose(1, 2).reshape(2, 16, 64)
    assert torch.allclose(y, ref, atol=1e-5)


def test_geglu():
    x = torch.randn(2, 8, 32)
    y = ops.geglu(x)
    a, b = x.chunk(2, -1)
    assert torch.allclose(y, a * F.gelu(b), atol=1e-6)


def test_pre_post_roundtrip():
    u8 = torch.randint(0, 256, (1, 16, 16, 3), dtype=torch.uint8)
    f = ops.preprocess_from_u8(u8, torch.float32)
    assert f.min() >= -1.0 and f.max() <= 1.0
    back = ops.postprocess_to_u8(f)
    assert (back.int() - u8.int()).abs().max() <= 1


def test_upsample_nearest2x():
    x = torch.arange(4.0).view(1, 2, 2, 1)
    y = ops.upsample_nearest2x_nhwc(x)
    assert y.shape == (1, 4, 4, 1)
    assert y[0, 0, 0, 0] == y[0, 1, 1, 0] == 0
    assert y[0, 2, 0, 0] == 2 and y[0, 2, 2, 0] == 3


def test_layer_norm():
    x = torch.randn(2, 5, 32)
    g, b = torch.randn(32), torch.randn(32)
    y = ops.layer_norm(x, g, b)
    assert torch.allclose(y, F.layer_norm(x, (32,), g, b), atol=1e-5)


def test_fused_gn_conv_cpu_parity():
    """CPU fallback: conv2d_nhwc(in_affine) == gn_silu -> conv."""
    import math

    import torch

    from ai_rtc_agent_amd import ops

    g = torch.Generator().manual_seed(0)
    x = torch.randn(2, 8, 8, 16, generator=g)
    w = torch.randn(24, 16, 3, 3, generator=g) / math.sqrt(16 * 9)
    gamma = torch.randn(16, generator=g) * 0.3 + 1
    beta = torch.randn(16, generator=g) * 0.2
    ref = ops.conv2d_nhwc(ops.group_norm_silu_nhwc(x, 4, gamma, beta, 1e-5, True), w)
    aff = ops.group_norm_coeffs(x, 4, gamma, beta, 1e-5)
    got = ops.conv2d_nhwc(x, w, in_affine=aff, in_act=ops.ACT_SILU)
    assert torch.allclose(got.float(), ref.float(), atol=1e-4), \
        (got - ref).abs().max()
