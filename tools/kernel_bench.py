#!/usr/bin/env python3
"""Per-op microbenchmarks at the SD shapes the frame actually runs.

    python tools/kernel_bench.py [--iters 50]

Reports per-op time and effective TFLOP/s (or GB/s for memory-bound ops)
with within-process interleaved repeats (guide §5.4 rule 24). GPU only.
"""
from __future__ import annotations

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from ai_rtc_agent_amd import ops

CONV_SHAPES = [
    # (name, B, H, W, IC, OC, k, stride)   — SD-Turbo/TAESD frame shapes
    ("unet 64x64x320 3x3", 1, 64, 64, 320, 320, 3, 1),
    ("unet 32x32x640 3x3", 1, 32, 32, 640, 640, 3, 1),
    ("unet 16x16x1280 3x3", 1, 16, 16, 1280, 1280, 3, 1),
    ("unet 8x8x1280 3x3", 1, 8, 8, 1280, 1280, 3, 1),
    ("unet up 32x32 1920->640", 1, 32, 32, 1920, 640, 3, 1),
    ("taesd 512x512x64 3x3", 1, 512, 512, 64, 64, 3, 1),
    ("taesd 256x256x64 3x3", 1, 256, 256, 64, 64, 3, 1),
    ("down 64->32 s2", 1, 64, 64, 320, 320, 3, 2),
    ("proj 64x64 320->640 1x1", 1, 64, 64, 320, 640, 1, 1),
]

ATTN_SHAPES = [
    # (name, B, Lq, Lk, C, heads)
    ("self 4096 c320 h5 (sd21@64x64)", 1, 4096, 4096, 320, 5),
    ("self 1024 c640 h10", 1, 1024, 1024, 640, 10),
    ("self 256 c1280 h20", 1, 256, 256, 1280, 20),
    ("cross 4096x77 c320", 1, 4096, 77, 320, 5),
    ("cross 1024x77 c640", 1, 1024, 77, 640, 10),
]


def timeit(fn, iters):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6  # us


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--iters", type=int, default=50)
    p.add_argument("--batch", type=int, default=1,
                   help="scale conv batch (the fbs>1 serving tier)")
    args = p.parse_args()
    assert torch.cuda.is_available(), "GPU microbench"
    dev = "cuda"

    print(f"== conv2d (NHWC implicit-GEMM MFMA) B={args.batch} ==")
    for name, b0, h, w, ic, oc, k, st in CONV_SHAPES:
        b = b0 * args.batch
        x = torch.randn(b, h, w, ic, device=dev).half()
        wt = (torch.randn(oc, ic, k, k, device=dev) * 0.02).half()
        bias = torch.randn(oc, device=dev).half()
        pad = k // 2
        fn = lambda: ops.conv2d_nhwc(x, wt, bias, stride=st, padding=pad)
        us = timeit(fn, args.iters)
        ho, wo = (h + 2 * pad - k) // st + 1, (w + 2 * pad - k) // st + 1
        fl = 2.0 * b * ho * wo * oc * ic * k * k
        print(f"  {name:32s} {us:8.1f} us  {fl/us/1e6:7.1f} TF/s")

    print(f"== conv2d fp8 (MX-scaled MFMA, per-OC weight scales) B={args.batch} ==")
    for name, b0, h, w, ic, oc, k, st in CONV_SHAPES:
        if ic % 64 != 0:
            continue
        b = b0 * args.batch
        x = torch.randn(b, h, w, ic, device=dev).half()
        wt = (torch.randn(oc, ic, k, k, device=dev) * 0.02).half()
        bias = torch.randn(oc, device=dev).float()
        pad = k // 2
        a_scale = x.float().abs().max().item() / ops.FP8_MAX
        fn = lambda: ops.conv2d_fp8_nhwc(x, wt, a_scale, None, stride=st,
                                         padding=pad)
        us = timeit(fn, args.iters)
        ho, wo = (h + 2 * pad - k) // st + 1, (w + 2 * pad - k) // st + 1
        fl = 2.0 * b * ho * wo * oc * ic * k * k
        print(f"  {name:32s} {us:8.1f} us  {fl/us/1e6:7.1f} TF/s")

    print(f"== conv2d fp8 pre-quantized input (GN-fp8 -> conv) B={args.batch} ==")
    for name, b0, h, w, ic, oc, k, st in CONV_SHAPES:
        if ic % 64 != 0:
            continue
        b = b0 * args.batch
        x = torch.randn(b, h, w, ic, device=dev).half()
        wt = (torch.randn(oc, ic, k, k, device=dev) * 0.02).half()
        pad = k // 2
        a_scale = x.float().abs().max().item() / ops.FP8_MAX
        xq = ((x.float() / a_scale).clamp(-448, 448)
              .to(torch.float8_e4m3fn).view(torch.uint8).contiguous())
        fn = lambda: ops.conv2d_fp8_nhwc(xq, wt, a_scale, None, stride=st,
                                         padding=pad)
        us = timeit(fn, args.iters)
        ho, wo = (h + 2 * pad - k) // st + 1, (w + 2 * pad - k) // st + 1
        fl = 2.0 * b * ho * wo * oc * ic * k * k
        print(f"  {name:32s} {us:8.1f} us  {fl/us/1e6:7.1f} TF/s")

    print(f"== attention (flash MFMA + tr_b16) B={args.batch} ==")
    for name, b0, lq, lk, c, hds in ATTN_SHAPES:
        b = b0 * args.batch
        q = torch.randn(b, lq, c, device=dev).half()
        kk = torch.randn(b, lk, c, device=dev).half()
        v = torch.randn(b, lk, c, device=dev).half()
        fn = lambda: ops.attention(q, kk, v, hds)
        us = timeit(fn, args.iters)
        fl = 2.0 * 2 * b * lq * lk * c
        print(f"  {name:32s} {us:8.1f} us  {fl/us/1e6:7.1f} TF/s")

    print("== norms / elementwise (GB/s = read+write traffic) ==")
    x = torch.randn(1, 64, 64, 320, device=dev).half()
    g = torch.randn(320, device=dev).float()
    be = torch.randn(320, device=dev).float()
    us = timeit(lambda: ops.group_norm_silu_nhwc(x, 32, g, be), args.iters)
    traffic = x.numel() * 2 * 3  # 2 reads (stats+apply) + 1 write
    print(f"  {'group_norm_silu 64x64x320':32s} {us:8.1f} us  {traffic/us/1e3:7.1f} GB/s")
    t = torch.randn(1, 4096, 320, device=dev).half()
    us = timeit(lambda: ops.layer_norm(t, g, be), args.iters)
    print(f"  {'layer_norm 4096x320':32s} {us:8.1f} us  {t.numel()*2*2/us/1e3:7.1f} GB/s")
    u8 = torch.randint(0, 256, (1, 512, 512, 3), dtype=torch.uint8, device=dev)
    us = timeit(lambda: ops.preprocess_from_u8(u8, torch.float16), args.iters)
    print(f"  {'preprocess 512x512':32s} {us:8.1f} us  {u8.numel()*3/us/1e3:7.1f} GB/s")


if __name__ == "__main__":
    main()
