"""Focused fp8 conv debug: encode-path probes + NaN topology of one conv.

Run on a GPU box; writes gpurun_out/fp8_debug.txt.
"""
import math
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

from ai_rtc_agent_amd import ops

lines = []


def log(s):
    print(s)
    lines.append(str(s))


def main():
    C = ops.hip_ext()
    dev = "cuda:0"
    g = torch.Generator().manual_seed(0)

    # 1. exact encode-path probe, both variants, normal + overflow values
    vals = torch.randn(16, generator=g).half() * 2.0
    vals[3] = 60000.0
    vals[7] = -60000.0
    for sa in (4.0 / 448.0, 1.0):
        ref = (vals.float().clamp(-448 * sa, 448 * sa) / sa).clamp(-448, 448) \
            .to(torch.float8_e4m3fn).view(torch.uint8).numpy()
        for variant in (0, 1):
            out = C.fp8_quant_probe(vals.to(dev), sa, variant)
            torch.cuda.synchronize()
            got = out.cpu().numpy()
            match = (got == ref).all()
            dec = out.cpu().view(torch.float8_e4m3fn).to(torch.float32)
            log(f"quant probe sa={sa:.5f} variant={variant}: match={match} "
                f"bytes={got.tolist()}")
            if not match:
                log(f"  expected {ref.tolist()}")
                log(f"  decoded  {(dec * sa).tolist()}")

    # 2. NaN topology of one small conv
    ic, oc, h = 64, 64, 16
    x = (torch.randn(1, h, h, ic, generator=g) * 2.0).half().to(dev)
    w = (torch.randn(oc, ic, 3, 3, generator=g) / math.sqrt(ic * 9)).half().to(dev)
    sa = x.float().abs().max().item() / ops.FP8_MAX
    y = ops.conv2d_fp8_nhwc(x, w, sa)
    torch.cuda.synchronize()
    yn = torch.isnan(y.float()).squeeze(0)  # (H, W, OC)
    log(f"conv ({ic},{oc},{h}): NaN count {int(yn.sum())} of {yn.numel()}")
    if yn.any():
        m_nan = yn.any(dim=2)
        log(f"  rows with NaN: {m_nan.any(dim=1).nonzero().flatten().tolist()}")
        log(f"  cols(oc) with NaN: {yn.any(dim=0).any(dim=0).nonzero().flatten().tolist()[:40]}")
        # pixel 0 channel profile
        log(f"  y[0,0,:8] = {y[0,0,0,:8].float().tolist()}")
        ref = ops.conv2d_fp8_nhwc(x.cpu(), w.cpu(), sa)
        log(f"  ref[0,0,:8] = {ref[0,0,0,:8].float().tolist()}")
        fin = torch.isfinite(y.float())
        err = ((y.float().cpu() - ref.float())[fin.cpu()]).abs()
        log(f"  finite-part max err = {err.max().item() if err.numel() else 'n/a'}")

    with open("gpurun_out/fp8_debug.txt", "w") as f:
        f.write("\n".join(lines) + "\n")


if __name__ == "__main__":
    main()
