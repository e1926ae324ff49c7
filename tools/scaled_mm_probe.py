"""Probe hipBLASLt fp8 GEMM availability via torch._scaled_mm on gfx950.

If e4m3 scaled_mm works and beats f16 F.linear at the transformer's
batched shapes, the fp8 tier can extend to the QKV/FF linears.
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def t(fn, iters=30):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6


def main():
    dev = "cuda"
    lines = []

    def log(s):
        print(s)
        lines.append(str(s))

    # correctness probe
    try:
        M, K, N = 256, 320, 640
        a = torch.randn(M, K, device=dev).half()
        b = torch.randn(N, K, device=dev).half()
        sa = a.abs().max().float() / 448.0
        sb = b.abs().max().float() / 448.0
        a8 = (a.float() / sa).clamp(-448, 448).to(torch.float8_e4m3fn)
        b8 = (b.float() / sb).clamp(-448, 448).to(torch.float8_e4m3fn)
        y = torch._scaled_mm(a8, b8.t(), scale_a=sa, scale_b=sb,
                             out_dtype=torch.float16)
        ref = a.float() @ b.float().t()
        err = (y.float() - ref).abs().mean().item() / ref.abs().mean().item()
        log(f"scaled_mm works: rel err {err:.4f}")
    except Exception as e:
        log(f"scaled_mm FAILED: {type(e).__name__}: {e}")
        with open("gpurun_out/scaled_mm_probe.txt", "w") as f:
            f.write("\n".join(lines) + "\n")
        return

    # perf at the transformer's batched (fbs=8) shapes
    shapes = [
        ("qkv c320 L4096*8", 32768, 320, 960),
        ("ff-in c320", 32768, 320, 2560),
        ("ff-out c320", 32768, 1280, 320),
        ("qkv c640 L1024*8", 8192, 640, 1920),
        ("qkv c1280 L256*8", 2048, 1280, 3840),
    ]
    for name, M, K, N in shapes:
        a = torch.randn(M, K, device=dev).half()
        w = torch.randn(N, K, device=dev).half()
        sa = a.abs().max().float() / 448.0
        sw = w.abs().max().float() / 448.0
        a8 = (a.float() / sa).clamp(-448, 448).to(torch.float8_e4m3fn)
        # mat2 must be column-major [K, N]: .t() of the row-major [N, K]
        w8t = (w.float() / sw).clamp(-448, 448).to(torch.float8_e4m3fn).t()
        us16 = t(lambda: torch.nn.functional.linear(a, w))
        us8 = t(lambda: torch._scaled_mm(a8, w8t, scale_a=sa, scale_b=sw,
                                         out_dtype=torch.float16))
        fl = 2.0 * M * K * N
        log(f"{name:22s} f16 {us16:7.1f} us {fl/us16/1e6:6.1f} TF | "
            f"fp8 {us8:7.1f} us {fl/us8/1e6:6.1f} TF | x{us16/us8:.2f}")

    with open("gpurun_out/scaled_mm_probe.txt", "w") as f:
        f.write("\n".join(lines) + "\n")


if __name__ == "__main__":
    main()
