#!/usr/bin/env python3
"""Sustained-serving soak: N frames through the engine, reporting fps over
time windows and device-memory growth (leak canary).

    python tools/soak.py [--frames 2000]
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from ai_rtc_agent_amd.config import sd_turbo_config
from ai_rtc_agent_amd.engine import StreamDiffusionEngine


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--frames", type=int, default=2000)
    args = p.parse_args()
    assert torch.cuda.is_available()

    eng = StreamDiffusionEngine(sd_turbo_config(device="cuda"))
    eng.prepare()
    g = torch.Generator().manual_seed(0)
    frames = [torch.randint(0, 256, (512, 512, 3), generator=g, dtype=torch.uint8).cuda()
              for _ in range(8)]
    # warmup + capture
    for i in range(20):
        eng(frames[i % 8])
    torch.cuda.synchronize()
    mem0 = torch.cuda.memory_allocated() / 1e6

    window = max(100, args.frames // 10)
    t0 = time.perf_counter()
    tw = t0
    for i in range(args.frames):
        eng(frames[i % 8])
        if (i + 1) % window == 0:
            torch.cuda.synchronize()
            now = time.perf_counter()
            print(f"frames {i+1-window}-{i+1}: {window/(now-tw):7.1f} fps  "
                  f"mem {torch.cuda.memory_allocated()/1e6:9.1f} MB", flush=True)
            tw = now
    torch.cuda.synchronize()
    total = time.perf_counter() - t0
    mem1 = torch.cuda.memory_allocated() / 1e6
    print(f"TOTAL: {args.frames} frames in {total:.1f}s = {args.frames/total:.1f} fps; "
          f"mem {mem0:.1f} -> {mem1:.1f} MB (delta {mem1-mem0:+.2f})")
    assert abs(mem1 - mem0) < 50, "memory growth under soak"
    print("SOAK OK")


if __name__ == "__main__":
    main()
