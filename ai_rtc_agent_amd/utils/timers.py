"""Per-stage timers + FPS / glass-to-glass latency stats.

The reference has no metrics; BASELINE.json's headline
metric (img2img FPS + p50 glass-to-glass ms) requires first-class counters,
so every pipeline stage (decode/preprocess/unet/vae/encode) reports here and
the agent exposes them at GET /stats.

GPU timing uses cuda events recorded on the stream (no host syncs in the hot
loop); events are reduced lazily when stats are read.
"""
from __future__ import annotations

import time
from collections import deque
from typing import Deque, Dict, List, Tuple

import torch


class _Percentile:
    def __init__(self, maxlen: int = 512):
        self.samples: Deque[float] = deque(maxlen=maxlen)

    def add(self, v: float) -> None:
        self.samples.append(v)

    def percentile(self, p: float) -> float:
        if not self.samples:
            return 0.0
        s = sorted(self.samples)
        i = min(len(s) - 1, int(p / 100.0 * len(s)))
        return s[i]

    def mean(self) -> float:
        return sum(self.samples) / len(self.samples) if self.samples else 0.0


class StageTimers:
    """Per-stage timing + rocTX ranges.

    On ROCm, torch.cuda.nvtx maps to rocTX: every stage() context also
    emits a range so rocprofv3 --marker-trace groups kernels by pipeline
    stage (SURVEY.md §5.1 — the reference has no tracing hooks at all).
    """

    def __init__(self, use_cuda: bool = False):
        self.use_cuda = use_cuda and torch.cuda.is_available()
        self.stages: Dict[str, _Percentile] = {}
        self._pending: List[Tuple[str, torch.cuda.Event, torch.cuda.Event]] = []
        self._frame_times: Deque[float] = deque(maxlen=512)
        self._last_frame_t: float | None = None
        self.frames = 0

    # -- stage timing ---------------------------------------------------
    class _Ctx:
        def __init__(self, parent: "StageTimers", name: str):
            self.parent, self.name = parent, name

        def __enter__(self):
            p = self.parent
            if p.use_cuda:
                torch.cuda.nvtx.range_push(f"airtc/{self.name}")  # rocTX
                self.e0 = torch.cuda.Event(enable_timing=True)
                self.e0.record()
            else:
                self.t0 = time.perf_counter()
            return self

        def __exit__(self, *exc):
            p = self.parent
            if p.use_cuda:
                e1 = torch.cuda.Event(enable_timing=True)
                e1.record()
                torch.cuda.nvtx.range_pop()
                p._pending.append((self.name, self.e0, e1))
            else:
                p.stages.setdefault(self.name, _Percentile()).add(
                    (time.perf_counter() - self.t0) * 1000.0
                )
            return False

    def stage(self, name: str) -> "StageTimers._Ctx":
        return StageTimers._Ctx(self, name)

    # -- frame accounting ----------------------------------------------
    def frame_done(self) -> None:
        self.frames += 1
        t = time.perf_counter()
        if self._last_frame_t is not None:
            self._frame_times.append(t - self._last_frame_t)
        self._last_frame_t = t

    def _drain(self) -> None:
        if not self._pending:
            return
        torch.cuda.synchronize()
        for name, e0, e1 in self._pending:
            self.stages.setdefault(name, _Percentile()).add(e0.elapsed_time(e1))
        self._pending.clear()

    def snapshot(self) -> dict:
        self._drain()
        fps = 0.0
        if self._frame_times:
            mean_dt = sum(self._frame_times) / len(self._frame_times)
            fps = 1.0 / mean_dt if mean_dt > 0 else 0.0
        return {
            "frames": self.frames,
            "fps": round(fps, 2),
            "stages_ms": {
                k: {"mean": round(v.mean(), 3), "p50": round(v.percentile(50), 3), "p90": round(v.percentile(90), 3)}
                for k, v in self.stages.items()
            },
        }
