"""Video track adapter — the per-frame pull loop.

Behavioural parity with reference lib/tracks.py:9-38:
- first WARMUP_FRAMES frames (env, default 10) are pulled through the
  pipeline and DISCARDED (engine warm-up / graph capture burn-in),
  returning the raw source frame instead (lib/tracks.py:21-25)
- DROP_FRAMES (env, default 0) source frames are skipped per output frame
  (lib/tracks.py:27-31)
- frames are torch.Tensors end-to-end (the GPU-resident path); the pts /
  time_base of the source frame are preserved onto the output
  (lib/pipeline.py:90-93 equivalent lives here since our frames are plain
  tensors + metadata)
"""
from __future__ import annotations

import asyncio
from dataclasses import dataclass
from typing import Callable

import torch

from .. import config


@dataclass
class VideoFrame:
    """A video frame: u8 RGB tensor + timing metadata."""

    tensor: torch.Tensor  # (H, W, 3) u8
    pts: int = 0
    time_base: float = 1.0 / 90000


class MediaStreamTrack:
    """Minimal async track interface (recv() -> VideoFrame)."""

    kind = "video"

    async def recv(self) -> VideoFrame:  # pragma: no cover - interface
        raise NotImplementedError

    def stop(self) -> None:
        pass


class VideoStreamTrack(MediaStreamTrack):
    """Wraps a source track; every pulled frame runs through the pipeline."""

    def __init__(self, track: MediaStreamTrack, pipeline: Callable[[torch.Tensor], torch.Tensor]):
        self.track = track
        self.pipeline = pipeline
        self.warmed_up = False
        self._warmup_left = config.warmup_frames()
        self._drop = config.drop_frames()

    async def _invoke(self, tensor: torch.Tensor) -> torch.Tensor:
        """Run the pipeline; batched-serving proxies return awaitables
        (parallel/batching.py), plain pipelines return tensors."""
        import inspect

        out = self.pipeline(tensor)
        if inspect.isawaitable(out):
            out = await out
        return out

    async def recv(self) -> VideoFrame:
        if self._warmup_left > 0:
            # burn-in: run the pipeline, discard the output
            self._warmup_left -= 1
            frame = await self.track.recv()
            try:
                _ = await self._invoke(frame.tensor)
            except asyncio.CancelledError:
                pass  # superseded by a newer frame (batched mode)
            if self._warmup_left == 0:
                self.warmed_up = True
            return frame
        for _ in range(self._drop):
            await self.track.recv()
        frame = await self.track.recv()
        out = await self._invoke(frame.tensor)
        return VideoFrame(tensor=out, pts=frame.pts, time_base=frame.time_base)


class QueueTrack(MediaStreamTrack):
    """A source track fed by push() — the receive side of an RTP session."""

    def __init__(self, maxsize: int = 4):
        self._q: asyncio.Queue[VideoFrame] = asyncio.Queue(maxsize=maxsize)

    def push(self, frame: VideoFrame) -> None:
        # real-time: drop the oldest frame rather than grow latency
        if self._q.full():
            try:
                self._q.get_nowait()
            except asyncio.QueueEmpty:
                pass
        self._q.put_nowait(frame)

    async def recv(self) -> VideoFrame:
        return await self._q.get()
