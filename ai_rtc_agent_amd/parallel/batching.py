"""Multi-stream batched serving: K WebRTC sessions through ONE engine.

The measured motivation (profiles/batching_ab.md): at B=1 the 64²-latent
kernels leave the MI355X mostly idle (MFMA util 4–5%); batching 8 streams
through one engine raises aggregate throughput +124% while every stream
stays real-time. This module is the serving-side mechanism: a
BatchedPipeline owns a pipeline whose engine runs frame_buffer_size=K
(stream-batch semantics: the batch dim is K independent streams), sessions
acquire slots, and an asyncio collation loop gathers the newest frame per
slot, launches one batched engine call, and resolves each session's future
with its own output — with per-stream latency accounting in stats().

A slot with no fresh frame this tick re-feeds its previous frame (its
stream-batch state keeps advancing, mirroring how the reference's shared
pipeline behaves when one publisher stalls) and its stale output is
discarded.
"""
from __future__ import annotations

import asyncio
import time
from collections import deque
from typing import Dict, List, Optional

import torch


class SlotProxy:
    """What a session holds: pipeline-shaped (callable + update surface)."""

    def __init__(self, owner: "BatchedPipeline", slot: int):
        self._owner = owner
        self.slot = slot

    def __call__(self, frame: torch.Tensor):
        # returns an awaitable; VideoStreamTrack awaits it
        return self._owner.submit(self.slot, frame)

    def update_prompt(self, prompt: str) -> None:
        self._owner.base.update_prompt(prompt)

    def update_t_index_list(self, t: list) -> None:
        self._owner.base.update_t_index_list(t)

    def stats(self) -> dict:
        return self._owner.stats()


class BatchedPipeline:
    def __init__(self, pipeline, slots: int):
        self.base = pipeline
        self.slots = slots
        h, w = pipeline.cfg.height, pipeline.cfg.width
        self._shape = (h, w, 3)
        self._active = [False] * slots
        self._frame: List[Optional[torch.Tensor]] = [None] * slots
        self._future: List[Optional[asyncio.Future]] = [None] * slots
        self._t_submit: List[float] = [0.0] * slots
        self._lat_ms: List[deque] = [deque(maxlen=128) for _ in range(slots)]
        self._new = asyncio.Event()
        self._task: Optional[asyncio.Task] = None
        self._sessions: Dict[str, int] = {}

    # pipeline-surface passthrough (pool.active() / config broadcasts)
    @property
    def cfg(self):
        return self.base.cfg

    def update_prompt(self, prompt: str) -> None:
        self.base.update_prompt(prompt)

    def update_t_index_list(self, t: list) -> None:
        self.base.update_t_index_list(t)

    # -- slot lifecycle --------------------------------------------------
    def acquire(self, stream_id: str) -> Optional[SlotProxy]:
        if stream_id in self._sessions:
            return SlotProxy(self, self._sessions[stream_id])
        for i in range(self.slots):
            if not self._active[i]:
                self._active[i] = True
                self._sessions[stream_id] = i
                if self._task is None or self._task.done():
                    self._task = asyncio.ensure_future(self._run())
                return SlotProxy(self, i)
        return None  # replica full

    def release(self, stream_id: str) -> None:
        i = self._sessions.pop(stream_id, None)
        if i is not None:
            self._active[i] = False
            fut = self._future[i]
            if fut is not None and not fut.done():
                fut.cancel()
            self._future[i] = None
            self._frame[i] = None

    @property
    def n_active(self) -> int:
        return sum(self._active)

    # -- frame path ------------------------------------------------------
    async def submit(self, slot: int, frame: torch.Tensor) -> torch.Tensor:
        # a newer frame supersedes the pending one (real-time semantics),
        # but any earlier waiter shares the same future — it simply gets
        # the next tick's output for its stream
        self._frame[slot] = frame
        self._t_submit[slot] = time.perf_counter()
        fut = self._future[slot]
        if fut is None or fut.done():
            fut = asyncio.get_event_loop().create_future()
            self._future[slot] = fut
        self._new.set()
        return await fut

    async def _run(self) -> None:
        zeros = torch.zeros(self._shape, dtype=torch.uint8)
        last: List[torch.Tensor] = [zeros] * self.slots
        try:
            while self.n_active > 0:
                await self._new.wait()
                self._new.clear()
                pending = [i for i in range(self.slots)
                           if self._future[i] is not None
                           and not self._future[i].done()]
                if not pending:
                    continue
                for i in range(self.slots):
                    if self._frame[i] is not None:
                        last[i] = self._frame[i]
                        self._frame[i] = None
                # frames may live on mixed devices (decoder output is CPU,
                # the engine uploads); normalise before stacking
                batch = torch.stack([t.cpu() for t in last])
                try:
                    out = self.base(batch)  # (K, H, W, 3)
                except Exception as e:  # engine fault: fail the waiters,
                    for i in pending:   # keep the loop alive for retries
                        fut = self._future[i]
                        if fut is not None and not fut.done():
                            fut.set_exception(RuntimeError(str(e)))
                            self._future[i] = None
                    continue
                now = time.perf_counter()
                for i in pending:
                    fut = self._future[i]
                    if fut is not None and not fut.done():
                        fut.set_result(out[i])
                        self._lat_ms[i].append((now - self._t_submit[i]) * 1e3)
                        self._future[i] = None
                # frames submitted DURING the engine call keep the event set
                if any(f is not None and not f.done() for f in self._future):
                    self._new.set()
        except asyncio.CancelledError:
            pass

    # -- observability ---------------------------------------------------
    def stats(self) -> dict:
        per_slot = []
        for i in range(self.slots):
            lat = sorted(self._lat_ms[i])
            per_slot.append({
                "active": self._active[i],
                "p50_ms": round(lat[len(lat) // 2], 2) if lat else None,
                "frames": len(lat),
            })
        base = self.base.stats() if hasattr(self.base, "stats") else {}
        return {**base, "batched_slots": self.slots,
                "active_streams": self.n_active, "per_stream": per_slot}
