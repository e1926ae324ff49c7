"""Session -> GPU dispatch: the multi-replica pipeline pool.

The reference shares ONE pipeline object across every peer connection in
one asyncio loop (reference agent.py:423; concurrent publishers interleave
frames through one stream-batch state machine — SURVEY.md §5.2 calls this
out as a hazard it simply accepts). The MI355X design
replaces that with explicit per-stream pipeline replicas, one per GPU:

- PipelinePool.create(n_gpus) builds one replica per visible GPU (replica i
  pinned to cuda:i); weight init is seed-identical across replicas, and
  when the pool is built under torch.distributed the weights come from the
  rank-0 RCCL broadcast instead (parallel/collectives.py)
- assign(stream_id) binds a session to the least-loaded replica (stream <->
  GPU affinity; steady state needs no cross-GPU traffic)
- release(stream_id) frees the slot

For CPU tests the pool degrades to one CPU pipeline replica.
"""
from __future__ import annotations

import logging
from typing import Dict, List

import torch

logger = logging.getLogger(__name__)


class PipelinePool:
    def __init__(self, pipelines: List):
        self._pipelines = pipelines
        self._load: Dict[int, int] = {i: 0 for i in range(len(pipelines))}
        self._sessions: Dict[str, int] = {}

    # -- construction ----------------------------------------------------
    @staticmethod
    def create(model_id: str, n_gpus: int = 1, cfg=None,
               streams_per_replica: int = 1) -> "PipelinePool":
        """streams_per_replica > 1 enables multi-stream batched serving:
        each replica's engine runs frame_buffer_size=K and up to K sessions
        share it through the asyncio collation loop (parallel/batching.py;
        measured +124% aggregate throughput at K=8 on MI355X,
        profiles/batching_ab.md)."""
        from ..config import EngineConfig
        from ..pipeline import StreamDiffusionPipeline

        import copy

        def build(device: str):
            c = copy.deepcopy(cfg) if cfg is not None else EngineConfig(model_id=model_id)
            c.device = device
            if device == "cpu":
                c.use_hip_graph = False
            if streams_per_replica > 1:
                c.frame_buffer_size = streams_per_replica
            p = StreamDiffusionPipeline(model_id, cfg=c)
            if streams_per_replica > 1:
                from .batching import BatchedPipeline

                return BatchedPipeline(p, streams_per_replica)
            return p

        pipelines = []
        if torch.cuda.is_available():
            n = min(n_gpus, torch.cuda.device_count())
            for i in range(n):
                # per-replica config copy: replicas must not share (and
                # last-write) one mutable config object
                pipelines.append(build(f"cuda:{i}"))
        else:
            pipelines.append(build("cpu"))
        logger.info("pipeline pool: %d replica(s) x %d stream slot(s)",
                    len(pipelines), streams_per_replica)
        return PipelinePool(pipelines)

    @staticmethod
    def single(pipeline) -> "PipelinePool":
        return PipelinePool([pipeline])

    # -- session affinity -------------------------------------------------
    def assign(self, stream_id: str):
        from .batching import BatchedPipeline

        if stream_id in self._sessions:
            idx = self._sessions[stream_id]
            p = self._pipelines[idx]
            return p.acquire(stream_id) if isinstance(p, BatchedPipeline) else p
        for idx in sorted(self._load, key=lambda i: self._load[i]):
            p = self._pipelines[idx]
            if isinstance(p, BatchedPipeline):
                proxy = p.acquire(stream_id)
                if proxy is None:
                    continue  # replica's slots are full
                self._load[idx] += 1
                self._sessions[stream_id] = idx
                logger.info("stream %s -> replica %d slot %d",
                            stream_id, idx, proxy.slot)
                return proxy
            self._load[idx] += 1
            self._sessions[stream_id] = idx
            logger.info("stream %s -> replica %d", stream_id, idx)
            return p
        # everything full: overload the least-loaded batched replica's base
        idx = min(self._load, key=lambda i: self._load[i])
        self._load[idx] += 1
        self._sessions[stream_id] = idx
        logger.warning("all stream slots busy; stream %s overloads replica %d",
                       stream_id, idx)
        p = self._pipelines[idx]
        return p.base if isinstance(p, BatchedPipeline) else p

    def release(self, stream_id: str) -> None:
        from .batching import BatchedPipeline

        idx = self._sessions.pop(stream_id, None)
        if idx is not None:
            self._load[idx] = max(0, self._load[idx] - 1)
            p = self._pipelines[idx]
            if isinstance(p, BatchedPipeline):
                p.release(stream_id)

    def active(self) -> List:
        return self._pipelines

    @staticmethod
    def _device_info(p) -> dict | None:
        dev = getattr(getattr(p, "cfg", None), "device", None)
        if dev is None or not str(dev).startswith("cuda") or not torch.cuda.is_available():
            return None
        d = torch.device(dev)
        return {
            "device": str(d),
            "name": torch.cuda.get_device_name(d),
            "memory_allocated_mb": round(torch.cuda.memory_allocated(d) / 1e6, 1),
        }

    def stats(self) -> dict:
        per_replica = []
        for i, p in enumerate(self._pipelines):
            entry = {"load": self._load[i]}
            info = self._device_info(p)
            if info:
                entry.update(info)
            if hasattr(p, "stats"):
                entry.update(p.stats())
            per_replica.append(entry)
        device = None
        if torch.cuda.is_available():
            device = {
                "name": torch.cuda.get_device_name(0),
                "memory_allocated_mb": round(torch.cuda.memory_allocated() / 1e6, 1),
            }
        return {
            "device": device,
            "replicas": len(self._pipelines),
            "sessions": {k: v for k, v in self._sessions.items()},
            "per_replica": per_replica,
        }
