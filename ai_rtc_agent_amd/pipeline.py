"""Frame pipeline — the glue between the media plane and the engine.

Parity with reference lib/pipeline.py (L3 of SURVEY.md §1): owns the engine,
exposes __call__(frame) -> frame plus the runtime update surface
(update_prompt / update_t_index_list, reference lib/pipeline.py:44-48).
Pre/post-processing live inside the engine (fused HIP kernels) rather than
as CV-CUDA calls (reference lib/pipeline.py:50-74).

Defaults reproduce the reference's production pipeline
(lib/pipeline.py:11-14): prompt "fireworks in the night sky",
t_index_list [18,26,35,45] of 50 steps, guidance 0.0 (cfg self).
"""
from __future__ import annotations

from typing import Optional, Sequence

import torch

from .config import EngineConfig
from .engine import StreamDiffusionEngine


class StreamDiffusionPipeline:
    def __init__(self, model_id: str = "lykon/dreamshaper-8", cfg: Optional[EngineConfig] = None):
        if cfg is None:
            cfg = EngineConfig(model_id=model_id)
            if not torch.cuda.is_available():
                cfg.device = "cpu"
                cfg.use_hip_graph = False
        self.cfg = cfg
        # load-else-build ladder (reference lib/wrapper.py:611-615): a
        # pre-built engine plan (python build.py) skips model init + LoRA
        # fusion; otherwise build fresh.
        from .engine.plan import load_plan

        eng = None
        try:
            eng = load_plan(cfg)
        except Exception:
            eng = None
        self.engine = eng if eng is not None else StreamDiffusionEngine(cfg)
        self.engine.prepare(
            prompt=cfg.prompt,
            num_inference_steps=cfg.num_inference_steps,
            guidance_scale=cfg.guidance_scale,
        )

    def __call__(self, frame: torch.Tensor) -> torch.Tensor:
        out = self.engine(frame)
        # media-plane consumers read the tensor on their own stream; fence it
        # against the engine's pipelined decode stream
        self.engine.sync_output()
        return out

    def update_prompt(self, prompt: str) -> None:
        self.engine.update_prompt(prompt)

    def update_t_index_list(self, t_index_list: Sequence[int]) -> None:
        self.engine.update_t_index_list(list(t_index_list))

    def stats(self) -> dict:
        return self.engine.stats()
