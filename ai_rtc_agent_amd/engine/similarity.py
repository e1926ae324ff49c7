"""Stochastic similarity filter.

Contract from the reference wrapper surface
(lib/wrapper.py:57-59 defaults threshold=0.98 / max_skip_frame=10;
enable path lib/wrapper.py:192-195): when consecutive inputs are nearly
identical, probabilistically skip inference and replay the previous output,
capped at max_skip_frame consecutive skips.

Skip probability ramps with similarity: p_skip = clamp((sim - threshold) /
(1 - threshold), 0, 1), so a static scene saves almost all compute while any
motion resumes instantly. The RNG is torch-seeded for reproducibility.
"""
from __future__ import annotations

import torch


class StochasticSimilarityFilter:
    def __init__(
        self,
        threshold: float = 0.98,
        max_skip_frame: int = 10,
        generator: torch.Generator | None = None,
    ) -> None:
        self.threshold = float(threshold)
        self.max_skip_frame = int(max_skip_frame)
        self.generator = generator
        self._prev: torch.Tensor | None = None
        self._skips = 0

    def reset(self) -> None:
        self._prev = None
        self._skips = 0

    def similarity(self, x: torch.Tensor) -> float:
        if self._prev is None:
            return 0.0
        a = x.flatten().float()
        b = self._prev.flatten().float()
        denom = a.norm() * b.norm()
        if denom == 0:
            return 1.0
        return float((a @ b) / denom)

    def should_skip(self, x: torch.Tensor) -> bool:
        """Decide, then remember x as the new reference frame."""
        sim = self.similarity(x)
        self._prev = x.detach().clone()
        if sim < self.threshold or self._skips >= self.max_skip_frame:
            self._skips = 0
            return False
        p_skip = min(1.0, (sim - self.threshold) / max(1e-6, 1.0 - self.threshold))
        r = torch.rand((), generator=self.generator, device="cpu").item()
        if r < p_skip:
            self._skips += 1
            return True
        self._skips = 0
        return False
