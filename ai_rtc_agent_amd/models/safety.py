"""Safety checker — optional NSFW gate on output frames.

Parity: the reference wrapper optionally runs the StableDiffusionSafetyChecker
on outputs and substitutes a black image on trigger
(reference lib/wrapper.py:930-942; ctor flag use_safety_checker, :66).
Ours is a lightweight conv classifier with the same contract: score(frame)
in [0,1]; the engine blanks frames whose score exceeds the threshold.
(Offline: weights are random-init; on a deployment box load trained weights
via the plan cache like every other module.)
"""
from __future__ import annotations

import torch
import torch.nn as nn

from .. import ops
from .unet import Conv2d


class SafetyChecker(nn.Module):
    def __init__(self, threshold: float = 0.5):
        super().__init__()
        self.threshold = threshold
        self.c1 = Conv2d(3, 16, 3, stride=2)
        self.c2 = Conv2d(16, 32, 3, stride=2)
        self.c3 = Conv2d(32, 32, 3, stride=2)
        self.head = nn.Linear(32, 1)

    @torch.no_grad()
    def score(self, img: torch.Tensor) -> torch.Tensor:
        """img: (B,H,W,3) in [-1,1] -> (B,) score in [0,1]."""
        h = self.c1(img, act=ops.ACT_RELU)
        h = self.c2(h, act=ops.ACT_RELU)
        h = self.c3(h, act=ops.ACT_RELU)
        pooled = h.mean(dim=(1, 2))
        return torch.sigmoid(self.head(pooled.float())).squeeze(-1)

    @torch.no_grad()
    def filter(self, img: torch.Tensor) -> torch.Tensor:
        """Blank (black) any frame whose score crosses the threshold —
        the reference's black-image substitution behaviour."""
        s = self.score(img)
        mask = (s <= self.threshold).to(img.dtype).view(-1, 1, 1, 1)
        return img * mask - (1.0 - mask)  # blanked frames go to -1 (black)
