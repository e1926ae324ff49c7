"""Residual Classifier-Free Guidance (RCFG).

From-scratch implementation of the RCFG option surface the reference exposes
through its wrapper: cfg_type in {none, full, self, initialize} with
guidance_scale and delta (reference lib/wrapper.py:61,124-127,217-219).

Semantics implemented here (our own math, locked by tests):

- none:       eps_out = eps_c. No extra UNet cost.
- full:       classic CFG — the UNet batch is doubled with uncond embeddings
              (handled by the engine); eps_out = eps_u + g (eps_c - eps_u).
- self:       Residual CFG Self-Negative: a running per-stage "stock noise"
              buffer stands in for the uncond prediction.
              eps_out = eps_c + (g - 1) (eps_c - delta * stock), and the
              stock buffer is updated from the conditional prediction as the
              stream shifts (stage i's eps becomes stage i+1's stock).
- initialize: like self, but the stock buffer is (re)seeded from a single
              uncond UNet pass on the first frame_buffer_size rows
              (engine adds those rows to the batch; see unet_batch law,
              reference lib/wrapper.py:159-163).
"""
from __future__ import annotations

import torch


class ResidualCFG:
    def __init__(self, cfg_type: str, guidance_scale: float, delta: float = 1.0):
        if cfg_type not in ("none", "full", "self", "initialize"):
            raise ValueError(f"unknown cfg_type {cfg_type!r}")
        self.cfg_type = cfg_type
        self.guidance_scale = float(guidance_scale)
        self.delta = float(delta)
        self.stock_noise: torch.Tensor | None = None

    @property
    def active(self) -> bool:
        return self.cfg_type != "none" and self.guidance_scale > 1.0

    def reset(self, init_noise: torch.Tensor) -> None:
        """Seed the stock-noise buffer (engine calls at prepare())."""
        self.stock_noise = init_noise.clone()

    def apply(self, eps: torch.Tensor, frame_buffer_size: int) -> torch.Tensor:
        """eps is the raw UNet output batch.

        For cfg_type "full" the batch is [uncond | cond] stacked on dim 0;
        for "initialize" the first frame_buffer_size rows are the uncond
        seed pass; otherwise eps is the conditional batch.
        Returns the guided eps with the stream-batch row count.
        """
        g = self.guidance_scale
        if self.cfg_type == "full" and self.active:
            half = eps.shape[0] // 2
            eps_u, eps_c = eps[:half], eps[half:]
            return eps_u + g * (eps_c - eps_u)

        if self.cfg_type == "initialize" and self.active:
            seed, eps_c = eps[:frame_buffer_size], eps[frame_buffer_size:]
            if self.stock_noise is None or self.stock_noise.shape != eps_c.shape:
                self.stock_noise = eps_c.detach().clone()
            # seed pass overwrites the first stage's stock rows (in place:
            # graph-capturable, stable address)
            self.stock_noise[:frame_buffer_size].copy_(seed)
            out = eps_c + (g - 1.0) * (eps_c - self.delta * self.stock_noise)
            self._shift_stock(eps_c, frame_buffer_size)
            return out

        if self.cfg_type == "self" and self.active:
            if self.stock_noise is None or self.stock_noise.shape != eps.shape:
                self.stock_noise = eps.detach().clone()
            out = eps + (g - 1.0) * (eps - self.delta * self.stock_noise)
            self._shift_stock(eps, frame_buffer_size)
            return out

        if self.cfg_type == "initialize":
            # inactive guidance: drop the seed rows
            return eps[frame_buffer_size:]
        return eps

    def _shift_stock(self, eps_c: torch.Tensor, fbs: int) -> None:
        """Stage i's conditional eps becomes stage i+1's negative residual.

        In-place (copy_) into the stable stock buffer so the update is
        hipGraph-capturable: the graph reads and writes the SAME address
        every replay (SURVEY.md §7 hard part #2)."""
        shifted = torch.cat([eps_c[:fbs], eps_c[:-fbs]], dim=0)
        self.stock_noise.copy_(shifted)
