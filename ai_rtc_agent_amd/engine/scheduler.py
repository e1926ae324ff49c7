"""Diffusion noise schedule + the t_index sub-sampling semantics.

From-scratch implementation of the schedule math the reference delegates to
its StreamDiffusion dependency. The *contract* it satisfies is visible in the
reference at lib/wrapper.py:394-407 (sub_timesteps = [timesteps[t] for t in
t_index_list], then repeat_interleave by frame_buffer_size) and
lib/wrapper.py:221-234 (prepare precomputes per-sub-timestep alpha/beta
coefficients used by the batched denoise step).

Schedule: standard Stable-Diffusion "scaled_linear" beta schedule
(beta in sqrt-space from 0.00085 to 0.012 over 1000 train steps), with an
LCM-style x0-prediction step. All coefficient tensors are precomputed at
prepare() time so the per-frame hot loop is pure tensor arithmetic (and can
be captured into a hipGraph).
"""
from __future__ import annotations

import math
from typing import List, Sequence

import torch


class StreamScheduler:
    """Precomputes per-sub-timestep coefficients for stream-batch denoising."""

    def __init__(
        self,
        num_train_timesteps: int = 1000,
        beta_start: float = 0.00085,
        beta_end: float = 0.012,
        num_inference_steps: int = 50,
    ) -> None:
        self.num_train_timesteps = num_train_timesteps
        self.num_inference_steps = num_inference_steps
        betas = (
            torch.linspace(
                beta_start ** 0.5, beta_end ** 0.5, num_train_timesteps,
                dtype=torch.float64,
            )
            ** 2
        )
        alphas = 1.0 - betas
        self.alphas_cumprod = torch.cumprod(alphas, dim=0).to(torch.float32)

        # Inference timetable: num_inference_steps descending timesteps over
        # the train range. t_index_list indexes into THIS table
        # (reference lib/wrapper.py:394-396: sub_timesteps = timesteps[t]).
        step = num_train_timesteps // num_inference_steps
        self.timesteps = torch.flip(
            torch.arange(0, num_inference_steps, dtype=torch.long) * step
            + (step - 1),
            dims=[0],
        )  # e.g. 50 steps -> [999, 979, ..., 19]

    # -- sub-timestep selection --------------------------------------------
    def sub_timesteps(self, t_index_list: Sequence[int]) -> List[int]:
        return [int(self.timesteps[t]) for t in t_index_list]

    def coefficients(
        self,
        t_index_list: Sequence[int],
        frame_buffer_size: int = 1,
        device: torch.device = torch.device("cpu"),
        dtype: torch.dtype = torch.float32,
    ) -> dict:
        """Per-stage coefficient tensors, repeat-interleaved by
        frame_buffer_size (reference lib/wrapper.py:398-407)."""
        subs = self.sub_timesteps(t_index_list)
        t = torch.tensor(subs, dtype=torch.long)
        a_prod = self.alphas_cumprod[t]
        alpha_sqrt = a_prod.sqrt()
        beta_sqrt = (1.0 - a_prod).sqrt()

        def ri(x: torch.Tensor) -> torch.Tensor:
            return x.repeat_interleave(frame_buffer_size).to(device=device, dtype=dtype)

        # LCM-style boundary-condition scalings (x0-prediction blend).
        # sigma_data = 0.5, timestep_scaling = 10 (standard LCM constants).
        sigma_data = 0.5
        scaled_t = t.to(torch.float32) * 10.0
        c_skip = sigma_data ** 2 / (scaled_t ** 2 + sigma_data ** 2)
        c_out = scaled_t / (scaled_t ** 2 + sigma_data ** 2).sqrt()

        def ri32(x: torch.Tensor) -> torch.Tensor:
            # flat f32 copies for the fused scheduler kernels
            # (ops.sched_add_noise / ops.sched_blend)
            return x.repeat_interleave(frame_buffer_size).to(
                device=device, dtype=torch.float32).contiguous()

        return {
            "sub_timesteps": subs,
            "sub_timesteps_tensor": ri(t.to(torch.float32)).to(torch.long),
            "alpha_prod_t_sqrt": ri(alpha_sqrt).view(-1, 1, 1, 1),
            "beta_prod_t_sqrt": ri(beta_sqrt).view(-1, 1, 1, 1),
            "c_skip": ri(c_skip).view(-1, 1, 1, 1),
            "c_out": ri(c_out).view(-1, 1, 1, 1),
            "alpha_f32": ri32(alpha_sqrt),
            "beta_f32": ri32(beta_sqrt),
            "c_skip_f32": ri32(c_skip),
            "c_out_f32": ri32(c_out),
        }

    # -- core math ----------------------------------------------------------
    @staticmethod
    def add_noise(
        x0: torch.Tensor,
        noise: torch.Tensor,
        alpha_sqrt: torch.Tensor,
        beta_sqrt: torch.Tensor,
    ) -> torch.Tensor:
        """q(x_t | x_0): x_t = sqrt(a)x0 + sqrt(1-a) eps."""
        return alpha_sqrt * x0 + beta_sqrt * noise

    @staticmethod
    def pred_x0(
        x_t: torch.Tensor,
        eps: torch.Tensor,
        alpha_sqrt: torch.Tensor,
        beta_sqrt: torch.Tensor,
    ) -> torch.Tensor:
        """Epsilon-parameterisation inversion: x0 = (x_t - sqrt(1-a) eps)/sqrt(a)."""
        return (x_t - beta_sqrt * eps) / alpha_sqrt

    def step_batch(
        self,
        eps: torch.Tensor,
        x_t: torch.Tensor,
        coeff: dict,
    ) -> torch.Tensor:
        """One denoise step for every in-flight stage simultaneously.

        Returns the LCM-style denoised estimate per stage:
        d = c_out * x0_pred + c_skip * x_t  (x0-prediction blend).
        The stream-batch machine shifts stage i's output to stage i+1's input
        (with fresh noise) — see stream_batch.py.
        """
        x0 = self.pred_x0(x_t, eps, coeff["alpha_prod_t_sqrt"], coeff["beta_prod_t_sqrt"])
        return coeff["c_out"] * x0 + coeff["c_skip"] * x_t
