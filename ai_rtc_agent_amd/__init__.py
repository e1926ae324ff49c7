"""ai_rtc_agent_amd — MI355X-native real-time video-diffusion agent.

A from-scratch rebuild of the capabilities of yondonfu/ai-rtc-agent
(reference layer map in SURVEY.md) designed MI355X-first:

- Diffusion engine (stream-batch denoising, RCFG, stochastic similarity
  filter, TAESD, LCM-LoRA fusing) implemented natively — no StreamDiffusion
  dependency (replaces reference L4/L5: lib/wrapper.py, streamdiffusion pkg).
- Per-frame hot path on hand-written HIP/CDNA4 kernels (MFMA implicit-GEMM
  conv, fused GroupNorm+SiLU, flash-style attention, fused pre/post-process),
  captured into a hipGraph (replaces TensorRT engines, reference
  lib/wrapper.py:409-512).
- Frame-level data parallelism across the 8 GPUs of one MI355X node with an
  RCCL weight broadcast over xGMI (new; the reference is single-GPU).
- WebRTC-style signalling API parity: /whip /whep /offer /config /
  (reference agent.py:466-472).
"""

__version__ = "0.1.0"

from . import config  # noqa: F401
