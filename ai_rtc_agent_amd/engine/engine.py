"""StreamDiffusionEngine — the per-frame diffusion engine.

From-scratch replacement for reference L4+L5 (lib/wrapper.py +
the external StreamDiffusion package). Public surface mirrors the
reference wrapper contract:

- ctor option surface            (reference lib/wrapper.py:34-132)
- prepare(prompt, steps, g)      (reference lib/wrapper.py:197-234)
- __call__/img2img/txt2img       (reference lib/wrapper.py:236-343)
- update_prompt                  (reference lib/pipeline.py:44-45)
- update_t_index_list            (reference lib/wrapper.py:389-407)
- stream-batch law B = len(t_index)*frame_buffer (lib/wrapper.py:159-163)

MI355X-native execution: the whole per-frame step (VAE encode -> stream-batch
UNet -> scheduler -> VAE decode) runs over static device buffers so it can be
captured ONCE into a hipGraph (torch.cuda.CUDAGraph is hipGraph on ROCm) and
replayed per frame — this replaces the reference's TensorRT engine loading
(lib/wrapper.py:409-512). Prompt and t_index updates write into
graph-external buffers in place; the graph is never re-captured
(SURVEY.md §7 hard part #2).
"""
from __future__ import annotations

from typing import Optional, Sequence

import logging
import math
import os

import torch

from ..config import EngineConfig
from ..models import TinyVAE, TextEncoder, UNet2DCondition, UNetConfig
from ..models.lora import fuse_lora_state_dict, load_lora_file, make_random_lora
from ..utils.timers import StageTimers
from .. import ops
from .rcfg import ResidualCFG
from .scheduler import StreamScheduler
from .similarity import StochasticSimilarityFilter


def _unet_config_for(family: str) -> UNetConfig:
    return {
        "sd15": UNetConfig.sd15,
        "sd21": UNetConfig.sd21,
        "sdxl": UNetConfig.sdxl,
        "tiny": UNetConfig.tiny,
        "tiny_xl": UNetConfig.tiny_xl,
    }[family]()


class StreamDiffusionEngine:
    def __init__(
        self,
        cfg: EngineConfig,
        unet: Optional[UNet2DCondition] = None,
        vae: Optional[TinyVAE] = None,
        text_encoder: Optional[TextEncoder] = None,
    ) -> None:
        self.cfg = cfg
        self.device = torch.device(cfg.device if torch.cuda.is_available() or cfg.device == "cpu" else "cpu")
        self.dtype = getattr(torch, cfg.dtype) if self.device.type == "cuda" else torch.float32
        torch.manual_seed(cfg.seed)

        self.scheduler = StreamScheduler(num_inference_steps=cfg.num_inference_steps)
        self.rcfg = ResidualCFG(cfg.cfg_type, cfg.guidance_scale, cfg.delta)
        self.sim_filter: Optional[StochasticSimilarityFilter] = None
        if cfg.similarity_filter.enabled:
            gen = torch.Generator().manual_seed(cfg.seed)
            self.sim_filter = StochasticSimilarityFilter(
                cfg.similarity_filter.threshold, cfg.similarity_filter.max_skip_frame, gen
            )

        ucfg = _unet_config_for(cfg.model_family)
        self.unet = unet if unet is not None else UNet2DCondition(ucfg)
        self.vae = vae if vae is not None else TinyVAE()
        # family conventions: SD1.5 = CLIP ViT-L/14 (quick-gelu, last
        # layer); SD2.x = OpenCLIP ViT-H (gelu, PENULTIMATE layer, 23
        # blocks); sdxl = the DUAL encoder (ViT-L + OpenCLIP bigG, both
        # penultimate, per-token concat to 2048, pooled from bigG)
        if text_encoder is not None:
            self.text_encoder = text_encoder
        elif ucfg.addition_embed_dim:
            from ..models.text_encoder import DualTextEncoder

            if cfg.model_family == "sdxl":
                self.text_encoder = DualTextEncoder()
            else:  # tiny_xl: small dual with the same contract
                half = ucfg.cross_attention_dim // 2
                self.text_encoder = DualTextEncoder(
                    hidden1=half, layers1=2,
                    hidden2=ucfg.cross_attention_dim - half, layers2=2,
                    vocab_size=512)
        else:
            if cfg.model_family == "tiny":
                te_layers, te_act, te_skip = 2, "quick_gelu", 0
            elif cfg.model_family == "sd21":
                te_layers, te_act, te_skip = 23, "gelu", 1
            else:
                te_layers, te_act, te_skip = 12, "quick_gelu", 0
            self.text_encoder = TextEncoder(
                hidden=ucfg.cross_attention_dim,
                layers=te_layers,
                act=te_act,
                clip_skip=te_skip,
            )
        self.ctx_dim = ucfg.cross_attention_dim
        self.addition_embed_dim = ucfg.addition_embed_dim

        self.controlnet = None
        if cfg.use_controlnet:
            from ..models.controlnet import ControlNet

            self.controlnet = ControlNet(ucfg)
        self.safety_checker = None
        if cfg.use_safety_checker:
            from ..models.safety import SafetyChecker

            self.safety_checker = SafetyChecker()

        # real weights when available: model_id as a local diffusers-style
        # dir, or an HF-cache snapshot (random init otherwise — offline)
        self._load_weights_if_present(cfg.model_id)

        # LoRA fusion happens BEFORE device placement / graph capture
        # (reference fuses before TRT compile, lib/wrapper.py:645-697).
        if cfg.use_lcm_lora:
            sd = (
                load_lora_file(cfg.lcm_lora_id)
                if cfg.lcm_lora_id
                else make_random_lora(self.unet, rank=4, seed=cfg.seed)
            )
            fuse_lora_state_dict(self.unet, sd, scale=1.0)
        if cfg.lora_dict:
            for path, scale in cfg.lora_dict.items():
                fuse_lora_state_dict(self.unet, load_lora_file(path), scale=scale)

        self.unet = self.unet.to(self.device, self.dtype).eval()
        self.vae = self.vae.to(self.device, self.dtype).eval()
        self.text_encoder = self.text_encoder.to(self.device).eval()
        if self.controlnet is not None:
            self.controlnet = self.controlnet.to(self.device, self.dtype).eval()
        if self.safety_checker is not None:
            self.safety_checker = self.safety_checker.to(self.device, self.dtype).eval()

        self.timers = StageTimers(use_cuda=self.device.type == "cuda")
        self._graph: Optional[torch.cuda.CUDAGraph] = None
        self._prepared = False
        self.prompt: str = cfg.prompt

        # fp8 serving tier (opt-in): calibrate per-layer activation scales
        # on the first real frames (eager), freeze + quality-gate, then the
        # graph captures the GN-fp8 -> MX-MFMA conv path. fp16 stays the
        # default; incompatible with the env-gated AIRTC_FUSE_GN experiment
        # (that path bypasses the GN apply kernel entirely).
        self._fp8_norms: list = []
        self._fp8_calib_left = 0
        self.fp8_active = False
        self.fp8_snr_db: Optional[float] = None
        self._fp8_vae: dict = {"convs": [], "outs": []}
        if cfg.use_fp8 and os.environ.get("AIRTC_FUSE_GN") != "1":
            from ..models.taesd import fp8_flag_convs
            from ..models.unet import fp8_eligible_norms

            self._fp8_norms = fp8_eligible_norms(self.unet)
            if self.controlnet is not None:
                # ControlNet shares the resnet blocks — same tier applies
                self._fp8_norms += fp8_eligible_norms(self.controlnet)
            for nrm in self._fp8_norms:
                nrm._fp8_calibrate = True
                nrm._fp8_amax = 0.0
            if cfg.use_tiny_vae:
                self._fp8_vae = fp8_flag_convs(self.vae)
                for c in self._fp8_vae["convs"]:
                    c._fp8_calibrate = True
                    c._fp8_in_amax = 0.0
                    c._fp8_out_amax = 0.0
            self._fp8_calib_left = max(1, cfg.fp8_calib_frames)

    def _load_weights_if_present(self, model_id: str) -> None:
        import glob
        import os

        from ..models.load import load_model_dir

        candidates = [model_id] if os.path.isdir(model_id) else []
        hub = os.environ.get("HF_HUB_CACHE", os.path.expanduser("~/.cache/huggingface/hub"))
        candidates += glob.glob(
            os.path.join(hub, "models--" + model_id.replace("/", "--"), "snapshots", "*")
        )
        for c in candidates:
            try:
                if load_model_dir(self, c):
                    return
            except Exception:  # corrupt cache entry: keep random init
                continue

    # ------------------------------------------------------------------
    # prepare
    # ------------------------------------------------------------------
    @torch.no_grad()
    def prepare(
        self,
        prompt: Optional[str] = None,
        num_inference_steps: Optional[int] = None,
        guidance_scale: Optional[float] = None,
        t_index_list: Optional[Sequence[int]] = None,
    ) -> None:
        cfg = self.cfg
        if prompt is not None:
            self.prompt = prompt
        if num_inference_steps is not None:
            cfg.num_inference_steps = num_inference_steps
            self.scheduler = StreamScheduler(num_inference_steps=num_inference_steps)
        if guidance_scale is not None:
            cfg.guidance_scale = guidance_scale
            self.rcfg = ResidualCFG(cfg.cfg_type, guidance_scale, cfg.delta)
        if t_index_list is not None:
            cfg.t_index_list = list(t_index_list)

        n = cfg.denoising_steps
        fbs = cfg.frame_buffer_size
        B = n * fbs
        lh, lw = cfg.latent_height, cfg.latent_width
        dev, dt = self.device, self.dtype

        # coefficients live in the engine compute dtype: scheduler tensor math
        # must not promote the latent out of f16 (the HIP dispatch dtype)
        self._coeff = self.scheduler.coefficients(cfg.t_index_list, fbs, dev, self.dtype)
        # NHWC coefficient views: (B,1,1,1) already broadcast over (B,h,w,c)
        self._embeds = self.text_encoder.encode(self.prompt, dev, dt)
        self._embeds_batch = self._embeds.expand(B, -1, -1).contiguous()
        if cfg.cfg_type == "full":
            neg = self.text_encoder.encode(cfg.negative_prompt, dev, dt)
            self._embeds_full = torch.cat([neg.expand(B, -1, -1), self._embeds_batch], dim=0).contiguous()

        # sdxl addition conditioning: pooled text (1280) + 6 sinusoidal
        # time-id embeddings of 256 (orig h/w, crop t/l, target h/w) = 2816
        self._added_cond = None
        if self.addition_embed_dim:
            self._added_cond = torch.zeros((B, self.addition_embed_dim), device=dev, dtype=dt)
            self._refresh_added_cond()

        g = torch.Generator(device="cpu").manual_seed(cfg.seed)
        self._init_noise = torch.randn((B, lh, lw, 4), generator=g).to(dev, dt)
        self._x_t_buffer = torch.zeros((max(0, B - fbs), lh, lw, 4), device=dev, dtype=dt)
        self.rcfg.reset(self._init_noise)

        # static I/O buffers (graph-stable addresses); _frame_in (u8) is the
        # only per-frame input — preprocess lives inside the graph
        self._frame_in = torch.zeros((fbs, cfg.height, cfg.width, 3), device=dev, dtype=torch.uint8)
        self._ts_batch = self._coeff["sub_timesteps_tensor"].to(dev)
        self._prev_out: Optional[torch.Tensor] = None
        self._graph = None
        self._pipelined = False
        self._last_done = None
        self._prepared = True
        self._refresh_static_kv()
        self._ts_unet_cache = None
        self._refresh_temb_static()

    def _cross_attn_modules(self):
        from ..models.unet import CrossAttention

        mods = []
        for root in (self.unet, self.controlnet):
            if root is None:
                continue
            for name, m in root.named_modules():
                if isinstance(m, CrossAttention) and name.endswith("attn2"):
                    mods.append(m)
        return mods

    @torch.no_grad()
    def _refresh_static_kv(self) -> None:
        """Precompute every cross-attention layer's K|V from the (static)
        text embeddings. K/V depend only on the prompt, so these GEMMs run
        at prepare()/update_prompt() instead of once per frame; the hot
        graph reads the static buffers (stable addresses, copy_ on update)."""
        ctx = self._unet_batch_embeds()
        for m in self._cross_attn_modules():
            kv = m.compute_kv(ctx)
            cur = getattr(m, "static_kv", None)
            if cur is None or cur.shape != kv.shape or cur.dtype != kv.dtype:
                m.static_kv = kv.contiguous()
            else:
                cur.copy_(kv)

    @torch.no_grad()
    def _refresh_added_cond(self) -> None:
        """(Re)compute the sdxl addition-embedding into its static buffer."""
        from ..models.unet import timestep_embedding

        cfg = self.cfg
        pooled = self.text_encoder.pooled(self.prompt, self.device, self.dtype)
        ids = torch.tensor(
            [cfg.height, cfg.width, 0, 0, cfg.height, cfg.width],
            dtype=torch.float32, device=self.device,
        )
        time_emb = timestep_embedding(ids, 256).flatten()[None].to(self.dtype)
        vec = torch.cat([pooled, time_emb], dim=-1)
        self._added_cond.copy_(vec.expand_as(self._added_cond))

    # ------------------------------------------------------------------
    # weight hot-swap (LoRA) — SURVEY.md §5.8 / N12: fuse, invalidate the
    # kernel-side weight caches, re-broadcast over RCCL when distributed,
    # and drop the hipGraphs (re-captured on the next frame)
    # ------------------------------------------------------------------
    @torch.no_grad()
    def load_lora(self, path_or_sd, scale: float = 1.0) -> int:
        sd = load_lora_file(path_or_sd) if isinstance(path_or_sd, str) else path_or_sd
        n = fuse_lora_state_dict(self.unet, sd, scale=scale)
        self.refresh_weights()
        return n

    @torch.no_grad()
    def refresh_weights(self) -> None:
        """After any in-place weight change: drop the lazily-built GPU weight
        transforms (conv GEMM layouts, fused QKV, f32 bias copies), re-sync
        replicas, and force a graph re-capture."""
        for module in (self.unet, self.vae, self.controlnet):
            if module is None:
                continue
            for m in module.modules():
                for attr in ("_wqkv", "_wkv", "_wq", "_wout"):
                    if hasattr(m, attr):
                        delattr(m, attr)
                for t in list(m.parameters(recurse=False)):
                    for attr in ("_airtc_wperm", "_airtc_b32", "_airtc_g32", "_airtc_w16", "_airtc_wpad32", "_airtc_wfp8", "_airtc_wdq"):
                        if hasattr(t, attr):
                            delattr(t, attr)
        from ..parallel.collectives import broadcast_engine_weights

        broadcast_engine_weights(self)  # no-op at world_size 1
        if self._prepared:
            self._refresh_static_kv()
            self._refresh_temb_static()  # time-emb MLP weights changed
        # fp8 tier: new weights shift the activation distributions — drop
        # back to calibration (a few eager frames) and re-gate
        if self.cfg.use_fp8 and (self._fp8_norms or self._fp8_vae["convs"]):
            for nrm in self._fp8_norms:
                nrm._fp8_calibrate = True
                nrm._fp8_scale = None
                nrm._fp8_amax = 0.0
            for c in self._fp8_vae["convs"]:
                c._fp8_calibrate = True
                c._fp8_in_scale = c._fp8_out_scale = None
                c._fp8_in_amax = c._fp8_out_amax = 0.0
            self.fp8_active = False
            self._fp8_calib_left = max(1, self.cfg.fp8_calib_frames)
        if self.device.type == "cuda":
            # quiesce in-flight replays before their graphs are dropped
            torch.cuda.synchronize()
        self._graph = None  # re-capture lazily with the new weights

    # ------------------------------------------------------------------
    # runtime config updates (POST /config + datachannel)
    # ------------------------------------------------------------------
    @torch.no_grad()
    def update_prompt(self, prompt: str) -> None:
        """Re-encode the prompt and overwrite the cached embeddings IN PLACE
        (graph-external buffer update; reference lib/pipeline.py:44-45)."""
        self.prompt = prompt
        emb = self.text_encoder.encode(prompt, self.device, self.dtype)
        self._embeds.copy_(emb)
        self._embeds_batch.copy_(emb.expand_as(self._embeds_batch))
        if self.cfg.cfg_type == "full":
            B = self._embeds_batch.shape[0]
            self._embeds_full[B:].copy_(self._embeds_batch)
        if self._added_cond is not None:
            self._refresh_added_cond()
            self._refresh_temb_static()  # added-cond feeds the static temb
        self._refresh_static_kv()

    @torch.no_grad()
    def update_t_index_list(self, t_index_list: Sequence[int]) -> None:
        """Contract of reference lib/wrapper.py:389-407: no-op when
        unchanged; length changes require prepare() (batch shape changes)."""
        t_index_list = list(t_index_list)
        if t_index_list == self.cfg.t_index_list:
            return
        if len(t_index_list) != len(self.cfg.t_index_list):
            # prepare() reallocates graph-captured buffers (_coeff, _x_t_buffer,
            # _frame_in) and drops the graphs; quiesce first so in-flight
            # pipelined replays on streams A/B are not reading freed memory
            # (same rule refresh_weights follows).
            if self.device.type == "cuda":
                torch.cuda.synchronize(self.device)
            self.cfg.t_index_list = t_index_list
            self.prepare()
            return
        self.cfg.t_index_list = t_index_list
        new = self.scheduler.coefficients(
            t_index_list, self.cfg.frame_buffer_size, self.device, self.dtype
        )
        for k in ("alpha_prod_t_sqrt", "beta_prod_t_sqrt", "c_skip", "c_out"):
            self._coeff[k].copy_(new[k])
        self._coeff["sub_timesteps"] = new["sub_timesteps"]
        self._ts_batch.copy_(new["sub_timesteps_tensor"].to(self.device))
        for k in ("alpha_f32", "beta_f32", "c_skip_f32", "c_out_f32"):
            self._coeff[k].copy_(new[k].to(self.device))
        # static time-embedding caches follow the new timesteps (in place —
        # any captured graph reads the same storage). The batched-ts cache
        # is a SEPARATE cat buffer for cfg full/initialize: rebuild it.
        self._ts_unet_cache = None
        self._refresh_temb_static()

    # ------------------------------------------------------------------
    # core step (graph-capturable: static shapes, static buffers)
    # ------------------------------------------------------------------
    def _unet_batch_input(self, x_t: torch.Tensor) -> torch.Tensor:
        cfg = self.cfg
        if cfg.cfg_type == "full" and self.rcfg.active:
            return torch.cat([x_t, x_t], dim=0)
        if cfg.cfg_type == "initialize" and self.rcfg.active:
            return torch.cat([x_t[: cfg.frame_buffer_size], x_t], dim=0)
        return x_t

    def _unet_batch_embeds(self) -> torch.Tensor:
        cfg = self.cfg
        if cfg.cfg_type == "full" and self.rcfg.active:
            return self._embeds_full
        if cfg.cfg_type == "initialize" and self.rcfg.active:
            return torch.cat(
                [self._embeds_batch[: cfg.frame_buffer_size], self._embeds_batch], dim=0
            )
        return self._embeds_batch

    def _unet_batch_timesteps(self) -> torch.Tensor:
        # cached: the static-timestep UNet fast path identity-checks this
        # exact tensor object (prepare()/t-index updates reset the cache)
        t = getattr(self, "_ts_unet_cache", None)
        if t is not None:
            return t
        cfg = self.cfg
        if cfg.cfg_type == "full" and self.rcfg.active:
            t = torch.cat([self._ts_batch, self._ts_batch], dim=0)
        elif cfg.cfg_type == "initialize" and self.rcfg.active:
            t = torch.cat([self._ts_batch[: cfg.frame_buffer_size], self._ts_batch], dim=0)
        else:
            t = self._ts_batch
        self._ts_unet_cache = t
        return t

    def _refresh_temb_static(self) -> None:
        """(Re)compute the UNet's static time-embedding caches — call after
        anything that changes timesteps, added-cond or weights. In-place
        when shapes match, so captured graphs keep reading the same
        storage."""
        cfg = self.cfg
        ts = self._unet_batch_timesteps()
        added = self._added_cond
        if added is not None and (cfg.cfg_type in ("full", "initialize") and self.rcfg.active):
            extra = added.shape[0] if cfg.cfg_type == "full" else cfg.frame_buffer_size
            added = torch.cat([added[:extra], added], dim=0)
        self.unet.precompute_time_embeddings(ts, added, dtype=self.dtype)

    @torch.no_grad()
    @torch.no_grad()
    def _fp8_quality_snr(self) -> float:
        """SNR (dB) of the fp8 UNet forward vs the f16 forward on the same
        inputs — a pure function of the UNet (no engine state mutated)."""
        unet_in = self._unet_batch_input(self._init_noise)
        ts = self._unet_batch_timesteps()
        emb = self._unet_batch_embeds()
        added = self._added_cond
        cfg = self.cfg
        if added is not None and (cfg.cfg_type in ("full", "initialize") and self.rcfg.active):
            extra = added.shape[0] if cfg.cfg_type == "full" else cfg.frame_buffer_size
            added = torch.cat([added[:extra], added], dim=0)
        scales = [n._fp8_scale for n in self._fp8_norms]
        for n in self._fp8_norms:
            n._fp8_scale = None
        ref = self.unet(unet_in, ts, emb, added_cond=added).float()
        for n, s in zip(self._fp8_norms, scales):
            n._fp8_scale = s
        got = self.unet(unet_in, ts, emb, added_cond=added).float()
        err = ((got - ref) ** 2).mean().item()
        sig = (ref ** 2).mean().item()
        return 10.0 * math.log10(sig / max(err, 1e-20))

    @torch.no_grad()
    def _fp8_vae_snr(self) -> float:
        """min SNR of the fp8 TAESD decode (random latent) and encode
        (random frame) vs f16 — pure functions of the VAE."""
        g = torch.Generator(device="cpu").manual_seed(self.cfg.seed + 7)
        lat = torch.randn((1, self.cfg.latent_height, self.cfg.latent_width, 4),
                          generator=g).to(self.device, self.dtype)
        img = (torch.rand((1, self.cfg.height, self.cfg.width, 3),
                          generator=g) * 2 - 1).to(self.device, self.dtype)
        scales = [(c, c._fp8_in_scale, c._fp8_out_scale)
                  for c in self._fp8_vae["convs"]]
        for c, _, _ in scales:
            c._fp8_in_scale = c._fp8_out_scale = None
        ref_d = self.vae.decode(lat).float()
        ref_e = self.vae.encode(img).float()
        for c, si, so in scales:
            c._fp8_in_scale, c._fp8_out_scale = si, so
        got_d = self.vae.decode(lat).float()
        got_e = self.vae.encode(img).float()

        def snr(got, ref):
            err = ((got - ref) ** 2).mean().item()
            return 10.0 * math.log10((ref ** 2).mean().item() / max(err, 1e-20))

        return min(snr(got_d, ref_d), snr(got_e, ref_e))

    def _fp8_freeze(self) -> None:
        """End calibration: freeze per-layer scales, run the quality gates
        (UNet forward + TAESD decode), fall back to f16 if either fails
        (SURVEY.md §6 quality guard)."""
        log = logging.getLogger("airtc.engine")
        cfg = self.cfg
        for nrm in self._fp8_norms:
            nrm._fp8_calibrate = False
            if nrm._fp8_amax > 0:
                nrm._fp8_scale = nrm._fp8_amax * cfg.fp8_margin / 448.0
        for c in self._fp8_vae["convs"]:
            c._fp8_calibrate = False
            if c._fp8_in_amax > 0:
                c._fp8_in_scale = c._fp8_in_amax * cfg.fp8_margin / 448.0
        for c in self._fp8_vae["outs"]:
            if c._fp8_out_amax > 0:
                c._fp8_out_scale = c._fp8_out_amax * cfg.fp8_margin / 448.0
        self.fp8_snr_db = self._fp8_quality_snr()
        vae_snr = self._fp8_vae_snr() if self._fp8_vae["convs"] else 1e9
        if self.fp8_snr_db < cfg.fp8_min_snr_db or vae_snr < cfg.fp8_min_snr_db:
            for nrm in self._fp8_norms:
                nrm._fp8_scale = None
            for c in self._fp8_vae["convs"]:
                c._fp8_in_scale = c._fp8_out_scale = None
            self.fp8_active = False
            log.warning(
                "fp8 quality gate FAILED (unet %.1f dB, vae %.1f dB, "
                "min %.1f) — serving f16",
                self.fp8_snr_db, vae_snr, cfg.fp8_min_snr_db)
        else:
            self.fp8_active = True
            log.info(
                "fp8 tier active: %d GN->conv pairs + %d VAE convs, "
                "unet %.1f dB / vae %.1f dB",
                len(self._fp8_norms), len(self._fp8_vae["convs"]),
                self.fp8_snr_db, vae_snr)

    @torch.no_grad()
    def _denoise_core(self) -> torch.Tensor:
        """Stream-batch round WITHOUT the VAE decode: _img_in -> denoised
        latent (fbs, lh, lw, 4). Split out so the decode half can run on a
        second HIP stream overlapped with the next frame's denoise."""
        cfg = self.cfg
        fbs = cfg.frame_buffer_size
        co = self._coeff

        if cfg.mode == "img2img":
            # preprocess lives INSIDE the graph: _frame_in (u8) is the only
            # per-frame input buffer
            self._img_proc = ops.preprocess_from_u8(self._frame_in, self.dtype)
            x0 = self.vae.encode(self._img_proc)
            x_t0 = ops.sched_add_noise(
                x0,
                self._init_noise[:fbs],
                co["alpha_f32"][:fbs],
                co["beta_f32"][:fbs],
            ) if cfg.do_add_noise else x0
        else:  # txt2img: stage-0 input is pure noise
            x_t0 = self._init_noise[:fbs]

        x_t = torch.cat([x_t0, self._x_t_buffer], dim=0) if self._x_t_buffer.shape[0] else x_t0

        added = self._added_cond
        if added is not None and (cfg.cfg_type in ("full", "initialize") and self.rcfg.active):
            extra = added.shape[0] if cfg.cfg_type == "full" else fbs
            added = torch.cat([added[:extra], added], dim=0)
        unet_in = self._unet_batch_input(x_t)
        unet_ts = self._unet_batch_timesteps()
        unet_emb = self._unet_batch_embeds()
        control = None
        if self.controlnet is not None and cfg.mode == "img2img":
            # hint = the current input frame (per in-flight stage we reuse
            # the newest frame's hint; per-stage hints would need a hint
            # FIFO mirroring the latent buffer)
            hint = self._img_proc.expand(unet_in.shape[0], -1, -1, -1).contiguous()
            control = self.controlnet(
                unet_in, unet_ts, unet_emb, hint, scale=cfg.controlnet_scale
            )
        eps = self.unet(unet_in, unet_ts, unet_emb, added_cond=added, control=control)
        eps = self.rcfg.apply(eps, fbs)
        # fused LCM step (one kernel; same math as scheduler.step_batch)
        denoised = ops.sched_blend(x_t, eps, co["alpha_f32"], co["beta_f32"],
                                   co["c_out_f32"], co["c_skip_f32"])

        if denoised.shape[0] > fbs:
            # shift: stage i output -> stage i+1 input at tau_{i+1}
            nxt = ops.sched_add_noise(
                denoised[:-fbs].contiguous(),
                self._init_noise[fbs:],
                co["alpha_f32"][fbs:],
                co["beta_f32"][fbs:],
            ) if cfg.do_add_noise else denoised[:-fbs]
            self._x_t_buffer.copy_(nxt)

        return denoised[-fbs:]

    @torch.no_grad()
    def _decode_core(self, latent: torch.Tensor) -> torch.Tensor:
        decoded = self.vae.decode(latent)
        if self.safety_checker is not None:
            decoded = self.safety_checker.filter(decoded)
        return ops.postprocess_to_u8(decoded)

    @torch.no_grad()
    def _step_core(self) -> torch.Tensor:
        """Sequential full step (CPU / eager GPU path)."""
        return self._decode_core(self._denoise_core())

    def _maybe_capture(self) -> None:
        """Capture the per-frame step into hipGraphs (torch.cuda.CUDAGraph is
        hipGraph on ROCm).

        PIPELINED mode (default): the step is split into two graphs —
        g1 = preprocess-side denoise (VAE encode -> stream-batch UNet ->
        scheduler -> latent handoff buffer) on stream A, and
        g2 = VAE decode + postprocess on stream B. g2(frame i) runs
        CONCURRENTLY with g1(frame i+1): steady-state throughput becomes
        max(g1, g2) instead of g1+g2. Latent handoff buffers are ping-pong
        (two g1/g2 captures sharing pools per stream); event chain:
          sA: wait done[pp] -> g1 -> record lat[pp]
          sB: wait lat[pp]  -> g2 -> record done[pp]
        The CALLER's stream is never made to wait (that would transitively
        serialise the next frame's g1 behind this frame's decode) —
        consumers that read the output tensor call sync_output() first.
        """
        if (
            self._graph is not None
            or self.device.type != "cuda"
            or not self.cfg.use_hip_graph
        ):
            return
        fbs = self.cfg.frame_buffer_size
        lh, lw = self.cfg.latent_height, self.cfg.latent_width
        # pipeline overlap composes with the similarity filter: a skipped
        # frame simply returns the persistent output buffer of the last
        # replay without touching the stream A/B event chain
        self._pipelined = self.cfg.pipeline_overlap
        sA = torch.cuda.Stream()
        sB = torch.cuda.Stream()
        self._sA, self._sB = sA, sB
        self._lat_out = [
            torch.zeros((fbs, lh, lw, 4), device=self.device, dtype=self.dtype)
            for _ in range(2)
        ]
        # warmup (weight-transform caches etc.) on a side stream
        sA.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(sA):
            for _ in range(2):
                lat = self._denoise_core()
                self._lat_out[0].copy_(lat)
                _ = self._decode_core(self._lat_out[0])
        torch.cuda.current_stream().wait_stream(sA)

        if not self._pipelined:
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g, stream=sA):
                self._graph_out = self._step_core()
            self._graph = g
            return

        self._g1, self._g2, self._out_u8 = [], [], []
        for pp in range(2):
            g1 = torch.cuda.CUDAGraph()
            kw = {"pool": self._g1[0].pool()} if pp else {}
            with torch.cuda.graph(g1, stream=sA, **kw):
                self._lat_out[pp].copy_(self._denoise_core())
            self._g1.append(g1)
            g2 = torch.cuda.CUDAGraph()
            kw = {"pool": self._g2[0].pool()} if pp else {}
            with torch.cuda.graph(g2, stream=sB, **kw):
                self._out_u8.append(self._decode_core(self._lat_out[pp]))
            self._g2.append(g2)
        self._ev_in = torch.cuda.Event()
        self._ev_lat = [torch.cuda.Event() for _ in range(2)]
        self._ev_done = [torch.cuda.Event() for _ in range(2)]
        for e in self._ev_done:
            e.record(sB)
        self._pp = 0
        self._graph = True  # sentinel: pipelined graphs ready

    def sync_output(self) -> None:
        """Make the caller's stream wait until the last returned output is
        fully produced (pipelined mode defers this so the next frame's
        denoise can overlap the decode)."""
        ev = getattr(self, "_last_done", None)
        if ev is not None:
            torch.cuda.current_stream().wait_event(ev)

    # ------------------------------------------------------------------
    # public frame API
    # ------------------------------------------------------------------
    @torch.no_grad()
    def __call__(self, frame_u8: torch.Tensor) -> torch.Tensor:
        """frame_u8: (H,W,3) or (fbs,H,W,3) uint8 RGB on any device.
        Returns stylised (H,W,3) / (fbs,H,W,3) uint8 RGB on self.device."""
        assert self._prepared, "call prepare() first"
        squeeze = frame_u8.dim() == 3
        if squeeze:
            frame_u8 = frame_u8.unsqueeze(0)

        with self.timers.stage("preprocess"):
            frame_u8 = frame_u8.to(self.device, non_blocking=True)

        if self.sim_filter is not None and self._prev_out is not None:
            # on-device pooled cosine (centering + 64x64 avg-pool inside
            # the filter); the only CPU sync is one scalar read that waits
            # on the frame upload, NOT on the pipelined streams
            if self.sim_filter.should_skip(frame_u8):
                out = self._prev_out
                self.timers.frame_done()
                return out[0] if squeeze else out

        with self.timers.stage("diffusion"):
            if (self.device.type == "cuda" and self.cfg.use_hip_graph
                    and not self._fp8_calib_left):
                self._maybe_capture()
                if self._pipelined:
                    cur = torch.cuda.current_stream()
                    pp = self._pp
                    self._pp ^= 1
                    self._ev_in.record(cur)
                    with torch.cuda.stream(self._sA):
                        self._sA.wait_event(self._ev_in)
                        self._sA.wait_event(self._ev_done[pp])
                        self._frame_in.copy_(frame_u8)
                        frame_u8.record_stream(self._sA)
                        self._g1[pp].replay()
                        self._ev_lat[pp].record()
                    with torch.cuda.stream(self._sB):
                        self._sB.wait_event(self._ev_lat[pp])
                        self._g2[pp].replay()
                        self._ev_done[pp].record()
                    self._last_done = self._ev_done[pp]
                    out = self._out_u8[pp]
                else:
                    self._frame_in.copy_(frame_u8)
                    self._graph.replay()
                    out = self._graph_out
            else:
                self._frame_in.copy_(frame_u8)
                out = self._step_core()
                if self._fp8_calib_left:
                    self._fp8_calib_left -= 1
                    if self._fp8_calib_left == 0:
                        self._fp8_freeze()

        self._prev_out = out
        self.timers.frame_done()
        return out[0] if squeeze else out

    # reference wrapper exposes explicit img2img/txt2img entry points
    @torch.no_grad()
    def img2img(self, frame_u8: torch.Tensor) -> torch.Tensor:
        assert self.cfg.mode == "img2img"
        return self(frame_u8)

    @torch.no_grad()
    def txt2img(self) -> torch.Tensor:
        assert self.cfg.mode == "txt2img"
        fbs = self.cfg.frame_buffer_size
        dummy = torch.zeros(
            (fbs, self.cfg.height, self.cfg.width, 3), dtype=torch.uint8, device=self.device
        )
        return self(dummy)

    def stats(self) -> dict:
        d = self.timers.snapshot()
        if self.cfg.use_fp8:
            d["fp8"] = {
                "active": self.fp8_active,
                "layers": len(self._fp8_norms),
                "vae_convs": len(self._fp8_vae["convs"]),
                "quality_snr_db": self.fp8_snr_db,
                "calibrating": self._fp8_calib_left > 0,
            }
        return d
