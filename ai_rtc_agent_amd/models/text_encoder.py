"""Prompt -> embedding (CLIP-class text encoder).

The reference runs CLIP ViT-L/14 in eager torch even on the TRT path
(reference lib/wrapper.py:468-473) because prompt encoding is cold —
it runs only at prepare()/update_prompt (reference lib/pipeline.py:44-45).
We keep the same design: eager PyTorch-ROCm (hipBLASLt under torch) is the
MI355X-native choice for this cold path.

Offline constraints: HF checkpoints/tokenizer files are unavailable in this
environment, so the encoder builds from-config (random init) when the cache
is empty, and a deterministic hash tokenizer stands in for the BPE vocab.
The contract the engine relies on — encode(prompt) -> (1, 77, ctx), stable
and prompt-sensitive — holds in both cases.
"""
from __future__ import annotations

import hashlib
import math

import torch
import torch.nn as nn


class HashTokenizer:
    """Deterministic stand-in tokenizer: word -> stable bucket id."""

    def __init__(self, vocab_size: int = 49408, max_length: int = 77):
        self.vocab_size = vocab_size
        self.max_length = max_length
        self.bos = 49406 % vocab_size
        self.eos = 49407 % vocab_size

    def __call__(self, text: str) -> torch.Tensor:
        ids = [self.bos]
        for w in text.lower().split():
            h = int.from_bytes(hashlib.sha1(w.encode()).digest()[:4], "little")
            ids.append(h % (self.vocab_size - 3))
        ids = ids[: self.max_length - 1] + [self.eos]
        ids += [self.eos] * (self.max_length - len(ids))
        return torch.tensor(ids, dtype=torch.long).unsqueeze(0)


class TextEncoder(nn.Module):
    """CLIP-style causal transformer text encoder (from-config)."""

    def __init__(
        self,
        hidden: int = 768,
        layers: int = 12,
        heads: int | None = None,
        vocab_size: int = 49408,
        max_length: int = 77,
        seed: int = 0,
        pooled_dim: int | None = None,
    ):
        super().__init__()
        if heads is None:
            heads = hidden // 64 if hidden % 64 == 0 else 8
        # sdxl: the addition-embedding consumes a 1280-dim pooled text vector
        self.pooled_proj = (
            nn.Linear(hidden, pooled_dim, bias=False) if pooled_dim else None
        )
        torch.manual_seed(seed)
        self.tokenizer = HashTokenizer(vocab_size, max_length)
        self.hidden = hidden
        self.token_emb = nn.Embedding(vocab_size, hidden)
        self.pos_emb = nn.Parameter(torch.randn(max_length, hidden) * 0.01)
        layer = nn.TransformerEncoderLayer(
            d_model=hidden,
            nhead=heads,
            dim_feedforward=hidden * 4,
            activation="gelu",
            batch_first=True,
            norm_first=True,
        )
        self.encoder = nn.TransformerEncoder(layer, num_layers=layers)
        self.final_ln = nn.LayerNorm(hidden)
        mask = torch.full((max_length, max_length), float("-inf"))
        self.register_buffer("causal_mask", torch.triu(mask, diagonal=1), persistent=False)

    @torch.no_grad()
    def encode(self, prompt: str, device=None, dtype=torch.float32) -> torch.Tensor:
        ids = self.tokenizer(prompt)
        if device is not None:
            ids = ids.to(device)
        x = self.token_emb(ids) + self.pos_emb[None]
        x = self.encoder(x, mask=self.causal_mask)
        x = self.final_ln(x)
        return x.to(dtype)

    @torch.no_grad()
    def pooled(self, prompt: str, device=None, dtype=torch.float32) -> torch.Tensor:
        """EOS-token pooled embedding (sdxl addition-embed path);
        projected to pooled_dim (1280 for sdxl) when configured."""
        emb = self.encode(prompt, device=device, dtype=torch.float32)
        p = emb[:, -1]
        if self.pooled_proj is not None:
            p = self.pooled_proj(p.to(self.pooled_proj.weight.dtype))
        return p.to(dtype)
