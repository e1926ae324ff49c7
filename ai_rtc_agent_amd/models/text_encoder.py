"""Prompt -> embedding: CLIP text encoder + BPE tokenizer.

The reference runs CLIP ViT-L/14 in eager torch even on the TRT path
(reference lib/wrapper.py:468-473) because prompt encoding is cold — it
runs only at prepare()/update_prompt (reference lib/pipeline.py:44-45).
We keep the same design: eager PyTorch-ROCm (hipBLASLt under torch) is the
MI355X-native choice for this cold path.

Two layers of fidelity, both first-party:
- ClipBpeTokenizer: the real CLIP byte-pair-encoding algorithm
  (byte-to-unicode table, end-of-word `</w>` merges, lowercase + clean),
  loading vocab.json/merges.txt from a checkpoint's tokenizer/ directory —
  exactly the files a diffusers snapshot (lykon/dreamshaper-8) ships.
- TextEncoder: the CLIPTextModel architecture (pre-LN transformer with
  separate q/k/v/out projections, quick-gelu MLPs, causal mask, final
  LayerNorm, EOS-pooled output, optional penultimate-layer "clip skip"
  for SD2.x) whose parameters load 1:1 from CLIPTextModel checkpoints
  (models/load.py: load_clip_text_encoder).

Offline, where no checkpoint exists, both degrade deterministically: a
hash tokenizer stands in for the vocab and the encoder stays random-init.
The engine contract — encode(prompt) -> (1, 77, ctx), stable and
prompt-sensitive — holds in every configuration.
"""
from __future__ import annotations

import gzip
import hashlib
import html
import json
import os
import re
from functools import lru_cache
from typing import Dict, List, Optional, Tuple

import torch
import torch.nn as nn
import torch.nn.functional as F


# ---------------------------------------------------------------------------
# CLIP BPE tokenizer (openai/CLIP simple_tokenizer algorithm)
# ---------------------------------------------------------------------------
@lru_cache()
def bytes_to_unicode() -> Dict[int, str]:
    """GPT-2/CLIP reversible byte <-> unicode mapping (printable chars)."""
    bs = (list(range(ord("!"), ord("~") + 1))
          + list(range(ord("\xa1"), ord("\xac") + 1))
          + list(range(ord("\xae"), ord("\xff") + 1)))
    cs = bs[:]
    n = 0
    for b in range(256):
        if b not in bs:
            bs.append(b)
            cs.append(256 + n)
            n += 1
    return dict(zip(bs, [chr(c) for c in cs]))


def _get_pairs(word: Tuple[str, ...]):
    return {(word[i], word[i + 1]) for i in range(len(word) - 1)}


_WORD_PAT = re.compile(
    r"<\|startoftext\|>|<\|endoftext\|>|'s|'t|'re|'ve|'m|'ll|'d"
    r"|[^\W\d_]+|\d|[^\s\w]+",
    re.IGNORECASE,
)


class ClipBpeTokenizer:
    """CLIP BPE over a vocab.json + merges.txt pair (HF tokenizer layout)."""

    def __init__(self, vocab: Dict[str, int], merges: List[Tuple[str, str]],
                 max_length: int = 77):
        self.encoder = vocab
        self.bpe_ranks = {m: i for i, m in enumerate(merges)}
        self.byte_encoder = bytes_to_unicode()
        self.max_length = max_length
        self.bos = vocab.get("<|startoftext|>", len(vocab) - 2)
        self.eos = vocab.get("<|endoftext|>", len(vocab) - 1)
        self.vocab_size = max(len(vocab), self.eos + 1)
        self._cache: Dict[str, List[str]] = {}

    @staticmethod
    def from_dir(path: str, max_length: int = 77) -> "ClipBpeTokenizer":
        """Load from a checkpoint dir: <path>/tokenizer/{vocab.json,
        merges.txt} or the files directly under <path>."""
        for base in (os.path.join(path, "tokenizer"), path):
            vj = os.path.join(base, "vocab.json")
            mt = os.path.join(base, "merges.txt")
            if os.path.exists(vj) and os.path.exists(mt):
                with open(vj, encoding="utf-8") as f:
                    vocab = json.load(f)
                opener = gzip.open if mt.endswith(".gz") else open
                with opener(mt, "rt", encoding="utf-8") as f:
                    lines = f.read().split("\n")
                merges = []
                for ln in lines:
                    ln = ln.strip()
                    if not ln or ln.startswith("#"):
                        continue
                    parts = ln.split()
                    if len(parts) == 2:
                        merges.append((parts[0], parts[1]))
                return ClipBpeTokenizer(vocab, merges, max_length)
        raise FileNotFoundError(f"no tokenizer files under {path}")

    def _bpe(self, token: str) -> List[str]:
        if token in self._cache:
            return self._cache[token]
        word = tuple(token[:-1]) + (token[-1] + "</w>",)
        pairs = _get_pairs(word)
        if not pairs:
            return [token + "</w>"]
        while True:
            bigram = min(pairs, key=lambda p: self.bpe_ranks.get(p, 1 << 30))
            if bigram not in self.bpe_ranks:
                break
            first, second = bigram
            new_word: List[str] = []
            i = 0
            while i < len(word):
                try:
                    j = word.index(first, i)
                except ValueError:
                    new_word.extend(word[i:])
                    break
                new_word.extend(word[i:j])
                i = j
                if i < len(word) - 1 and word[i] == first and word[i + 1] == second:
                    new_word.append(first + second)
                    i += 2
                else:
                    new_word.append(word[i])
                    i += 1
            word = tuple(new_word)
            if len(word) == 1:
                break
            pairs = _get_pairs(word)
        out = list(word)
        self._cache[token] = out
        return out

    def __call__(self, text: str) -> torch.Tensor:
        text = html.unescape(html.unescape(text)).strip().lower()
        text = re.sub(r"\s+", " ", text)
        ids = [self.bos]
        for tok in _WORD_PAT.findall(text):
            tok = "".join(self.byte_encoder[b] for b in tok.encode("utf-8"))
            for piece in self._bpe(tok):
                ids.append(self.encoder.get(piece, self.eos))
        # CLIP/SD padding: truncate, close with EOS, pad with EOS
        ids = ids[: self.max_length - 1] + [self.eos]
        ids += [self.eos] * (self.max_length - len(ids))
        return torch.tensor(ids, dtype=torch.long).unsqueeze(0)


class HashTokenizer:
    """Deterministic stand-in tokenizer: word -> stable bucket id
    (offline environments without vocab files)."""

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


# ---------------------------------------------------------------------------
# CLIPTextModel architecture (loads 1:1 from diffusers checkpoints)
# ---------------------------------------------------------------------------
class _ClipBlock(nn.Module):
    def __init__(self, hidden: int, heads: int, act: str):
        super().__init__()
        self.heads = heads
        self.ln1 = nn.LayerNorm(hidden)
        self.q_proj = nn.Linear(hidden, hidden)
        self.k_proj = nn.Linear(hidden, hidden)
        self.v_proj = nn.Linear(hidden, hidden)
        self.out_proj = nn.Linear(hidden, hidden)
        self.ln2 = nn.LayerNorm(hidden)
        self.fc1 = nn.Linear(hidden, hidden * 4)
        self.fc2 = nn.Linear(hidden * 4, hidden)
        self.act = act

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        B, L, C = x.shape
        h = self.ln1(x)
        q = self.q_proj(h).view(B, L, self.heads, -1).transpose(1, 2)
        k = self.k_proj(h).view(B, L, self.heads, -1).transpose(1, 2)
        v = self.v_proj(h).view(B, L, self.heads, -1).transpose(1, 2)
        a = F.scaled_dot_product_attention(q, k, v, is_causal=True)
        a = a.transpose(1, 2).reshape(B, L, C)
        x = x + self.out_proj(a)
        h = self.fc1(self.ln2(x))
        h = h * torch.sigmoid(1.702 * h) if self.act == "quick_gelu" else F.gelu(h)
        return x + self.fc2(h)


class TextEncoder(nn.Module):
    """CLIP text encoder with the CLIPTextModel module tree.

    clip_skip=1 returns the penultimate layer's hidden states (still
    final-LayerNormed) — the SD2.x convention; 0 is the SD1.5 default.
    """

    def __init__(
        self,
        hidden: int = 768,
        layers: int = 12,
        heads: Optional[int] = None,
        vocab_size: int = 49408,
        max_length: int = 77,
        seed: int = 0,
        pooled_dim: Optional[int] = None,
        act: str = "quick_gelu",
        clip_skip: int = 0,
    ):
        super().__init__()
        if heads is None:
            heads = hidden // 64 if hidden % 64 == 0 else 8
        torch.manual_seed(seed)
        self.hidden = hidden
        self.max_length = max_length
        self.clip_skip = clip_skip
        self.tokenizer = HashTokenizer(vocab_size, max_length)
        self.token_emb = nn.Embedding(vocab_size, hidden)
        self.pos_emb = nn.Embedding(max_length, hidden)
        nn.init.normal_(self.token_emb.weight, std=0.02)
        nn.init.normal_(self.pos_emb.weight, std=0.01)
        self.blocks = nn.ModuleList(
            [_ClipBlock(hidden, heads, act) for _ in range(layers)]
        )
        self.final_ln = nn.LayerNorm(hidden)
        # sdxl: the addition-embedding consumes a pooled text vector
        # (CLIP text_projection)
        self.pooled_proj = (
            nn.Linear(hidden, pooled_dim, bias=False) if pooled_dim else None
        )

    def load_tokenizer_dir(self, path: str) -> bool:
        """Swap in the real BPE vocab from a checkpoint directory."""
        try:
            self.tokenizer = ClipBpeTokenizer.from_dir(path, self.max_length)
            return True
        except (FileNotFoundError, json.JSONDecodeError):
            return False

    def _hidden_states(self, ids: torch.Tensor) -> List[torch.Tensor]:
        pos = torch.arange(ids.shape[1], device=ids.device)
        x = self.token_emb(ids) + self.pos_emb(pos)[None]
        states = []
        for blk in self.blocks:
            x = blk(x)
            states.append(x)
        return states

    @torch.no_grad()
    def encode(self, prompt: str, device=None, dtype=torch.float32) -> torch.Tensor:
        ids = self.tokenizer(prompt)
        if device is not None:
            ids = ids.to(device)
        states = self._hidden_states(ids)
        x = states[-1 - self.clip_skip]
        return self.final_ln(x).to(dtype)

    @torch.no_grad()
    def pooled(self, prompt: str, device=None, dtype=torch.float32) -> torch.Tensor:
        """EOS-token pooled embedding (CLIP convention: the position of the
        highest token id, i.e. the first EOS); projected through
        text_projection when configured (sdxl addition-embed path)."""
        ids = self.tokenizer(prompt)
        if device is not None:
            ids = ids.to(device)
        x = self.final_ln(self._hidden_states(ids)[-1])
        eos_pos = ids.argmax(dim=-1)
        p = x[torch.arange(x.shape[0], device=x.device), eos_pos]
        if self.pooled_proj is not None:
            p = self.pooled_proj(p.to(self.pooled_proj.weight.dtype))
        return p.to(dtype)


class DualTextEncoder(nn.Module):
    """SDXL text conditioning: CLIP ViT-L/14 (768-d, quick-gelu) and
    OpenCLIP ViT-bigG (1280-d, gelu, text_projection for the pooled
    vector), both read at the PENULTIMATE layer per the sdxl convention;
    per-token features concatenate to the UNet's 2048-d context
    (reference: diffusers loads text_encoder/ + text_encoder_2/ for
    stabilityai/sdxl-turbo; the wrapper consumes the same two encoders).

    Exposes the same encode()/pooled() surface as TextEncoder, so the
    engine treats both interchangeably."""

    def __init__(
        self,
        hidden1: int = 768,
        layers1: int = 12,
        hidden2: int = 1280,
        layers2: int = 32,
        vocab_size: int = 49408,
        max_length: int = 77,
        seed: int = 0,
        pooled_dim: int = 1280,
    ):
        super().__init__()
        self.enc1 = TextEncoder(hidden=hidden1, layers=layers1, act="quick_gelu",
                                clip_skip=1, vocab_size=vocab_size,
                                max_length=max_length, seed=seed)
        self.enc2 = TextEncoder(hidden=hidden2, layers=layers2, act="gelu",
                                clip_skip=1, vocab_size=vocab_size,
                                max_length=max_length, seed=seed + 1,
                                pooled_dim=pooled_dim)
        self.hidden = hidden1 + hidden2

    def load_tokenizer_dir(self, path: str) -> bool:
        """sdxl snapshots ship tokenizer/ (ViT-L) and tokenizer_2 (bigG)."""
        ok1 = self.enc1.load_tokenizer_dir(os.path.join(path, "tokenizer")) \
            or self.enc1.load_tokenizer_dir(path)
        ok2 = self.enc2.load_tokenizer_dir(os.path.join(path, "tokenizer_2")) \
            or self.enc2.load_tokenizer_dir(path)
        return ok1 or ok2

    @torch.no_grad()
    def encode(self, prompt: str, device=None, dtype=torch.float32) -> torch.Tensor:
        e1 = self.enc1.encode(prompt, device=device, dtype=dtype)
        e2 = self.enc2.encode(prompt, device=device, dtype=dtype)
        return torch.cat([e1, e2], dim=-1)

    @torch.no_grad()
    def pooled(self, prompt: str, device=None, dtype=torch.float32) -> torch.Tensor:
        return self.enc2.pooled(prompt, device=device, dtype=dtype)
