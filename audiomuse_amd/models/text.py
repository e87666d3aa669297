"""Text encoders: CLAP text tower + GTE-style embedder, plus a
deterministic first-party tokenizer.

Reference capability (/root/reference/tasks/clap_analyzer.py:527-560 —
LAION CLAP text ONNX, RoBERTa tokenizer from transformers; and
/root/reference/lyrics/gte_onnx.py:55-157 — GTE-multilingual INT8,
CLS-pool 768-d): text -> fixed-dim embedding in the shared audio/text
space (512-d, L2-normed) or the lyrics space (768-d).

These are our own bidirectional transformer encoders (pre-LN, GELU,
FusedLayerNorm on GPU, bf16-ready; head_dim 64). No pretrained vocab
files exist in-image, so the tokenizer is a deterministic hashed
word-piece scheme: stable ids for a fixed vocab budget, exact special
tokens — the published-model tokenizers drop in via the same interface
when weights are available.
"""

from __future__ import annotations

import hashlib
import re
from dataclasses import dataclass
from typing import List, Optional, Sequence, Tuple

import torch
import torch.nn as nn
import torch.nn.functional as F

from audiomuse_amd.ops.norms import FusedLayerNorm

PAD, CLS, SEP, UNK = 0, 1, 2, 3
_SPECIAL = 4
_WORD_RE = re.compile(r"[a-z0-9']+|[^\sa-z0-9']", re.IGNORECASE)


class HashTokenizer:
    """Deterministic hashed tokenizer: word -> stable id in [4, vocab)."""

    def __init__(self, vocab_size: int = 50000, max_len: int = 128):
        self.vocab_size = vocab_size
        self.max_len = max_len

    def _id(self, token: str) -> int:
        h = hashlib.blake2s(token.lower().encode(), digest_size=4).digest()
        return _SPECIAL + int.from_bytes(h, "big") % (self.vocab_size - _SPECIAL)

    def encode(self, text: str, max_len: Optional[int] = None) -> List[int]:
        max_len = max_len or self.max_len
        toks = _WORD_RE.findall(text or "")
        ids = [CLS] + [self._id(t) for t in toks[: max_len - 2]] + [SEP]
        return ids

    def batch(self, texts: Sequence[str], max_len: Optional[int] = None
              ) -> Tuple[torch.Tensor, torch.Tensor]:
        rows = [self.encode(t, max_len) for t in texts]
        L = max(len(r) for r in rows)
        ids = torch.full((len(rows), L), PAD, dtype=torch.long)
        mask = torch.zeros(len(rows), L, dtype=torch.bool)
        for i, r in enumerate(rows):
            ids[i, : len(r)] = torch.tensor(r)
            mask[i, : len(r)] = True
        return ids, mask


class EncoderLayer(nn.Module):
    def __init__(self, dim: int, heads: int, mlp_ratio: float = 4.0):
        super().__init__()
        self.heads = heads
        self.norm1 = FusedLayerNorm(dim)
        self.qkv = nn.Linear(dim, 3 * dim)
        self.proj = nn.Linear(dim, dim)
        self.norm2 = FusedLayerNorm(dim)
        hidden = int(dim * mlp_ratio)
        self.mlp = nn.Sequential(nn.Linear(dim, hidden), nn.GELU(),
                                 nn.Linear(hidden, dim))

    def forward(self, x: torch.Tensor, mask: torch.Tensor) -> torch.Tensor:
        B, L, D = x.shape
        h = self.heads
        xn = self.norm1(x)
        qkv = self.qkv(xn).view(B, L, 3, h, D // h).permute(2, 0, 3, 1, 4)
        q, k, v = qkv.unbind(0)
        attn_mask = torch.where(mask, 0.0, float("-inf"))[:, None, None, :]
        # explicit math attention: torch's SDPA backend selection picked
        # the AOTriton path (unsupported on gfx950) intermittently under
        # threaded workers; this encoder is small and never the hot path
        scale = (D // h) ** -0.5
        s = (q @ k.transpose(-2, -1)) * scale + attn_mask.to(q.dtype)
        out = torch.softmax(s, dim=-1) @ v
        out = out.transpose(1, 2).reshape(B, L, D)
        x = x + self.proj(out)
        return x + self.mlp(self.norm2(x))


@dataclass
class TextEncoderConfig:
    vocab_size: int = 50000
    dim: int = 768
    layers: int = 12
    heads: int = 12
    max_len: int = 128
    out_dim: int = 512          # projection dim; 0 = raw CLS
    pool: str = "cls"           # cls | mean
    tokenizer_json: str = ""    # path to a published tokenizer.json
                                # (RoBERTa/GTE interop; "" = hashed)


class TextEncoder(nn.Module):
    def __init__(self, cfg: TextEncoderConfig):
        super().__init__()
        self.cfg = cfg
        self.tok = nn.Embedding(cfg.vocab_size, cfg.dim, padding_idx=PAD)
        self.pos = nn.Embedding(cfg.max_len, cfg.dim)
        self.layers = nn.ModuleList(
            EncoderLayer(cfg.dim, cfg.heads) for _ in range(cfg.layers))
        self.norm = FusedLayerNorm(cfg.dim)
        self.head = (nn.Linear(cfg.dim, cfg.out_dim)
                     if cfg.out_dim else nn.Identity())

    def forward(self, ids: torch.Tensor, mask: torch.Tensor) -> torch.Tensor:
        L = ids.shape[1]
        pos = torch.arange(L, device=ids.device).clamp(max=self.cfg.max_len - 1)
        x = self.tok(ids) + self.pos(pos)[None]
        for layer in self.layers:
            x = layer(x, mask)
        x = self.norm(x)
        if self.cfg.pool == "mean":
            m = mask.unsqueeze(-1).to(x.dtype)
            pooled = (x * m).sum(dim=1) / m.sum(dim=1).clamp(min=1.0)
        else:
            pooled = x[:, 0]
        return self.head(pooled)


def clap_text_config() -> TextEncoderConfig:
    """LAION-CLAP-class text tower: 768/12L -> 512-d projection."""
    return TextEncoderConfig(dim=768, layers=12, heads=12, out_dim=512,
                             pool="cls", max_len=77)


def gte_config() -> TextEncoderConfig:
    """GTE-class embedder: 768/12L, CLS-pooled 768-d (gte_onnx.py:127)."""
    from audiomuse_amd import config as C
    return TextEncoderConfig(dim=768, layers=12, heads=12, out_dim=0,
                             pool="cls", max_len=C.LYRICS_GTE_MAX_TOKENS)


class TrainedTokenizer:
    """Adapter for a published BPE/WordPiece vocabulary via the
    `tokenizers` wheel (the reference loads RoBERTa/GTE vocab files,
    clap_analyzer.py:534 / gte_onnx.py:55). Drop a HuggingFace
    tokenizer.json next to the checkpoint and pass its path through
    TextEncoderConfig.tokenizer_json (or AUDIOMUSE_TOKENIZER_JSON) —
    the hashed stand-in is only the no-vocab-files fallback."""

    def __init__(self, path: str, vocab_size: int, max_len: int = 128):
        from tokenizers import Tokenizer
        self.tk = Tokenizer.from_file(path)
        if self.tk.get_vocab_size() > vocab_size:
            raise ValueError(
                f"tokenizer vocab {self.tk.get_vocab_size()} exceeds the "
                f"model's embedding table ({vocab_size}); re-init the "
                "model with vocab_size >= the tokenizer's")
        self.max_len = max_len

    def encode(self, text: str, max_len: Optional[int] = None) -> List[int]:
        max_len = max_len or self.max_len
        ids = self.tk.encode(text or "").ids[: max_len - 2]
        return [CLS] + ids + [SEP]

    def batch(self, texts: Sequence[str], max_len: Optional[int] = None
              ) -> Tuple[torch.Tensor, torch.Tensor]:
        rows = [self.encode(t, max_len) for t in texts]
        L = max(len(r) for r in rows)
        ids = torch.full((len(rows), L), PAD, dtype=torch.long)
        mask = torch.zeros(len(rows), L, dtype=torch.bool)
        for i, r in enumerate(rows):
            ids[i, : len(r)] = torch.tensor(r)
            mask[i, : len(r)] = True
        return ids, mask


class TextEmbedder:
    """Model + tokenizer wrapper: texts -> L2-normed embeddings
    (reference: get_text_embeddings_batch, clap_analyzer.py:534)."""

    def __init__(self, cfg: TextEncoderConfig, device: str = "cpu",
                 dtype: torch.dtype = torch.float32, seed: int = 0):
        from audiomuse_amd import config as C
        torch.manual_seed(seed)
        self.model = TextEncoder(cfg).to(device=device, dtype=dtype).eval()
        tok_path = (getattr(cfg, "tokenizer_json", "")
                    or getattr(C, "TOKENIZER_JSON", ""))
        if tok_path:
            self.tokenizer = TrainedTokenizer(tok_path, cfg.vocab_size,
                                              cfg.max_len)
        else:
            self.tokenizer = HashTokenizer(cfg.vocab_size, cfg.max_len)
        self.device = device

    @torch.inference_mode()
    def embed(self, texts: Sequence[str]) -> torch.Tensor:
        ids, mask = self.tokenizer.batch(list(texts))
        ids, mask = ids.to(self.device), mask.to(self.device)
        emb = self.model(ids, mask).float()
        return emb / emb.norm(dim=1, keepdim=True).clamp(min=1e-9)
