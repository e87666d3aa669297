"""Whisper-style speech-to-text encoder/decoder with KV-cache decode.

Reference capability (/root/reference/lyrics/whisper_onnx.py:217-738):
30 s log-mel chunks -> encoder hidden states; autoregressive decoder
with merged KV cache, greedy or beam search (LYRICS_ASR_BEAM_SIZE),
repetition penalty, no-repeat-ngram blocking and language-token
detection. The reference drives opaque ONNX graphs; this is a
first-party PyTorch-ROCm implementation of the same decode loop with
the model re-designed for bf16 MI355X inference (pre-LN transformer,
head_dim 64).

The decode loop is the latency-critical part (SURVEY.md §7 hard part
#1); the KV cache is preallocated per beam so decode steps are
fixed-shape (hipGraph-capturable later).
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import List, Optional, Tuple

import torch
import torch.nn as nn
import torch.nn.functional as F

from audiomuse_amd import config as C
from audiomuse_amd.ops.norms import FusedLayerNorm

# special tokens (our own vocabulary layout)
TOK_PAD = 0
TOK_SOT = 1          # start of transcript
TOK_EOT = 2          # end of transcript
TOK_LANG_BASE = 10   # language tokens occupy [10, 10+n_langs)
LANGS = ["en", "es", "fr", "de", "it", "pt", "nl", "ja", "zh", "ko", "ru",
         "other"]
TOK_TEXT_BASE = 10 + len(LANGS)


@dataclass
class WhisperConfig:
    n_mels: int = 80
    n_frames: int = 3000          # 30 s at hop 160 / 16 kHz
    dim: int = 768
    enc_layers: int = 12
    dec_layers: int = 12
    heads: int = 12
    vocab_size: int = 51200
    max_tokens: int = 224


class _SelfAttn(nn.Module):
    def __init__(self, dim: int, heads: int, causal: bool):
        super().__init__()
        self.heads = heads
        self.causal = causal
        self.qkv = nn.Linear(dim, 3 * dim)
        self.proj = nn.Linear(dim, dim)

    def forward(self, x: torch.Tensor,
                kv_cache: Optional[Tuple[torch.Tensor, torch.Tensor]] = None,
                cache_len: int = 0):
        B, L, D = x.shape
        h = self.heads
        q, k, v = self.qkv(x).view(B, L, 3, h, D // h).permute(2, 0, 3, 1, 4).unbind(0)
        if kv_cache is not None:
            ck, cv = kv_cache                  # (B, h, T_max, d)
            ck[:, :, cache_len : cache_len + L] = k
            cv[:, :, cache_len : cache_len + L] = v
            k = ck[:, :, : cache_len + L]
            v = cv[:, :, : cache_len + L]
            if L == 1:
                out = F.scaled_dot_product_attention(q, k, v)
            else:
                # causal over the new tokens, offset by the cache length
                i = torch.arange(L, device=x.device).unsqueeze(1)
                j = torch.arange(cache_len + L, device=x.device).unsqueeze(0)
                amask = torch.where(j <= cache_len + i, 0.0, float("-inf"))
                out = F.scaled_dot_product_attention(q, k, v,
                                                     attn_mask=amask.to(q.dtype))
        else:
            out = F.scaled_dot_product_attention(q, k, v, is_causal=self.causal)
        return self.proj(out.transpose(1, 2).reshape(B, L, D))

    def forward_static(self, x: torch.Tensor, kv_cache, len_t: torch.Tensor,
                       arange_T: torch.Tensor):
        """Fixed-shape single-token step for hipGraph capture: the write
        position and attention extent come from the device tensor `len_t`
        (read at replay time), so every replay has identical shapes and
        kernel arguments."""
        B, L, D = x.shape                      # L == 1
        h = self.heads
        q, k, v = self.qkv(x).view(B, L, 3, h, D // h).permute(2, 0, 3, 1, 4).unbind(0)
        ck, cv = kv_cache                      # (B, h, T_max, d)
        idx = len_t.view(1)
        ck.index_copy_(2, idx, k)
        cv.index_copy_(2, idx, v)
        # attend over the FULL preallocated cache; positions > len masked
        amask = torch.where(arange_T.view(1, 1, 1, -1) <= len_t, 0.0,
                            float("-inf")).to(q.dtype)
        out = F.scaled_dot_product_attention(q, ck, cv, attn_mask=amask)
        return self.proj(out.transpose(1, 2).reshape(B, L, D))


class _CrossAttn(nn.Module):
    def __init__(self, dim: int, heads: int):
        super().__init__()
        self.heads = heads
        self.q = nn.Linear(dim, dim)
        self.kv = nn.Linear(dim, 2 * dim)
        self.proj = nn.Linear(dim, dim)

    def precompute(self, enc: torch.Tensor):
        B, L, D = enc.shape
        h = self.heads
        k, v = self.kv(enc).view(B, L, 2, h, D // h).permute(2, 0, 3, 1, 4).unbind(0)
        return k, v

    def forward(self, x: torch.Tensor, kv: Tuple[torch.Tensor, torch.Tensor]):
        B, L, D = x.shape
        h = self.heads
        q = self.q(x).view(B, L, h, D // h).transpose(1, 2)
        out = F.scaled_dot_product_attention(q, kv[0], kv[1])
        return self.proj(out.transpose(1, 2).reshape(B, L, D))


class _MLP(nn.Sequential):
    def __init__(self, dim: int):
        super().__init__(nn.Linear(dim, 4 * dim), nn.GELU(),
                         nn.Linear(4 * dim, dim))


class EncoderBlock(nn.Module):
    def __init__(self, dim: int, heads: int):
        super().__init__()
        self.norm1 = FusedLayerNorm(dim)
        self.attn = _SelfAttn(dim, heads, causal=False)
        self.norm2 = FusedLayerNorm(dim)
        self.mlp = _MLP(dim)

    def forward(self, x):
        x = x + self.attn(self.norm1(x))
        return x + self.mlp(self.norm2(x))


class DecoderBlock(nn.Module):
    def __init__(self, dim: int, heads: int):
        super().__init__()
        self.norm1 = FusedLayerNorm(dim)
        self.self_attn = _SelfAttn(dim, heads, causal=True)
        self.norm2 = FusedLayerNorm(dim)
        self.cross = _CrossAttn(dim, heads)
        self.norm3 = FusedLayerNorm(dim)
        self.mlp = _MLP(dim)

    def forward(self, x, self_cache, cache_len, cross_kv):
        x = x + self.self_attn(self.norm1(x), self_cache, cache_len)
        x = x + self.cross(self.norm2(x), cross_kv)
        return x + self.mlp(self.norm3(x))

    def forward_static(self, x, self_cache, len_t, arange_T, cross_kv):
        x = x + self.self_attn.forward_static(self.norm1(x), self_cache,
                                              len_t, arange_T)
        x = x + self.cross(self.norm2(x), cross_kv)
        return x + self.mlp(self.norm3(x))


class WhisperModel(nn.Module):
    def __init__(self, cfg: WhisperConfig | None = None):
        super().__init__()
        self.cfg = cfg = cfg or WhisperConfig()
        self.conv1 = nn.Conv1d(cfg.n_mels, cfg.dim, 3, padding=1)
        self.conv2 = nn.Conv1d(cfg.dim, cfg.dim, 3, stride=2, padding=1)
        self.enc_pos = nn.Parameter(
            torch.randn(cfg.n_frames // 2, cfg.dim) * 0.01)
        self.enc_blocks = nn.ModuleList(
            EncoderBlock(cfg.dim, cfg.heads) for _ in range(cfg.enc_layers))
        self.enc_norm = FusedLayerNorm(cfg.dim)

        self.tok_emb = nn.Embedding(cfg.vocab_size, cfg.dim)
        self.dec_pos = nn.Parameter(torch.randn(cfg.max_tokens, cfg.dim) * 0.01)
        self.dec_blocks = nn.ModuleList(
            DecoderBlock(cfg.dim, cfg.heads) for _ in range(cfg.dec_layers))
        self.dec_norm = FusedLayerNorm(cfg.dim)

    # -- encoder -----------------------------------------------------------

    def encode(self, mel: torch.Tensor) -> torch.Tensor:
        """mel (B, n_mels, T<=n_frames) -> (B, T//2, dim)."""
        x = F.gelu(self.conv1(mel))
        x = F.gelu(self.conv2(x))
        x = x.transpose(1, 2)
        x = x + self.enc_pos[: x.shape[1]][None]
        for blk in self.enc_blocks:
            x = blk(x)
        return self.enc_norm(x)

    # -- decoder -----------------------------------------------------------

    def make_caches(self, B: int, device, dtype):
        cfg = self.cfg
        d = cfg.dim // cfg.heads
        return [
            (torch.zeros(B, cfg.heads, cfg.max_tokens, d, device=device, dtype=dtype),
             torch.zeros(B, cfg.heads, cfg.max_tokens, d, device=device, dtype=dtype))
            for _ in range(cfg.dec_layers)
        ]

    def decode_step(self, tokens: torch.Tensor, cache_len: int,
                    caches, cross_kvs) -> torch.Tensor:
        """tokens (B, L_new) -> logits (B, L_new, vocab); caches updated."""
        pos = torch.arange(cache_len, cache_len + tokens.shape[1],
                           device=tokens.device)
        x = self.tok_emb(tokens) + self.dec_pos[pos][None]
        for blk, cache, ckv in zip(self.dec_blocks, caches, cross_kvs):
            x = blk(x, cache, cache_len, ckv)
        x = self.dec_norm(x)
        return x @ self.tok_emb.weight.T

    def cross_kvs(self, enc: torch.Tensor):
        return [blk.cross.precompute(enc) for blk in self.dec_blocks]

    def decode_step_static(self, tok: torch.Tensor, len_t: torch.Tensor,
                           caches, cross_kvs, arange_T) -> torch.Tensor:
        """Fixed-shape (B, 1) decode step, hipGraph-capturable."""
        x = self.tok_emb(tok) + self.dec_pos.index_select(0, len_t.view(1))[None]
        for blk, cache, ckv in zip(self.dec_blocks, caches, cross_kvs):
            x = blk.forward_static(x, cache, len_t, arange_T, ckv)
        x = self.dec_norm(x)
        return x @ self.tok_emb.weight.T


class GraphedDecoder:
    """hipGraph-captured single-token decode loop (greedy, B=1).

    The eager decode_step launches ~150 tiny kernels per token (12 blocks
    x GEMMs/LNs/SDPA at M=1) — pure launch latency. This captures ONE
    fixed-shape step (decode_step_static) into a hipGraph; each token is
    then tok/len update + one graph replay. The reference's ONNX decoder
    re-runs a full session per token (whisper_onnx.py:332-528); this is
    the MI355X-native answer to SURVEY hard part #1.
    """

    def __init__(self, model: WhisperModel, enc: torch.Tensor):
        cfg = model.cfg
        device, dtype = enc.device, enc.dtype
        self.model = model
        self.enc_shape = tuple(enc.shape)
        self.caches = model.make_caches(1, device, dtype)
        self.ckv = model.cross_kvs(enc)
        self.tok = torch.zeros(1, 1, dtype=torch.long, device=device)
        self.len_t = torch.zeros((), dtype=torch.long, device=device)
        self.arange_T = torch.arange(cfg.max_tokens, device=device)
        # warm up the exact op sequence on a side stream, then capture
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):
                self.model.decode_step_static(self.tok, self.len_t,
                                              self.caches, self.ckv,
                                              self.arange_T)
        torch.cuda.current_stream().wait_stream(s)
        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            self.logits = self.model.decode_step_static(
                self.tok, self.len_t, self.caches, self.ckv, self.arange_T)
        # the warmup/capture wrote junk at position 0; the caller's first
        # eager prompt step overwrites it and the mask hides the rest
        for ck, cv in self.caches:
            ck.zero_()
            cv.zero_()

    def reset(self, enc: torch.Tensor) -> None:
        """Re-arm the captured graph for a NEW chunk: the fresh cross
        K/V are copied into the captured buffers and the self caches are
        zeroed. Capture cost (~2 s) is paid once per (model, enc shape);
        every later chunk is a copy + replays."""
        new = self.model.cross_kvs(enc)
        for (kb, vb), (k, v) in zip(self.ckv, new):
            kb.copy_(k)
            vb.copy_(v)
        for ck, cv in self.caches:
            ck.zero_()
            cv.zero_()

    def step(self, token: int, pos: int) -> torch.Tensor:
        """Returns logits (1, 1, vocab) for `token` written at `pos`."""
        self.tok.fill_(token)
        self.len_t.fill_(pos)
        self.graph.replay()
        return self.logits


def _graphed_decoder_for(model: WhisperModel,
                         enc: torch.Tensor) -> Optional[GraphedDecoder]:
    """Per-model cached GraphedDecoder, re-armed per chunk."""
    gdec = getattr(model, "_graphed_decoder", None)
    if gdec is not None and gdec.enc_shape == tuple(enc.shape):
        gdec.reset(enc)
        return gdec
    try:
        gdec = GraphedDecoder(model, enc)
    except Exception:  # noqa: BLE001 — capture failure falls back to eager
        return None
    model._graphed_decoder = gdec
    return gdec


def _block_repeats(logits: torch.Tensor, seq: List[int],
                   repetition_penalty: float, no_repeat_ngram: int) -> None:
    """In-place logit adjustments (whisper_onnx decode loop semantics)."""
    if repetition_penalty > 1.0 and seq:
        idx = torch.tensor(sorted(set(seq)), device=logits.device)
        vals = logits[idx]
        logits[idx] = torch.where(vals > 0, vals / repetition_penalty,
                                  vals * repetition_penalty)
    if no_repeat_ngram > 1 and len(seq) >= no_repeat_ngram - 1:
        prefix = tuple(seq[-(no_repeat_ngram - 1):])
        n = no_repeat_ngram
        for i in range(len(seq) - n + 1):
            if tuple(seq[i : i + n - 1]) == prefix:
                logits[seq[i + n - 1]] = float("-inf")


@torch.inference_mode()
def detect_language(model: WhisperModel, enc: torch.Tensor) -> int:
    """One decode step from SOT; argmax over language tokens
    (whisper_onnx.py:364)."""
    caches = model.make_caches(1, enc.device, enc.dtype)
    ckv = model.cross_kvs(enc)
    logits = model.decode_step(
        torch.tensor([[TOK_SOT]], device=enc.device), 0, caches, ckv)
    lang_logits = logits[0, -1, TOK_LANG_BASE : TOK_LANG_BASE + len(LANGS)]
    return int(lang_logits.argmax())


@torch.inference_mode()
def greedy_decode(model: WhisperModel, mel: torch.Tensor, *,
                  max_tokens: Optional[int] = None,
                  repetition_penalty: Optional[float] = None,
                  no_repeat_ngram: Optional[int] = None,
                  use_graph: Optional[bool] = None,
                  return_logprob: bool = False):
    """Greedy KV-cache decode of one chunk. mel (n_mels, T).

    On GPU the per-token step runs as one hipGraph replay
    (GraphedDecoder) unless use_graph=False; CPU always runs eager."""
    cfg = model.cfg
    if repetition_penalty is None:
        repetition_penalty = C.WHISPER_REPETITION_PENALTY
    if no_repeat_ngram is None:
        no_repeat_ngram = C.WHISPER_NO_REPEAT_NGRAM
    max_tokens = min(max_tokens or min(C.WHISPER_MAX_NEW_TOKENS,
                                       cfg.max_tokens - 4),
                     cfg.max_tokens - 4)
    enc = model.encode(mel.unsqueeze(0))
    if use_graph is None:
        use_graph = enc.is_cuda
    gdec: Optional[GraphedDecoder] = None
    if use_graph and enc.is_cuda:
        gdec = _graphed_decoder_for(model, enc)
    if gdec is not None:
        caches, ckv = gdec.caches, gdec.ckv
    else:
        caches = model.make_caches(1, enc.device, enc.dtype)
        ckv = model.cross_kvs(enc)
    lang = detect_language(model, enc)
    prompt = [TOK_SOT, TOK_LANG_BASE + lang]
    logits = model.decode_step(
        torch.tensor([prompt], device=enc.device), 0, caches, ckv)
    seq: List[int] = []
    logprob_sum = 0.0
    cache_len = len(prompt)
    step_logits = logits[0, -1].float()
    for _ in range(max_tokens):
        _block_repeats(step_logits, seq, repetition_penalty, no_repeat_ngram)
        nxt = int(step_logits.argmax())
        if nxt == TOK_EOT:
            break
        if return_logprob:
            # avg token logprob feeds the reference's ASR confidence
            # gates (LYRICS_ASR_MIN_AVG_LOGPROB)
            logprob_sum += float(torch.log_softmax(step_logits, dim=-1)[nxt])
        seq.append(nxt)
        if gdec is not None:
            out = gdec.step(nxt, cache_len)
        else:
            out = model.decode_step(
                torch.tensor([[nxt]], device=enc.device), cache_len, caches,
                ckv)
        cache_len += 1
        step_logits = out[0, -1].float()
    if return_logprob:
        return seq, (logprob_sum / max(len(seq), 1))
    return seq


@torch.inference_mode()
def beam_decode(model: WhisperModel, mel: torch.Tensor, beam: int = 2, *,
                max_tokens: Optional[int] = None,
                repetition_penalty: float = 1.2,
                no_repeat_ngram: int = 3) -> List[int]:
    """Beam-search decode with per-beam KV caches
    (whisper_onnx.py:332-528 beam path)."""
    if beam <= 1:
        return greedy_decode(model, mel, max_tokens=max_tokens,
                             repetition_penalty=repetition_penalty,
                             no_repeat_ngram=no_repeat_ngram)
    cfg = model.cfg
    if repetition_penalty is None:
        repetition_penalty = C.WHISPER_REPETITION_PENALTY
    if no_repeat_ngram is None:
        no_repeat_ngram = C.WHISPER_NO_REPEAT_NGRAM
    max_tokens = min(max_tokens or min(C.WHISPER_MAX_NEW_TOKENS,
                                       cfg.max_tokens - 4),
                     cfg.max_tokens - 4)
    enc = model.encode(mel.unsqueeze(0)).expand(beam, -1, -1).contiguous()
    caches = model.make_caches(beam, enc.device, enc.dtype)
    ckv = model.cross_kvs(enc)
    lang = detect_language(model, enc[:1])
    prompt = [TOK_SOT, TOK_LANG_BASE + lang]
    toks = torch.tensor([prompt] * beam, device=enc.device)
    logits = model.decode_step(toks, 0, caches, ckv)[:, -1].float()
    cache_len = len(prompt)
    seqs: List[List[int]] = [[] for _ in range(beam)]
    scores = torch.zeros(beam, device=enc.device)
    finished: List[Tuple[float, List[int]]] = []
    # first expansion: take top-beam from beam 0 only (identical states)
    logp = F.log_softmax(logits[0], dim=-1)
    top = torch.topk(logp, beam)
    for b in range(beam):
        seqs[b] = [int(top.indices[b])]
        scores[b] = top.values[b]
    nxt_tokens = torch.tensor([[s[-1]] for s in seqs], device=enc.device)
    for _ in range(max_tokens - 1):
        logits = model.decode_step(nxt_tokens, cache_len, caches, ckv)[:, -1].float()
        cache_len += 1
        all_cand = []
        for b in range(beam):
            lb = logits[b]
            _block_repeats(lb, seqs[b], repetition_penalty, no_repeat_ngram)
            lp = F.log_softmax(lb, dim=-1)
            t = torch.topk(lp, beam)
            for j in range(beam):
                all_cand.append((float(scores[b] + t.values[j]), b,
                                 int(t.indices[j])))
        all_cand.sort(key=lambda c: -c[0])
        new_seqs, new_scores, new_tokens, reorder = [], [], [], []
        for sc, b, tok in all_cand:
            if tok == TOK_EOT:
                finished.append((sc / max(len(seqs[b]), 1), seqs[b]))
                continue
            new_seqs.append(seqs[b] + [tok])
            new_scores.append(sc)
            new_tokens.append([tok])
            reorder.append(b)
            if len(new_seqs) == beam:
                break
        if not new_seqs:
            break
        ridx = torch.tensor(reorder, device=enc.device)
        for ck, cv in caches:
            ck.copy_(ck[ridx])
            cv.copy_(cv[ridx])
        seqs = new_seqs
        scores = torch.tensor(new_scores, device=enc.device)
        nxt_tokens = torch.tensor(new_tokens, device=enc.device)
    for b in range(len(seqs)):
        finished.append((float(scores[b]) / max(len(seqs[b]), 1), seqs[b]))
    finished.sort(key=lambda c: -c[0])
    return finished[0][1] if finished else []
