"""Lyrics pipeline: transcription -> embedding -> axis scores.

Reference: /root/reference/lyrics/lyrics_transcriber.py (analyze_lyrics
:1137 — 9 stages): provided/server lyrics win; otherwise VAD gates the
audio, Whisper transcribes, language/quality gates filter junk, GTE
embeds the text, and 27 thematic axes get temperature-softmax scores
(T=0.1) of label-embedding similarity (_score_axes :744). Instrumental
tracks take a sentinel embedding + axis fill.
"""

from __future__ import annotations

import re
from dataclasses import dataclass, field
from typing import Dict, Optional, Sequence

import numpy as np
import torch

from audiomuse_amd import config as C
from audiomuse_amd.models.text import TextEmbedder
from audiomuse_amd.models.vad import (SileroStyleVAD, speech_probabilities,
                                      speech_ratio)

INSTRUMENTAL_AXIS_FILL = 0.0
_WORD_RE = re.compile(r"[a-zA-Z']+")

# tiny stopword tables for a first-party language gate (reference uses
# langdetect; only the gating behavior matters for capability parity)
_LANG_STOPWORDS = {
    "en": {"the", "and", "you", "for", "that", "with", "this", "have"},
    "es": {"que", "los", "las", "por", "con", "para", "una", "del"},
    "fr": {"les", "des", "que", "pour", "dans", "avec", "une", "est"},
    "de": {"und", "der", "die", "das", "nicht", "mit", "ein", "ich"},
    "it": {"che", "per", "con", "del", "della", "una", "sono", "non"},
    "pt": {"que", "com", "para", "uma", "mais", "por", "das", "dos"},
}


def detect_language(text: str) -> str:
    words = set(w.lower() for w in _WORD_RE.findall(text or ""))
    best, best_n = "unknown", 0
    for lang, sw in _LANG_STOPWORDS.items():
        n = len(words & sw)
        if n > best_n:
            best, best_n = lang, n
    return best if best_n >= 2 else "unknown"


def quality_gate(text: str, min_words: int = 8,
                 max_repeat_ratio: float = 0.6) -> bool:
    """Reject junk transcripts: too short, or dominated by one token
    (reference quality gates)."""
    words = [w.lower() for w in _WORD_RE.findall(text or "")]
    if len(words) < min_words:
        return False
    top = max(words.count(w) for w in set(words))
    return top / len(words) <= max_repeat_ratio


def score_axes(embedding: np.ndarray, axis_label_embeddings: Dict[str, np.ndarray],
               temperature: Optional[float] = None) -> Dict[str, float]:
    """Per-axis temperature softmax over label-embedding similarities
    (_score_axes, lyrics_transcriber.py:744; the per-axis positive-label
    probability is the axis score)."""
    temperature = temperature or C.LYRICS_AXIS_TEMPERATURE
    emb = np.asarray(embedding, dtype=np.float32)
    out: Dict[str, float] = {}
    for axis, matrix in axis_label_embeddings.items():
        m = np.asarray(matrix, dtype=np.float32)
        if m.size == 0:
            out[axis] = 0.0
            continue
        sims = m @ emb
        z = sims / max(temperature, 1e-6)
        z -= z.max()
        p = np.exp(z)
        p /= p.sum()
        out[axis] = float(p[0])     # first row = the axis's positive label
    return out


@dataclass
class LyricsResult:
    text: str = ""
    language: str = ""
    instrumental: bool = False
    embedding: Optional[np.ndarray] = None
    axis_scores: Dict[str, float] = field(default_factory=dict)
    source: str = "none"            # provided | asr | instrumental


class LyricsPipeline:
    def __init__(self, embedder: TextEmbedder,
                 vad: Optional[SileroStyleVAD] = None,
                 asr_fn=None, axis_labels: Optional[Sequence[str]] = None,
                 vad_speech_threshold: float = 0.15):
        """asr_fn: audio(16k tensor) -> transcript string (Whisper decode
        wired in by the analysis task; None disables ASR)."""
        self.embedder = embedder
        self.vad = vad
        self.asr_fn = asr_fn
        self.vad_speech_threshold = vad_speech_threshold
        labels = list(axis_labels or C.LYRICS_AXES)
        self.axis_labels = labels
        # positive + contrast prompts per axis; row 0 is the positive label
        self._axis_emb: Dict[str, np.ndarray] = {}
        texts = []
        for a in labels:
            texts.append(f"a song about {a}")
            texts.append(f"a song not about {a}")
        embs = self.embedder.embed(texts).cpu().numpy()
        for i, a in enumerate(labels):
            self._axis_emb[a] = embs[2 * i : 2 * i + 2]

    def _instrumental(self) -> LyricsResult:
        dim = C.LYRICS_EMBEDDING_DIMENSION
        return LyricsResult(
            instrumental=True, source="instrumental",
            embedding=np.zeros(dim, dtype=np.float32),
            axis_scores={a: INSTRUMENTAL_AXIS_FILL for a in self.axis_labels})

    def analyze(self, audio: Optional[torch.Tensor] = None,
                provided_lyrics: Optional[str] = None) -> LyricsResult:
        """analyze_lyrics (:1137): provided text wins; else VAD -> ASR ->
        gates -> embed + axes."""
        text: Optional[str] = None
        source = "none"
        if provided_lyrics and quality_gate(provided_lyrics, min_words=4):
            text, source = provided_lyrics, "provided"
        elif audio is not None and self.asr_fn is not None:
            if audio.shape[-1] > C.LYRICS_MAX_AUDIO_SECONDS * 16000:
                audio = audio[..., : C.LYRICS_MAX_AUDIO_SECONDS * 16000]
            if self.vad is not None:
                vdev = next(self.vad.parameters()).device
                probs = speech_probabilities(self.vad,
                                             audio.to(vdev, torch.float32))
                if speech_ratio(probs) < self.vad_speech_threshold:
                    return self._instrumental()
            transcript = self.asr_fn(audio) or ""
            words = _WORD_RE.findall(transcript)
            transcript = " ".join(words[:300])   # reference 300-word cap
            if quality_gate(transcript):
                text, source = transcript, "asr"
        if not text:
            return self._instrumental()
        emb_t = self.embedder.embed([text])[0]
        emb = emb_t.cpu().numpy().astype(np.float32)
        return LyricsResult(
            text=text, language=detect_language(text), instrumental=False,
            embedding=emb, axis_scores=score_axes(emb, self._axis_emb),
            source=source)
