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


_CJK_RANGES = ((0x3040, 0x30FF),    # hiragana/katakana
               (0x4E00, 0x9FFF),    # CJK unified
               (0xAC00, 0xD7AF),    # hangul
               (0x3400, 0x4DBF))


def _cjk_ratio(text: str) -> float:
    chars = [c for c in text if not c.isspace()]
    if not chars:
        return 0.0
    n = sum(1 for c in chars
            if any(lo <= ord(c) <= hi for lo, hi in _CJK_RANGES))
    return n / len(chars)


def detect_language(text: str, with_confidence: bool = False):
    """Script-aware language gate (reference: langdetect +
    LYRICS_CJK_SCRIPT_MIN_RATIO / LYRICS_LANG_CONFIDENCE_MIN gates,
    lyrics_transcriber.py language stage). CJK scripts identify by
    codepoint ratio; Latin languages by stopword evidence with a
    confidence = hit fraction among known stopwords."""
    ratio = _cjk_ratio(text or "")
    if ratio >= C.LYRICS_CJK_SCRIPT_MIN_RATIO:
        han = sum(1 for c in text if 0x4E00 <= ord(c) <= 0x9FFF)
        kana = sum(1 for c in text if 0x3040 <= ord(c) <= 0x30FF)
        hangul = sum(1 for c in text if 0xAC00 <= ord(c) <= 0xD7AF)
        lang = ("ja" if kana > 0 else "ko" if hangul > han else "zh")
        return (lang, 1.0) if with_confidence else lang
    words = [w.lower() for w in _WORD_RE.findall(text or "")]
    uniq = set(words)
    scores = {lang: len(uniq & sw) for lang, sw in _LANG_STOPWORDS.items()}
    ranked = sorted(scores.items(), key=lambda kv: -kv[1])
    best, hits = ranked[0] if ranked else ("unknown", 0)
    runner_up = ranked[1][1] if len(ranked) > 1 else 0
    # confidence = margin over the runner-up language (shared stopwords
    # like es/pt "que" depress it exactly when the call is ambiguous)
    conf = hits / max(hits + runner_up, 1) if hits >= 2 else 0.0
    lang = best if conf >= C.LYRICS_LANG_CONFIDENCE_MIN and hits >= 2 \
        else "unknown"
    return (lang, conf) if with_confidence else lang


def compression_ratio(text: str) -> float:
    """zlib length ratio — ASR loops compress extremely well (reference
    LYRICS_TEXT_MAX_COMPRESSION_RATIO gate)."""
    import zlib
    data = (text or "").encode("utf-8")
    if not data:
        return 0.0
    return len(data) / max(len(zlib.compress(data, 6)), 1)


def quality_gate(text: str, min_words: int = 8,
                 max_repeat_ratio: float = 0.6) -> bool:
    """Reject junk transcripts: too short, dominated by one token, or
    degenerate-repetitive (compression gate; reference quality gates +
    LYRICS_TEXT_MAX_COMPRESSION_RATIO)."""
    words = [w.lower() for w in _WORD_RE.findall(text or "")]
    if len(words) < min_words:
        return False
    top = max(words.count(w) for w in set(words))
    if top / len(words) > max_repeat_ratio:
        return False
    return compression_ratio(text) <= C.LYRICS_TEXT_MAX_COMPRESSION_RATIO


def fetch_external_lyrics(title: str, artist: str,
                          http_get=None) -> Optional[str]:
    """External lyrics APIs, tried in order before ASR (reference:
    lyrics_transcriber stages 1-2 + LYRICS_API_{1,2}_* config).
    URL templates may carry {artist}/{title} placeholders; otherwise the
    configured param names ride the query string. SSRF-guarded."""
    if not C.LYRICS_API_ENABLE:
        return None
    from audiomuse_amd.utils.logging_utils import validate_outbound_url
    if http_get is None:
        import requests
        http_get = requests.get
    for n in (1, 2):
        tmpl = getattr(C, f"LYRICS_API_{n}_URL_TEMPLATE", "")
        if not tmpl:
            continue
        field_path = getattr(C, f"LYRICS_API_{n}_LYRICS_FIELD")
        params = {}
        url = tmpl
        if "{artist}" in tmpl or "{title}" in tmpl:
            from urllib.parse import quote
            url = tmpl.replace("{artist}", quote(artist)).replace(
                "{title}", quote(title))
        else:
            params[getattr(C, f"LYRICS_API_{n}_ARTIST_PARAM")] = artist
            params[getattr(C, f"LYRICS_API_{n}_TITLE_PARAM")] = title
        key_param = getattr(C, f"LYRICS_API_{n}_APIKEY_PARAM")
        if key_param:
            params[key_param] = getattr(C, f"LYRICS_API_{n}_APIKEY_VALUE")
        try:
            validate_outbound_url(url)
            r = http_get(url, params=params,
                         timeout=getattr(C, f"LYRICS_API_{n}_TIMEOUT"))
            if r.status_code != 200:
                continue
            body = r.json()
            value = body
            for part in field_path.split("."):
                if not isinstance(value, dict):
                    value = None
                    break
                value = value.get(part)
            if isinstance(value, str) and value.strip():
                return value.strip()
        except Exception:
            continue
    return None


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
                provided_lyrics: Optional[str] = None,
                title: str = "", artist: str = "",
                http_get=None) -> LyricsResult:
        """analyze_lyrics (:1137): provided/server text wins; else the
        external lyrics APIs (stages 1-2); else VAD -> ASR -> gates ->
        embed + axes."""
        text: Optional[str] = None
        source = "none"
        if provided_lyrics and quality_gate(provided_lyrics, min_words=4):
            text, source = provided_lyrics, "provided"
        if text is None and (title or artist):
            api_text = fetch_external_lyrics(title, artist,
                                             http_get=http_get)
            if api_text and quality_gate(api_text, min_words=4):
                text, source = api_text, "api"
        if text is None and audio is not None and self.asr_fn is not None:
            if audio.shape[-1] > C.LYRICS_MAX_AUDIO_SECONDS * 16000:
                audio = audio[..., : C.LYRICS_MAX_AUDIO_SECONDS * 16000]
            if self.vad is not None and C.VAD_VOICE_RECOGNITION:
                vdev = next(self.vad.parameters()).device
                probs = speech_probabilities(self.vad,
                                             audio.to(vdev, torch.float32))
                if speech_ratio(probs) < self.vad_speech_threshold:
                    return self._instrumental()
            asr_out = self.asr_fn(audio) or ""
            # ASR confidence gate (reference LYRICS_ASR_MIN_AVG_LOGPROB /
            # NON_ENGLISH_MIN_LOGPROB): asr_fn may return
            # (text, avg_logprob); low-confidence decodes are junk
            avg_logprob = None
            if isinstance(asr_out, tuple):
                transcript, avg_logprob = asr_out[0] or "", asr_out[1]
            else:
                transcript = asr_out
            if avg_logprob is not None:
                lang_guess = detect_language(transcript)
                floor = (C.LYRICS_ASR_MIN_AVG_LOGPROB if lang_guess == "en"
                         else C.LYRICS_ASR_NON_ENGLISH_MIN_LOGPROB)
                if avg_logprob < floor:
                    return self._instrumental()
            words = _WORD_RE.findall(transcript)
            transcript = " ".join(words[:C.LYRICS_MAX_WORDS])
            # ASR junk guard: embed only transcripts of real length
            # (reference LYRICS_MIN_CHARS_FOR_EMBEDDING; provided/API
            # lyrics are trusted sources and skip this gate)
            if quality_gate(transcript) and \
                    len(transcript) >= C.LYRICS_MIN_CHARS_FOR_EMBEDDING:
                text, source = transcript, "asr"
        if not text:
            return self._instrumental()
        emb_t = self.embedder.embed([text])[0]
        emb = emb_t.cpu().numpy().astype(np.float32)
        return LyricsResult(
            text=text, language=detect_language(text), instrumental=False,
            embedding=emb, axis_scores=score_axes(emb, self._axis_emb),
            source=source)
