"""Per-track analysis runtime: decode -> DSP -> encoders -> features.

Reference: /root/reference/tasks/analysis/song.py (analyze_track :478)
+ clap_analyzer.analyze_audio_file (:432) + the staged per-track
pipeline in tasks/analysis/album.py:290. The ONNX sessions are replaced
by our torch/HIP models; on a GPU the whole album batches through the
encoders at once instead of the reference's
chunked per-track loops.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Optional

import numpy as np
import torch

from audiomuse_amd import config as C
from audiomuse_amd.models.htsat import (HTSATConfig, HTSATEncoder,
                                        clap_track_embedding)
from audiomuse_amd.models.musicnn import (MusiCNNEmbedding, MusiCNNPrediction,
                                          aggregate_track)
from audiomuse_amd.models.text import TextEmbedder, clap_text_config
from audiomuse_amd.ops import dsp, features, hip_ops
from audiomuse_amd.ops.audio_io import load_audio
from audiomuse_amd.utils.resources import oom_retry


@dataclass
class TrackAnalysis:
    tempo: float = 0.0
    energy: float = 0.0
    key: str = "C"
    scale: str = "major"
    duration: float = 0.0
    embedding: Optional[np.ndarray] = None          # 200-d MusiCNN
    moods: Dict[str, float] = field(default_factory=dict)
    clap_embedding: Optional[np.ndarray] = None     # 512-d
    other_features: Dict[str, float] = field(default_factory=dict)
    chromaprint: Optional[bytes] = None             # compressed fingerprint


class AnalysisRuntime:
    """Holds the models; one instance per worker process/GPU rank."""

    def __init__(self, device: str = "cpu", seed: int = 0,
                 enable_clap: Optional[bool] = None):
        self.device = torch.device(device)
        # GPU_DTYPE (PARAMETERS.md): bf16 default; f16 for experiments
        self.dtype = ((torch.float16 if C.GPU_DTYPE == "f16"
                       else torch.bfloat16)
                      if self.device.type == "cuda" else torch.float32)
        self.enable_clap = C.CLAP_ENABLED if enable_clap is None else enable_clap
        torch.manual_seed(seed)
        self.musicnn_emb = MusiCNNEmbedding().to(self.device).eval()
        self.musicnn_pred = MusiCNNPrediction().to(self.device).eval()
        self.htsat = (HTSATEncoder(
                          HTSATConfig(out_dim=C.CLAP_EMBEDDING_DIMENSION))
                      .to(self.device, self.dtype).eval()
                      if self.enable_clap else None)
        self._clap_text: Optional[TextEmbedder] = None
        self._other_label_embs: Optional[np.ndarray] = None
        self._lyrics = None
        self._seed = seed

    def recycle_models(self) -> None:
        """PER_Song_MODEL_RELOAD compatibility switch (reference
        PER_SONG_MODEL_RELOAD: small-VRAM deployments drop and reload
        models between songs; with 288 GB HBM the default keeps them
        resident). Called by the analysis task after each album batch
        when the flag is on."""
        if not C.PER_SONG_MODEL_RELOAD:
            return
        import gc
        torch.manual_seed(self._seed)
        self.musicnn_emb = MusiCNNEmbedding().to(self.device).eval()
        self.musicnn_pred = MusiCNNPrediction().to(self.device).eval()
        if self.enable_clap:
            self.htsat = HTSATEncoder(
                HTSATConfig(out_dim=C.CLAP_EMBEDDING_DIMENSION)
            ).to(self.device, self.dtype).eval()
        self._lyrics = None
        gc.collect()
        if self.device.type == "cuda":
            torch.cuda.empty_cache()

    def lyrics_pipeline(self):
        """Lazy lyrics pipeline (GTE embedder + VAD; ASR off by default —
        random-init Whisper transcripts never pass the quality gate, so
        enabling it only burns compute until trained weights exist)."""
        if self._lyrics is None:
            from audiomuse_amd.engines.lyrics import LyricsPipeline
            from audiomuse_amd.models.text import TextEmbedder, gte_config
            from audiomuse_amd.models.vad import SileroStyleVAD

            embedder = TextEmbedder(gte_config(), device=str(self.device),
                                    dtype=self.dtype)
            vad = SileroStyleVAD().to(self.device).eval()
            asr_fn = self._make_asr_fn() if C.LYRICS_ASR_ENABLED else None
            self._lyrics = LyricsPipeline(embedder, vad=vad, asr_fn=asr_fn)
        return self._lyrics

    def _make_asr_fn(self):
        """Whisper greedy/beam decode over 30 s chunks
        (whisper_onnx.py:173-178 chunking; random-init weights produce
        gate-rejected transcripts until trained weights are loaded)."""
        from audiomuse_amd.models.whisper import (WhisperModel, beam_decode,
                                                  greedy_decode)
        from audiomuse_amd.ops.dsp import whisper_mel_config

        model = WhisperModel().to(self.device, self.dtype).eval()
        mel_cfg = whisper_mel_config()

        def asr(audio16k: torch.Tensor):
            chunk = C.WHISPER_CHUNK_SECONDS * C.WHISPER_SAMPLE_RATE
            tokens = []
            logprobs = []
            for s0 in range(0, audio16k.shape[-1], chunk):
                seg = audio16k[s0 : s0 + chunk].to(self.device)
                mel = hip_ops.mel_spectrogram(seg, mel_cfg,
                                              force_reference=True)
                mel = mel.to(self.dtype)
                if C.LYRICS_ASR_BEAM_SIZE > 1:
                    toks = beam_decode(model, mel,
                                       beam=C.LYRICS_ASR_BEAM_SIZE)
                else:
                    toks = greedy_decode(model, mel, return_logprob=True)
                if isinstance(toks, tuple):
                    toks, lp = toks
                    logprobs.append(lp)
                tokens.extend(toks)
            # token ids -> placeholder wordpieces (no trained vocab in-image)
            text = " ".join(f"tok{t}" for t in tokens)
            avg_lp = (sum(logprobs) / len(logprobs)) if logprobs else None
            # confidence rides along for the reference's
            # LYRICS_ASR_MIN_AVG_LOGPROB gate (engines/lyrics.py)
            return (text, avg_lp) if avg_lp is not None else text

        return asr

    # -- label text embeddings (clap_analyzer.py:580: cached .npz) -------

    def other_feature_label_embeddings(self) -> np.ndarray:
        if self._other_label_embs is None:
            if self._clap_text is None:
                self._clap_text = TextEmbedder(clap_text_config(),
                                               device=str(self.device),
                                               dtype=self.dtype)
            texts = [f"this song is {lbl}" for lbl in C.OTHER_FEATURE_LABELS]
            self._other_label_embs = self._clap_text.embed(texts).cpu().numpy()
        return self._other_label_embs

    # -- single-track analysis ------------------------------------------

    @torch.inference_mode()
    def analyze_track(self, audio_source, sr: Optional[int] = None
                      ) -> Optional[TrackAnalysis]:
        """audio_source: wav bytes/path, or float tensor (with sr)."""
        if isinstance(audio_source, torch.Tensor):
            audio, in_sr = audio_source, sr or 44100
        else:
            audio, in_sr = load_audio(audio_source)
            if audio is None:
                return None
        out = TrackAnalysis(duration=audio.shape[-1] / in_sr)

        from audiomuse_amd.ops.audio_io import resample

        audio = audio.to(self.device)          # resample runs on-device
        a16 = resample(audio, in_sr, C.MUSICNN_SAMPLE_RATE)
        out.tempo, out.energy, out.key, out.scale = \
            features.extract_basic_features(a16, C.MUSICNN_SAMPLE_RATE)
        if C.CHROMAPRINT_COLLECTION_ENABLED:
            # acoustic fingerprint for the identity confirm gate
            # (reference: _stage_collect_chromaprint, album.py:120)
            from audiomuse_amd.engines import chromaprint as cp
            try:
                # stays on-device: only the ~(frames,12) chroma matrix
                # crosses back to the host
                out.chromaprint = cp.compute(a16, C.MUSICNN_SAMPLE_RATE)
            except Exception:  # noqa: BLE001 — fingerprint is best-effort
                out.chromaprint = None
        if C.LYRICS_MUSICNN_SKIP and C.LYRICS_ENABLED:
            # lyrics-only pass (reference LYRICS_MUSICNN_SKIP): keep the
            # cheap scalar features, skip the MusiCNN/CLAP embeddings
            return out

        # MusiCNN patches (song.py:240-256): 187-frame log-mel windows
        mel = hip_ops.mel_spectrogram(a16, dsp.musicnn_mel_config())
        frames = mel.shape[-1]
        P = C.MUSICNN_PATCH_FRAMES
        if frames >= P:
            patches = torch.stack([mel[:, i : i + P]
                                   for i in range(0, frames - P + 1, P)])
            patches = patches.transpose(1, 2)          # (N, 187, 96)
            # chunked inference (reference MUSICNN_BATCH_SIZE,
            # song.py:374-390 — bounded activations on small GPUs; the
            # 288 GB default makes the chunks large no-ops)
            bs = max(int(C.MUSICNN_BATCH_SIZE), 1)
            embs = [self.musicnn_emb(patches[i : i + bs].float())
                    for i in range(0, patches.shape[0], bs)]
            emb = torch.cat(embs, dim=0)
            logits = self.musicnn_pred(emb)
            track_emb, moods = aggregate_track(emb, logits)
            out.embedding = track_emb.cpu().numpy().astype(np.float32)
            out.moods = moods

        if self.htsat is not None:
            a48 = resample(audio, in_sr, C.CLAP_SAMPLE_RATE)
            out.clap_embedding = self._clap_embed(a48)
            out.other_features = self._score_other_features(out.clap_embedding)
        return out

    def _clap_embed(self, a48: torch.Tensor) -> np.ndarray:
        """Segments -> fused mel -> HTSAT -> mean + L2
        (clap_analyzer.py:432-511; int16 round-trip fused in kernel)."""
        seg_len = int(C.CLAP_SEGMENT_SECONDS * C.CLAP_SAMPLE_RATE)
        segs = dsp.segment_audio(a48, seg_len, C.CLAP_SEGMENT_HOP_SAMPLES)
        mel = hip_ops.mel_spectrogram(segs, dsp.clap_mel_config(),
                                      quantize_int16=True)
        emb = self.htsat(mel.to(self.dtype)).float()
        return clap_track_embedding(emb).cpu().numpy().astype(np.float32)

    def _score_other_features(self, clap_emb: np.ndarray) -> Dict[str, float]:
        """cosine vs the 6 cached label embeddings (clap_analyzer.py:641)."""
        labels = self.other_feature_label_embeddings()
        v = clap_emb / (np.linalg.norm(clap_emb) + 1e-9)
        sims = labels @ v
        return {lbl: float(s) for lbl, s in zip(C.OTHER_FEATURE_LABELS, sims)}

    # -- batched album analysis (GPU path) --------------------------------

    @torch.inference_mode()
    def analyze_album_batch(self, wav_blobs: List[bytes]
                            ) -> List[Optional[TrackAnalysis]]:
        """All tracks of an album; CLAP segments of every track batch
        into one encoder pass."""
        results: List[Optional[TrackAnalysis]] = []
        seg_batches: List[torch.Tensor] = []
        seg_owner: List[int] = []
        patch_batches: List[torch.Tensor] = []
        patch_owner: List[int] = []
        feat_audio: List[torch.Tensor] = []
        feat_owner: List[int] = []
        from audiomuse_amd.ops.audio_io import resample

        for i, blob in enumerate(wav_blobs):
            audio, in_sr = load_audio(blob)
            if audio is None:
                results.append(None)
                continue
            audio = audio.to(self.device)      # resample + DSP on-device
            res = TrackAnalysis(duration=audio.shape[-1] / in_sr)
            a16 = resample(audio, in_sr, C.MUSICNN_SAMPLE_RATE)
            # tempo/energy/key batch across the album after this loop
            # (per-track DSP was kernel-launch bound)
            feat_audio.append(a16)
            feat_owner.append(i)
            if C.CHROMAPRINT_COLLECTION_ENABLED:
                from audiomuse_amd.engines import chromaprint as cp
                try:
                    res.chromaprint = cp.compute(a16,
                                                 C.MUSICNN_SAMPLE_RATE)
                except Exception:  # noqa: BLE001 — best-effort
                    res.chromaprint = None
            mel = hip_ops.mel_spectrogram(a16, dsp.musicnn_mel_config())
            P = C.MUSICNN_PATCH_FRAMES
            if mel.shape[-1] >= P:
                patches = torch.stack(
                    [mel[:, j : j + P]
                     for j in range(0, mel.shape[-1] - P + 1, P)]
                ).transpose(1, 2)              # (n, 187, 96)
                patch_batches.append(patches)
                patch_owner.extend([i] * patches.shape[0])
            results.append(res)
            if self.htsat is not None:
                a48 = resample(audio, in_sr, C.CLAP_SAMPLE_RATE)
                segs = dsp.segment_audio(
                    a48, int(C.CLAP_SEGMENT_SECONDS * C.CLAP_SAMPLE_RATE),
                                         C.CLAP_SEGMENT_HOP_SAMPLES)
                seg_batches.append(segs)
                seg_owner.extend([i] * segs.shape[0])

        # one batched DSP pass for tempo/energy/key (exact per-track math)
        if feat_audio:
            feats = features.extract_basic_features_batch(
                feat_audio, C.MUSICNN_SAMPLE_RATE)
            for owner, (tempo, energy, key, scale) in zip(feat_owner, feats):
                res = results[owner]
                res.tempo, res.energy, res.key, res.scale = (
                    tempo, energy, key, scale)

        # one MusiCNN pass for the whole album
        if patch_batches:
            all_patches = torch.cat(patch_batches, dim=0).float()
            emb = oom_retry(self.musicnn_emb, all_patches)
            logits = oom_retry(self.musicnn_pred, emb)
            owner = torch.tensor(patch_owner)
            for i, res in enumerate(results):
                if res is None:
                    continue
                mine = owner == i
                if not bool(mine.any()):
                    continue
                track_emb, moods = aggregate_track(emb[mine], logits[mine])
                res.embedding = track_emb.cpu().numpy().astype(np.float32)
                res.moods = moods
        if self.htsat is not None and seg_batches:
            all_segs = torch.cat(seg_batches, dim=0)
            mel = hip_ops.mel_spectrogram(all_segs, dsp.clap_mel_config(),
                                          quantize_int16=True)
            embs = oom_retry(lambda m: self.htsat(m.to(self.dtype)).float(),
                             mel)
            owner = torch.tensor(seg_owner)
            for i, res in enumerate(results):
                if res is None:
                    continue
                mine = embs[owner == i]
                if mine.shape[0] == 0:
                    continue
                res.clap_embedding = clap_track_embedding(mine).cpu().numpy()
                res.other_features = self._score_other_features(res.clap_embedding)
        return results

    @torch.inference_mode()
    def analyze_track_base(self, audio: torch.Tensor, in_sr: int
                           ) -> TrackAnalysis:
        """Features + MusiCNN only (no CLAP) — used by the batched path."""
        from audiomuse_amd.ops.audio_io import resample

        out = TrackAnalysis(duration=audio.shape[-1] / in_sr)
        a16 = resample(audio, in_sr, C.MUSICNN_SAMPLE_RATE).to(self.device)
        out.tempo, out.energy, out.key, out.scale = \
            features.extract_basic_features(a16, C.MUSICNN_SAMPLE_RATE)
        mel = hip_ops.mel_spectrogram(a16, dsp.musicnn_mel_config())
        P = C.MUSICNN_PATCH_FRAMES
        if mel.shape[-1] >= P:
            patches = torch.stack([mel[:, i : i + P]
                                   for i in range(0, mel.shape[-1] - P + 1, P)])
            patches = patches.transpose(1, 2)
            emb = self.musicnn_emb(patches.float())
            logits = self.musicnn_pred(emb)
            track_emb, moods = aggregate_track(emb, logits)
            out.embedding = track_emb.cpu().numpy().astype(np.float32)
            out.moods = moods
        return out
