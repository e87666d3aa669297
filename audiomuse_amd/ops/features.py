"""Basic track features: tempo / energy / key+scale.

Reference behavior: /root/reference/tasks/analysis/song.py:150-231
(librosa beat_track + rms + chroma_cqt + Krumhansl-Schmuckler key
correlation). librosa is not in this image; these are first-party torch
implementations of the same features, GPU-capable for batched analysis:

- energy: frame RMS -> dB (ref=1) -> clip (-60, 0) -> [0, 1] -> mean.
  Matches librosa.feature.rms semantics (frame 2048, hop 512, centered).
- tempo: onset-strength envelope (positive log-mel spectral flux) ->
  autocorrelation tempogram -> log-normal prior around 120 BPM ->
  octave-fold into [TEMPO_MIN_BPM, TEMPO_MAX_BPM] (song.py:186-199).
- key/scale: chroma from a log-frequency (pitch-class) projection of the
  STFT magnitude (the reference uses chroma_cqt; this projection is the
  STFT-domain equivalent), then the exact reference correlation: z-scored
  dot products against 12 rotations of the Krumhansl major/minor profiles.
"""

from __future__ import annotations

import math
from typing import Tuple

import numpy as np
import torch

from audiomuse_amd import config as C
from audiomuse_amd.ops.dsp import mel_filterbank, power_spectrogram

_KEYS = ["C", "C#", "D", "D#", "E", "F", "F#", "G", "G#", "A", "A#", "B"]

_KS_MAJOR = torch.tensor([6.35, 2.23, 3.48, 2.33, 4.38, 4.09,
                          2.52, 5.19, 2.39, 3.66, 2.29, 2.88])
_KS_MINOR = torch.tensor([6.33, 2.68, 3.52, 5.38, 2.60, 3.53,
                          2.54, 4.75, 3.98, 2.69, 3.34, 3.17])

TEMPO_MIN_BPM = 40.0
TEMPO_MAX_BPM = 200.0

# constant per-(sr, fft, device) tensors, built once: rebuilding the
# mel filterbank / chroma projection per call was a host-side matrix
# build + H2D upload for EVERY track of the album loop
_FB_CACHE: dict = {}
_CHROMA_W_CACHE: dict = {}


def _onset_fb(sr: int, n_fft: int, n_mels: int,
              device: torch.device) -> torch.Tensor:
    key = (sr, n_fft, n_mels, str(device))
    fb = _FB_CACHE.get(key)
    if fb is None:
        fb = torch.from_numpy(
            mel_filterbank(sr, n_fft, n_mels, 0.0, sr / 2.0)).to(device)
        _FB_CACHE[key] = fb
    return fb


def _chroma_weights(sr: int, n_fft: int,
                    device: torch.device) -> torch.Tensor:
    key = (sr, n_fft, str(device))
    wt = _CHROMA_W_CACHE.get(key)
    if wt is None:
        freqs = np.linspace(0, sr / 2.0, 1 + n_fft // 2)
        with np.errstate(divide="ignore"):
            midi = 69.0 + 12.0 * np.log2(np.maximum(freqs, 1e-9) / 440.0)
        pc = np.mod(midi, 12.0)
        w = np.zeros((12, len(freqs)), dtype=np.float32)
        usable = (freqs >= 30.0) & (freqs <= 5000.0)
        for c in range(12):
            d = np.minimum(np.abs(pc - c), 12.0 - np.abs(pc - c))
            w[c] = np.exp(-0.5 * (d / 1.0) ** 2) * usable
        wt = torch.from_numpy(w).to(device)
        _CHROMA_W_CACHE[key] = wt
    return wt


def estimate_energy(audio: torch.Tensor, frame: int = 2048,
                    hop: int = 512) -> float:
    """song.py:202-210: mean of clip((rms_db + 60)/60, 0, 1)."""
    if audio.numel() == 0:
        return 0.0
    audio = audio.float().flatten()
    pad = frame // 2
    x = torch.nn.functional.pad(audio.unsqueeze(0), (pad, pad)).squeeze(0)
    n_frames = 1 + (x.shape[0] - frame) // hop
    if n_frames < 1:
        return 0.0
    frames = x.unfold(0, frame, hop)[:n_frames]
    rms = frames.square().mean(dim=1).sqrt()
    rms_db = 20.0 * torch.log10(torch.clamp(rms, min=1e-9))
    energy = torch.clamp((rms_db + 60.0) / 60.0, 0.0, 1.0)
    return float(energy.mean())


def onset_envelope(audio: torch.Tensor, sr: int, n_fft: int = 2048,
                   hop: int = 512, n_mels: int = 128) -> torch.Tensor:
    """Spectral-flux onset strength (librosa.onset.onset_strength analog)."""
    power = power_spectrogram(audio.float().flatten(), n_fft, hop, center=True)
    fb = _onset_fb(sr, n_fft, n_mels, audio.device)
    mel = fb @ power
    log_mel = torch.log10(torch.clamp(mel, min=1e-10))
    flux = torch.clamp(log_mel[:, 1:] - log_mel[:, :-1], min=0.0)
    env = flux.mean(dim=0)
    return env - env.mean()


def estimate_tempo(audio: torch.Tensor, sr: int, hop: int = 512) -> float:
    """Autocorrelation tempogram + log-normal 120 BPM prior, folded into
    [40, 200] BPM (song.py:186-199)."""
    if audio.numel() <= 1024:   # shorter than the reflect pad: no tempo
        return 0.0
    env = onset_envelope(audio, sr, hop=hop)
    n = env.shape[0]
    if n < 4:
        return 0.0
    # full autocorrelation via FFT
    f = torch.fft.rfft(env, n=2 * n)
    ac = torch.fft.irfft(f * f.conj(), n=2 * n)[:n]
    ac = ac / (ac[0] + 1e-12)
    fps = sr / hop
    lags = torch.arange(1, n, device=audio.device, dtype=torch.float32)
    bpm = 60.0 * fps / lags
    # log-normal prior around 120 BPM, one-octave std (librosa default)
    prior = torch.exp(-0.5 * ((torch.log2(bpm) - math.log2(120.0)) / 1.0) ** 2)
    valid = (bpm >= 20.0) & (bpm <= 400.0)
    score = ac[1:] * prior * valid
    # one D2H transfer for (best score, best bpm) instead of three
    best = score.argmax()
    pair = torch.stack([score[best], bpm[best]]).cpu()
    if float(pair[0]) <= 0:
        return 0.0
    tempo = float(pair[1])
    if tempo <= 0:
        return 0.0
    while tempo < TEMPO_MIN_BPM:
        tempo *= 2.0
    while tempo > TEMPO_MAX_BPM:
        tempo /= 2.0
    return tempo


def chroma_from_stft(audio: torch.Tensor, sr: int, n_fft: int = 4096,
                     hop: int = 1024) -> torch.Tensor:
    """(12, frames) chroma via pitch-class projection of |STFT|."""
    power = power_spectrogram(audio.float().flatten(), n_fft, hop, center=True)
    mag = power.sqrt()
    # gaussian weighting of each bin onto its nearest pitch classes
    # (constant per (sr, n_fft, device) — cached)
    wt = _chroma_weights(sr, n_fft, audio.device)
    chroma = wt @ mag
    return chroma


def _key_from_chroma_mean(cm: torch.Tensor) -> Tuple[str, str]:
    """Krumhansl-Schmuckler correlation from a mean chroma vector
    (song.py:213-231, exact math). cm: (12,) on any device."""
    if float(cm.sum()) <= 0:
        return "C", "major"
    c = (cm / (cm.norm() + 1e-9)).cpu()
    maj = torch.stack([torch.dot(c, torch.roll(_KS_MAJOR, i)) for i in range(12)])
    mnr = torch.stack([torch.dot(c, torch.roll(_KS_MINOR, i)) for i in range(12)])
    maj = (maj - maj.mean()) / (maj.std(unbiased=True) + 1e-9)
    mnr = (mnr - mnr.mean()) / (mnr.std(unbiased=True) + 1e-9)
    mi, ni = int(maj.argmax()), int(mnr.argmax())
    if float(maj[mi]) > float(mnr[ni]):
        return _KEYS[mi], "major"
    return _KEYS[ni], "minor"


def estimate_key_scale(audio: torch.Tensor, sr: int) -> Tuple[str, str]:
    """Krumhansl-Schmuckler correlation (song.py:213-231, exact math)."""
    if audio.numel() <= 2048:   # shorter than the reflect pad: no key
        return "C", "major"
    chroma = chroma_from_stft(audio, sr)
    return _key_from_chroma_mean(chroma.mean(dim=1))


def extract_basic_features(audio: torch.Tensor, sr: int):
    """song.py:233-238: (tempo, energy, key, scale)."""
    tempo = estimate_tempo(audio, sr)
    energy = estimate_energy(audio)
    key, scale = estimate_key_scale(audio, sr)
    return tempo, energy, key, scale


def _reflect_rows(audios, lens, pad: int, device) -> torch.Tensor:
    """Stack variable-length tracks into rows pre-padded with each
    track's OWN reflection (so batched center=False framing reproduces
    the per-track center=True frames exactly); zeros beyond."""
    L = max(lens)
    rows = torch.zeros(len(audios), L + 2 * pad, device=device)
    for i, a in enumerate(audios):
        rows[i, : lens[i] + 2 * pad] = torch.nn.functional.pad(
            a[None], (pad, pad), mode="reflect")[0]
    return rows


def extract_basic_features_batch(audios, sr: int):
    """Batched (tempo, energy, key, scale) for an album's tracks.

    Same math as the per-track functions — framing is reproduced
    exactly (reflect pre-pad per track; masked means; zero-padded
    linear autocorrelation is lag-wise identical) — but one
    STFT/FFT/matmul launch set per album instead of per track
    (the per-track loop was kernel-launch bound on GPU).
    """
    audios = [a.float().flatten() for a in audios]
    if not audios:
        return []
    lens = [a.numel() for a in audios]
    # tiny or single tracks: the per-track path handles edge cases
    if len(audios) == 1 or min(lens) <= 4096:
        return [extract_basic_features(a, sr) for a in audios]
    device = audios[0].device
    B = len(audios)
    L = max(lens)
    nT = torch.tensor(lens, device=device)[:, None]

    # -- energy (frame 2048, hop 512, zero pad — zeros beyond each
    # track ARE the per-track end padding) --
    frame, hop = 2048, 512
    p = frame // 2
    xb = torch.zeros(B, L, device=device)
    for i, a in enumerate(audios):
        xb[i, : lens[i]] = a
    frames = torch.nn.functional.pad(xb, (p, p)).unfold(1, frame, hop)
    rms = frames.square().mean(dim=2).sqrt()
    rms_db = 20.0 * torch.log10(torch.clamp(rms, min=1e-9))
    en = torch.clamp((rms_db + 60.0) / 60.0, 0.0, 1.0)
    n_en = 1 + (nT + 2 * p - frame) // hop
    eidx = torch.arange(en.shape[1], device=device)[None, :]
    emask = (eidx < n_en).float()
    energy = ((en * emask).sum(1) / emask.sum(1).clamp(min=1.0))

    # -- tempo (onset flux 2048/512/128-mel -> autocorr + prior) --
    pf = 2048 // 2
    rows = _reflect_rows(audios, lens, pf, device)
    pw = power_spectrogram(rows, 2048, 512, center=False)
    fb = _onset_fb(sr, 2048, 128, device)
    log_mel = torch.log10(torch.clamp(fb @ pw, min=1e-10))
    flux = torch.clamp(log_mel[:, :, 1:] - log_mel[:, :, :-1], min=0.0)
    env = flux.mean(dim=1)                       # (B, Tmax-1)
    n_env = (1 + torch.div(nT, 512, rounding_mode="floor")) - 1
    vmask = (torch.arange(env.shape[1], device=device)[None, :] < n_env)
    env = env * vmask
    env = (env - (env.sum(1, keepdim=True)
                  / n_env.clamp(min=1).float())) * vmask
    n2 = 2 * env.shape[1]
    f = torch.fft.rfft(env, n=n2)
    ac = torch.fft.irfft(f * f.conj(), n=n2)[:, : env.shape[1]]
    ac = ac / (ac[:, :1] + 1e-12)
    fps = sr / 512
    lags = torch.arange(1, env.shape[1], device=device, dtype=torch.float32)
    bpm = 60.0 * fps / lags
    prior = torch.exp(-0.5 * ((torch.log2(bpm) - math.log2(120.0)) / 1.0) ** 2)
    score = ac[:, 1:] * prior * ((bpm >= 20.0) & (bpm <= 400.0))
    best = score.argmax(dim=1)
    smax = score.gather(1, best[:, None]).squeeze(1)

    # -- key (chroma 4096/1024, masked frame mean) --
    pc = 4096 // 2
    rows2 = _reflect_rows(audios, lens, pc, device)
    mag = power_spectrogram(rows2, 4096, 1024, center=False).sqrt()
    wt = _chroma_weights(sr, 4096, device)
    ch = wt @ mag                                # (B, 12, T2)
    n_ch = 1 + torch.div(nT, 1024, rounding_mode="floor")
    cmask = (torch.arange(ch.shape[2], device=device)[None, :]
             < n_ch).float()[:, None, :]
    cm = (ch * cmask).sum(2) / cmask.sum(2).clamp(min=1.0)

    # one D2H for everything, then cheap per-track CPU tail
    packed = torch.cat([energy[:, None], smax[:, None], bpm[best][:, None],
                        n_env.float(), cm], dim=1).cpu()
    out = []
    for i in range(B):
        row = packed[i]
        e = float(row[0])
        s, t, nv = float(row[1]), float(row[2]), int(row[3])
        if nv < 4 or s <= 0 or t <= 0:
            tempo = 0.0
        else:
            tempo = t
            while tempo < TEMPO_MIN_BPM:
                tempo *= 2.0
            while tempo > TEMPO_MAX_BPM:
                tempo /= 2.0
        key, scale = _key_from_chroma_mean(row[4:16])
        out.append((tempo, e, key, scale))
    return out
