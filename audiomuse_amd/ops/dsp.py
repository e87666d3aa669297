"""Audio DSP front-end: librosa-equivalent mel spectrograms in PyTorch.

This module is the *numerics reference* (fp32, CPU or GPU via plain torch
ops) for the fused HIP kernels in ops/csrc/mel.hip, and the CPU fallback
path. Behavior matches the reference implementations:

- CLAP mel:    /root/reference/tasks/clap_analyzer.py:394-430
               (librosa melspectrogram, 48 kHz, n_fft=2048, hop=480,
               hann, center=True reflect, power=2, slaney fbank,
               power_to_db ref=1 amin=1e-10 top_db=None)
- MusiCNN mel: /root/reference/tasks/analysis/song.py:240-256
               (16 kHz, n_fft=512, hop=256, n_mels=96, center=False,
               slaney norm, log10(1+10000*mel))
- Whisper mel: /root/reference/lyrics/whisper_onnx.py:156-199
               (16 kHz, n_fft=400, hop=160, 80 mels, log10 clamp,
               max-8 clamp, (x+4)/4)

The mel filterbank is first-party (no librosa in the image): Slaney-style
triangular filters with 'slaney' area normalization, identical math to
librosa.filters.mel(htk=False, norm='slaney').
"""

from __future__ import annotations

import math
from dataclasses import dataclass

import numpy as np
import torch


# --------------------------------------------------------------------------
# Mel filterbank (librosa.filters.mel equivalent, htk=False, norm='slaney')
# --------------------------------------------------------------------------

_F_SP = 200.0 / 3.0          # linear region: mels per Hz below 1 kHz
_MIN_LOG_HZ = 1000.0
_MIN_LOG_MEL = _MIN_LOG_HZ / _F_SP
_LOGSTEP = math.log(6.4) / 27.0


def hz_to_mel(freq: np.ndarray) -> np.ndarray:
    freq = np.asarray(freq, dtype=np.float64)
    mels = freq / _F_SP
    log_t = freq >= _MIN_LOG_HZ
    mels = np.where(log_t, _MIN_LOG_MEL + np.log(np.maximum(freq, 1e-30) / _MIN_LOG_HZ) / _LOGSTEP, mels)
    return mels


def mel_to_hz(mels: np.ndarray) -> np.ndarray:
    mels = np.asarray(mels, dtype=np.float64)
    freqs = mels * _F_SP
    log_t = mels >= _MIN_LOG_MEL
    freqs = np.where(log_t, _MIN_LOG_HZ * np.exp(_LOGSTEP * (mels - _MIN_LOG_MEL)), freqs)
    return freqs


def mel_filterbank(sr: int, n_fft: int, n_mels: int, fmin: float, fmax: float,
                   norm: str = "slaney") -> np.ndarray:
    """Triangular mel filterbank, shape (n_mels, 1 + n_fft//2), float32."""
    if fmax is None or fmax <= 0:
        fmax = sr / 2.0
    n_freqs = 1 + n_fft // 2
    fftfreqs = np.linspace(0.0, sr / 2.0, n_freqs, dtype=np.float64)
    mel_pts = np.linspace(hz_to_mel(np.array(fmin)), hz_to_mel(np.array(fmax)), n_mels + 2)
    mel_f = mel_to_hz(mel_pts)

    fdiff = np.diff(mel_f)
    ramps = mel_f.reshape(-1, 1) - fftfreqs.reshape(1, -1)

    lower = -ramps[:-2] / fdiff[:-1].reshape(-1, 1)
    upper = ramps[2:] / fdiff[1:].reshape(-1, 1)
    weights = np.maximum(0.0, np.minimum(lower, upper))

    if norm == "slaney":
        enorm = 2.0 / (mel_f[2 : n_mels + 2] - mel_f[:n_mels])
        weights *= enorm.reshape(-1, 1)
    return weights.astype(np.float32)


# --------------------------------------------------------------------------
# STFT power spectrogram (librosa-compatible framing)
# --------------------------------------------------------------------------

def power_spectrogram(audio: torch.Tensor, n_fft: int, hop: int,
                      center: bool, win_length: int | None = None) -> torch.Tensor:
    """|STFT|^2. audio: (..., T) float32. Returns (..., n_freqs, n_frames).

    Matches librosa: periodic hann window, reflect padding when center=True.
    """
    if win_length is None:
        win_length = n_fft
    window = torch.hann_window(win_length, periodic=True, dtype=audio.dtype, device=audio.device)
    shape = audio.shape
    flat = audio.reshape(-1, shape[-1])
    spec = torch.stft(flat, n_fft=n_fft, hop_length=hop, win_length=win_length,
                      window=window, center=center, pad_mode="reflect",
                      return_complex=True)
    power = spec.real.square() + spec.imag.square()
    return power.reshape(*shape[:-1], *power.shape[-2:])


def power_to_db(power: torch.Tensor, ref: float = 1.0, amin: float = 1e-10) -> torch.Tensor:
    """librosa.power_to_db with top_db=None (clap_analyzer.py:419)."""
    log_spec = 10.0 * torch.log10(torch.clamp(power, min=amin))
    log_spec = log_spec - 10.0 * math.log10(max(amin, ref))
    return log_spec


# --------------------------------------------------------------------------
# Named front-end configs
# --------------------------------------------------------------------------

@dataclass(frozen=True)
class MelConfig:
    sr: int
    n_fft: int
    hop: int
    n_mels: int
    fmin: float
    fmax: float
    center: bool
    log_mode: str  # "db" | "log10_1p10k" | "whisper"

    @property
    def n_freqs(self) -> int:
        return 1 + self.n_fft // 2


def clap_mel_config() -> MelConfig:
    from audiomuse_amd import config as C
    return MelConfig(sr=C.CLAP_SAMPLE_RATE, n_fft=C.CLAP_AUDIO_N_FFT,
                     hop=C.CLAP_AUDIO_HOP_LENGTH, n_mels=C.CLAP_AUDIO_N_MELS,
                     fmin=C.CLAP_AUDIO_FMIN, fmax=C.CLAP_AUDIO_FMAX,
                     center=True, log_mode="db")


def musicnn_mel_config() -> MelConfig:
    from audiomuse_amd import config as C
    return MelConfig(sr=C.MUSICNN_SAMPLE_RATE, n_fft=C.MUSICNN_N_FFT,
                     hop=C.MUSICNN_HOP, n_mels=C.MUSICNN_N_MELS,
                     fmin=0.0, fmax=C.MUSICNN_SAMPLE_RATE / 2.0,
                     center=False, log_mode="log10_1p10k")


def whisper_mel_config() -> MelConfig:
    from audiomuse_amd import config as C
    return MelConfig(sr=C.WHISPER_SAMPLE_RATE, n_fft=C.WHISPER_N_FFT,
                     hop=C.WHISPER_HOP, n_mels=C.WHISPER_N_MELS,
                     fmin=0.0, fmax=C.WHISPER_SAMPLE_RATE / 2.0,
                     center=True, log_mode="whisper")


def _apply_log(mel: torch.Tensor, mode: str) -> torch.Tensor:
    if mode == "db":
        return power_to_db(mel)
    if mode == "log10_1p10k":
        return torch.log10(1.0 + 10000.0 * torch.clamp(mel, min=0.0))
    if mode == "whisper":
        log_spec = torch.log10(torch.clamp(mel, min=1e-10))
        # whisper clamps per-chunk at max - 8 then scales
        maxv = log_spec.amax(dim=(-2, -1), keepdim=True)
        log_spec = torch.maximum(log_spec, maxv - 8.0)
        return (log_spec + 4.0) / 4.0
    raise ValueError(f"unknown log mode {mode}")


class MelFrontend:
    """Reference mel front-end. fp32 torch ops, CPU or GPU.

    The fused HIP kernel (ops.hip_ops.mel_spectrogram) is numerically
    validated against this class (tests/test_mel.py).
    """

    def __init__(self, cfg: MelConfig, device: str | torch.device = "cpu"):
        self.cfg = cfg
        self.device = torch.device(device)
        fb = mel_filterbank(cfg.sr, cfg.n_fft, cfg.n_mels, cfg.fmin, cfg.fmax)
        self.fbank = torch.from_numpy(fb).to(self.device)  # (n_mels, n_freqs)

    def __call__(self, audio: torch.Tensor) -> torch.Tensor:
        """audio: (B, T) or (T,) fp32 -> (B, n_mels, n_frames) log-mel fp32."""
        single = audio.dim() == 1
        if single:
            audio = audio.unsqueeze(0)
        audio = audio.to(self.device, torch.float32)
        power = power_spectrogram(audio, self.cfg.n_fft, self.cfg.hop, self.cfg.center)
        mel = torch.matmul(self.fbank, power)  # (B, n_mels, frames)
        out = _apply_log(mel, self.cfg.log_mode)
        return out[0] if single else out


def int16_roundtrip(audio: torch.Tensor) -> torch.Tensor:
    """Bit-exact reproduction of the reference's training-time quantization
    (clap_analyzer.py:453-455): clip +-1, scale to int16, back to float."""
    clipped = torch.clamp(audio, -1.0, 1.0)
    q = (clipped * 32767.0).to(torch.int16)
    return q.to(torch.float32) / 32767.0


def segment_audio(audio: torch.Tensor, segment_len: int, hop: int) -> torch.Tensor:
    """Reference CLAP segmentation (clap_analyzer.py:460-475): full windows
    at `hop` stride, plus a tail window aligned to the end; short audio is
    zero-padded to one window. Returns (n_segments, segment_len)."""
    total = audio.shape[-1]
    if total <= segment_len:
        pad = torch.zeros(segment_len - total, dtype=audio.dtype, device=audio.device)
        return torch.cat([audio, pad]).unsqueeze(0)
    segs = [audio[s : s + segment_len] for s in range(0, total - segment_len + 1, hop)]
    last_start = len(segs) * hop
    if last_start < total:
        segs.append(audio[-segment_len:])
    return torch.stack(segs)
