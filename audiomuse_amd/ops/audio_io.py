"""Audio IO + resampling.

The reference decodes with librosa/audioread and falls back to PyAV
(/root/reference/tasks/analysis/song.py:270-360); neither ships in this
image. First-party path: stdlib WAV (PCM16/PCM32/float) decode, and a
windowed-sinc polyphase resampler in torch (GPU-capable) replacing
soxr (song.py:333). Compressed formats route through an optional
`ffmpeg` binary when present; absent that, WAV-only (synthetic data is
WAV everywhere in this build).
"""

from __future__ import annotations

import io
import math
import os
import shutil
import subprocess
import wave
from typing import Optional, Tuple

import numpy as np
import torch


def load_wav(source) -> Tuple[torch.Tensor, int]:
    """source: path or bytes. Returns (mono float32 tensor in [-1,1], sr)."""
    if isinstance(source, (bytes, bytearray)):
        fh = io.BytesIO(bytes(source))
    else:
        fh = source
    with wave.open(fh, "rb") as w:
        sr = w.getframerate()
        n = w.getnframes()
        ch = w.getnchannels()
        width = w.getsampwidth()
        raw = w.readframes(n)
    if width == 2:
        data = np.frombuffer(raw, dtype=np.int16).astype(np.float32) / 32768.0
    elif width == 4:
        data = np.frombuffer(raw, dtype=np.int32).astype(np.float32) / 2147483648.0
    elif width == 1:
        data = (np.frombuffer(raw, dtype=np.uint8).astype(np.float32) - 128.0) / 128.0
    else:
        raise ValueError(f"unsupported WAV sample width {width}")
    if ch > 1:
        data = data.reshape(-1, ch).mean(axis=1)
    return torch.from_numpy(data.copy()), sr


def save_wav(path: str, audio: torch.Tensor, sr: int) -> None:
    a = torch.clamp(audio.detach().cpu().float(), -1.0, 1.0)
    pcm = (a * 32767.0).to(torch.int16).numpy()
    with wave.open(path, "wb") as w:
        w.setnchannels(1)
        w.setsampwidth(2)
        w.setframerate(sr)
        w.writeframes(pcm.tobytes())


def load_audio(source, target_sr: Optional[int] = None
               ) -> Tuple[Optional[torch.Tensor], int]:
    """Robust load (reference: robust_load_audio_with_fallback,
    song.py:312): WAV directly; other formats via ffmpeg when available."""
    try:
        audio, sr = load_wav(source)
    except Exception:
        audio, sr = _load_via_ffmpeg(source, target_sr)
        if audio is None:
            return None, 0
    if target_sr and sr != target_sr:
        audio = resample(audio, sr, target_sr)
        sr = target_sr
    return audio, sr


def _load_via_ffmpeg(source, target_sr):
    if shutil.which("ffmpeg") is None or not isinstance(source, str):
        return None, 0
    sr = target_sr or 48000
    try:
        out = subprocess.run(
            ["ffmpeg", "-v", "quiet", "-i", source, "-f", "f32le", "-ac", "1",
             "-ar", str(sr), "-"], capture_output=True, timeout=300, check=True)
        data = np.frombuffer(out.stdout, dtype=np.float32)
        return torch.from_numpy(data.copy()), sr
    except Exception:
        return None, 0


_KERNEL_CACHE = {}


def _resample_kernel(up: int, down: int, zeros: int, device
                     ) -> Tuple[torch.Tensor, int]:
    """Polyphase windowed-sinc filterbank: (up, 1, taps) kernel applied
    with conv stride=down; phase i produces output sample j*up+i."""
    rolloff = 0.945
    # cutoff in input-sample units: f_c = rolloff * min(1, up/down) / 2
    fc = 0.5 * rolloff * min(1.0, up / down)
    width = int(math.ceil(zeros / (2 * fc)))
    # output sample (frame j, phase i) sits at input time j*down + i*down/up,
    # so phase i's taps cover idx in [-width, width + down)
    idx = torch.arange(-width, width + down, dtype=torch.float64)
    phases = []
    for i in range(up):
        t = idx - (i * down / up)
        h = 2 * fc * torch.special.sinc(2 * fc * t)
        win = torch.cos(math.pi * t / (2 * width)).clamp(min=0.0) ** 2
        win = torch.where(t.abs() <= width, win, torch.zeros_like(win))
        phases.append(h * win)
    kernel = torch.stack(phases).unsqueeze(1).to(torch.float32).to(device)
    return kernel, width


def resample(audio: torch.Tensor, sr_in: int, sr_out: int,
             zeros: int = 16) -> torch.Tensor:
    """Windowed-sinc polyphase resampler (soxr replacement). Works on CPU
    and GPU; audio (..., T) float32."""
    if sr_in == sr_out:
        return audio
    g = math.gcd(sr_in, sr_out)
    up, down = sr_out // g, sr_in // g
    key = (up, down, zeros, str(audio.device))
    got = _KERNEL_CACHE.get(key)
    if got is None:
        got = _KERNEL_CACHE[key] = _resample_kernel(up, down, zeros,
                                                    audio.device)
    kernel, width = got
    single = audio.dim() == 1
    x = audio.float().reshape(1 if single else -1, 1, audio.shape[-1])
    n = x.shape[-1]
    x = torch.nn.functional.pad(x, (width, width + down))
    y = torch.nn.functional.conv1d(x, kernel, stride=down)  # (B, up, frames)
    y = y.transpose(1, 2).reshape(y.shape[0], -1)           # interleave phases
    out_len = int(math.ceil(n * up / down))
    y = y[:, :out_len]
    return y.squeeze(0) if single else y


def synthetic_track(seed: int, seconds: float = 12.0, sr: int = 44100
                    ) -> torch.Tensor:
    """Deterministic synthetic music-like audio: chord pad + beat +
    melody (used by the synthetic media provider and tests)."""
    g = torch.Generator().manual_seed(seed)
    t = torch.arange(int(seconds * sr), dtype=torch.float64) / sr
    root = 110.0 * 2 ** (float(torch.randint(0, 12, (1,), generator=g)) / 12)
    chord = sum(torch.sin(2 * math.pi * root * r * t)
                for r in (1.0, 1.25, 1.5))
    bpm = float(torch.randint(80, 160, (1,), generator=g))
    beat_period = 60.0 / bpm
    beat_phase = (t % beat_period) / beat_period
    beat = torch.exp(-beat_phase * 30.0) * torch.sin(2 * math.pi * 60.0 * t)
    melody_f = root * 2 * 2 ** (torch.floor(t / 0.5) % 7 / 7)
    melody = 0.4 * torch.sin(2 * math.pi * melody_f * t)
    noise = torch.randn(t.shape[0], generator=g) * 0.01
    mix = 0.25 * chord + 0.5 * beat + melody + noise
    return (mix / mix.abs().max() * 0.8).to(torch.float32)
