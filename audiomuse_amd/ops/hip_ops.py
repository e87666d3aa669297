"""Dispatch layer: fused HIP kernels on GPU, torch reference on CPU.

Per-(config, device) constant tensors (hann window, FFT twiddle table,
CSR mel filterbank) are cached on device; kernels launch on the current
torch HIP stream so they compose with streams/graphs.
"""

from __future__ import annotations

import math
from typing import Dict, Tuple

import numpy as np
import torch

from audiomuse_amd.ops import _ext
from audiomuse_amd.ops.dsp import MelConfig, MelFrontend, mel_filterbank

_LOG_MODE = {"db": 0, "log10_1p10k": 1, "raw": 2}

_PLAN_CACHE: Dict[Tuple, dict] = {}
_REF_CACHE: Dict[Tuple, MelFrontend] = {}


def _mel_plan(cfg: MelConfig, device: torch.device) -> dict:
    key = (cfg, str(device))
    plan = _PLAN_CACHE.get(key)
    if plan is not None:
        return plan
    n = cfg.n_fft
    window = torch.hann_window(n, periodic=True, dtype=torch.float32)
    j = np.arange(n // 2, dtype=np.float64)
    ang = 2.0 * math.pi * j / n
    twiddle = np.stack([np.cos(ang), -np.sin(ang)], axis=1).astype(np.float32)
    fb = mel_filterbank(cfg.sr, n, cfg.n_mels, cfg.fmin, cfg.fmax)  # (m, f)
    rowptr = [0]
    bins: list[int] = []
    weights: list[float] = []
    for m in range(cfg.n_mels):
        nz = np.nonzero(fb[m])[0]
        bins.extend(int(k) for k in nz)
        weights.extend(float(fb[m, k]) for k in nz)
        rowptr.append(len(bins))
    plan = {
        "window": window.to(device),
        "twiddle": torch.from_numpy(twiddle).contiguous().to(device),
        "rowptr": torch.tensor(rowptr, dtype=torch.int32, device=device),
        "bin": torch.tensor(bins, dtype=torch.int32, device=device),
        "w": torch.tensor(weights, dtype=torch.float32, device=device),
    }
    _PLAN_CACHE[key] = plan
    return plan


def _reference(cfg: MelConfig, device: torch.device) -> MelFrontend:
    key = (cfg, str(device))
    ref = _REF_CACHE.get(key)
    if ref is None:
        ref = MelFrontend(cfg, device)
        _REF_CACHE[key] = ref
    return ref


def mel_spectrogram(audio: torch.Tensor, cfg: MelConfig,
                    force_reference: bool = False,
                    quantize_int16: bool = False) -> torch.Tensor:
    """Log-mel spectrogram. audio (B, T) or (T,) fp32 -> (B, n_mels, frames).

    GPU inputs run the fused HIP kernel (ops/csrc/mel.hip); CPU inputs (or
    force_reference=True) run the torch reference (ops/dsp.MelFrontend).
    """
    single = audio.dim() == 1
    if single:
        audio = audio.unsqueeze(0)
    audio = audio.to(torch.float32).contiguous()

    use_native = (audio.is_cuda and not force_reference
                  and cfg.log_mode in _LOG_MODE
                  and cfg.n_fft in (256, 512, 1024, 2048, 4096))
    if use_native:
        ext = _ext.native_or_none()
        if ext is not None:
            plan = _mel_plan(cfg, audio.device)
            out = ext.mel_fwd(audio, plan["window"], plan["twiddle"],
                              plan["rowptr"], plan["bin"], plan["w"],
                              cfg.hop, cfg.n_fft, cfg.center,
                              _LOG_MODE[cfg.log_mode], quantize_int16)
            return out[0] if single else out
    if quantize_int16:
        from audiomuse_amd.ops.dsp import int16_roundtrip
        audio = int16_roundtrip(audio)
    out = _reference(cfg, audio.device)(audio)
    return out[0] if single else out
