"""Acoustic fingerprint (chromaprint-class).

Reference: /root/reference/tasks/chromaprint.py shells out to a vendored
`fpcalc` binary and compares zlib-compressed raw int fingerprints by
alignment/overlap bit-match. No fpcalc binary ships in this image, so
this is a first-party chroma fingerprint with the same role and
comparison semantics: 12-bin chroma frames -> temporal/spectral gradient
binarization -> packed uint32 frames; agreement = best-aligned bit-match
ratio over the overlap (chromaprint.py:67-115 behavior).
"""

from __future__ import annotations

import zlib
from typing import Optional

import numpy as np

from audiomuse_amd import config as C
import torch

from audiomuse_amd.ops.features import chroma_from_stft

FP_SR = 11025
_FRAME_BITS = 24          # 12 temporal + 12 spectral gradient bits


def compute(audio: torch.Tensor, sr: int, max_seconds: float = 120.0
            ) -> bytes:
    """zlib-compressed packed fingerprint of (<= max_seconds of) audio."""
    from audiomuse_amd.ops.audio_io import resample

    audio = audio.float().flatten()[: int(sr * max_seconds)]
    a = resample(audio, sr, FP_SR)
    chroma = chroma_from_stft(a, FP_SR, n_fft=4096, hop=1365)  # ~8 fps
    c = chroma.T.cpu().numpy()                                  # (frames, 12)
    if c.shape[0] < 3:
        return b""
    dt = (c[1:] > c[:-1]).astype(np.uint32)                     # temporal grad
    ds = (c[1:, :] > np.roll(c[1:, :], 1, axis=1)).astype(np.uint32)
    words = np.zeros(dt.shape[0], dtype=np.uint32)
    for b in range(12):
        words |= dt[:, b] << b
        words |= ds[:, b] << (12 + b)
    return zlib.compress(words.tobytes())


def _unpack(fp: bytes) -> Optional[np.ndarray]:
    if not fp:
        return None
    try:
        raw = zlib.decompress(fp)
    except zlib.error:
        return None
    return np.frombuffer(raw, dtype=np.uint32)


def bit_match_ratio(fp_a: bytes, fp_b: bytes,
                    max_offset: Optional[int] = None) -> float:
    """Best aligned per-bit agreement over the overlap (the reference's
    align/overlap/bit-match comparison). Alignment range and minimum
    overlap come from config (CHROMAPRINT_ALIGN_RANGE /
    CHROMAPRINT_MIN_OVERLAP)."""
    if max_offset is None:
        max_offset = C.CHROMAPRINT_ALIGN_RANGE // 2
    a = _unpack(fp_a)
    b = _unpack(fp_b)
    if a is None or b is None or a.size == 0 or b.size == 0:
        return 0.0
    best = 0.0
    for off in range(-max_offset, max_offset + 1):
        if off >= 0:
            aa, bb = a[off:], b
        else:
            aa, bb = a, b[-off:]
        n = min(aa.size, bb.size)
        if n < max(8, C.CHROMAPRINT_MIN_OVERLAP // 8):
            continue
        diff = np.bitwise_xor(aa[:n], bb[:n])
        bits = np.unpackbits(diff.view(np.uint8)).sum()
        ratio = 1.0 - bits / (n * 32.0)
        # only 24 of 32 bits carry signal; rescale agreement over them
        ratio = max(0.0, (ratio * 32.0 - 8.0) / 24.0)
        best = max(best, ratio)
    return best


def chromaprints_agree(fp_a: bytes, fp_b: bytes,
                       threshold: float = 0.85) -> bool:
    return bit_match_ratio(fp_a, fp_b) >= threshold
