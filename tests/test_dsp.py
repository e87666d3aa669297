"""Numerics tests for the audio front-end.

Strategy: every kernel is validated against a plain fp32
reference of the same op. Here:
- the torch reference (ops/dsp.py) is validated against an independent
  numpy STFT implementation,
- the HIP kernel's algorithm (bit-reversal radix-2 FFT + CSR mel) is
  mirrored in numpy and validated against np.fft,
- the HIP kernel itself is validated against the torch reference on GPU
  (marked gpu).
"""

import math

import numpy as np
import pytest
import torch

from audiomuse_amd.ops import dsp, hip_ops


def test_mel_scale_round_trip():
    f = np.array([0.0, 250.0, 999.0, 1000.0, 4000.0, 14000.0, 24000.0])
    m = dsp.hz_to_mel(f)
    back = dsp.mel_to_hz(m)
    np.testing.assert_allclose(back, f, rtol=1e-10, atol=1e-6)
    # Slaney anchors: linear below 1 kHz at 200/3 mel/Hz
    assert math.isclose(float(dsp.hz_to_mel(np.array(1000.0))), 15.0)
    assert math.isclose(float(dsp.hz_to_mel(np.array(500.0))), 7.5)


def test_mel_filterbank_shape_and_triangles():
    fb = dsp.mel_filterbank(48000, 2048, 128, 0.0, 14000.0)
    assert fb.shape == (128, 1025)
    assert np.all(fb >= 0)
    # every filter is unimodal (one contiguous support run)
    for m in range(128):
        nz = np.nonzero(fb[m])[0]
        assert nz.size > 0
        assert np.all(np.diff(nz) == 1), f"filter {m} support not contiguous"
        peak = np.argmax(fb[m])
        left, right = fb[m][: peak + 1], fb[m][peak:]
        assert np.all(np.diff(left[left > 0]) >= -1e-9)
        assert np.all(np.diff(right[right > 0]) <= 1e-9)
    # slaney normalization: peak ~= 2/width(hz)
    mel_pts = np.linspace(dsp.hz_to_mel(np.array(0.0)),
                          dsp.hz_to_mel(np.array(14000.0)), 130)
    hz = dsp.mel_to_hz(mel_pts)
    widths = hz[2:] - hz[:-2]
    peaks = fb.max(axis=1)
    apex = 2.0 / widths  # continuous triangle apex under slaney norm
    assert np.all(peaks <= apex * (1 + 1e-6))
    # sampled peak sits below the apex when the filter is ~1 bin wide,
    # but never by more than the bin-quantization allows
    assert np.all(peaks >= apex * 0.3)


def _numpy_stft_power(x, n_fft, hop, center):
    """Independent STFT power implementation (numpy, no torch)."""
    win = 0.5 - 0.5 * np.cos(2 * np.pi * np.arange(n_fft) / n_fft)
    if center:
        pad = n_fft // 2
        x = np.pad(x, pad, mode="reflect")
    n_frames = 1 + (len(x) - n_fft) // hop
    frames = np.stack([x[i * hop : i * hop + n_fft] * win for i in range(n_frames)])
    spec = np.fft.rfft(frames, axis=1)
    return (np.abs(spec) ** 2).T.astype(np.float64)  # (n_freqs, n_frames)


@pytest.mark.parametrize("cfg_fn", [dsp.clap_mel_config, dsp.musicnn_mel_config])
def test_torch_reference_matches_numpy(cfg_fn):
    cfg = cfg_fn()
    rng = np.random.default_rng(0)
    x = rng.standard_normal(cfg.sr).astype(np.float32) * 0.3
    ref = dsp.MelFrontend(cfg)(torch.from_numpy(x)).numpy()

    power = _numpy_stft_power(x.astype(np.float64), cfg.n_fft, cfg.hop, cfg.center)
    fb = dsp.mel_filterbank(cfg.sr, cfg.n_fft, cfg.n_mels, cfg.fmin, cfg.fmax).astype(np.float64)
    mel = fb @ power
    if cfg.log_mode == "db":
        expect = 10.0 * np.log10(np.maximum(mel, 1e-10))
    else:
        expect = np.log10(1.0 + 10000.0 * np.maximum(mel, 0.0))
    assert ref.shape == expect.shape
    np.testing.assert_allclose(ref, expect, rtol=1e-4, atol=1e-3)


def _kernel_mirror_fft(x):
    """Numpy mirror of the DIT FFT in ops/csrc/mel.hip: bit-reversal,
    then merged radix-8 passes (three radix-2 stages per LDS round-trip)
    with radix-4 / radix-2 leftovers for log2(N) % 3."""
    n = len(x)
    log2n = n.bit_length() - 1
    assert 1 << log2n == n
    j = np.arange(n // 2)
    tw = np.cos(2 * np.pi * j / n) - 1j * np.sin(2 * np.pi * j / n)
    rev = np.array([int(f"{i:0{log2n}b}"[::-1], 2) for i in range(n)])
    z = np.zeros(n, dtype=np.complex128)
    z[rev] = x
    s = 1
    while s + 2 <= log2n:
        h = 1 << (s - 1)
        for q in range(n // 8):
            grp, jj = q >> (s - 1), q & (h - 1)
            i0 = (grp << (s + 2)) + jj
            w1 = tw[jj * (n >> s)]
            w2 = tw[jj * (n >> (s + 1))]
            w4 = tw[jj * (n >> (s + 2))]
            xs = [z[i0 + m * h] for m in range(8)]
            b = [0.0] * 8
            for m in range(0, 8, 2):
                t = w1 * xs[m + 1]
                b[m], b[m + 1] = xs[m] + t, xs[m] - t
            d = [0.0] * 8
            w3 = -1j * w2
            for g in (0, 4):
                t = w2 * b[g + 2]
                d[g], d[g + 2] = b[g] + t, b[g] - t
                t = w3 * b[g + 3]
                d[g + 1], d[g + 3] = b[g + 1] + t, b[g + 1] - t
            for m in range(4):
                t = w4 * np.exp(-1j * np.pi * m / 4) * d[m + 4]
                z[i0 + m * h] = d[m] + t
                z[i0 + (m + 4) * h] = d[m] - t
        s += 3
    while s + 1 <= log2n:
        h = 1 << (s - 1)
        for q in range(n // 4):
            grp, jj = q >> (s - 1), q & (h - 1)
            i0 = (grp << (s + 1)) + jj
            w1 = tw[jj * (n >> s)]
            w2 = tw[jj * (n >> (s + 1))]
            x0, x1, x2, x3 = z[i0], z[i0 + h], z[i0 + 2 * h], z[i0 + 3 * h]
            a0, a1 = x0 + w1 * x1, x0 - w1 * x1
            a2, a3 = x2 + w1 * x3, x2 - w1 * x3
            w3 = -1j * w2
            z[i0], z[i0 + 2 * h] = a0 + w2 * a2, a0 - w2 * a2
            z[i0 + h], z[i0 + 3 * h] = a1 + w3 * a3, a1 - w3 * a3
        s += 2
    while s <= log2n:
        half = 1 << (s - 1)
        step = n >> s
        for bf in range(n // 2):
            grp, jj = bf >> (s - 1), bf & (half - 1)
            i0 = (grp << s) + jj
            i1 = i0 + half
            w = tw[jj * step]
            a, c = z[i0], z[i1]
            z[i0], z[i1] = a + w * c, a - w * c
        s += 1
    return z


@pytest.mark.parametrize("n", [256, 512, 1024, 2048, 4096])
def test_kernel_fft_algorithm_matches_numpy(n):
    rng = np.random.default_rng(1)
    x = rng.standard_normal(n)
    got = _kernel_mirror_fft(x)
    expect = np.fft.fft(x)
    np.testing.assert_allclose(got, expect, rtol=1e-9, atol=1e-9)


def test_int16_roundtrip_matches_numpy_semantics():
    x = torch.tensor([0.0, 0.5, -0.5, 1.2, -1.2, 1.0 / 32767.0, 0.99999])
    got = dsp.int16_roundtrip(x)
    xn = np.clip(x.numpy(), -1, 1)
    expect = (xn * 32767.0).astype(np.int16).astype(np.float32) / 32767.0
    np.testing.assert_array_equal(got.numpy(), expect)


def test_segment_audio_matches_reference_windowing():
    # reference: clap_analyzer.py:460-475 (10 s windows, 5 s hop, end-aligned tail)
    seg, hop = 480000, 240000
    short = torch.arange(1000, dtype=torch.float32)
    out = dsp.segment_audio(short, seg, hop)
    assert out.shape == (1, seg)
    assert torch.equal(out[0, :1000], short) and out[0, 1000:].abs().sum() == 0

    t = torch.arange(30 * 48000, dtype=torch.float32)  # 30 s
    out = dsp.segment_audio(t, seg, hop)
    # 5 full windows (starts 0..20 s) + end-aligned tail (5*hop < total)
    assert out.shape == (6, seg)
    assert out[0, 0] == 0 and out[1, 0] == hop and out[5, -1] == t[-1]

    t = torch.arange(int(23.5 * 48000), dtype=torch.float32)
    out = dsp.segment_audio(t, seg, hop)
    # 3 full windows (starts 0, 5, 10 s) + end-aligned tail
    assert out.shape == (4, seg)
    assert out[-1, -1] == t[-1]


@pytest.mark.gpu
@pytest.mark.parametrize("cfg_fn", [dsp.clap_mel_config, dsp.musicnn_mel_config])
def test_hip_mel_matches_reference(cfg_fn):
    cfg = cfg_fn()
    torch.manual_seed(0)
    audio = (torch.randn(3, cfg.sr * 10, device="cuda") * 0.2).clamp(-1, 1)
    native = hip_ops.mel_spectrogram(audio, cfg)
    ref = hip_ops.mel_spectrogram(audio, cfg, force_reference=True)
    assert native.shape == ref.shape
    # log-domain values: absolute tolerance dominates near the floor
    torch.testing.assert_close(native, ref, rtol=1e-3, atol=2e-3)


@pytest.mark.gpu
def test_hip_mel_long_batch():
    cfg = dsp.clap_mel_config()
    audio = torch.randn(8, 480000, device="cuda") * 0.1
    out = hip_ops.mel_spectrogram(audio, cfg)
    assert out.shape == (8, cfg.n_mels, 1001)
    assert torch.isfinite(out).all()
