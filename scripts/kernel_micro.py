"""Standalone micro-benchmarks of the custom HIP kernels (for rocprofv3
PMC runs and quick A/B timing). Run on an MI355X box:

  python scripts/kernel_micro.py            # timings
  rocprofv3 --pmc MfmaUtil VALUBusy OccupancyPercent SQ_LDS_BANK_CONFLICT \
      --kernel-trace -d out -- python scripts/kernel_micro.py --short
"""

import argparse
import sys
import time

import torch

sys.path.insert(0, __file__.rsplit("/", 2)[0])

from audiomuse_amd.models.htsat import HTSATConfig, HTSATEncoder  # noqa: E402
from audiomuse_amd.ops import dsp, hip_ops  # noqa: E402
from audiomuse_amd.ops import _ext  # noqa: E402


def timeit(fn, iters, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--short", action="store_true")
    args = ap.parse_args()
    iters = 2 if args.short else 10
    ext = _ext.require()
    dev = "cuda"

    # mel: 128 clips of 10 s @ 48 kHz
    audio = torch.randn(128, 480000, device=dev) * 0.2
    cfg = dsp.clap_mel_config()
    t = timeit(lambda: hip_ops.mel_spectrogram(audio, cfg, quantize_int16=True),
               iters)
    print(f"mel_fwd        : {t*1000:8.2f} ms /128 clips "
          f"({128/t:8.0f} clips/s)")

    # window attention: stage-1 shape (B=128, 32x256 grid, C=128, h=4)
    B, H, W, C, heads = 128, 32, 256, 128, 4
    qkv = torch.randn(B, H, W, 3 * C, device=dev, dtype=torch.bfloat16)
    bias = torch.randn(heads, 64, 64, device=dev).to(torch.bfloat16)
    t = timeit(lambda: ext.window_attn_fwd(qkv, bias, heads, 0, 0.176), iters)
    toks = B * H * W
    print(f"window_attn s1 : {t*1000:8.2f} ms ({toks/t/1e6:6.1f} Mtok/s)")
    # stage-3 shape (C=512, h=16, 8x64 grid)
    qkv3 = torch.randn(B, 8, 64, 3 * 512, device=dev, dtype=torch.bfloat16)
    bias3 = torch.randn(16, 64, 64, device=dev).to(torch.bfloat16)
    t = timeit(lambda: ext.window_attn_fwd(qkv3, bias3, 16, 4, 0.176), iters)
    print(f"window_attn s3 : {t*1000:8.2f} ms")
    # fp8-ingest variant, same shapes (A/B isolating the kernel from the
    # fp8-D GEMM and scale-state plumbing)
    qs = torch.full((), 0.03, device=dev, dtype=torch.float32)
    qkv8 = (qkv.float() / qs).clamp(-448, 448).to(
        torch.float8_e4m3fn).contiguous()
    t = timeit(lambda: ext.window_attn_fp8_fwd(qkv8, bias, qs, heads, 0,
                                               0.176), iters)
    print(f"window_attn s1 fp8: {t*1000:5.2f} ms ({toks/t/1e6:6.1f} Mtok/s)")
    qkv38 = (qkv3.float() / qs).clamp(-448, 448).to(
        torch.float8_e4m3fn).contiguous()
    t = timeit(lambda: ext.window_attn_fp8_fwd(qkv38, bias3, qs, 16, 4,
                                               0.176), iters)
    print(f"window_attn s3 fp8: {t*1000:5.2f} ms")
    # stage-4 shape: 4x32 grid, window 4, C=1024, h=32 (window_attn4)
    qkv4 = torch.randn(B, 4, 32, 3 * 1024, device=dev, dtype=torch.bfloat16)
    bias4 = torch.randn(32, 16, 16, device=dev).to(torch.bfloat16)
    t = timeit(lambda: ext.window_attn4_fwd(qkv4, bias4, 32, 2, 0.176),
               iters)
    print(f"window_attn4 s4: {t*1000:8.2f} ms")

    # layernorm: stage-1 tokens
    x = torch.randn(B * 8192, 128, device=dev, dtype=torch.bfloat16)
    w = torch.randn(128, device=dev, dtype=torch.bfloat16)
    bb = torch.randn(128, device=dev, dtype=torch.bfloat16)
    t = timeit(lambda: ext.layernorm_bf16(x, w, bb, 1e-5), iters)
    gb = x.numel() * 2 * 2 / 1e9
    print(f"layernorm d128 : {t*1000:8.2f} ms ({gb/t:6.2f} GB/s eff)")

    # IVF i8 scan: 1M x 512, 64 queries, nprobe 64
    if not args.short:
        from audiomuse_amd.index.ivf import IVFIndex

        xb = torch.randn(1_000_000, 512)
        idx = IVFIndex.build(xb, metric="angular", storage="i8",
                             device=dev, seed=0)
        q = xb[:64].to(dev)
        t = timeit(lambda: idx.scan(q, nprobe=64), 5)
        print(f"ivf_scan i8    : {t*1000:8.2f} ms (64 queries, nprobe=64, "
              f"1M x 512)")

    # full encoder forward
    model = HTSATEncoder(HTSATConfig()).to(dev, torch.bfloat16).eval()
    mel = torch.randn(256, 128, 1001, device=dev, dtype=torch.bfloat16)
    with torch.inference_mode():
        t = timeit(lambda: model(mel), max(iters // 2, 1))
    print(f"htsat fwd b256 : {t*1000:8.2f} ms ({256/t:8.0f} clips/s encoder-only)")


if __name__ == "__main__":
    main()
