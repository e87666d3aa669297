"""Is fp8 worth it at the ENCODER's shapes? (M huge, K/N small.)

The first fp8_probe measured +39% at M=8192,K=512,N=1536. The encoder's
GEMMs are M=0.5-2.1M, K=128-1024, N=128-4096 — memory-bound on the
activation read, where fp8's entire win is halved operand bytes. An
unfused quantize pass re-reads what the GEMM saves, so this probe times
(a) the bf16 GEMM, (b) the pre-quantized fp8 GEMM alone (upper bound on
the win), and (c) the quantize step in f32 vs bf16 arithmetic — which
bounds what a producer-fused fp8 epilogue could reach.
"""

import time

import torch

E = torch.float8_e4m3fn


def t(fn, n=20):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1e3


def main():
    dev = "cuda"
    fmax = torch.finfo(E).max
    # (M, K, N) per stage: qkv / proj / mlp0 / mlp2 at B=256
    shapes = [
        (2_097_152, 128, 384), (2_097_152, 128, 128),
        (2_097_152, 128, 512), (2_097_152, 512, 128),
        (524_288, 256, 768), (524_288, 1024, 256),
        (131_072, 512, 1536), (131_072, 2048, 512),
        (32_768, 1024, 3072), (32_768, 4096, 1024),
    ]
    import sys as _sys
    _sys.path.insert(0, "/root/repo")
    import audiomuse_amd._C as ext
    print(f"{'M':>9} {'K':>5} {'N':>5} | {'bf16':>7} {'fp8mm':>7} "
          f"{'fp8ext':>7} {'q_bf16':>7} | fp8ext+q_bf16 vs bf16")
    for M, K, N in shapes:
        x = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
        w = torch.randn(N, K, device=dev, dtype=torch.bfloat16) * 0.05
        xs = (x.abs().amax().float() / fmax).clamp(min=1e-12)
        ws = (w.abs().amax().float() / fmax).clamp(min=1e-12)
        xq = (x.float() / xs).clamp(-fmax, fmax).to(E)
        wq = (w.float() / ws).clamp(-fmax, fmax).to(E).contiguous()
        tb = t(lambda: x @ w.t())
        tf8 = t(lambda: torch._scaled_mm(xq, wq.t(), scale_a=xs, scale_b=ws,
                                         out_dtype=torch.bfloat16))
        # accuracy check once, then time the searched ext path
        got = ext.linear_fp8(xq, wq, xs, ws)
        ref = torch._scaled_mm(xq, wq.t(), scale_a=xs, scale_b=ws,
                               out_dtype=torch.bfloat16)
        rel = (got.float() - ref.float()).abs().mean() / (
            ref.float().abs().mean() + 1e-9)
        assert rel < 2e-2, f"ext fp8 mismatch {rel}"
        text = t(lambda: ext.linear_fp8(xq, wq, xs, ws))
        inv = (1.0 / xs).to(torch.bfloat16)
        tqbf = t(lambda: x.mul(inv).clamp(-fmax.__float__(),
                                          fmax.__float__()).to(E))
        tot = text + tqbf
        print(f"{M:>9} {K:>5} {N:>5} | {tb:7.3f} {tf8:7.3f} "
              f"{text:7.3f} {tqbf:7.3f} | {tot:7.3f} ({tb / tot:4.2f}x)")


if __name__ == "__main__":
    main()
