"""Probe fp8 (OCP e4m3) GEMM support on this torch/ROCm build.

gfx950 runs fp8 MFMA at 2x the bf16 rate (dense ~5 PF); if
torch._scaled_mm works here, the encoder's Linear layers can take a
quantized inference path. Run on an MI355X box.
"""

import time

import torch


def main():
    dev = "cuda"
    ok = hasattr(torch, "_scaled_mm")
    print("has _scaled_mm:", ok)
    if not ok:
        return
    M, K, N = 8192, 512, 1536
    a = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
    b = torch.randn(N, K, device=dev, dtype=torch.bfloat16)
    try:
        a8 = a.to(torch.float8_e4m3fn)
        b8 = b.to(torch.float8_e4m3fn).T
        sa = torch.tensor(1.0, device=dev)
        sb = torch.tensor(1.0, device=dev)
        out = torch._scaled_mm(a8, b8, scale_a=sa, scale_b=sb,
                               out_dtype=torch.bfloat16)
        ref = a @ b.T
        rel = (out.float() - ref.float()).abs().mean() / ref.float().abs().mean()
        print(f"fp8 scaled_mm OK; mean rel err {float(rel):.4f}")
        # timing vs bf16
        for name, fn in [("bf16", lambda: a @ b.T),
                         ("fp8", lambda: torch._scaled_mm(
                             a8, b8, scale_a=sa, scale_b=sb,
                             out_dtype=torch.bfloat16))]:
            for _ in range(5):
                fn()
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(50):
                fn()
            torch.cuda.synchronize()
            dt = (time.perf_counter() - t0) / 50
            tf = 2 * M * K * N / dt / 1e12
            print(f"{name}: {dt*1e6:8.1f} us  {tf:7.1f} TF")
    except Exception as exc:  # noqa: BLE001
        print("fp8 path failed:", type(exc).__name__, exc)


if __name__ == "__main__":
    main()
