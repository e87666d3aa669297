"""10M-resident k-NN evidence (BASELINE bar (a): the reference calls
sub-second CPU search at ~1M 'well under a second'; this measures build
time, single-query latency percentiles, and batched throughput at 10x
that scale, fully HBM-resident)."""

import sys
import time

import torch

sys.path.insert(0, __file__.rsplit("/", 2)[0])

from audiomuse_amd.index.ivf import IVFIndex  # noqa: E402


def main(n=10_000_000, d=512):
    dev = "cuda"
    g = torch.Generator(device=dev).manual_seed(0)
    centers = torch.randn(2048, d, generator=g, device=dev)
    assign = torch.randint(0, 2048, (n,), generator=g, device=dev)
    x = centers[assign] + torch.randn(n, d, generator=g, device=dev) * 0.3
    del centers, assign
    torch.cuda.synchronize()

    t0 = time.perf_counter()
    idx = IVFIndex.build(x, metric="angular", storage="i8", device=dev,
                         keep_f32=True)
    torch.cuda.synchronize()
    print(f"build {n/1e6:.0f}M x {d} i8: {time.perf_counter() - t0:.1f} s "
          f"(nlist {idx.nlist})")

    q = x[:512] + torch.randn(512, d, generator=g, device=dev) * 0.05
    # single-query latency (warm up first: the cold call pays kernel
    # module load + allocator growth and is not steady-state latency)
    for i in range(3):
        idx.query(q[i], k=20)
    torch.cuda.synchronize()
    lats = []
    for i in range(50):
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        idx.query(q[i], k=20)
        torch.cuda.synchronize()
        lats.append((time.perf_counter() - t0) * 1000)
    lats.sort()
    print(f"single-query latency: p50 {lats[25]:.1f} ms  "
          f"p90 {lats[45]:.1f} ms  p99 {lats[-1]:.1f} ms  (nprobe 1024)")

    # batched throughput
    for _ in range(2):
        idx.query(q, k=20)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    reps = 5
    for _ in range(reps):
        idx.query(q, k=20)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    print(f"batched q=512: {512 * reps / dt:.0f} queries/s")


if __name__ == "__main__":
    main(n=int(sys.argv[1]) if len(sys.argv) > 1 else 10_000_000)
