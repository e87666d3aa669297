"""Clustering GPU-vs-CPU speedup (BASELINE bar (c): the reference quotes
cuML at 10-30x over sklearn; here the SAME first-party math runs on both
devices — identical algorithms, measured head-to-head)."""

import sys
import time

import torch

sys.path.insert(0, __file__.rsplit("/", 2)[0])

from audiomuse_amd.cluster import algorithms as alg  # noqa: E402


def timeit(fn, n=3):
    fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n


def main():
    g = torch.Generator().manual_seed(0)
    n, d, k = 50_000, 200, 40
    centers = torch.randn(k, d, generator=g) * 4
    assign = torch.randint(0, k, (n,), generator=g)
    x = centers[assign] + torch.randn(n, d, generator=g) * 0.4
    xg = x.cuda()

    rows = []
    for name, cpu_fn, gpu_fn, reps in [
        ("kmeans k=40", lambda: alg.kmeans_fit(x, k, seed=0),
         lambda: alg.kmeans_fit(xg, k, seed=0), 3),
        ("gmm k=20", lambda: alg.gmm_fit(x[:20000], 20, seed=0),
         lambda: alg.gmm_fit(xg[:20000], 20, seed=0), 3),
        ("dbscan 8k", lambda: alg.dbscan_fit(x[:8000], 1.5, 5),
         lambda: alg.dbscan_fit(xg[:8000], 1.5, 5), 1),
        ("spectral 4k", lambda: alg.spectral_fit(x[:4000], 10, seed=0),
         lambda: alg.spectral_fit(xg[:4000], 10, seed=0), 1),
        ("pca d=32", lambda: alg.pca_fit_transform(x, 32),
         lambda: alg.pca_fit_transform(xg, 32), 3),
    ]:
        t_gpu = timeit(gpu_fn, reps)
        t0 = time.perf_counter()
        for _ in range(max(reps // 3, 1)):
            cpu_fn()
        t_cpu = (time.perf_counter() - t0) / max(reps // 3, 1)
        rows.append((name, t_cpu, t_gpu, t_cpu / t_gpu))
        print(f"{name:14s} cpu {t_cpu*1000:9.0f} ms  gpu {t_gpu*1000:8.1f} ms"
              f"  speedup {t_cpu/t_gpu:6.1f}x")


if __name__ == "__main__":
    main()
