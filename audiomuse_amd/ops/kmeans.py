"""Mini-batch k-means for the IVF coarse quantizer.

Reference behavior: sklearn MiniBatchKMeans with init='random', batch
10 000, <=25 iters (/root/reference/tasks/paged_ivf.py:1411-1460 — the
code comments there record that k-means++ seeding was the cost, hence
random init).

MI355X-native re-expression: assignment is a chunked GEMM
(torch.matmul -> rocBLAS) + argmin; the centroid update is an
index_add_ scatter reduction. On multi-GPU builds each rank feeds its
shard and the per-iteration (sum, count) partials are all-reduced over
RCCL — handled by the caller passing a process
group.
"""

from __future__ import annotations

from typing import Optional

import torch


def _assign_chunked(x: torch.Tensor, centroids: torch.Tensor,
                    chunk: int = 65536) -> torch.Tensor:
    """argmin_j ||x_i - c_j||^2 via the |x|^2 - 2 x.c + |c|^2 expansion."""
    c_sq = centroids.square().sum(dim=1)  # (k,)
    out = torch.empty(x.shape[0], dtype=torch.long, device=x.device)
    for s in range(0, x.shape[0], chunk):
        xb = x[s : s + chunk]
        scores = xb @ centroids.T * -2.0 + c_sq
        out[s : s + chunk] = scores.argmin(dim=1)
    return out


def minibatch_kmeans(x: torch.Tensor, k: int, iters: int = 25,
                     batch: int = 10000, seed: int = 0,
                     group: Optional[object] = None) -> torch.Tensor:
    """Train k centroids on x (N, d) f32. Returns (k, d) f32 on x.device.

    group: optional torch.distributed process group; when given, every
    rank must call with its own shard and identical k/seed — centroid
    state stays replicated via all-reduced (sum, count) partials.
    """
    n, d = x.shape
    k = min(k, n) if group is None else k
    g = torch.Generator(device="cpu").manual_seed(seed)
    init_idx = torch.randperm(n, generator=g)[:k]
    centroids = x[init_idx.to(x.device)].clone().float()
    if group is not None:
        import torch.distributed as dist

        dist.broadcast(centroids, src=0, group=group)

    counts_ema = torch.zeros(k, device=x.device)
    for it in range(iters):
        bidx = torch.randint(0, n, (min(batch, n),), generator=g).to(x.device)
        xb = x[bidx].float()
        assign = _assign_chunked(xb, centroids)
        sums = torch.zeros_like(centroids)
        cnts = torch.zeros(k, device=x.device)
        sums.index_add_(0, assign, xb)
        cnts.index_add_(0, assign, torch.ones_like(assign, dtype=torch.float))
        if group is not None:
            import torch.distributed as dist

            dist.all_reduce(sums, group=group)
            dist.all_reduce(cnts, group=group)
        counts_ema += cnts
        nz = cnts > 0
        lr = (cnts[nz] / counts_ema[nz].clamp(min=1.0)).unsqueeze(1)
        centroids[nz] = centroids[nz] * (1 - lr) + (sums[nz] / cnts[nz].unsqueeze(1)) * lr
    return centroids


def assign_to_centroids(x: torch.Tensor, centroids: torch.Tensor,
                        chunk: int = 20000) -> torch.Tensor:
    """Full-library assignment in chunks (reference: paged_ivf.py:1462-1465
    predicts in 20 000-row chunks)."""
    return _assign_chunked(x.float(), centroids.float(), chunk=chunk)
