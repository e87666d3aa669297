"""Clustering algorithms on torch (GPU via rocBLAS GEMMs on ROCm).

Replaces both the reference's sklearn CPU path
(/root/reference/tasks/clustering_helper.py:615-688) and its optional
RAPIDS cuML wrappers (/root/reference/tasks/clustering_gpu.py:93-364 —
GPUKMeans/GPUDBSCAN/GPUPCA/GPUGaussianMixture/GPUSpectralClustering)
with first-party implementations that run on MI355X through PyTorch-ROCm.
The distance cores are GEMM-shaped (|x|^2 - 2xc + |c|^2 expansions ->
hipBLASLt); scatter reductions use index_add_. Everything accepts a
`device` and falls back to CPU transparently (used by the unit tests).

API mirrors sklearn's fit_predict contract so the evolutionary driver
(cluster/evolve.py) can swap algorithms by name, like the reference's
get_clustering_model factory (clustering_gpu.py:364).
"""

from __future__ import annotations

import math
from dataclasses import dataclass
from typing import Dict, Optional, Tuple

import torch


def _pairwise_sq(x: torch.Tensor, y: torch.Tensor, chunk: int = 65536
                 ) -> torch.Tensor:
    """Full (n, m) squared distances; callers chunk when n*m is large."""
    x_sq = x.square().sum(dim=1, keepdim=True)
    y_sq = y.square().sum(dim=1)
    return (x_sq - 2.0 * (x @ y.T) + y_sq).clamp_(min=0.0)


@dataclass
class KMeansResult:
    labels: torch.Tensor            # (n,) long
    centers: torch.Tensor           # (k, d) f32
    inertia: float


def kmeans_fit(x: torch.Tensor, k: int, iters: int = 50, seed: int = 0,
               tol: float = 1e-5) -> KMeansResult:
    """Full-batch Lloyd k-means with k-means++-lite seeding (distant
    sampling), GEMM assignment + index_add update."""
    x = x.float()
    n, d = x.shape
    k = max(1, min(k, n))
    g = torch.Generator(device="cpu").manual_seed(seed)
    # seeding: random first, then farthest-point sampling on a subsample
    first = int(torch.randint(0, n, (1,), generator=g))
    centers = [x[first]]
    sub = x[torch.randperm(n, generator=g)[: min(n, 4096)].to(x.device)]
    dist = (sub - centers[0]).square().sum(dim=1)
    for _ in range(1, k):
        centers.append(sub[int(dist.argmax())])
        dist = torch.minimum(dist, (sub - centers[-1]).square().sum(dim=1))
    C = torch.stack(centers)
    prev_inertia = float("inf")
    labels = torch.zeros(n, dtype=torch.long, device=x.device)
    for _ in range(iters):
        dists = _pairwise_sq(x, C)
        labels = dists.argmin(dim=1)
        inertia = float(dists.gather(1, labels.unsqueeze(1)).sum())
        sums = torch.zeros_like(C)
        cnts = torch.zeros(C.shape[0], device=x.device)
        sums.index_add_(0, labels, x)
        cnts.index_add_(0, labels, torch.ones_like(labels, dtype=torch.float))
        nz = cnts > 0
        C[nz] = sums[nz] / cnts[nz].unsqueeze(1)
        # re-seed empty clusters at the farthest points
        if (~nz).any():
            far = dists.gather(1, labels.unsqueeze(1)).squeeze(1).argsort(descending=True)
            C[~nz] = x[far[: int((~nz).sum())]]
        if abs(prev_inertia - inertia) <= tol * max(1.0, abs(prev_inertia)):
            break
        prev_inertia = inertia
    # final assignment against the updated centers
    dists = _pairwise_sq(x, C)
    labels = dists.argmin(dim=1)
    inertia = float(dists.gather(1, labels.unsqueeze(1)).sum())
    return KMeansResult(labels=labels, centers=C, inertia=inertia)


def dbscan_fit(x: torch.Tensor, eps: float, min_samples: int,
               chunk: int = 8192) -> torch.Tensor:
    """DBSCAN via chunked GEMM region queries + BFS core expansion.
    Returns (n,) long labels, -1 = noise (sklearn semantics)."""
    x = x.float()
    n = x.shape[0]
    eps_sq = eps * eps
    device = x.device
    # neighbor counts + core mask (chunked full scan)
    neighbor_rows = []
    counts = torch.zeros(n, dtype=torch.long, device=device)
    for s in range(0, n, chunk):
        d = _pairwise_sq(x[s : s + chunk], x)
        mask = d <= eps_sq
        counts[s : s + chunk] = mask.sum(dim=1)
        neighbor_rows.append(mask)
    adj = torch.cat(neighbor_rows, dim=0)      # (n, n) bool; subset-sized inputs
    core = counts >= min_samples
    # connected components over core-core edges via min-label propagation
    # with path halving — fully vectorized (the per-cluster BFS frontier
    # loop was the CPU-parity bottleneck: profiles/r01_kernel_pmc.md)
    INF = n
    labels = torch.where(core, torch.arange(n, device=device),
                         torch.full((n,), INF, device=device))
    adj_cc = adj & core.unsqueeze(0) & core.unsqueeze(1)
    big = torch.full((n, n), INF, device=device, dtype=torch.long)
    for _ in range(n):  # converges in O(log diameter) with halving
        nb = torch.where(adj_cc, labels.unsqueeze(0).expand(n, n), big)
        new = torch.minimum(labels, nb.min(dim=1).values)
        safe = torch.clamp(new, max=n - 1)
        new = torch.minimum(new, torch.where(new < INF, new[safe], new))
        if torch.equal(new, labels):
            break
        labels = new
    # border points take a core neighbor's component; isolated stay noise
    nb_core = torch.where(adj & core.unsqueeze(0),
                          labels.unsqueeze(0).expand(n, n), big)
    border_lab = nb_core.min(dim=1).values
    labels = torch.where(core, labels,
                         torch.where(border_lab < INF, border_lab,
                                     torch.full_like(labels, INF)))
    # compress to sklearn-style 0..k-1 in first-appearance order; noise -1
    out = torch.full((n,), -1, dtype=torch.long, device=device)
    valid = labels < INF
    if bool(valid.any()):
        roots = labels[valid]
        uniq = []
        seen = set()
        for r in roots.tolist():
            if r not in seen:
                seen.add(r)
                uniq.append(r)
        remap = torch.full((n,), -1, dtype=torch.long, device=device)
        remap[torch.tensor(uniq, device=device)] = torch.arange(
            len(uniq), device=device)
        out[valid] = remap[roots]
    return out


def pca_fit_transform(x: torch.Tensor, n_components: int
                      ) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """(projected, components, mean). Small inputs use exact SVD
    (rocSOLVER); large ones use randomized subspace iteration (all-GEMM,
    Halko et al.) — the dense SVD was the GPU PCA bottleneck."""
    x = x.float()
    mean = x.mean(dim=0)
    xc = x - mean
    q = min(n_components, min(xc.shape) - 1)
    n, d = xc.shape
    if n * d <= 1_000_000 or q >= d // 2:
        U, S, Vh = torch.linalg.svd(xc, full_matrices=False)
        comps = Vh[:q]
        return xc @ comps.T, comps, mean
    # randomized range finder: oversample + 2 power iterations
    g = torch.Generator(device="cpu").manual_seed(0)
    p = min(d, q + 8)
    omega = torch.randn(d, p, generator=g).to(xc.device)
    y = xc @ omega
    for _ in range(2):
        y = xc @ (xc.T @ y)
        y, _ = torch.linalg.qr(y)
    b = y.T @ xc                                  # (p, d)
    _, _, Vh = torch.linalg.svd(b, full_matrices=False)
    comps = Vh[:q]
    return xc @ comps.T, comps, mean


@dataclass
class GMMResult:
    labels: torch.Tensor
    means: torch.Tensor            # (k, d)
    variances: torch.Tensor        # (k, d) diagonal
    weights: torch.Tensor          # (k,)
    log_likelihood: float

    def bic(self, n: int) -> float:
        k, d = self.means.shape
        n_params = k * d * 2 + (k - 1)
        return -2.0 * self.log_likelihood + n_params * math.log(max(n, 2))


def gmm_fit(x: torch.Tensor, k: int, iters: int = 60, seed: int = 0,
            reg: float = 1e-6, tol: float = 1e-4,
            covariance_type: str = "diag") -> GMMResult:
    """Diagonal-covariance EM (reference: artist_gmm_manager.fit_best_gmm
    uses sklearn diag GMM; clustering uses GMM option). Batched over all
    components: responsibilities are one GEMM-shaped log-prob pass.
    covariance_type: only the reference default "diag" is implemented
    (GMM_COVARIANCE_TYPE); anything else fails loudly."""
    if covariance_type != "diag":
        raise ValueError(f"covariance_type {covariance_type!r} not "
                         "implemented (diag only, the reference default)")
    x = x.float()
    n, d = x.shape
    k = max(1, min(k, n))
    km = kmeans_fit(x, k, iters=10, seed=seed)
    means = km.centers.clone()
    var = x.var(dim=0, unbiased=False).clamp(min=reg).repeat(k, 1)
    weights = torch.full((k,), 1.0 / k, device=x.device)
    prev_ll = -float("inf")
    ll = prev_ll
    for _ in range(iters):
        # log N(x | m, diag v) = -0.5 [ d log 2pi + sum log v + sum (x-m)^2/v ]
        inv = 1.0 / var                                   # (k, d)
        x_sq = x.square() @ inv.T                         # (n, k)
        cross = x @ (means * inv).T                       # (n, k)
        m_sq = (means.square() * inv).sum(dim=1)          # (k,)
        quad = x_sq - 2 * cross + m_sq
        logdet = torch.log(var).sum(dim=1)
        logp = -0.5 * (d * math.log(2 * math.pi) + logdet + quad)
        logp = logp + torch.log(weights + 1e-12)
        lse = torch.logsumexp(logp, dim=1)
        ll = float(lse.mean())
        resp = torch.exp(logp - lse.unsqueeze(1))         # (n, k)
        nk = resp.sum(dim=0).clamp(min=1e-10)
        means = (resp.T @ x) / nk.unsqueeze(1)
        ex2 = (resp.T @ x.square()) / nk.unsqueeze(1)
        var = (ex2 - means.square()).clamp(min=reg)
        weights = nk / n
        if abs(ll - prev_ll) < tol:
            break
        prev_ll = ll
    labels = (torch.log(weights + 1e-12) - 0.5 * (
        torch.log(var).sum(dim=1) + _mahalanobis_diag(x, means, var))).argmax(dim=1)
    return GMMResult(labels=labels, means=means, variances=var,
                     weights=weights, log_likelihood=ll * n)


def gmm_fit_many(xs: "list[torch.Tensor]", k: int, iters: int = 60,
                 seed: int = 0, reg: float = 1e-6,
                 device: "Optional[str]" = None):
    """Diagonal-covariance EM for MANY small datasets AT ONCE.

    The artist-similarity index fits one BIC-selected GMM per artist
    (reference: artist_gmm_manager.py:59-177 on a process pool). Per-
    artist GPU fits are kernel-launch bound — each EM step on a
    (50, 200) dataset is microseconds of math. This runs every
    dataset's EM in one masked batch ("batched GPU GMM
    EM kernel across all artists at once"): all per-iteration work is
    three bmm's + elementwise over (A, N, k).

    xs: list of (n_i, d) f32 tensors, every n_i >= k.
    Returns (means (A,k,d), weights (A,k), bic (A,)) on `device`.
    """
    A = len(xs)
    d = xs[0].shape[1]
    dev = device or xs[0].device
    N = max(int(x.shape[0]) for x in xs)
    X = torch.zeros(A, N, d, device=dev)
    M = torch.zeros(A, N, dtype=torch.bool, device=dev)
    for i, x in enumerate(xs):
        X[i, : x.shape[0]] = x.float().to(dev)
        M[i, : x.shape[0]] = True
    counts = M.sum(dim=1)                                   # (A,)
    fM = M.float()

    # init: deterministic row sample per dataset, then masked Lloyd
    g = torch.Generator().manual_seed(seed)
    sel = torch.stack([
        torch.randperm(int(c), generator=g)[:k] if int(c) >= k
        else torch.zeros(k, dtype=torch.int64)
        for c in counts.cpu()]).to(dev)                     # (A, k)
    means = torch.gather(X, 1, sel.unsqueeze(-1).expand(A, k, d)).clone()
    for _ in range(10):
        dist = (X.square().sum(-1, keepdim=True)
                - 2 * X @ means.transpose(1, 2)
                + means.square().sum(-1).unsqueeze(1))      # (A, N, k)
        dist = dist.masked_fill(~M.unsqueeze(-1), float("inf"))
        onehot = torch.nn.functional.one_hot(
            dist.argmin(-1), k).float() * fM.unsqueeze(-1)
        cnt = onehot.sum(dim=1)                             # (A, k)
        new = (onehot.transpose(1, 2) @ X) / cnt.clamp(min=1).unsqueeze(-1)
        means = torch.where((cnt > 0).unsqueeze(-1), new, means)

    var = (((X - (X * fM.unsqueeze(-1)).sum(1, keepdim=True)
             / counts.clamp(min=1).reshape(A, 1, 1)).square()
            * fM.unsqueeze(-1)).sum(1)
           / counts.clamp(min=1).unsqueeze(-1)).clamp(min=reg)
    var = var.unsqueeze(1).expand(A, k, d).contiguous()     # (A, k, d)
    weights = torch.full((A, k), 1.0 / k, device=dev)
    log2pi = math.log(2 * math.pi)
    ll_per = torch.zeros(A, device=dev)
    for _ in range(iters):
        inv = 1.0 / var
        x_sq = X.square() @ inv.transpose(1, 2)             # (A, N, k)
        cross = X @ (means * inv).transpose(1, 2)
        m_sq = (means.square() * inv).sum(-1)               # (A, k)
        quad = x_sq - 2 * cross + m_sq.unsqueeze(1)
        logdet = torch.log(var).sum(-1)                     # (A, k)
        logp = (-0.5 * (d * log2pi + logdet.unsqueeze(1) + quad)
                + torch.log(weights + 1e-12).unsqueeze(1))
        lse = torch.logsumexp(logp, dim=-1)                 # (A, N)
        resp = torch.exp(logp - lse.unsqueeze(-1)) * fM.unsqueeze(-1)
        nk = resp.sum(dim=1).clamp(min=1e-10)               # (A, k)
        means = (resp.transpose(1, 2) @ X) / nk.unsqueeze(-1)
        ex2 = (resp.transpose(1, 2) @ X.square()) / nk.unsqueeze(-1)
        var = (ex2 - means.square()).clamp(min=reg)
        weights = nk / counts.clamp(min=1).unsqueeze(1)
        ll_per = (lse * fM).sum(dim=1)
    n_params = k * d * 2 + (k - 1)
    bic = -2.0 * ll_per + n_params * torch.log(
        counts.clamp(min=2).float())
    return means, weights, bic


def _mahalanobis_diag(x, means, var):
    inv = 1.0 / var
    return (x.square() @ inv.T - 2 * (x @ (means * inv).T)
            + (means.square() * inv).sum(dim=1))


def spectral_fit(x: torch.Tensor, k: int, n_neighbors: int = 10,
                 seed: int = 0) -> torch.Tensor:
    """Spectral clustering: kNN affinity -> normalized Laplacian ->
    eigenvectors (rocSOLVER eigh) -> k-means on the embedding."""
    x = x.float()
    n = x.shape[0]
    d = _pairwise_sq(x, x)
    knn = d.topk(min(n_neighbors + 1, n), dim=1, largest=False)
    sigma = knn.values[:, -1].sqrt().clamp(min=1e-6)
    aff = torch.zeros_like(d)
    scale = (sigma.unsqueeze(1) * sigma.unsqueeze(0)).clamp(min=1e-12)
    w = torch.exp(-d / scale)
    aff.scatter_(1, knn.indices, w.gather(1, knn.indices))
    aff = torch.maximum(aff, aff.T)
    deg = aff.sum(dim=1).clamp(min=1e-12)
    dm12 = deg.rsqrt()
    lap = torch.eye(n, device=x.device) - dm12.unsqueeze(1) * aff * dm12.unsqueeze(0)
    evals, evecs = torch.linalg.eigh(lap)
    emb = evecs[:, :k]
    emb = emb / emb.norm(dim=1, keepdim=True).clamp(min=1e-12)
    return kmeans_fit(emb, k, seed=seed).labels


def fit_predict(algorithm: str, x: torch.Tensor, params: Dict) -> Tuple[torch.Tensor, Optional[torch.Tensor]]:
    """Factory (reference: clustering_gpu.get_clustering_model).
    Returns (labels, centers-or-None)."""
    if algorithm == "kmeans":
        r = kmeans_fit(x, int(params.get("n_clusters", 8)),
                       seed=int(params.get("seed", 0)))
        return r.labels, r.centers
    if algorithm == "dbscan":
        labels = dbscan_fit(x, float(params.get("eps", 0.5)),
                            int(params.get("min_samples", 5)))
        return labels, _centers_from_labels(x, labels)
    if algorithm == "gmm":
        from audiomuse_amd import config as C
        r = gmm_fit(x, int(params.get("n_components", 8)),
                    seed=int(params.get("seed", 0)),
                    covariance_type=params.get("covariance_type",
                                               C.GMM_COVARIANCE_TYPE))
        return r.labels, r.means
    if algorithm == "spectral":
        from audiomuse_amd import config as C
        labels = spectral_fit(x, int(params.get("n_clusters", 8)),
                              n_neighbors=int(params.get(
                                  "n_neighbors", C.SPECTRAL_N_NEIGHBORS)),
                              seed=int(params.get("seed", 0)))
        return labels, _centers_from_labels(x, labels)
    raise ValueError(f"unknown algorithm {algorithm!r}")


def _centers_from_labels(x: torch.Tensor, labels: torch.Tensor) -> torch.Tensor:
    ids = labels[labels >= 0].unique()
    if ids.numel() == 0:
        return torch.zeros(0, x.shape[1], device=x.device)
    return torch.stack([x[labels == c].mean(dim=0) for c in ids.tolist()])
