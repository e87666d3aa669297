"""Artist similarity via per-artist diagonal GMMs.

Reference: /root/reference/tasks/artist_gmm_manager.py — per artist, fit
a diagonal GMM on the artist's track embeddings with the component count
(2..10) chosen by BIC (fit_best_gmm :59); artist-to-artist similarity is
a soft Chamfer distance over component-mean sets weighted by mixture
weights (gmm_soft_chamfer_distance :204); an IVF over flattened artist
descriptors accelerates candidate lookup (build_and_store_artist_index
:365, find_similar_artists :701).

Reference fits artists in a process pool; here artists batch onto the
GPU through cluster.algorithms.gmm_fit (torch EM), so one device fits
the whole catalogue.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, List, Optional, Tuple

import numpy as np
import torch

from audiomuse_amd import config as C
from audiomuse_amd.cluster.algorithms import gmm_fit


@dataclass
class ArtistModel:
    name: str
    means: np.ndarray          # (k, d)
    weights: np.ndarray        # (k,)
    n_tracks: int


def fit_best_gmm(x: torch.Tensor, min_k: Optional[int] = None,
                 max_k: Optional[int] = None, seed: int = 0):
    """BIC-selected diagonal GMM (artist_gmm_manager.py:59)."""
    n = x.shape[0]
    min_k = min_k or C.ARTIST_GMM_MIN_COMPONENTS
    max_k = max_k or C.ARTIST_GMM_MAX_COMPONENTS
    best = None
    best_bic = float("inf")
    for k in range(min_k, min(max_k, max(n, 1)) + 1):
        try:
            r = gmm_fit(x, k, seed=seed)
        except Exception:
            continue
        b = r.bic(n)
        if b < best_bic:
            best, best_bic = r, b
    if best is None:
        r = gmm_fit(x, 1, seed=seed)
        return r
    return best


def fit_artist(name: str, embeddings: np.ndarray, seed: int = 0) -> ArtistModel:
    x = torch.as_tensor(np.asarray(embeddings, dtype=np.float32))
    if x.shape[0] < 2:
        return ArtistModel(name=name, means=x.numpy().reshape(-1, x.shape[-1]),
                           weights=np.ones(max(x.shape[0], 1), dtype=np.float32),
                           n_tracks=x.shape[0])
    r = fit_best_gmm(x, seed=seed)
    return ArtistModel(name=name, means=r.means.cpu().numpy(),
                       weights=r.weights.cpu().numpy(), n_tracks=x.shape[0])


def soft_chamfer_distance(a: ArtistModel, b: ArtistModel) -> float:
    """Bidirectional weighted min cosine distance over component means
    (artist_gmm_manager.py:198-218)."""
    ma = np.asarray(a.means, dtype=np.float64)
    mb = np.asarray(b.means, dtype=np.float64)
    if ma.size == 0 or mb.size == 0:
        return 2.0
    na = ma / (np.linalg.norm(ma, axis=1, keepdims=True) + 1e-12)
    nb = mb / (np.linalg.norm(mb, axis=1, keepdims=True) + 1e-12)
    dist = 1.0 - np.clip(na @ nb.T, -1.0, 1.0)       # (ka, kb)
    wa = np.asarray(a.weights, dtype=np.float64)
    wa = wa / (wa.sum() + 1e-12)
    wb = np.asarray(b.weights, dtype=np.float64)
    wb = wb / (wb.sum() + 1e-12)
    fwd = float((dist.min(axis=1) * wa).sum())
    bwd = float((dist.min(axis=0) * wb).sum())
    return 0.5 * (fwd + bwd)


class ArtistSimilarity:
    """Catalogue of fitted artist models + similarity queries."""

    def __init__(self):
        self.models: Dict[str, ArtistModel] = {}
        self._centroids: Optional[np.ndarray] = None
        self._names: List[str] = []

    def fit_catalogue(self, per_artist: Dict[str, np.ndarray],
                      seed: int = 0) -> None:
        """Per-artist BIC-selected GMMs. On CPU the independent fits run
        on an INDEX_BUILD_WORKERS thread pool (the reference's process
        pool, artist_gmm_manager.py:219-336); on GPU all artists batch
        through ONE masked EM per k — per-artist GPU
        fits are kernel-launch bound at these shapes."""
        import torch as _torch

        from audiomuse_amd import config as C

        items = [(n, e) for n, e in per_artist.items() if len(e) > 0]
        if items and _torch.cuda.is_available() and len(items) >= 8:
            self._fit_catalogue_batched(items, seed=seed, device="cuda")
        elif items and not _torch.cuda.is_available() \
                and C.INDEX_BUILD_WORKERS > 1 and len(items) > 2:
            from concurrent.futures import ThreadPoolExecutor
            with ThreadPoolExecutor(
                    max_workers=C.INDEX_BUILD_WORKERS) as pool:
                fitted = list(pool.map(
                    lambda ne: (ne[0], fit_artist(ne[0], ne[1], seed=seed)),
                    items))
            for name, model in fitted:
                self.models[name] = model
        else:
            for name, embs in items:
                self.models[name] = fit_artist(name, embs, seed=seed)
        self._names = list(self.models)
        cents = [m.means.mean(axis=0) for m in self.models.values()]
        self._centroids = (np.stack(cents).astype(np.float32)
                           if cents else None)

    def _fit_catalogue_batched(self, items, seed: int = 0,
                               device: str = "cuda",
                               chunk_size: int = 512) -> None:
        """BIC selection with one gmm_fit_many call per (chunk, k):
        artists sorted by track count and chunked so padding stays
        bounded; every artist keeps its lowest-BIC k."""
        import torch as _torch

        from audiomuse_amd import config as C
        from audiomuse_amd.cluster.algorithms import gmm_fit_many

        singles = [(n, e) for n, e in items if len(e) < 2]
        for name, e in singles:
            self.models[name] = fit_artist(name, e, seed=seed)
        big = sorted(((n, np.asarray(e, dtype=np.float32))
                      for n, e in items if len(e) >= 2),
                     key=lambda ne: ne[1].shape[0])
        min_k, max_k = (C.ARTIST_GMM_MIN_COMPONENTS,
                        C.ARTIST_GMM_MAX_COMPONENTS)
        for c0 in range(0, len(big), chunk_size):
            chunk = big[c0 : c0 + chunk_size]
            names = [n for n, _ in chunk]
            xs = [_torch.from_numpy(e) for _, e in chunk]
            ns = [int(x.shape[0]) for x in xs]
            best: Dict[int, Tuple[float, np.ndarray, np.ndarray]] = {}
            for k in range(min_k, max_k + 1):
                eligible = [i for i, n_i in enumerate(ns) if n_i >= k]
                if not eligible:
                    break
                means, weights, bic = gmm_fit_many(
                    [xs[i] for i in eligible], k, seed=seed, device=device)
                mc, wc, bc = (means.cpu().numpy(), weights.cpu().numpy(),
                              bic.cpu().numpy())
                for j, i in enumerate(eligible):
                    if i not in best or float(bc[j]) < best[i][0]:
                        best[i] = (float(bc[j]), mc[j], wc[j])
            # artists whose n is below min_k entirely: k=1 fit
            leftover = [i for i in range(len(chunk)) if i not in best]
            if leftover:
                means, weights, bic = gmm_fit_many(
                    [xs[i] for i in leftover], 1, seed=seed, device=device)
                mc, wc, bc = (means.cpu().numpy(), weights.cpu().numpy(),
                              bic.cpu().numpy())
                for j, i in enumerate(leftover):
                    best[i] = (float(bc[j]), mc[j], wc[j])
            for i, (_, m, w) in best.items():
                self.models[names[i]] = ArtistModel(
                    name=names[i], means=m, weights=w, n_tracks=ns[i])

    def find_similar_artists(self, name: str, n: int = 10,
                             candidates: int = 100) -> List[Tuple[str, float]]:
        """Coarse centroid-cosine prefilter, exact soft-Chamfer re-rank
        (find_similar_artists :701)."""
        model = self.models.get(name)
        if model is None or self._centroids is None:
            return []
        q = model.means.mean(axis=0)
        cn = self._centroids / (np.linalg.norm(self._centroids, axis=1,
                                               keepdims=True) + 1e-12)
        qn = q / (np.linalg.norm(q) + 1e-12)
        coarse = 1.0 - cn @ qn
        order = np.argsort(coarse)[: candidates + 1]
        scored = []
        for i in order:
            other = self._names[int(i)]
            if other == name:
                continue
            scored.append((other, soft_chamfer_distance(model, self.models[other])))
        scored.sort(key=lambda t: t[1])
        return scored[:n]
