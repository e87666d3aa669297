"""Song alchemy: vector arithmetic playlists.

Reference: /root/reference/tasks/song_alchemy.py (1112 LoC) — add /
subtract centroids built from songs, artists (GMM component means),
moods, playlists and saved anchors; query the audio IVF with the
combined vector(s); filter results inside the subtract radius;
temperature-softmax sampling for variety (docs/ALGORITHM.md:1466-1471).
"""

from __future__ import annotations

from typing import Dict, List, Optional, Sequence

import numpy as np
import torch

from audiomuse_amd.engines.similarity import SimilarityEngine


def combine_vectors(add: Sequence[np.ndarray],
                    subtract: Sequence[np.ndarray] = (),
                    subtract_weight: float = 1.0) -> Optional[np.ndarray]:
    """Mean(add) - subtract_weight * mean(subtract), L2-normalized."""
    if not add:
        return None
    v = np.mean(np.stack([np.asarray(a, dtype=np.float32) for a in add]), axis=0)
    if subtract:
        s = np.mean(np.stack([np.asarray(a, dtype=np.float32) for a in subtract]),
                    axis=0)
        v = v - subtract_weight * s
    norm = float(np.linalg.norm(v))
    if norm <= 0:
        return None
    return v / norm


def subtract_radius_filter(results: List[Dict], engine: SimilarityEngine,
                           subtract: Sequence[np.ndarray],
                           radius: float) -> List[Dict]:
    """Drop results whose vector falls within `radius` (cosine distance)
    of any subtracted centroid (song_alchemy subtract-radius filter)."""
    if not subtract or radius <= 0:
        return results
    subs = [np.asarray(s, dtype=np.float32) for s in subtract]
    out = []
    for r in results:
        vec = engine.vector_for_id(r["item_id"])
        if vec is None:
            continue
        v = vec.cpu().numpy()
        near = False
        for s in subs:
            denom = float(np.linalg.norm(v) * np.linalg.norm(s))
            cos = float(np.dot(v, s)) / denom if denom > 0 else 0.0
            if 1.0 - cos < radius:
                near = True
                break
        if not near:
            out.append(r)
    return out


def temperature_sample(results: List[Dict], n: int, temperature: float,
                       seed: Optional[int] = None) -> List[Dict]:
    """softmax(-distance / T) weighted draw without replacement
    (docs/ALGORITHM.md:1466-1471). T <= 0 -> deterministic top-n."""
    if temperature <= 0 or len(results) <= n:
        return results[:n]
    d = np.array([r["distance"] for r in results], dtype=np.float64)
    logits = -d / temperature
    logits -= logits.max()
    p = np.exp(logits)
    p /= p.sum()
    rng = np.random.default_rng(seed)
    idx = rng.choice(len(results), size=n, replace=False, p=p)
    picked = [results[i] for i in sorted(idx, key=lambda i: d[i])]
    return picked


def alchemy_query(engine: SimilarityEngine, add: Sequence[np.ndarray],
                  subtract: Sequence[np.ndarray] = (), *,
                  n: Optional[int] = None,
                  subtract_radius: Optional[float] = None,
                  temperature: Optional[float] = None,
                  exclude: Sequence[str] = (), seed: Optional[int] = None,
                  **filters) -> List[Dict]:
    """Full alchemy pipeline: combine -> multi-query -> subtract-radius
    -> temperature sample. Defaults from config (reference ALCHEMY_*)."""
    from audiomuse_amd import config as C
    n = min(n if n is not None else C.ALCHEMY_DEFAULT_N_RESULTS,
            C.ALCHEMY_MAX_N_RESULTS)
    if subtract_radius is None:
        subtract_radius = C.ALCHEMY_SUBTRACT_RADIUS if subtract else 0.0
    if temperature is None:
        temperature = C.ALCHEMY_TEMPERATURE
    combined = combine_vectors(add, subtract)
    if combined is None:
        return []
    fetch = max(n * 3, 30)
    results = engine.find_similar_by_vector(
        torch.from_numpy(combined), fetch, exclude=exclude, **filters)
    results = subtract_radius_filter(results, engine, subtract, subtract_radius)
    return temperature_sample(results, n, temperature, seed=seed)
