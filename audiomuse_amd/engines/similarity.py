"""Audio-index query engine.

Reference: /root/reference/tasks/ivf_manager.py — query-side feature
logic over the IVF index: over-fetch + exact-f32 re-rank (inside
IVFIndex.query here), near-duplicate filtering with a lookback window,
mood filtering, per-artist caps, radius-walk mode, multi-query union,
and a small result cache. This engine is shared by similar-song, song
path, alchemy, SemGrove and sonic-fingerprint features.
"""

from __future__ import annotations

import threading
import time
from collections import OrderedDict
from typing import Callable, Dict, List, Optional, Sequence, Tuple

import numpy as np
import torch

from audiomuse_amd import config as C
from audiomuse_amd.engines.radius_walk import execute_radius_walk
from audiomuse_amd.index.ivf import IVFIndex


class ResultCache:
    """TTL LRU cache (reference: ivf_manager._ResultCache :73)."""

    def __init__(self, max_items: Optional[int] = None,
                 ttl: Optional[float] = None):
        if max_items is None:
            max_items = C.IVF_RESULT_CACHE_MAX
        if ttl is None:
            ttl = C.IVF_RESULT_CACHE_SECONDS
        self.max_items = max_items
        self.ttl = ttl
        self._data: OrderedDict = OrderedDict()
        self._lock = threading.Lock()

    def get(self, key):
        with self._lock:
            hit = self._data.get(key)
            if hit is None:
                return None
            ts, value = hit
            if time.monotonic() - ts > self.ttl:
                del self._data[key]
                return None
            self._data.move_to_end(key)
            return value

    def put(self, key, value):
        with self._lock:
            self._data[key] = (time.monotonic(), value)
            self._data.move_to_end(key)
            while len(self._data) > self.max_items:
                self._data.popitem(last=False)

    def clear(self):
        with self._lock:
            self._data.clear()


MetaFn = Callable[[str], Optional[Dict]]   # item_id -> {title, author, mood_vector}


class SimilarityEngine:
    def __init__(self, index: IVFIndex, ids: Sequence[str],
                 meta_fn: Optional[MetaFn] = None):
        """index: built IVFIndex whose int64 ids are positions into `ids`
        (the canonical string item_ids)."""
        self.index = index
        self.item_ids = list(ids)
        self.pos = {s: i for i, s in enumerate(self.item_ids)}
        self.meta_fn = meta_fn or (lambda _id: None)
        self.cache = ResultCache()

    # -- vector resolution (ivf_manager._resolve_neighbor_query_vector) ---

    def vector_for_id(self, item_id: str) -> Optional[torch.Tensor]:
        p = self.pos.get(item_id)
        if p is None:
            return None
        return self.index.vector_for_id(p)

    def max_distance_for_id(self, item_id: str) -> Optional[Dict]:
        """Distance to the farthest catalogue track (reference:
        get_max_distance_for_id ivf_manager.py:1177 — UI slider
        normalization). Cached per item."""
        hit = self.cache.get(("maxd", item_id))
        if hit is not None:
            return dict(hit)
        vec = self.vector_for_id(item_id)
        if vec is None:
            return None
        d, fid = self.index.max_distance(vec)
        out = {"max_distance": d, "farthest_item_id": self.item_ids[fid]}
        self.cache.put(("maxd", item_id), dict(out))
        return out

    # -- core query ------------------------------------------------------

    def _query_candidates(self, vec: torch.Tensor, fetch: int,
                          nprobe: Optional[int] = None
                          ) -> List[Tuple[str, float]]:
        dist, ids = self.index.query(vec, k=fetch, nprobe=nprobe)
        out = []
        for d, i in zip(dist.tolist(), ids.tolist()):
            if i < 0 or not np.isfinite(d):
                continue
            out.append((self.item_ids[int(i)], float(d)))
        return out

    def _apply_filters(self, cands: List[Tuple[str, float]], n: int, *,
                       exclude: Sequence[str] = (),
                       eliminate_duplicates: Optional[bool] = None,
                       max_per_artist: Optional[int] = None,
                       mood_filter: Optional[str] = None
                       ) -> List[Tuple[str, float]]:
        """Near-dup lookback filter + artist cap + mood filter
        (ivf_manager.py:419-502, 652, 935)."""
        if eliminate_duplicates is None:
            eliminate_duplicates = C.SIMILARITY_ELIMINATE_DUPLICATES_DEFAULT
        lookback = C.DUPLICATE_DISTANCE_CHECK_LOOKBACK
        # threshold follows the index metric (reference keeps separate
        # cosine/euclidean knobs, ivf_manager.py:419)
        thresh = (C.DUPLICATE_DISTANCE_THRESHOLD_EUCLIDEAN
                  if self.index.metric == "euclidean"
                  else C.DUPLICATE_DISTANCE_THRESHOLD_COSINE)
        exclude_set = set(exclude)
        accepted: List[Tuple[str, float]] = []
        accepted_vecs: List[torch.Tensor] = []
        artist_counts: Dict[str, int] = {}
        # seed-mood similarity gate (reference MOOD_SIMILARITY_ENABLE:
        # neighbors must share the seed's mood profile, not just its
        # embedding neighborhood)
        seed_moods = None
        if C.MOOD_SIMILARITY_ENABLE and getattr(self, "_seed_moods", None):
            seed_moods = self._seed_moods
            seed_top = max(seed_moods, key=seed_moods.get)
            top_keys = sorted(seed_moods, key=seed_moods.get)[-5:]
        for item_id, dist in cands:
            if item_id in exclude_set:
                continue
            meta = self.meta_fn(item_id) or {}
            if mood_filter:
                moods = meta.get("mood_vector") or {}
                if moods and moods.get(mood_filter, 0.0) <= 0.0:
                    continue
            if seed_moods:
                moods = meta.get("mood_vector") or {}
                if moods:
                    if moods.get(seed_top, 0.0) < C.MOOD_SCORE_MATCH_THRESHOLD:
                        continue
                    drift = sum(abs(moods.get(m, 0.0) - seed_moods[m])
                                for m in top_keys) / max(len(top_keys), 1)
                    if drift > C.MOOD_SIMILARITY_THRESHOLD:
                        continue
            author = (meta.get("author") or "").strip().lower()
            cap = max_per_artist if max_per_artist is not None else C.MAX_SONGS_PER_ARTIST
            if cap and author and artist_counts.get(author, 0) >= cap:
                continue
            if eliminate_duplicates:
                vec = self.vector_for_id(item_id)
                dup = False
                if vec is not None:
                    for prev in accepted_vecs[-lookback:]:
                        cos = torch.dot(vec, prev) / (
                            vec.norm() * prev.norm() + 1e-12)
                        if 1.0 - float(cos) < thresh:
                            dup = True
                            break
                if dup:
                    continue
                if vec is not None:
                    accepted_vecs.append(vec)
            if author:
                artist_counts[author] = artist_counts.get(author, 0) + 1
            accepted.append((item_id, dist))
            if len(accepted) >= n:
                break
        return accepted

    def find_similar_by_vector(self, vec: torch.Tensor, n: int, *,
                               exclude: Sequence[str] = (),
                               eliminate_duplicates: Optional[bool] = None,
                               max_per_artist: Optional[int] = None,
                               mood_filter: Optional[str] = None,
                               radius: bool = False,
                               nprobe: Optional[int] = None
                               ) -> List[Dict]:
        fetch = max(n * 4 + len(exclude), 32)
        cands = self._query_candidates(vec, fetch, nprobe=nprobe)
        if radius:
            nprobe = nprobe or C.IVF_MAX_DISTANCE_NPROBE  # wider probe
            cands = self._query_candidates(vec, fetch, nprobe=nprobe)
            cdata = []
            for item_id, dist in cands:
                if item_id in set(exclude):
                    continue
                v = self.vector_for_id(item_id)
                if v is None:
                    continue
                meta = self.meta_fn(item_id) or {}
                cdata.append({"item_id": item_id, "vector": v.cpu().numpy(),
                              "dist_anchor": dist,
                              "author": meta.get("author")})
            walked = execute_radius_walk(
                cdata, n, eliminate_duplicates=eliminate_duplicates,
                max_songs_per_artist=max_per_artist or C.MAX_SONGS_PER_ARTIST
                or None)
            return walked
        picked = self._apply_filters(
            cands, n, exclude=exclude,
            eliminate_duplicates=eliminate_duplicates,
            max_per_artist=max_per_artist, mood_filter=mood_filter)
        return [{"item_id": i, "distance": d} for i, d in picked]

    def find_similar_by_id(self, item_id: str, n: int, **kw) -> List[Dict]:
        """reference: find_nearest_neighbors_by_id (ivf_manager.py:994)."""
        key = (item_id, n, tuple(sorted(kw.items())))
        hit = self.cache.get(key)
        if hit is not None:
            return hit
        vec = self.vector_for_id(item_id)
        if vec is None:
            return []
        kw.setdefault("exclude", (item_id,))
        # stash the seed's mood profile for the MOOD_SIMILARITY_ENABLE
        # gate (id-anchored queries only — raw vectors carry no moods)
        self._seed_moods = (self.meta_fn(item_id) or {}).get("mood_vector") \
            if C.MOOD_SIMILARITY_ENABLE else None
        try:
            out = self.find_similar_by_vector(vec, n, **kw)
        finally:
            self._seed_moods = None
        self.cache.put(key, out)
        return out

    def multi_query(self, vectors: Sequence[torch.Tensor], n: int, *,
                    exclude: Sequence[str] = (), **kw) -> List[Dict]:
        """Union of per-vector queries, best distance per id, re-sorted
        (reference: multi_query_ids, ivf_manager.py:389)."""
        best: Dict[str, float] = {}
        for vec in vectors:
            for item_id, dist in self._query_candidates(vec, n * 4):
                if dist < best.get(item_id, float("inf")):
                    best[item_id] = dist
        merged = sorted(best.items(), key=lambda kv: kv[1])
        picked = self._apply_filters(merged, n, exclude=exclude, **kw)
        return [{"item_id": i, "distance": d} for i, d in picked]


def build_engine_from_matrix(matrix: np.ndarray, ids: Sequence[str],
                             metric: str = "angular",
                             storage: Optional[str] = None,
                             device: str = "cpu",
                             meta_fn: Optional[MetaFn] = None,
                             nlist: Optional[int] = None) -> SimilarityEngine:
    x = torch.as_tensor(np.asarray(matrix, dtype=np.float32))
    index = IVFIndex.build(x, metric=metric, storage=storage, device=device,
                           nlist=nlist)
    return SimilarityEngine(index, ids, meta_fn=meta_fn)
