"""Radius walk: greedy anchored playlist walk.

Reference behavior (/root/reference/tasks/radius_walk_helper.py:204-349):
candidates sorted by distance-to-anchor are split into buckets of
BUCKET_SIZE=50; the walk consumes buckets near-to-far (window of
max(3, ceil(n/50)) buckets, doubling when starved), inside each bucket
greedily stepping to the candidate nearest the previous pick, with an
optional per-artist cap; finally a no-3-same-artist-in-a-row pass
reorders adjacent runs.
"""

from __future__ import annotations

import math
from typing import Callable, Dict, List, Optional

import numpy as np

BUCKET_SIZE = 50  # legacy constant; RADIUS_WALK_BUCKETS overrides


def _bucket_size(n_cands: int) -> int:
    """Candidates split into ~RADIUS_WALK_BUCKETS buckets (reference
    RADIUS_WALK_BUCKETS; the 50-item bucket is its shape at the default
    candidate count)."""
    from audiomuse_amd import config as C
    k = max(int(C.RADIUS_WALK_BUCKETS), 1)
    return max(1, (n_cands + k - 1) // k)


def _cosine_distance(a: np.ndarray, b: np.ndarray) -> float:
    denom = float(np.linalg.norm(a) * np.linalg.norm(b))
    if denom <= 0:
        return 1.0
    return float(np.clip(1.0 - float(np.dot(a, b)) / denom, 0.0, 2.0))


def avoid_triple_adjacent(ids: List[str],
                          id_to_author: Dict[str, Optional[str]]) -> List[str]:
    """Break runs of 3+ same-artist tracks by swapping ahead."""
    out = list(ids)
    for i in range(2, len(out)):
        a = id_to_author.get(out[i])
        if a is None:
            continue
        if id_to_author.get(out[i - 1]) == a and id_to_author.get(out[i - 2]) == a:
            for j in range(i + 1, len(out)):
                if id_to_author.get(out[j]) != a:
                    out[i], out[j] = out[j], out[i]
                    break
    return out


def execute_radius_walk(candidates: List[Dict], n: int,
                        eliminate_duplicates: bool = False,
                        max_songs_per_artist: Optional[int] = None,
                        get_distance_fn: Optional[Callable] = None
                        ) -> List[Dict]:
    """candidates: [{item_id, vector (np f32), dist_anchor, author?}].
    Returns [{item_id, distance}] of length <= n."""
    if not candidates:
        return []
    dist_fn = get_distance_fn or _cosine_distance
    cands = sorted(candidates, key=lambda c: c["dist_anchor"])
    bs = _bucket_size(len(cands))
    buckets = [cands[i : i + bs]
               for i in range(0, len(cands), bs)]

    cap_active = bool(eliminate_duplicates and max_songs_per_artist
                      and max_songs_per_artist > 0)
    picked: List[str] = [cands[0]["item_id"]]
    used = {cands[0]["item_id"]}
    prev_vec = np.asarray(cands[0]["vector"], dtype=np.float32)
    artist_counts: Dict[str, int] = {}
    first_author = cands[0].get("author")
    if first_author:
        artist_counts[first_author] = 1

    window = max(3, math.ceil(n / max(bs, 1)))
    processed = 0
    # RADIUS_INSTRUMENTATION (reference radius_walk_helper.py:18-32):
    # per-bucket pick counts logged for walk tuning
    from audiomuse_amd import config as C
    instrument = C.RADIUS_INSTRUMENTATION
    bucket_picks: Dict[int, int] = {}
    while len(picked) < n and processed < len(buckets):
        target = min(len(buckets), window)
        for bi in range(processed, target):
            for _ in range(len(buckets[bi])):
                best = None
                best_d = float("inf")
                for c in buckets[bi]:
                    if c["item_id"] in used:
                        continue
                    author = c.get("author")
                    if cap_active and author and \
                            artist_counts.get(author, 0) >= max_songs_per_artist:
                        continue
                    d = dist_fn(prev_vec, np.asarray(c["vector"], dtype=np.float32))
                    if d < best_d:
                        best, best_d = c, d
                if best is None:
                    break
                picked.append(best["item_id"])
                if instrument:
                    bucket_picks[bi] = bucket_picks.get(bi, 0) + 1
                used.add(best["item_id"])
                prev_vec = np.asarray(best["vector"], dtype=np.float32)
                author = best.get("author")
                if author:
                    artist_counts[author] = artist_counts.get(author, 0) + 1
                if len(picked) >= n:
                    break
            processed += 1
            if len(picked) >= n:
                break
        if len(picked) < n and target < len(buckets):
            window = min(len(buckets), max(window + 1, window * 2))

    if instrument:
        import logging
        logging.getLogger(__name__).info(
            "radius walk: %d picked over %d/%d buckets (size %d): %s",
            len(picked), len(bucket_picks), len(buckets), bs,
            dict(sorted(bucket_picks.items())))
    id_to_author = {c["item_id"]: c.get("author") for c in cands}
    picked = avoid_triple_adjacent(picked, id_to_author)[:n]
    dist_map = {c["item_id"]: float(c["dist_anchor"]) for c in cands}
    return [{"item_id": i, "distance": dist_map[i]} for i in picked]
