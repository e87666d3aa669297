"""Song path: interpolated journey between two songs.

Reference: /root/reference/tasks/path_manager.py (707 LoC;
docs/ALGORITHM.md:1363-1365) — linear or slerp interpolation between the
endpoint vectors produces k waypoint centroids; each waypoint is
resolved to its nearest unused track (dedupe + artist caps); adjacent
duplicates merge.
"""

from __future__ import annotations

import math
from typing import Dict, List, Optional

import numpy as np
import torch

from audiomuse_amd.engines.similarity import SimilarityEngine


def interpolate(a: np.ndarray, b: np.ndarray, k: int,
                mode: str = "slerp") -> np.ndarray:
    """k interior waypoints between unit vectors a and b (k, d)."""
    a = np.asarray(a, dtype=np.float64)
    b = np.asarray(b, dtype=np.float64)
    an = a / (np.linalg.norm(a) + 1e-12)
    bn = b / (np.linalg.norm(b) + 1e-12)
    ts = np.linspace(0.0, 1.0, k + 2)[1:-1]
    if mode == "linear":
        pts = np.stack([(1 - t) * an + t * bn for t in ts])
    else:
        cos = float(np.clip(np.dot(an, bn), -1.0, 1.0))
        omega = math.acos(cos)
        if omega < 1e-6:
            pts = np.stack([an for _ in ts])
        else:
            so = math.sin(omega)
            pts = np.stack([
                (math.sin((1 - t) * omega) / so) * an
                + (math.sin(t * omega) / so) * bn for t in ts])
    pts /= np.linalg.norm(pts, axis=1, keepdims=True) + 1e-12
    return pts.astype(np.float32)


def find_path(engine: SimilarityEngine, start_id: str, end_id: str,
              length: Optional[int] = None, mode: Optional[str] = None,
              max_per_artist: Optional[int] = None) -> List[Dict]:
    """Path of ~`length` tracks from start to end (path_manager entry).
    Defaults from config: PATH_DEFAULT_LENGTH; interpolation follows
    PATH_DISTANCE_METRIC (angular -> slerp on the unit sphere,
    euclidean/dot -> linear); PATH_FIX_SIZE backfills starved waypoints
    so the playlist comes out at exactly `length` (the reference's
    merge-on-starve behavior)."""
    from audiomuse_amd import config as C
    if length is None:
        length = C.PATH_DEFAULT_LENGTH
    if mode is None:
        mode = "slerp" if C.PATH_DISTANCE_METRIC == "angular" else "linear"
    va = engine.vector_for_id(start_id)
    vb = engine.vector_for_id(end_id)
    if va is None or vb is None:
        return []
    k = max(0, length - 2)
    waypoints = interpolate(va.cpu().numpy(), vb.cpu().numpy(), k, mode=mode)
    used = {start_id, end_id}
    artist_counts: Dict[str, int] = {}
    path = [{"item_id": start_id, "distance": 0.0}]
    for wp in waypoints:
        cands = engine.find_similar_by_vector(
            torch.from_numpy(wp), 10, exclude=tuple(used))
        picked = None
        for c in cands:
            meta = engine.meta_fn(c["item_id"]) or {}
            author = (meta.get("author") or "").strip().lower()
            if max_per_artist and author and \
                    artist_counts.get(author, 0) >= max_per_artist:
                continue
            picked = c
            if author:
                artist_counts[author] = artist_counts.get(author, 0) + 1
            break
        if picked is None:
            continue  # waypoint merges into its neighbor
        used.add(picked["item_id"])
        path.append(picked)
    path.append({"item_id": end_id, "distance": 0.0})
    # PATH_FIX_SIZE: starved waypoints shortened the path — backfill by
    # re-querying the midpoints of the largest gaps until `length` is
    # met (reference: merge-on-starve keeps the playlist size fixed)
    if C.PATH_FIX_SIZE:
        guard = 0
        while len(path) < length and guard < length * 2:
            guard += 1
            grew = False
            for i in range(len(path) - 1):
                a = engine.vector_for_id(path[i]["item_id"])
                b = engine.vector_for_id(path[i + 1]["item_id"])
                if a is None or b is None:
                    continue
                mid = interpolate(a.cpu().numpy(), b.cpu().numpy(), 1,
                                  mode=mode)[0]
                cands = engine.find_similar_by_vector(
                    torch.from_numpy(mid), 5, exclude=tuple(used))
                if cands:
                    used.add(cands[0]["item_id"])
                    path.insert(i + 1, cands[0])
                    grew = True
                    if len(path) >= length:
                        break
            if not grew:
                break
    return path
