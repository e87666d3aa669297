"""Hyperbolic explorer tree cache: build, skeleton, lazy warm/unload.

Reference: /root/reference/tasks/hyperbolic_manager.py:613-1495 — the
genre/mood hierarchy over the Poincare projection is precomputed into a
node tree (root -> genre folders -> cluster leaf folders -> track
items), persisted as TWO blobs: the FULL tree (every leaf's track
items) and a SKELETON (folder nodes only). Flask loads just the
skeleton at boot; the first request that needs leaf items lazily warms
the full tree, and a warm-cache timer (the same shape as the CLAP
text-model warm cycle) unloads it again after
HYPERBOLIC_TREE_WARMUP_DURATION so a 100k-track tree does not sit in
RAM between explorer sessions.

Structure re-expressed here: the catalogue has no genre tags, so the
first level is the predominant-mood folder (the reference falls back to
mood folders the same way, _build_mood_root_items); big folders split
into leaf clusters via k-means over the embedding space
(HYPERBOLIC_TARGET_LEAF_SIZE / HYPERBOLIC_MIN_CLUSTER_SIZE).
"""

from __future__ import annotations

import json
import threading
import time
import zlib
from typing import Callable, Dict, List, Optional

import torch

from audiomuse_amd import config as C
from audiomuse_amd.db.store import load_index_blob, store_index_blob
from audiomuse_amd.engines.hyperbolic import HyperbolicSpace

TREE_BLOB = "hyperbolic_tree_full"
SKELETON_BLOB = "hyperbolic_tree_skeleton"
TREE_VERSION = 2


def _leaf(node_id: str, name: str, members: List[Dict], mean_radius: float,
          nodes: Dict, flat_ids: Dict, kind: str = "cluster") -> Dict:
    node = {"id": node_id, "name": name, "type": "folder", "leaf": True,
            "kind": kind, "children_count": len(members),
            "summary": {"track_count": len(members),
                        "mean_radius": round(mean_radius, 4)},
            "items": members}
    nodes[node_id] = node
    flat_ids[node_id] = [m["item_id"] for m in members]
    return node


def build_tree(embeddings: torch.Tensor, item_ids: List[str],
               meta_fn: Callable[[str], Optional[Dict]]) -> Dict:
    """Full tree dict {version, n_bands, nodes, flat_ids, track_count}."""
    from audiomuse_amd.ops.kmeans import assign_to_centroids, minibatch_kmeans

    space = HyperbolicSpace(embeddings)
    radii = space.points.norm(dim=1)

    by_mood: Dict[str, List[int]] = {}
    metas = []
    for pos, item_id in enumerate(item_ids):
        meta = meta_fn(item_id) or {}
        metas.append(meta)
        moods = meta.get("mood_vector") or {}
        top = max(moods, key=moods.get) if moods else "unknown"
        by_mood.setdefault(top, []).append(pos)

    # HYPERBOLIC_RADIAL_SPREAD: display-space exponent that spreads the
    # dense outer shell of the ball (< 1 pushes mid-radii outward)
    spread = max(C.HYPERBOLIC_RADIAL_SPREAD, 1e-3)

    def track_item(pos: int) -> Dict:
        meta = metas[pos]
        return {"item_id": item_ids[pos], "type": "track",
                "title": meta.get("title", ""),
                "author": meta.get("author", ""),
                "radius": round(float(radii[pos]) ** spread, 4)}

    nodes: Dict[str, Dict] = {}
    flat_ids: Dict[str, List[str]] = {}
    root_items = []
    target = max(C.HYPERBOLIC_TARGET_LEAF_SIZE, 2)
    min_cluster = max(C.HYPERBOLIC_MIN_CLUSTER_SIZE, 1)

    for mood, members in sorted(by_mood.items(), key=lambda kv: -len(kv[1])):
        mood_id = f"mood:{mood}"
        if len(members) <= target:
            leaves = [members]
        else:
            k = min(max(2, (len(members) + target - 1) // target), 64)
            sub = embeddings[torch.tensor(members)].float()
            cents = minibatch_kmeans(sub, k, iters=8, seed=13)
            assign = assign_to_centroids(sub, cents)
            leaves, misc = [], []
            for c in range(k):
                idxs = [members[i] for i in
                        (assign == c).nonzero(as_tuple=True)[0].tolist()]
                (leaves if len(idxs) >= min_cluster else misc).append(idxs)
            leaves = [l for l in leaves if l]
            misc_flat = [i for l in misc for i in l]
            if misc_flat:
                leaves.append(misc_flat)
        child_summaries = []
        for li, idxs in enumerate(leaves):
            leaf_id = f"{mood_id}:{li}"
            items = sorted((track_item(p) for p in idxs),
                           key=lambda t: t["radius"])
            mean_r = float(radii[torch.tensor(idxs)].mean())
            _leaf(leaf_id, f"{mood.title()} · {li + 1}", items, mean_r,
                  nodes, flat_ids)
            child_summaries.append(
                {"id": leaf_id, "type": "folder", "leaf": True,
                 "name": f"{mood.title()} · {li + 1}",
                 "track_count": len(items)})
        nodes[mood_id] = {
            "id": mood_id, "name": mood.title(), "type": "folder",
            "leaf": False, "kind": "mood",
            "children_count": len(child_summaries),
            "summary": {"track_count": len(members)},
            "items": child_summaries}
        flat_ids[mood_id] = []
        root_items.append({"id": mood_id, "type": "folder", "leaf": False,
                           "name": mood.title(),
                           "track_count": len(members)})

    nodes["root"] = {"id": "root", "name": "Hyperbolic Explorer",
                     "type": "folder", "leaf": False, "kind": "root",
                     "children_count": len(root_items),
                     "summary": {"track_count": len(item_ids)},
                     "items": root_items}
    flat_ids["root"] = []
    return {"version": TREE_VERSION, "n_bands": len(root_items),
            "nodes": nodes, "flat_ids": flat_ids,
            "track_count": len(item_ids)}


def skeleton_of(tree: Dict) -> Dict:
    """Folder-only view (reference _skeleton_tree :582): non-leaf nodes
    keep their child summaries; leaf items stay in the full blob."""
    nodes = {nid: n for nid, n in (tree.get("nodes") or {}).items()
             if n.get("type") == "folder" and not n.get("leaf")}
    return {"version": tree.get("version"), "n_bands": tree.get("n_bands"),
            "nodes": nodes, "flat_ids": {},
            "track_count": tree.get("track_count") or 0}


def persist_tree(conn, tree: Dict) -> None:
    """Full + skeleton blobs (zlib json), via the segmented blob store."""
    for name, payload in ((TREE_BLOB, tree), (SKELETON_BLOB,
                                              skeleton_of(tree))):
        blob = zlib.compress(json.dumps(payload).encode(), level=6)
        store_index_blob(conn, name, blob, meta={"version": TREE_VERSION})


def _load_blob(conn, name: str) -> Optional[Dict]:
    got = load_index_blob(conn, name)
    if got is None:
        return None
    blob, _meta = got
    try:
        payload = json.loads(zlib.decompress(blob))
    except Exception:
        return None
    if payload.get("version") != TREE_VERSION:
        return None  # stale schema: rebuilt by the next analysis run
    return payload


class TreeCache:
    """Skeleton-resident, full-tree-on-demand cache with an unload
    timer (reference hyperbolic_manager.py:832-897)."""

    def __init__(self, warm_seconds: Optional[float] = None):
        self.warm_seconds = (warm_seconds if warm_seconds is not None
                             else C.HYPERBOLIC_TREE_WARMUP_DURATION)
        self._lock = threading.Lock()
        self._skeleton: Optional[Dict] = None
        self._full: Optional[Dict] = None
        self._timer: Optional[threading.Timer] = None
        self._expires_at = 0.0

    # ---- lifecycle ----

    def load_skeleton(self, conn) -> bool:
        with self._lock:
            self._skeleton = _load_blob(conn, SKELETON_BLOB)
            return self._skeleton is not None

    def _unload_full(self) -> None:
        with self._lock:
            self._full = None
            self._timer = None

    def _touch_warm(self) -> None:
        """(Re)start the unload countdown — caller holds the lock."""
        if self._timer is not None:
            self._timer.cancel()
        self._expires_at = time.time() + self.warm_seconds
        self._timer = threading.Timer(self.warm_seconds, self._unload_full)
        self._timer.daemon = True
        self._timer.start()

    def ensure_full(self, conn) -> Optional[Dict]:
        with self._lock:
            if self._full is None:
                self._full = _load_blob(conn, TREE_BLOB)
            if self._full is not None:
                self._touch_warm()
            return self._full

    def status(self) -> Dict:
        with self._lock:
            return {"skeleton_loaded": self._skeleton is not None,
                    "full_loaded": self._full is not None,
                    "warm_seconds_left": max(
                        0.0, round(self._expires_at - time.time(), 1))
                    if self._full is not None else 0.0,
                    "track_count": (self._skeleton or {}).get(
                        "track_count", 0)}

    # ---- queries ----

    def node(self, conn, node_id: str) -> Optional[Dict]:
        """Folder nodes serve from the skeleton (no warm); leaf items
        warm the full tree lazily."""
        with self._lock:
            skel = self._skeleton
        if skel is None:
            if not self.load_skeleton(conn):
                return None
            skel = self._skeleton
        hit = (skel.get("nodes") or {}).get(node_id)
        if hit is not None:
            return hit
        full = self.ensure_full(conn)
        if full is None:
            return None
        return (full.get("nodes") or {}).get(node_id)

    def flat_ids(self, conn, node_id: str) -> List[str]:
        full = self.ensure_full(conn)
        if full is None:
            return []
        return (full.get("flat_ids") or {}).get(node_id, [])
