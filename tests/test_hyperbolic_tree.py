"""Hyperbolic explorer tree cache (engines/hyperbolic_tree.py):
build structure, skeleton split, persist round trip, lazy warm/unload
lifecycle (reference: hyperbolic_manager.py:613-897)."""

import time

import numpy as np
import pytest
import torch

from audiomuse_amd.db import connect
from audiomuse_amd.db.schema import init_db
from audiomuse_amd.engines.hyperbolic_tree import (SKELETON_BLOB, TREE_BLOB,
                                                   TreeCache, build_tree,
                                                   persist_tree, skeleton_of)

MOODS = ["rock", "jazz", "pop"]


def _meta_fn(item_ids):
    def fn(item_id):
        i = int(item_id.split("_")[1])
        return {"title": f"Song {i}", "author": f"Artist {i % 7}",
                "mood_vector": {MOODS[i % 3]: 0.9}}
    return fn


def _build(n=120, dim=16, seed=0):
    torch.manual_seed(seed)
    emb = torch.randn(n, dim)
    ids = [f"fp_{i}" for i in range(n)]
    return build_tree(emb, ids, _meta_fn(ids)), ids


def test_tree_structure():
    tree, ids = _build()
    nodes = tree["nodes"]
    root = nodes["root"]
    assert root["children_count"] == 3          # one folder per mood
    assert tree["track_count"] == len(ids)
    for child in root["items"]:
        folder = nodes[child["id"]]
        assert folder["type"] == "folder" and not folder["leaf"]
        total = 0
        for leaf_ref in folder["items"]:
            leaf = nodes[leaf_ref["id"]]
            assert leaf["leaf"] and leaf["items"]
            assert tree["flat_ids"][leaf["id"]] == [
                t["item_id"] for t in leaf["items"]]
            # items sorted by hyperbolic radius
            radii = [t["radius"] for t in leaf["items"]]
            assert radii == sorted(radii)
            total += len(leaf["items"])
        assert total == folder["summary"]["track_count"]
    # every track appears exactly once across leaves
    all_ids = [i for leaf_ids in tree["flat_ids"].values() for i in leaf_ids]
    assert sorted(all_ids) == sorted(ids)


def test_big_folder_splits_into_leaf_clusters(monkeypatch):
    from audiomuse_amd import config as C

    monkeypatch.setattr(C, "HYPERBOLIC_TARGET_LEAF_SIZE", 20)
    monkeypatch.setattr(C, "HYPERBOLIC_MIN_CLUSTER_SIZE", 3)
    tree, _ = _build(n=300)
    root = tree["nodes"]["root"]
    folder = tree["nodes"][root["items"][0]["id"]]
    assert folder["children_count"] >= 2        # 100 tracks / 20 target


def test_skeleton_has_no_track_items():
    tree, _ = _build()
    skel = skeleton_of(tree)
    assert skel["flat_ids"] == {}
    assert "root" in skel["nodes"]
    assert all(not n.get("leaf") for n in skel["nodes"].values())
    # mood folders keep their child summaries so the UI can render
    # without warming the full tree
    mood = [n for nid, n in skel["nodes"].items() if nid != "root"][0]
    assert mood["items"] and all(c["leaf"] for c in mood["items"])


def test_persist_and_lazy_warm_unload(tmp_db_url):
    conn = connect(tmp_db_url)
    init_db(conn)
    tree, _ = _build()
    persist_tree(conn, tree)

    cache = TreeCache(warm_seconds=0.5)
    assert cache.load_skeleton(conn)
    st = cache.status()
    assert st["skeleton_loaded"] and not st["full_loaded"]

    # folder node: skeleton only, no warm
    root = cache.node(conn, "root")
    assert root is not None and not cache.status()["full_loaded"]

    # leaf node: lazily warms the full tree
    leaf_id = next(iter(tree["flat_ids"].keys() - {"root"}))
    leaf_id = [nid for nid, n in tree["nodes"].items() if n.get("leaf")][0]
    leaf = cache.node(conn, leaf_id)
    assert leaf is not None and leaf["items"]
    assert cache.status()["full_loaded"]
    assert cache.flat_ids(conn, leaf_id) == tree["flat_ids"][leaf_id]

    # warm timer unloads the full tree but keeps the skeleton
    time.sleep(0.8)
    st = cache.status()
    assert st["skeleton_loaded"] and not st["full_loaded"]
    # access re-warms
    assert cache.node(conn, leaf_id) is not None
    assert cache.status()["full_loaded"]
    conn.close()


def test_stale_version_discarded(tmp_db_url):
    import json
    import zlib

    from audiomuse_amd.db.store import store_index_blob

    conn = connect(tmp_db_url)
    init_db(conn)
    old = {"version": 1, "nodes": {"root": {}}, "flat_ids": {},
           "track_count": 5}
    store_index_blob(conn, SKELETON_BLOB,
                     zlib.compress(json.dumps(old).encode()))
    store_index_blob(conn, TREE_BLOB,
                     zlib.compress(json.dumps(old).encode()))
    cache = TreeCache()
    assert not cache.load_skeleton(conn)     # stale schema rejected
    assert cache.node(conn, "root") is None
    conn.close()
